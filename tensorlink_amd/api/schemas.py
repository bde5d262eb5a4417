"""Wire schemas — same request surface as the reference
(``tensorlink/api/models.py:9-84``)."""

from __future__ import annotations

from typing import Dict, List, Optional

from pydantic import BaseModel


class JobRequest(BaseModel):
    hf_name: str
    model_type: Optional[str] = None
    training: bool = False
    payment: int = 0
    # serving options (beyond the reference's schema)
    continuous: bool = False          # continuous batching
    max_slots: int = 16
    max_ctx: int = 4096
    prefill_chunk: Optional[int] = None
    prefix_caching: bool = False
    quantize: Optional[str] = None    # fp8 | fp8-dense | fp4-dense


class GenerationRequest(BaseModel):
    hf_name: str
    message: str = ""
    prompt: Optional[str] = None      # reference alias for message
    model_type: Optional[str] = None  # architecture hint (auto-detected)
    is_chat_completion: bool = False  # reference: openai response shape
    max_length: int = 2048
    max_new_tokens: int = 256
    temperature: float = 0.7
    top_p: float = 1.0
    top_k: int = 0
    do_sample: bool = True
    num_beams: int = 1
    n: int = 1                        # OpenAI multiple completions
    speculative: bool = False         # prompt-lookup speculative decode
    seed: Optional[int] = None        # deterministic sampling seed
    logprobs: bool = False            # return chosen-token logprobs
    presence_penalty: float = 0.0     # OpenAI range [-2, 2]
    frequency_penalty: float = 0.0
    stream: bool = False
    reasoning: bool = False
    stop: Optional[List[str]] = None
    history: Optional[List[Dict[str, str]]] = None
    input_format: str = "chat"       # chat | raw
    output_format: str = "simple"    # simple | openai | raw


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: str
    messages: List[ChatMessage]
    max_tokens: Optional[int] = 256
    max_completion_tokens: Optional[int] = None
    temperature: float = 0.7
    top_p: float = 1.0
    stream: bool = False
    n: int = 1
    stop: Optional[List[str]] = None
    user: Optional[str] = None


class ModelStatusResponse(BaseModel):
    model: str
    status: str
    requests_served: int = 0
    tokens_generated: int = 0
