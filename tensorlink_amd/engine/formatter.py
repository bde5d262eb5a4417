"""Generation-arg normalization, chat templating and OpenAI-style response
shaping.

Functional port of the reference's ``tensorlink/ml/formatter.py`` surface
(same behaviors, re-written):
- ``normalize_generate_args`` — pad/eos fallbacks, max_new_tokens clamping,
  temperature clamp 0.01-2.0, beam/sample conflict resolution
  (``formatter.py:7-115``);
- ``format_chat_prompt`` — tokenizer chat template with manual Qwen/Llama
  fallbacks (``formatter.py:161-235,238-324``);
- ``extract_reasoning_and_answer`` — ``<think>`` block split
  (``formatter.py:118-158``);
- ``ResponseFormatter`` — simple/openai non-streaming + SSE chunk shaping
  (``formatter.py:327-550``).
"""

from __future__ import annotations

import json
import time
import uuid
from typing import Dict, List, Optional, Tuple

MAX_NEW_TOKENS_CAP = 2048
DEFAULT_MAX_LENGTH = 2048


def normalize_generate_args(args: Dict, tokenizer=None,
                            model_max_length: int = DEFAULT_MAX_LENGTH) -> Dict:
    """Clamp and default the user-supplied generation arguments."""
    out = dict(args or {})

    mnt = out.get("max_new_tokens", 256)
    try:
        mnt = int(mnt)
    except (TypeError, ValueError):
        mnt = 256
    out["max_new_tokens"] = max(1, min(mnt, MAX_NEW_TOKENS_CAP))

    temp = out.get("temperature", 0.7)
    try:
        temp = float(temp)
    except (TypeError, ValueError):
        temp = 0.7
    out["temperature"] = min(max(temp, 0.01), 2.0)

    top_p = out.get("top_p", 1.0)
    try:
        top_p = float(top_p)
    except (TypeError, ValueError):
        top_p = 1.0
    out["top_p"] = min(max(top_p, 0.0), 1.0)

    do_sample = bool(out.get("do_sample", True))
    num_beams = int(out.get("num_beams", 1) or 1)
    for k in ("presence_penalty", "frequency_penalty"):
        out[k] = max(-2.0, min(2.0, float(out.get(k, 0.0) or 0.0)))
    if num_beams > 1 and do_sample:
        # beam search and sampling conflict: sampling wins (reference
        # formatter.py:94-99 resolves the same way)
        num_beams = 1
    out["do_sample"] = do_sample
    out["num_beams"] = num_beams
    if not do_sample:
        out["temperature"] = 0.0     # greedy

    if tokenizer is not None:
        if out.get("eos_token_id") is None:
            out["eos_token_id"] = getattr(tokenizer, "eos_token_id", None)
        if out.get("pad_token_id") is None:
            out["pad_token_id"] = (getattr(tokenizer, "pad_token_id", None)
                                   or out.get("eos_token_id"))
    return out


def format_chat_prompt(messages: List[Dict], tokenizer=None,
                       model_name: str = "", add_generation_prompt: bool = True
                       ) -> str:
    """Render a chat history to a prompt string.

    Uses the tokenizer's chat template when available; otherwise falls back
    to manual ChatML (Qwen) or Llama-3 templates keyed on the model name.
    """
    if tokenizer is not None and getattr(tokenizer, "chat_template", None):
        try:
            return tokenizer.apply_chat_template(
                messages, tokenize=False,
                add_generation_prompt=add_generation_prompt)
        except Exception:
            pass
    name = (model_name or "").lower()
    if "llama" in name:
        parts = ["<|begin_of_text|>"]
        for m in messages:
            parts.append(f"<|start_header_id|>{m['role']}<|end_header_id|>"
                         f"\n\n{m['content']}<|eot_id|>")
        if add_generation_prompt:
            parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
        return "".join(parts)
    # default: ChatML (Qwen convention)
    parts = []
    for m in messages:
        parts.append(f"<|im_start|>{m['role']}\n{m['content']}<|im_end|>\n")
    if add_generation_prompt:
        parts.append("<|im_start|>assistant\n")
    return "".join(parts)


def extract_reasoning_and_answer(text: str) -> Tuple[str, str]:
    """Split '<think>...</think>answer' into (reasoning, answer)."""
    start = text.find("<think>")
    if start == -1:
        return "", text.strip()
    end = text.find("</think>", start)
    if end == -1:
        # unterminated think block: everything after the tag is reasoning
        return text[start + len("<think>"):].strip(), ""
    reasoning = text[start + len("<think>"):end].strip()
    answer = (text[:start] + text[end + len("</think>"):]).strip()
    return reasoning, answer


class ResponseFormatter:
    """Shapes generation output as simple/openai/raw JSON and SSE chunks."""

    def __init__(self, model_name: str = "", output_format: str = "openai"):
        self.model_name = model_name
        self.output_format = output_format
        self.request_id = f"cmpl-{uuid.uuid4().hex[:24]}"
        self.created = int(time.time())

    # ---- non-streaming ----
    def format_response(self, text: str, *, prompt_tokens: int = 0,
                        completion_tokens: int = 0, reasoning: bool = False,
                        processing_time: Optional[float] = None,
                        extra_texts: Optional[list] = None) -> Dict:
        """extra_texts: additional sampled completions (OpenAI ``n>1``
        — the reference declares ``n`` in its schema,
        ``api/models.py:68``, but never produces extra choices)."""
        if reasoning:
            think, answer = extract_reasoning_and_answer(text)
        else:
            think, answer = "", text
        if self.output_format == "raw":
            return {"text": text}
        if self.output_format == "simple":
            out = {"response": answer, "model": self.model_name}
            if think:
                out["reasoning"] = think
            if processing_time is not None:
                out["processing_time"] = processing_time
            return out
        message = {"role": "assistant", "content": answer}
        if think:
            message["reasoning_content"] = think
        choices = [{"index": 0, "message": message,
                    "finish_reason": "stop"}]
        for i, extra in enumerate(extra_texts or [], start=1):
            choices.append({"index": i,
                            "message": {"role": "assistant",
                                        "content": extra},
                            "finish_reason": "stop"})
        return {
            "id": self.request_id,
            "object": "chat.completion",
            "created": self.created,
            "model": self.model_name,
            "choices": choices,
            "usage": {"prompt_tokens": prompt_tokens,
                      "completion_tokens": completion_tokens,
                      "total_tokens": prompt_tokens + completion_tokens},
        }

    # ---- streaming (SSE) ----
    def format_stream_chunk(self, token_text: str,
                            first: bool = False) -> str:
        if self.output_format == "simple":
            payload = {"token": token_text}
        else:
            delta = {"content": token_text}
            if first:
                delta["role"] = "assistant"
            payload = {
                "id": self.request_id,
                "object": "chat.completion.chunk",
                "created": self.created,
                "model": self.model_name,
                "choices": [{"index": 0, "delta": delta,
                             "finish_reason": None}],
            }
        return f"data: {json.dumps(payload)}\n\n"

    def format_final_chunk(self, finish_reason: str = "stop") -> str:
        if self.output_format == "simple":
            return "data: [DONE]\n\n"
        payload = {
            "id": self.request_id,
            "object": "chat.completion.chunk",
            "created": self.created,
            "model": self.model_name,
            "choices": [{"index": 0, "delta": {},
                         "finish_reason": finish_reason}],
        }
        return f"data: {json.dumps(payload)}\n\ndata: [DONE]\n\n"

    def format_error(self, message: str, code: int = 500) -> Dict:
        return {"error": {"message": message, "type": "server_error",
                          "code": code}}
