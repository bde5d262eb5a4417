"""Tokenizer loading with an offline fallback.

The reference loads HF AutoTokenizer from the hub per hosted job
(``tensorlink/ml/validator.py:1006-1024``). This node has no network, so:
local checkpoint dir -> HF tokenizer; otherwise a deterministic byte-level
tokenizer so the serving stack (templates, SSE, decode loop) runs
end-to-end on synthetic/random-weight models.
"""

from __future__ import annotations

import os
from typing import List


class ByteTokenizer:
    """Reversible byte-level tokenizer: token = byte value (+ specials)."""

    vocab_size = 260
    bos_token_id = 256
    eos_token_id = 257
    pad_token_id = 258
    chat_template = None

    def __call__(self, text, return_tensors=None, **kw):
        ids = self.encode(text)
        if return_tensors == "pt":
            import torch
            return {"input_ids": torch.tensor([ids], dtype=torch.long)}
        return {"input_ids": ids}

    def encode(self, text: str, **kw) -> List[int]:
        return list(text.encode("utf-8", errors="replace"))

    def decode(self, ids, skip_special_tokens: bool = True, **kw) -> str:
        out = bytes(i for i in _to_list(ids) if 0 <= i < 256)
        return out.decode("utf-8", errors="replace")

    def apply_chat_template(self, messages, tokenize=False,
                            add_generation_prompt=True):
        from tensorlink_amd.engine.formatter import format_chat_prompt
        return format_chat_prompt(messages, tokenizer=None)


def _to_list(ids):
    try:
        return ids.tolist()
    except AttributeError:
        return list(ids)


def load_tokenizer(model_name_or_dir: str):
    """HF tokenizer from a local checkpoint dir, else ByteTokenizer."""
    if os.path.isdir(model_name_or_dir):
        try:
            from transformers import AutoTokenizer
            return AutoTokenizer.from_pretrained(model_name_or_dir,
                                                 local_files_only=True)
        except Exception:
            pass
    return ByteTokenizer()
