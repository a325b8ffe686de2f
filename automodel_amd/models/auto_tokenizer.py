"""Tokenizer wrapper over HF transformers (local files; no hub access needed).

Reference behavior: nemo_automodel/_transformers/auto_tokenizer.py
(NeMoAutoTokenizer: AutoTokenizer + chat-template helpers + pad handling).
"""

from __future__ import annotations

from typing import Any


class AutoTokenizer:
    @staticmethod
    def from_pretrained(path: str, **kw) -> Any:
        from transformers import AutoTokenizer as HFTok

        tok = HFTok.from_pretrained(path, local_files_only=True, **kw)
        if tok.pad_token_id is None and tok.eos_token_id is not None:
            tok.pad_token = tok.eos_token
        return tok


def apply_chat_template(tokenizer, messages: list[dict], add_generation_prompt: bool = False):
    """Chat-template application with a plain-text fallback when the
    tokenizer ships no template."""
    if getattr(tokenizer, "chat_template", None):
        return tokenizer.apply_chat_template(
            messages, add_generation_prompt=add_generation_prompt, tokenize=True)
    text = "".join(f"<|{m['role']}|>{m['content']}\n" for m in messages)
    if add_generation_prompt:
        text += "<|assistant|>"
    return tokenizer.encode(text)
