"""Tokenizer adapters: transformers AutoTokenizer or the built-in byte
tokenizer (synthetic/CI models — no network, any vocab size).

Reference counterpart: mlx_lm load_tokenizer in
src/dnet/api/model_manager.py:170-173 + the incremental detokenizer used by
InferenceManager.
"""
from __future__ import annotations

from typing import Optional


class ByteTokenizer:
    """Reversible byte-level tokenizer: token = byte value; ids >= 256 are
    specials. Works with any model vocab >= 258."""

    def __init__(self, vocab_size: int = 512):
        self.vocab_size = vocab_size
        # specials live above the byte range when the vocab allows, else at
        # the top of the vocab (synthetic models with tiny vocabs)
        self.BOS = 256 if vocab_size >= 258 else vocab_size - 2
        self.EOS = 257 if vocab_size >= 258 else vocab_size - 1
        self.eos_token_id = self.EOS
        self.bos_token_id = self.BOS

    def apply_chat_template(self, messages, add_generation_prompt=True,
                            tokenize=False):
        text = ""
        for m in messages:
            content = m["content"] if isinstance(m, dict) else m.content
            role = m["role"] if isinstance(m, dict) else m.role
            text += f"<{role}>{content}</{role}>\n"
        if add_generation_prompt:
            text += "<assistant>"
        return text

    def encode(self, text: str, add_special_tokens: bool = True) -> list[int]:
        ids = [min(i, self.vocab_size - 3 if self.vocab_size < 258 else 255)
               for i in text.encode("utf-8", errors="replace")]
        return ([self.BOS] if add_special_tokens else []) + ids

    def decode(self, ids) -> str:
        # ids beyond the byte range (synthetic models sample the full model
        # vocab) render as a placeholder so streams stay visible
        out: list[str] = []
        buf: list[int] = []
        for i in ids:
            if i in (self.BOS, self.EOS):
                continue
            if 0 <= i < 256:
                buf.append(i)
            else:
                if buf:
                    out.append(bytes(buf).decode("utf-8", errors="replace"))
                    buf = []
                out.append("·")
        if buf:
            out.append(bytes(buf).decode("utf-8", errors="replace"))
        return "".join(out)


class Detokenizer:
    """Incremental detokenizer: add_token -> last_segment string delta."""

    def __init__(self, tokenizer):
        self.tokenizer = tokenizer
        self.ids: list[int] = []
        self._text = ""

    def add_token(self, tid: int) -> str:
        self.ids.append(tid)
        full = self.tokenizer.decode(self.ids)
        # hold back while the tail may be an incomplete utf-8 / merge point
        if full.endswith("�"):
            return ""
        delta = full[len(self._text):]
        self._text = full
        return delta

    @property
    def text(self) -> str:
        return self._text


def load_tokenizer(entry_tokenizer: str, model_path: str,
                   vocab_size: int = 512):
    if entry_tokenizer == "byte":
        return ByteTokenizer(vocab_size)
    try:
        from transformers import AutoTokenizer
        return AutoTokenizer.from_pretrained(model_path)
    except Exception:
        return ByteTokenizer(vocab_size)


def stop_token_ids(tokenizer) -> list[int]:
    ids = []
    for attr in ("eos_token_id",):
        v = getattr(tokenizer, attr, None)
        if v is None:
            continue
        ids.extend(v if isinstance(v, (list, tuple)) else [v])
    return ids
