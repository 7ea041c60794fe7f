"""Tokenizer abstraction.

Real checkpoints: HF tokenizers via transformers (tokenizer files on the
model PVC). Synthetic/random-init presets (no tokenizer files — this
environment has no network): a deterministic byte-level fallback so the HTTP
path works end to end.
"""

from __future__ import annotations

import os


class ByteTokenizer:
    """Deterministic byte-level tokenizer: token id = byte value + offset.
    Reserves ids 0..3 for special tokens."""

    OFFSET = 4

    def __init__(self, vocab_size: int, eos_token_id: int):
        self.vocab_size = vocab_size
        self.eos_token_id = eos_token_id

    def encode(self, text: str) -> list[int]:
        return [min(b + self.OFFSET, self.vocab_size - 1) for b in text.encode("utf-8")]

    def decode(self, ids: list[int]) -> str:
        bs = bytes(
            max(0, min(255, i - self.OFFSET))
            for i in ids
            if i >= self.OFFSET and i != self.eos_token_id
        )
        return bs.decode("utf-8", errors="replace")

    def apply_chat_template(self, messages: list[dict]) -> str:
        parts = [f"{m['role']}: {m['content']}" for m in messages]
        parts.append("assistant:")
        return "\n".join(parts)


class HFTokenizer:
    def __init__(self, model_path: str):
        from transformers import AutoTokenizer

        self.tok = AutoTokenizer.from_pretrained(model_path)
        self.eos_token_id = self.tok.eos_token_id

    def encode(self, text: str) -> list[int]:
        return self.tok.encode(text)

    def decode(self, ids: list[int]) -> str:
        return self.tok.decode(ids, skip_special_tokens=True)

    def apply_chat_template(self, messages: list[dict]) -> str:
        try:
            return self.tok.apply_chat_template(
                messages, tokenize=False, add_generation_prompt=True
            )
        except Exception:
            parts = [f"{m['role']}: {m['content']}" for m in messages]
            parts.append("assistant:")
            return "\n".join(parts)


def load_tokenizer(model_path: str | None, vocab_size: int, eos_token_id: int):
    if model_path and (
        os.path.exists(os.path.join(model_path, "tokenizer.json"))
        or os.path.exists(os.path.join(model_path, "tokenizer_config.json"))
    ):
        try:
            return HFTokenizer(model_path)
        except Exception:
            pass
    return ByteTokenizer(vocab_size, eos_token_id)


class IncrementalDetokenizer:
    """Streaming-safe detokenization: decoding token-by-token glues
    sentencepiece/BPE pieces wrongly (missing spaces, broken multibyte
    runes). vLLM-style offsets: decode a short suffix window and emit only
    complete text, holding back incomplete runes (U+FFFD at the cut)."""

    def __init__(self, tokenizer):
        self.tok = tokenizer
        self.ids: list[int] = []
        self.prefix_offset = 0
        self.read_offset = 0

    MAX_HOLD = 4  # never withhold more than this many tokens

    def feed(self, token_id: int) -> str:
        self.ids.append(token_id)
        prefix_text = self.tok.decode(self.ids[self.prefix_offset:self.read_offset])
        new_text = self.tok.decode(self.ids[self.prefix_offset:])
        complete = not new_text.endswith("\ufffd")
        overdue = len(self.ids) - self.read_offset >= self.MAX_HOLD
        if len(new_text) > len(prefix_text) and (complete or overdue):
            piece = new_text[len(prefix_text):]
            self.prefix_offset = self.read_offset
            self.read_offset = len(self.ids)
            return piece
        return ""

    def flush(self) -> str:
        """Emit whatever is still held (final chunk: incomplete runes too)."""
        prefix_text = self.tok.decode(self.ids[self.prefix_offset:self.read_offset])
        new_text = self.tok.decode(self.ids[self.prefix_offset:])
        self.prefix_offset = self.read_offset = len(self.ids)
        return new_text[len(prefix_text):]
