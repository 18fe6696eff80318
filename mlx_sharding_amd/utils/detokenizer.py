"""Incremental streaming detokenizer over a HF tokenizer.

Replacement for mlx_lm's TokenizerWrapper/streaming detokenizer the
reference relies on (/root/reference/generate.py:96-109): emits text
deltas as tokens arrive, holding back segments that end in an
incomplete UTF-8 sequence (U+FFFD) until more tokens resolve them.
"""

from __future__ import annotations

from typing import List


class StreamingDetokenizer:
    def __init__(self, tokenizer):
        self.tokenizer = tokenizer
        self.tokens: List[int] = []
        self._emitted = ""

    def reset(self):
        self.tokens = []
        self._emitted = ""

    def add_token(self, token_id: int) -> str:
        """Add one token; return newly-finalized text (may be '')."""
        self.tokens.append(token_id)
        text = self.tokenizer.decode(self.tokens)
        if text.endswith("�"):
            return ""
        delta = text[len(self._emitted):]
        self._emitted = text
        return delta

    def finalize(self) -> str:
        text = self.tokenizer.decode(self.tokens)
        delta = text[len(self._emitted):]
        self._emitted = text
        return delta

    @property
    def text(self) -> str:
        return self._emitted
