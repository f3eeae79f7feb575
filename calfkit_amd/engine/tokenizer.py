"""Deterministic synthetic tokenizer.

There is no network for real tokenizer downloads and BASELINE runs on
synthetic data / random-init weights, so tokenization just needs to be
deterministic, stable across processes, and shaped like the real thing
(BOS/EOS, roughly word-per-token).
"""

from __future__ import annotations

import hashlib
import re
from functools import lru_cache

__all__ = ["SyntheticTokenizer"]

_WORD = re.compile(r"\S+")


@lru_cache(maxsize=65536)
def _word_id(word: str, span: int) -> int:
    # cached: agent prompts repeat words heavily (a 512-turn rollover
    # tokenizes ~65k words; the distinct set is a few hundred)
    h = hashlib.blake2s(word.encode(), digest_size=4).digest()
    return int.from_bytes(h, "little") % span


class SyntheticTokenizer:
    BOS = 1
    EOS = 2
    #: tool-call grammar specials (ops.G_TCALL / G_TEND — the engine's
    #: constrained sampler steers generation through
    #: ``TCALL <name> <args> TEND``; see engine/local.py)
    TCALL = 5
    TEND = 6
    _RESERVED = 16  # ids 0..15 reserved for specials

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size

    def word_id(self, word: str) -> int:
        """Token id a single word encodes to (tool names are single words)."""
        return self._RESERVED + _word_id(word, self.vocab_size - self._RESERVED)

    def encode(self, text: str, *, bos: bool = True) -> list[int]:
        ids = [self.BOS] if bos else []
        span = self.vocab_size - self._RESERVED
        for m in _WORD.finditer(text):
            ids.append(self._RESERVED + _word_id(m.group(), span))
        return ids

    def decode(self, ids: list[int]) -> str:
        return " ".join(f"t{i}" for i in ids if i >= self._RESERVED)
