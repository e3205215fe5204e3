"""L1 fixed-boundary prefix cache for tokenization (reference:
crates/tokenizer/src/cache/l1.rs (855 LoC); cache/mod.rs L0+L1 design).

Long shared prompt prefixes (system prompts, few-shot headers) are encoded
once: the cache stores token ids for text prefixes cut at fixed boundaries,
and an encode call re-tokenizes only the suffix.  Correctness: a prefix is
only reused when the cut falls on a whitespace boundary in BOTH the cached
text and the query (byte-level BPE never merges across the pre-tokenizer's
whitespace splits, so token sequences compose at those cuts).
"""
from __future__ import annotations

import hashlib
from typing import Dict, List, Optional, Tuple


def _h(text: str) -> bytes:
    return hashlib.blake2b(text.encode(), digest_size=16).digest()


class L1PrefixCache:
    BOUNDARIES = (256, 512, 1024, 2048, 4096, 8192)

    def __init__(self, max_memory: int = 64 << 20):
        self.max_memory = max_memory
        self._entries: Dict[bytes, Tuple[List[int], int]] = {}  # hash -> (tokens, chars)
        self._memory = 0
        self._order: List[bytes] = []  # LRU
        self.hits = 0
        self.misses = 0

    @staticmethod
    def _safe_cut(text: str, boundary: int) -> Optional[int]:
        """Largest whitespace-aligned cut <= boundary (cut BEFORE the space so
        the suffix keeps its leading-space pre-token)."""
        if boundary >= len(text):
            return None
        i = boundary
        while i > 0 and not text[i].isspace():
            i -= 1
        return i if i > 0 else None

    def lookup(self, text: str) -> Optional[Tuple[List[int], int]]:
        """(prefix_tokens, chars_consumed) for the longest cached prefix."""
        for b in reversed(self.BOUNDARIES):
            cut = self._safe_cut(text, b)
            if cut is None:
                continue
            key = _h(text[:cut])
            entry = self._entries.get(key)
            if entry is not None:
                self.hits += 1
                try:
                    self._order.remove(key)
                except ValueError:
                    pass
                self._order.append(key)
                return entry
        self.misses += 1
        return None

    def store(self, text: str, encode_fn) -> None:
        """Cache whitespace-aligned prefixes of `text` at each boundary."""
        for b in self.BOUNDARIES:
            cut = self._safe_cut(text, b)
            if cut is None:
                break
            key = _h(text[:cut])
            if key in self._entries:
                continue
            tokens = encode_fn(text[:cut])
            cost = len(tokens) * 8 + cut
            self._entries[key] = (tokens, cut)
            self._order.append(key)
            self._memory += cost
        while self._memory > self.max_memory and self._order:
            victim = self._order.pop(0)
            toks, chars = self._entries.pop(victim)
            self._memory -= len(toks) * 8 + chars


class L1CachedTokenizer:
    """Wraps any tokenizer with the L1 prefix cache (+ pass-through decode)."""

    def __init__(self, inner, max_memory: int = 64 << 20, store_threshold: int = 256):
        self.inner = inner
        self.l1 = L1PrefixCache(max_memory)
        self.store_threshold = store_threshold
        self.name = getattr(inner, "name", "l1")
        self.vocab_size = getattr(inner, "vocab_size", None)
        self.model_max_length = getattr(inner, "model_max_length", 1 << 20)

    def encode(self, text: str) -> List[int]:
        hit = self.l1.lookup(text)
        if hit is not None:
            prefix_tokens, chars = hit
            return prefix_tokens + self.inner.encode(text[chars:])
        out = self.inner.encode(text)
        if len(text) >= self.store_threshold:
            self.l1.store(text, self.inner.encode)
        return out

    def encode_batch(self, texts: List[str]) -> List[List[int]]:
        return [self.encode(t) for t in texts]

    def decode(self, ids: List[int]) -> str:
        return self.inner.decode(ids)

    def decode_incremental(self, ids: List[int], prefix_len: int) -> str:
        return self.inner.decode_incremental(ids, prefix_len)
