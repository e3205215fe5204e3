"""Tokenizer registry with L0 exact-match cache (reference: crates/tokenizer —
registry.rs TokenizerRegistry, cache/mod.rs L0 whole-string cache l0.rs,
huggingface.rs encode_batch :485).

Backends:
  * HuggingFace `tokenizers` (Rust core, available in-image) for real vocabs;
  * MockTokenizer (deterministic hash tokenization) for tests (mock.rs);
  * the gfx950 batch-BPE HIP kernel (smg_amd._core) services encode_batch on
    MI355X (ops/bpe kernel) with this registry as the host-side vocabulary
    loader — wired in via GpuBpeTokenizer when the extension is built.
"""
from __future__ import annotations

import collections
import threading
from typing import Dict, List, Optional


class L0Cache:
    """Exact-match encode cache: whole input string -> token ids (zero-copy hits)."""

    def __init__(self, max_entries: int = 8192):
        self.max_entries = max_entries
        self._map: "collections.OrderedDict[str, List[int]]" = collections.OrderedDict()
        self.hits = 0
        self.misses = 0
        self._lock = threading.Lock()

    def get(self, text: str) -> Optional[List[int]]:
        with self._lock:
            ids = self._map.get(text)
            if ids is not None:
                self.hits += 1
                self._map.move_to_end(text)
                return ids
            self.misses += 1
            return None

    def put(self, text: str, ids: List[int]) -> None:
        with self._lock:
            self._map[text] = ids
            self._map.move_to_end(text)
            while len(self._map) > self.max_entries:
                self._map.popitem(last=False)


class MockTokenizer:
    """Deterministic hash tokenizer (reference mock.rs): 1 token / 4 chars."""

    vocab_size = 32768
    model_max_length = 131072
    name = "mock"

    def __init__(self, name: str = "mock"):
        self.name = name

    def encode(self, text: str) -> List[int]:
        return [(hash(text[i : i + 4]) & 0x7FFF) for i in range(0, len(text), 4)]

    def encode_batch(self, texts: List[str]) -> List[List[int]]:
        return [self.encode(t) for t in texts]

    def decode(self, ids: List[int]) -> str:
        return "".join(f" tok{i}" for i in ids)

    def decode_incremental(self, ids: List[int], prefix_len: int) -> str:
        return "".join(f" tok{i}" for i in ids[prefix_len:])


class HFTokenizer:
    """HuggingFace-backed tokenizer with the L0 cache in front."""

    def __init__(self, path: str, name: Optional[str] = None, l0_entries: int = 8192):
        from tokenizers import Tokenizer

        import os

        if os.path.isdir(path):
            path = os.path.join(path, "tokenizer.json")
        self._tk = Tokenizer.from_file(path)
        self.name = name or path
        self.vocab_size = self._tk.get_vocab_size()
        self.model_max_length = 1 << 20
        self.l0 = L0Cache(l0_entries)

    def encode(self, text: str) -> List[int]:
        cached = self.l0.get(text)
        if cached is not None:
            return cached
        ids = self._tk.encode(text, add_special_tokens=False).ids
        self.l0.put(text, ids)
        return ids

    def encode_batch(self, texts: List[str]) -> List[List[int]]:
        out: List[Optional[List[int]]] = [self.l0.get(t) for t in texts]
        missing = [(i, t) for i, t in enumerate(texts) if out[i] is None]
        if missing:
            encs = self._tk.encode_batch([t for _, t in missing], add_special_tokens=False)
            for (i, t), e in zip(missing, encs):
                out[i] = e.ids
                self.l0.put(t, e.ids)
        return out  # type: ignore[return-value]

    def decode(self, ids: List[int]) -> str:
        return self._tk.decode(ids, skip_special_tokens=False)

    def decode_incremental(self, ids: List[int], prefix_len: int) -> str:
        # standard two-window incremental detokenization
        full = self._tk.decode(ids, skip_special_tokens=False)
        prev = self._tk.decode(ids[:prefix_len], skip_special_tokens=False)
        return full[len(prev):]


class TokenizerRegistry:
    """Runtime add/remove of tokenizers (reference registry.rs, REST-managed)."""

    def __init__(self):
        self._tokenizers: Dict[str, object] = {}
        self._default: Optional[str] = None

    def load(self, name: str, path: str, chat_template: Optional[str] = None):
        if path == "mock":
            tok = MockTokenizer(name)
        else:
            # hub-aware family dispatch: HF cache resolution + tiktoken-format
            # detection (incl. Kimi-K2), else the tokenizer.json loader
            from .hub import load_tokenizer

            tok = load_tokenizer(path, name)
        if chat_template:
            tok.chat_template = chat_template
        self._tokenizers[name] = tok
        if self._default is None:
            self._default = name
        return tok

    def add(self, name: str, tok) -> None:
        self._tokenizers[name] = tok
        if self._default is None:
            self._default = name

    def get(self, name: Optional[str] = None):
        if name and name in self._tokenizers:
            return self._tokenizers[name]
        if self._default is not None:
            return self._tokenizers[self._default]
        return None

    def remove(self, name: str) -> bool:
        if name in self._tokenizers:
            del self._tokenizers[name]
            if self._default == name:
                self._default = next(iter(self._tokenizers), None)
            return True
        return False

    def list(self) -> List[dict]:
        return [
            {"id": n, "vocab_size": getattr(t, "vocab_size", None)} for n, t in self._tokenizers.items()
        ]

    def info(self, name: str) -> Optional[dict]:
        t = self._tokenizers.get(name)
        if t is None:
            return None
        return {"id": name, "vocab_size": getattr(t, "vocab_size", None), "status": "ready"}
