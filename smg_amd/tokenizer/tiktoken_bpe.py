"""Tiktoken-format BPE tokenizer (reference: crates/tokenizer/src/tiktoken.rs
+ kimi_k2_tokenizer.rs).

Loads `.tiktoken` vocab files (base64-token<space>rank per line), pre-splits
with the cl100k_base pattern (or the Kimi-K2 Han-aware pattern when the model
dir references tokenization_kimi), and byte-pair-encodes each piece with the
standard greedy lowest-rank merge.  Special tokens come from
tokenizer_config.json's added_tokens_decoder.  No tiktoken package needed —
the algorithm is ~60 lines once the vocab is a bytes->rank map."""
from __future__ import annotations

import base64
import json
import os
from typing import Dict, List, Optional, Sequence, Tuple

import regex

# reference tiktoken.rs:41 — correct for OpenAI + most open tiktoken models
CL100K_BASE_PATTERN = (
    r"(?i:'s|'t|'re|'ve|'m|'ll|'d)|[^\r\n\p{L}\p{N}]?\p{L}+|\p{N}{1,3}"
    r"| ?[^\s\p{L}\p{N}]+[\r\n]*|\s*[\r\n]+|\s+(?!\S)|\s+"
)

# reference kimi_k2_tokenizer.rs:26 — Han runs split out, case-aware Latin.
# The Rust `[\p{Lu}..&&[^\p{Han}]]` intersection classes are written with the
# python `regex` module's V1 set operations.
KIMI_K2_PATTERN = (
    r"[\p{Han}]+"
    r"|[^\r\n\p{L}\p{N}]?[[\p{Lu}\p{Lt}\p{Lm}\p{Lo}\p{M}]--[\p{Han}]]*"
    r"[[\p{Ll}\p{Lm}\p{Lo}\p{M}]--[\p{Han}]]+(?i:'s|'t|'re|'ve|'m|'ll|'d)?"
    r"|[^\r\n\p{L}\p{N}]?[[\p{Lu}\p{Lt}\p{Lm}\p{Lo}\p{M}]--[\p{Han}]]+"
    r"[[\p{Ll}\p{Lm}\p{Lo}\p{M}]--[\p{Han}]]*(?i:'s|'t|'re|'ve|'m|'ll|'d)?"
    r"|\p{N}{1,3}| ?[^\s\p{L}\p{N}]+[\r\n]*|\s*[\r\n]+|\s+(?!\S)|\s+"
)


def load_tiktoken_bpe(path: str) -> Dict[bytes, int]:
    """Parse a .tiktoken file: one `base64(token) rank` pair per line."""
    encoder: Dict[bytes, int] = {}
    with open(path, "rb") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            tok_b64, rank = line.split()
            encoder[base64.b64decode(tok_b64)] = int(rank)
    return encoder


class CoreBPE:
    """The tiktoken merge core: greedy lowest-rank adjacent-pair merging."""

    def __init__(self, encoder: Dict[bytes, int], special_tokens: Dict[str, int],
                 pattern: str = CL100K_BASE_PATTERN):
        self.encoder = encoder
        self.decoder = {v: k for k, v in encoder.items()}
        self.special_tokens = dict(special_tokens)
        self.special_decoder = {v: k.encode() for k, v in special_tokens.items()}
        self.pat = regex.compile(pattern, regex.V1)
        self.special_pat = (
            regex.compile("|".join(regex.escape(s) for s in
                                   sorted(special_tokens, key=len, reverse=True)))
            if special_tokens else None
        )

    def _bpe(self, piece: bytes) -> List[int]:
        enc = self.encoder
        rank = enc.get(piece)
        if rank is not None:
            return [rank]
        # parts[i] = (start, rank of merging parts[i] with parts[i+1])
        parts: List[Tuple[int, int]] = []
        n = len(piece)
        INF = 1 << 60
        for i in range(n - 1):
            parts.append((i, enc.get(piece[i: i + 2], INF)))
        parts.append((n - 1, INF))
        parts.append((n, INF))

        def pair_rank(i: int) -> int:
            if i + 2 >= len(parts):
                return INF
            return enc.get(piece[parts[i][0]: parts[i + 2][0]], INF)

        while len(parts) > 2:
            best_i, best_r = -1, INF
            for i in range(len(parts) - 1):
                if parts[i][1] < best_r:
                    best_r = parts[i][1]
                    best_i = i
            if best_r == INF:
                break
            del parts[best_i + 1]
            parts[best_i] = (parts[best_i][0], pair_rank(best_i))
            if best_i > 0:
                parts[best_i - 1] = (parts[best_i - 1][0], pair_rank(best_i - 1))
        return [enc[piece[parts[i][0]: parts[i + 1][0]]] for i in range(len(parts) - 1)]

    def encode_ordinary(self, text: str) -> List[int]:
        out: List[int] = []
        for m in self.pat.finditer(text):
            out.extend(self._bpe(m.group().encode("utf-8")))
        return out

    def encode(self, text: str) -> List[int]:
        """Special tokens in the text are recognized and emitted as their
        ids (reference tiktoken.rs:469 — chat-template tokens like
        <|media_pad|> stay intact)."""
        if self.special_pat is None:
            return self.encode_ordinary(text)
        out: List[int] = []
        pos = 0
        for m in self.special_pat.finditer(text):
            out.extend(self.encode_ordinary(text[pos: m.start()]))
            out.append(self.special_tokens[m.group()])
            pos = m.end()
        out.extend(self.encode_ordinary(text[pos:]))
        return out

    def decode_bytes(self, ids: Sequence[int]) -> bytes:
        parts = []
        for i in ids:
            b = self.decoder.get(i)
            if b is None:
                b = self.special_decoder.get(i, b"")
            parts.append(b)
        return b"".join(parts)

    def decode(self, ids: Sequence[int]) -> str:
        return self.decode_bytes(ids).decode("utf-8", errors="replace")


def _find_tiktoken_file(d: str) -> Optional[str]:
    cands = [f for f in os.listdir(d) if f.endswith(".tiktoken") or f == "tiktoken.model"]
    return os.path.join(d, sorted(cands)[0]) if cands else None


def is_tiktoken_dir(d: str) -> bool:
    return os.path.isdir(d) and _find_tiktoken_file(d) is not None


def _is_kimi_config(cfg: Optional[dict]) -> bool:
    """kimi_k2_tokenizer.rs:35 — tokenizer_config references
    tokenization_kimi via auto_map / tokenizer_class."""
    if not cfg:
        return False
    blob = json.dumps(cfg)
    return "tokenization_kimi" in blob or "TikTokenTokenizer" in blob and "kimi" in blob.lower()


class TiktokenTokenizer:
    """Registry-compatible tokenizer over CoreBPE (encode/encode_batch/
    decode/decode_incremental), with special tokens + eos from
    tokenizer_config.json."""

    def __init__(self, core: CoreBPE, name: str = "tiktoken", eos_ids: Sequence[int] = ()):
        self._core = core
        self.name = name
        self.vocab_size = (max(core.decoder) + 1) if core.decoder else 0
        if core.special_tokens:
            self.vocab_size = max(self.vocab_size, max(core.special_tokens.values()) + 1)
        self.model_max_length = 1 << 20
        self.eos_token_ids = list(eos_ids)

    # ---- loaders ----------------------------------------------------------
    @classmethod
    def from_file(cls, path: str, name: Optional[str] = None,
                  special_tokens: Optional[Dict[str, int]] = None,
                  pattern: str = CL100K_BASE_PATTERN) -> "TiktokenTokenizer":
        enc = load_tiktoken_bpe(path)
        return cls(CoreBPE(enc, special_tokens or {}, pattern), name or os.path.basename(path))

    @classmethod
    def from_dir(cls, d: str, name: Optional[str] = None) -> "TiktokenTokenizer":
        tk_path = _find_tiktoken_file(d)
        if tk_path is None:
            raise FileNotFoundError(f"no .tiktoken file in {d}")
        cfg = None
        cfg_path = os.path.join(d, "tokenizer_config.json")
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                cfg = json.load(f)
        special: Dict[str, int] = {}
        eos_ids: List[int] = []
        if cfg:
            for tid, meta in (cfg.get("added_tokens_decoder") or {}).items():
                if isinstance(meta, dict) and meta.get("content"):
                    special[meta["content"]] = int(tid)
            eos = cfg.get("eos_token")
            if isinstance(eos, dict):
                eos = eos.get("content")
            if isinstance(eos, str) and eos in special:
                eos_ids.append(special[eos])
        pattern = KIMI_K2_PATTERN if _is_kimi_config(cfg) else CL100K_BASE_PATTERN
        enc = load_tiktoken_bpe(tk_path)
        return cls(CoreBPE(enc, special, pattern), name or os.path.basename(d.rstrip("/")),
                   eos_ids)

    # ---- registry surface --------------------------------------------------
    def encode(self, text: str) -> List[int]:
        return self._core.encode(text)

    def encode_batch(self, texts: List[str]) -> List[List[int]]:
        return [self._core.encode(t) for t in texts]

    def decode(self, ids: List[int]) -> str:
        return self._core.decode(ids)

    def decode_incremental(self, ids: List[int], prefix_len: int) -> str:
        full = self._core.decode(ids)
        prev = self._core.decode(ids[:prefix_len])
        return full[len(prev):]
