"""GPU batch BPE tokenizer (gfx950 kernel in csrc/bpe.hip).

Loads a HuggingFace tokenizer.json (byte-level BPE family: GPT-2/Llama-3/
Qwen-style), builds the merge-rank hash table on the device, and services
encode_batch with one kernel launch over all pre-tokenized pieces.

Pre-tokenization uses the HF tokenizer's own Rust pre_tokenizer (exact), with
the C++ scanner in _core as fallback; BPE merges then run on the GPU.  The
host C++ path runs the identical merge loop, so GPU-vs-host is differentially
testable, and GPU-vs-HF validates end-to-end token equality.
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional

import numpy as np


def _byte_unicode_table() -> Dict[str, int]:
    """The GPT-2 byte<->unicode bijection (public algorithm): printable bytes
    map to themselves, the rest to U+0100.. offsets."""
    bs = list(range(ord("!"), ord("~") + 1)) + list(range(0xA1, 0xAD)) + list(range(0xAE, 0x100))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {chr(c): b for b, c in zip(bs, cs)}


class GpuBpeTokenizer:
    def __init__(self, tokenizer_json_path: str, use_gpu: bool = True, name: Optional[str] = None):
        try:
            import torch  # noqa: F401 — HIP runtime must load before _core.so
        except ImportError:
            pass
        from .. import _core

        with open(tokenizer_json_path) as f:
            spec = json.load(f)
        model = spec.get("model", {})
        if model.get("type") != "BPE":
            raise ValueError(f"only BPE tokenizer.json supported (got {model.get('type')})")
        self.vocab: Dict[str, int] = model["vocab"]
        self.vocab_size = len(self.vocab)
        self.name = name or tokenizer_json_path
        self.model_max_length = 1 << 20
        merges = model["merges"]
        self._char2byte = _byte_unicode_table()
        self._id2token = {v: k for k, v in self.vocab.items()}

        # byte -> initial token id
        byte_to_tok = np.zeros(256, dtype=np.uint32)
        byte2char = {b: c for c, b in self._char2byte.items()}
        for b in range(256):
            ch = byte2char[b]
            tid = self.vocab.get(ch)
            if tid is None:
                tid = 0
            byte_to_tok[b] = tid

        keys = np.empty(len(merges), dtype=np.uint64)
        vals = np.empty(len(merges), dtype=np.uint64)
        for rank, m in enumerate(merges):
            if isinstance(m, str):
                left, right = m.split(" ", 1)
            else:
                left, right = m
            a = self.vocab.get(left)
            b = self.vocab.get(right)
            merged = self.vocab.get(left + right)
            if a is None or b is None or merged is None:
                a, b, merged = 0, 0, 0
            keys[rank] = (np.uint64(a) << np.uint64(32)) | np.uint64(b)
            vals[rank] = (np.uint64(rank) << np.uint64(32)) | np.uint64(merged)
        self._bpe = _core.Bpe(keys, vals, byte_to_tok, use_gpu=use_gpu)
        self.on_gpu = self._bpe.on_gpu()

        # exact pre-tokenizer when HF is importable
        self._hf_pretok = None
        try:
            from tokenizers import Tokenizer

            self._hf = Tokenizer.from_file(tokenizer_json_path)
            self._hf_pretok = self._hf.pre_tokenizer
        except Exception:
            self._hf = None

        # decode support
        self._byte_of_char = self._char2byte

    # ---- encode ------------------------------------------------------------
    def _pieces_to_bytes(self, piece: str) -> bytes:
        return bytes(self._byte_of_char.get(ch, 0) for ch in piece)

    def encode_batch(self, texts: List[str]) -> List[List[int]]:
        piece_bytes: List[bytes] = []
        text_piece_counts: List[int] = []
        for text in texts:
            if self._hf_pretok is not None:
                pieces = [p for p, _span in self._hf_pretok.pre_tokenize_str(text)]
                bts = [self._pieces_to_bytes(p) for p in pieces]
            else:
                raw = text.encode("utf-8")
                offs = self._bpe.pretokenize(raw)
                bts = [raw[offs[i]: offs[i + 1]] for i in range(len(offs) - 1)]
            piece_bytes.extend(bts)
            text_piece_counts.append(len(bts))
        if not piece_bytes:
            return [[] for _ in texts]
        flat = b"".join(piece_bytes)
        offsets = np.zeros(len(piece_bytes) + 1, dtype=np.uint32)
        np.cumsum([len(b) for b in piece_bytes], out=offsets[1:])
        tokens_flat, counts = self._bpe.encode_pieces(flat, offsets)
        out: List[List[int]] = []
        pi = 0
        ti = 0
        counts = counts.tolist()
        tokens_flat = tokens_flat.tolist()
        for n_pieces in text_piece_counts:
            n_tok = sum(counts[pi: pi + n_pieces])
            out.append(tokens_flat[ti: ti + n_tok])
            pi += n_pieces
            ti += n_tok
        return out

    def encode(self, text: str) -> List[int]:
        return self.encode_batch([text])[0]

    # ---- decode ------------------------------------------------------------
    def decode(self, ids: List[int]) -> str:
        chars = "".join(self._id2token.get(i, "") for i in ids)
        data = bytes(self._byte_of_char.get(ch, 0x20) for ch in chars)
        return data.decode("utf-8", "replace")

    def decode_incremental(self, ids: List[int], prefix_len: int) -> str:
        return self.decode(ids[prefix_len:])
