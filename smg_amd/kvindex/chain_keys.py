"""Pure-Python twin of the device chain-key schedule (csrc/gpu_tree.hip) —
used to resolve stored entries to their table keys host-side (KV-event
removal).  Arithmetic is u64-wrapping and must stay bit-identical."""
from __future__ import annotations

from typing import List, Sequence

M = (1 << 64) - 1
CHAIN_W = 0xA24BAED4963EE407
CHAIN_GOLD = 0x9E3779B97F4A7C15
CHAIN_SALT = 0x5851F42D4C957F2D
C_PRIME = 0x100000001B3


def mix64(x: int) -> int:
    x = (x + CHAIN_GOLD) & M
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & M
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & M
    return x ^ (x >> 31)


def page_hash(tokens: Sequence[int]) -> int:
    h = 0
    pw = 1
    for t in tokens:
        h = (h + mix64((t + CHAIN_SALT) & M) * pw) & M
        pw = (pw * C_PRIME) & M
    return mix64(h ^ len(tokens))


def chain_keys_scalar(tokens: Sequence[int], page_size: int) -> List[int]:
    """Reference scalar form (bit-identical to the device schedule)."""
    n = (len(tokens) // page_size) * page_size
    chain = 0
    wpow = 1
    out: List[int] = []
    for p in range(n // page_size):
        ph = page_hash(tokens[p * page_size: (p + 1) * page_size])
        chain = (chain + ph * wpow) & M
        wpow = (wpow * CHAIN_W) & M
        k = mix64(chain ^ ((p + 1) * CHAIN_GOLD) & M)
        out.append(k + 2 if k < 2 else k)
    return out


# ---- vectorized twin (numpy uint64 wraps mod 2^64 like the device) --------
import numpy as _np

_POW_CACHE: dict = {}


def _mix64_np(x):
    x = x + _np.uint64(CHAIN_GOLD)
    x = (x ^ (x >> _np.uint64(30))) * _np.uint64(0xBF58476D1CE4E5B9)
    x = (x ^ (x >> _np.uint64(27))) * _np.uint64(0x94D049BB133111EB)
    return x ^ (x >> _np.uint64(31))


def _pows(base: int, n: int):
    key = (base, n)
    got = _POW_CACHE.get(key)
    if got is None or len(got) < n:
        out = _np.empty(n, dtype=_np.uint64)
        v = 1
        for i in range(n):
            out[i] = v
            v = (v * base) & M
        _POW_CACHE[key] = got = out
    return got[:n]


def chain_keys(tokens: Sequence[int], page_size: int) -> List[int]:
    """The table key of every page depth of this token path.  One vectorized
    O(n) pass (this runs per admission in the engine's prefix cache and per
    routed request host-side, so the scalar loop was multiple ms per serving
    tick at depth ~500 tokens x 32 admissions)."""
    n = (len(tokens) // page_size) * page_size
    if n == 0:
        return []
    t = _np.asarray(tokens[:n], dtype=_np.uint64).reshape(-1, page_size)
    npages = t.shape[0]
    m = _mix64_np(t + _np.uint64(CHAIN_SALT))
    ph = _np.add.reduce(m * _pows(C_PRIME, page_size), axis=1, dtype=_np.uint64)
    ph = _mix64_np(ph ^ _np.uint64(page_size))
    chain = _np.cumsum(ph * _pows(CHAIN_W, npages), dtype=_np.uint64)
    idx = _np.arange(1, npages + 1, dtype=_np.uint64)
    k = _mix64_np(chain ^ (idx * _np.uint64(CHAIN_GOLD)))
    k = _np.where(k < 2, k + _np.uint64(2), k)
    return k.tolist()
