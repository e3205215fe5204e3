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


def chain_keys(tokens: Sequence[int], page_size: int) -> List[int]:
    """The table key of every page depth of this token path."""
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
