"""Consistent-hash ring (reference: model_gateway/src/worker/hash_ring.rs:29).

160 virtual nodes per worker on a 64-bit ring, binary-search lookup.
"""
from __future__ import annotations

import bisect
import hashlib
from typing import List, Optional


def _hash64(data: bytes) -> int:
    return int.from_bytes(hashlib.blake2b(data, digest_size=8).digest(), "little")


class HashRing:
    VNODES = 160

    def __init__(self, urls: Optional[List[str]] = None):
        self._points: List[int] = []
        self._owners: List[str] = []
        self._urls: set = set()
        if urls:
            # bulk build: collect then sort once (O(n·v·log) vs insort O((n·v)^2))
            pts = []
            for url in urls:
                if url in self._urls:
                    continue
                self._urls.add(url)
                for v in range(self.VNODES):
                    pts.append((_hash64(f"{url}#{v}".encode()), url))
            pts.sort()
            self._points = [p for p, _ in pts]
            self._owners = [o for _, o in pts]

    def add(self, url: str) -> None:
        if url in self._urls:
            return
        self._urls.add(url)
        for v in range(self.VNODES):
            h = _hash64(f"{url}#{v}".encode())
            idx = bisect.bisect_left(self._points, h)
            self._points.insert(idx, h)
            self._owners.insert(idx, url)

    def remove(self, url: str) -> None:
        if url not in self._urls:
            return
        self._urls.discard(url)
        keep_p, keep_o = [], []
        for p, o in zip(self._points, self._owners):
            if o != url:
                keep_p.append(p)
                keep_o.append(o)
        self._points, self._owners = keep_p, keep_o

    def lookup(self, key: str | bytes) -> Optional[str]:
        if not self._points:
            return None
        if isinstance(key, str):
            key = key.encode()
        h = _hash64(key)
        idx = bisect.bisect_right(self._points, h)
        if idx == len(self._points):
            idx = 0
        return self._owners[idx]

    def __len__(self) -> int:
        return len(self._urls)
