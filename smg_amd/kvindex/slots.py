"""Worker URL <-> tenant slot (0..63) mapping shared by the native trees.
The device tree stores tenants as a 64-bit bitmask, matching the reference's
practical scale (xGMI node = 8 workers; 64 covers multi-node DP fleets)."""
from __future__ import annotations

from typing import Dict, Optional


class SlotMap:
    MAX_SLOTS = 64

    def __init__(self) -> None:
        self.url_to_slot: Dict[str, int] = {}
        self.slot_to_url: Dict[int, str] = {}
        self._free = list(range(self.MAX_SLOTS - 1, -1, -1))

    def slot_of(self, url: str) -> int:
        slot = self.url_to_slot.get(url)
        if slot is None:
            if not self._free:
                raise RuntimeError("tenant slot space exhausted (64 workers max per tree)")
            slot = self._free.pop()
            self.url_to_slot[url] = slot
            self.slot_to_url[slot] = url
        return slot

    def existing_slot(self, url: str) -> Optional[int]:
        return self.url_to_slot.get(url)

    def url_of(self, slot: int) -> Optional[str]:
        if slot is None or slot < 0:
            return None
        return self.slot_to_url.get(slot)

    def release(self, url: str) -> None:
        slot = self.url_to_slot.pop(url, None)
        if slot is not None:
            self.slot_to_url.pop(slot, None)
            self._free.append(slot)

    def healthy_mask(self, urls) -> int:
        mask = 0
        for u in urls:
            mask |= 1 << self.slot_of(u)
        return mask
