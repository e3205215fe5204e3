"""C++ host tree wrapper — same API as pytree.PagedRadixTree, backed by
smg_amd._core.HostTokenTree.  Tenants are strings at this layer; the C++ core
works in worker slots 0..63, mapped here."""
from __future__ import annotations

from typing import Dict, Sequence

import numpy as np

from .pytree import MatchResult
from .slots import SlotMap


class HostTokenTree:
    def __init__(self, page_size: int = 16):
        from .. import _core  # torch-first guard lives in kvindex.__init__

        self._tree = _core.HostTokenTree(page_size=page_size)
        self.page_size = page_size
        self.slots = SlotMap()

    def _arr(self, tokens: Sequence) -> np.ndarray:
        return np.asarray(tokens, dtype=np.uint32)

    def match(self, tokens: Sequence) -> MatchResult:
        slot, matched, total = self._tree.match(self._arr(tokens), True)
        return MatchResult(self.slots.url_of(slot), matched, total)

    def insert(self, tokens: Sequence, tenant: str) -> int:
        return self._tree.insert(self._arr(tokens), self.slots.slot_of(tenant))

    def match_and_insert(self, tokens: Sequence, choose_tenant):
        result = self.match(tokens)
        tenant = choose_tenant(result)
        if tenant is not None:
            self.insert(tokens, tenant)
        return result, tenant

    def remove_tenant(self, tenant: str) -> None:
        slot = self.slots.existing_slot(tenant)
        if slot is not None:
            self._tree.remove_tenant(slot)
            self.slots.release(tenant)

    def evict(self, max_nodes: int) -> int:
        return self._tree.evict(max_nodes)

    def clear(self) -> None:
        self._tree.clear()

    def __len__(self) -> int:
        return len(self._tree)

    @property
    def tenant_token_count(self) -> Dict[str, int]:
        return {
            url: int(self._tree.tenant_tokens(slot))
            for url, slot in self.slots.url_to_slot.items()
        }


class HostTextTree(HostTokenTree):
    """Byte-paged text variant (HTTP routing path)."""

    def __init__(self, page_size: int = 8):
        super().__init__(page_size=page_size)

    @staticmethod
    def _bytes(text: str) -> np.ndarray:
        return np.frombuffer(text.encode("utf-8", "ignore"), dtype=np.uint8).astype(np.uint32)

    def match_text(self, text: str) -> MatchResult:
        return self.match(self._bytes(text))

    def insert_text(self, text: str, tenant: str) -> int:
        return self.insert(self._bytes(text), tenant)

    def match_and_insert_text(self, text: str, choose_tenant):
        return self.match_and_insert(self._bytes(text), choose_tenant)
