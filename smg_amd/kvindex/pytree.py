"""Reference (pure-Python) paged radix prefix tree.

Semantics follow the reference TokenTree (crates/kv_index/src/token_tree.rs:303,
match_prefix_with_counts :620, match_and_insert :754):

* input truncated to a page boundary before lookup (page = `page_size` tokens);
* a match advances page by page; a node whose tenants were all evicted stops
  the match; the routed tenant is the deepest matched node's most-recently
  touched tenant, and matching touches that tenant's LRU stamp;
* match+insert is one descent: the match result is resolved against the
  pre-insert tree, then the path (matched + new) is attributed to the chosen
  tenant;
* LRU eviction removes stale (leaf-node, tenant) attributions until the tree
  is within budget; a node with no tenants left blocks matches through it.

Design deviation from the reference (deliberate, GPU-first): nodes hold exactly
ONE page instead of variable-length runs, and children are a single global
open-addressed map keyed by (parent_id, page).  Match results are identical
(matches are page-aligned either way); this is the same layout the gfx950 HIP
kernel uses in device memory, so host and device trees are differentially
testable.  This class is the numerics reference for `_core`'s C++/HIP trees.
"""
from __future__ import annotations

import heapq
from typing import Dict, List, Optional, Sequence, Tuple


class _Node:
    __slots__ = ("node_id", "parent", "depth", "tenants", "child_count")

    def __init__(self, node_id: int, parent: int, depth: int):
        self.node_id = node_id
        self.parent = parent
        self.depth = depth  # in pages
        self.tenants: Dict[str, int] = {}  # tenant -> last-access stamp
        self.child_count = 0


class MatchResult:
    __slots__ = ("tenant", "matched_token_count", "input_token_count")

    def __init__(self, tenant: Optional[str], matched: int, input_count: int):
        self.tenant = tenant
        self.matched_token_count = matched
        self.input_token_count = input_count

    @property
    def match_rate(self) -> float:
        if self.input_token_count == 0:
            return 0.0
        return self.matched_token_count / self.input_token_count


class PagedRadixTree:
    """Paged token-prefix tree over an arbitrary hashable page unit."""

    def __init__(self, page_size: int = 16):
        self.page_size = page_size
        self._children: Dict[Tuple[int, Tuple], int] = {}  # (parent, page) -> node_id
        self._nodes: Dict[int, _Node] = {0: _Node(0, -1, 0)}
        self._keys: Dict[int, Tuple[int, Tuple]] = {}  # node_id -> its key (for eviction)
        self._next_id = 1
        self._clock = 0
        self.tenant_token_count: Dict[str, int] = {}

    # ---- core ops --------------------------------------------------------
    def _tick(self) -> int:
        self._clock += 1
        return self._clock

    def _pages(self, tokens: Sequence) -> List[Tuple]:
        n = (len(tokens) // self.page_size) * self.page_size
        return [tuple(tokens[i : i + self.page_size]) for i in range(0, n, self.page_size)]

    def match(self, tokens: Sequence) -> MatchResult:
        """Longest page-aligned prefix match; touches the matched tenant."""
        pages = self._pages(tokens)
        stamp = self._tick()
        cur = 0
        matched = 0
        tenant: Optional[str] = None
        for page in pages:
            nid = self._children.get((cur, page))
            if nid is None:
                break
            node = self._nodes[nid]
            if not node.tenants:
                break  # all attributions evicted: prefix no longer cached anywhere
            t = max(node.tenants, key=lambda k: node.tenants[k])
            node.tenants[t] = stamp
            tenant = t
            matched += self.page_size
            cur = nid
        return MatchResult(tenant, matched, len(tokens))

    def insert(self, tokens: Sequence, tenant: str) -> int:
        """Attribute the page-aligned prefix path to `tenant`; returns tokens newly
        attributed to that tenant."""
        pages = self._pages(tokens)
        stamp = self._tick()
        cur = 0
        added = 0
        for page in pages:
            key = (cur, page)
            nid = self._children.get(key)
            if nid is None:
                nid = self._next_id
                self._next_id += 1
                node = _Node(nid, cur, self._nodes[cur].depth + 1)
                self._nodes[nid] = node
                self._children[key] = nid
                self._keys[nid] = key
                self._nodes[cur].child_count += 1
            else:
                node = self._nodes[nid]
            if tenant not in node.tenants:
                added += self.page_size
            node.tenants[tenant] = stamp
            cur = nid
        if added:
            self.tenant_token_count[tenant] = self.tenant_token_count.get(tenant, 0) + added
        return added

    def match_and_insert(self, tokens: Sequence, choose_tenant) -> Tuple[MatchResult, Optional[str]]:
        """One descent: match against the pre-insert tree, let `choose_tenant`
        (a callable of the MatchResult) pick the tenant, then insert for it.
        `choose_tenant` returning None skips the insert (reference
        match_and_insert_with, token_tree.rs:754)."""
        result = self.match(tokens)
        tenant = choose_tenant(result)
        if tenant is not None:
            self.insert(tokens, tenant)
        return result, tenant

    # ---- maintenance -----------------------------------------------------
    def __len__(self) -> int:
        return len(self._nodes) - 1

    def total_tokens(self) -> int:
        return (len(self._nodes) - 1) * self.page_size

    def remove_tenant(self, tenant: str) -> None:
        empty: List[int] = []
        for nid, node in self._nodes.items():
            if tenant in node.tenants:
                del node.tenants[tenant]
        self.tenant_token_count.pop(tenant, None)
        # nodes with no tenants stay until eviction (they block matches already)

    def evict(self, max_nodes: int) -> int:
        """LRU-evict (leaf, tenant) attributions, then prune empty leaves,
        until the node count fits `max_nodes`.  Returns nodes removed."""
        removed = 0
        if len(self) <= max_nodes:
            return 0
        heap: List[Tuple[int, int, str]] = []
        for nid, node in self._nodes.items():
            if nid != 0 and node.child_count == 0:
                for t, ts in node.tenants.items():
                    heapq.heappush(heap, (ts, nid, t))
                if not node.tenants:
                    heapq.heappush(heap, (0, nid, ""))
        while len(self) > max_nodes and heap:
            _, nid, t = heapq.heappop(heap)
            node = self._nodes.get(nid)
            if node is None or node.child_count != 0:
                continue
            if t and t in node.tenants:
                node.tenants.pop(t)
                self.tenant_token_count[t] = max(0, self.tenant_token_count.get(t, 0) - self.page_size)
            if not node.tenants:
                self._prune(nid)
                removed += 1
                parent = self._nodes.get(node.parent)
                if parent is not None and parent.node_id != 0 and parent.child_count == 0:
                    for pt, pts in parent.tenants.items():
                        heapq.heappush(heap, (pts, parent.node_id, pt))
                    if not parent.tenants:
                        heapq.heappush(heap, (0, parent.node_id, ""))
        return removed

    def clear(self) -> None:
        self.__init__(self.page_size)

    # ---- snapshot / replay (reference kv_index/src/snapshot.rs:18-59 —
    # TreeSnapshot of SnapshotNodes serialized for mesh join / persistence) --
    def snapshot(self) -> bytes:
        """Serialize nodes in BFS order so parents precede children; each
        entry is (local_idx, parent_idx, page, {tenant: stamp})."""
        import msgpack

        by_parent: Dict[int, List[Tuple[Tuple, int]]] = {}
        for (parent, page), cid in self._children.items():
            by_parent.setdefault(parent, []).append((page, cid))
        order: Dict[int, int] = {0: 0}
        entries = []
        queue = [0]
        while queue:
            nid = queue.pop(0)
            for page, cid in by_parent.get(nid, ()):
                order[cid] = len(order)
                child = self._nodes[cid]
                entries.append(
                    (order[nid], list(page), {t: ts for t, ts in child.tenants.items()})
                )
                queue.append(cid)
        return msgpack.packb(
            {"page_size": self.page_size, "clock": self._clock, "nodes": entries},
            use_bin_type=True,
        )

    @classmethod
    def from_snapshot(cls, blob: bytes) -> "PagedRadixTree":
        import msgpack

        d = msgpack.unpackb(blob, raw=False, strict_map_key=False)
        tree = cls(d["page_size"])
        idx_to_nid = {0: 0}
        for i, (parent_idx, page, tenants) in enumerate(d["nodes"], start=1):
            parent_nid = idx_to_nid[parent_idx]
            nid = tree._next_id
            tree._next_id += 1
            node = _Node(nid, parent_nid, tree._nodes[parent_nid].depth + 1)
            node.tenants = dict(tenants)
            key = (parent_nid, tuple(page))
            tree._nodes[nid] = node
            tree._children[key] = nid
            tree._keys[nid] = key
            tree._nodes[parent_nid].child_count += 1
            idx_to_nid[i] = nid
            for t in tenants:
                tree.tenant_token_count[t] = tree.tenant_token_count.get(t, 0) + tree.page_size
        tree._clock = d.get("clock", 0)
        return tree

    def _prune(self, nid: int) -> None:
        node = self._nodes.pop(nid)
        key = self._keys.pop(nid)
        self._children.pop(key, None)
        parent = self._nodes.get(node.parent)
        if parent is not None:
            parent.child_count -= 1


class TokenTree(PagedRadixTree):
    """Token-id tree (gRPC/tokenized path)."""

    def __init__(self, page_size: int = 16):
        super().__init__(page_size)


class StringTree(PagedRadixTree):
    """Byte-paged text tree (HTTP path).  The reference StringTree
    (string_tree.rs) matches per-character; this uses byte pages of
    `page_size` (default 8) — match counts are page-quantized, which only
    coarsens the match_rate signal, never the routed tenant."""

    def __init__(self, page_size: int = 8):
        super().__init__(page_size)

    def match_text(self, text: str) -> MatchResult:
        return self.match(text.encode("utf-8", "ignore"))

    def insert_text(self, text: str, tenant: str) -> int:
        return self.insert(text.encode("utf-8", "ignore"), tenant)

    def match_and_insert_text(self, text: str, choose_tenant):
        return self.match_and_insert(text.encode("utf-8", "ignore"), choose_tenant)
