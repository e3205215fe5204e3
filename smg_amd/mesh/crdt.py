"""CRDT key-value store for gateway HA sync (reference: crates/mesh —
MeshKV with per-namespace merge engines: default last-writer-wins,
EpochMaxWins for rate-limit shards with tombstones (crdt_kv/epoch_max_wins.rs),
op-log + per-peer watermarks (crdt_kv/operation.rs, watermark.rs)).
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional, Tuple


@dataclass
class Op:
    ns: str
    key: str
    value: Any  # None = tombstone
    hlc: Tuple[int, int, str]  # (wall_us, counter, node) — hybrid logical clock
    seq: int = 0  # local op-log sequence (assigned by the emitting node)

    def to_dict(self):
        return {"ns": self.ns, "key": self.key, "value": self.value, "hlc": list(self.hlc), "seq": self.seq}

    @classmethod
    def from_dict(cls, d):
        return cls(d["ns"], d["key"], d.get("value"), tuple(d["hlc"]), d.get("seq", 0))


class HLC:
    """Hybrid logical clock: (wall_us, counter, node_id)."""

    def __init__(self, node_id: str):
        self.node_id = node_id
        self._last = (0, 0)

    def now(self) -> Tuple[int, int, str]:
        wall = int(time.time() * 1e6)
        lw, lc = self._last
        if wall > lw:
            self._last = (wall, 0)
        else:
            self._last = (lw, lc + 1)
        return (*self._last, self.node_id)

    def observe(self, hlc: Tuple[int, int, str]) -> None:
        w, c, _ = hlc
        lw, lc = self._last
        if (w, c) > (lw, lc):
            self._last = (w, c)


def lww_merge(existing: Optional[Op], incoming: Op) -> bool:
    """Last-writer-wins by HLC.  Returns True if incoming should be applied."""
    return existing is None or incoming.hlc > existing.hlc


def epoch_max_wins_merge(existing: Optional[Op], incoming: Op) -> bool:
    """Rate-limit shard merge (reference epoch_max_wins.rs): values are
    {epoch, used}; higher epoch wins, same epoch takes max(used); tombstones
    (None) win by HLC."""
    if existing is None:
        return True
    if incoming.value is None or existing.value is None:
        return incoming.hlc > existing.hlc
    ee, ie = existing.value.get("epoch", 0), incoming.value.get("epoch", 0)
    if ie != ee:
        return ie > ee
    return incoming.value.get("used", 0) > existing.value.get("used", 0)


class MeshKV:
    def __init__(self, node_id: str):
        self.node_id = node_id
        self.hlc = HLC(node_id)
        self._data: Dict[str, Dict[str, Op]] = {}  # ns -> key -> Op
        self._log: List[Op] = []
        self._seq = 0
        self._mergers: Dict[str, Callable[[Optional[Op], Op], bool]] = {}
        self._watchers: Dict[str, List[Callable[[Op], None]]] = {}
        self.watermarks: Dict[str, int] = {}  # peer node -> max seq applied

    def register_namespace(self, ns: str, merger: Optional[Callable] = None) -> None:
        self._mergers[ns] = merger or lww_merge

    def watch(self, ns: str, fn: Callable[[Op], None]) -> None:
        self._watchers.setdefault(ns, []).append(fn)

    # ---- local writes -----------------------------------------------------
    def put(self, ns: str, key: str, value: Any) -> Op:
        self._seq += 1
        op = Op(ns, key, value, self.hlc.now(), self._seq)
        self._apply(op, notify=False)
        self._log.append(op)
        return op

    def delete(self, ns: str, key: str) -> Op:
        return self.put(ns, key, None)

    # ---- remote application -----------------------------------------------
    def apply_remote(self, op: Op, from_node: str) -> bool:
        self.hlc.observe(op.hlc)
        if op.seq > self.watermarks.get(from_node, 0):
            self.watermarks[from_node] = op.seq
        return self._apply(op, notify=True)

    def _apply(self, op: Op, notify: bool) -> bool:
        ns_map = self._data.setdefault(op.ns, {})
        merger = self._mergers.get(op.ns, lww_merge)
        if not merger(ns_map.get(op.key), op):
            return False
        ns_map[op.key] = op
        if notify:
            for fn in self._watchers.get(op.ns, []):
                fn(op)
        return True

    # ---- reads / sync -----------------------------------------------------
    def get(self, ns: str, key: str) -> Optional[Any]:
        op = self._data.get(ns, {}).get(key)
        return op.value if op is not None else None

    def items(self, ns: str) -> Dict[str, Any]:
        return {k: op.value for k, op in self._data.get(ns, {}).items() if op.value is not None}

    def ops_since(self, seq: int) -> List[Op]:
        return [op for op in self._log if op.seq > seq]

    def local_seq(self) -> int:
        return self._seq

    def snapshot_ops(self) -> List[Op]:
        """Full-state ops for a joining peer (compaction substitute)."""
        out = []
        for ns_map in self._data.values():
            out.extend(ns_map.values())
        return out

    def namespaces(self) -> List[str]:
        return list(self._mergers.keys())

    def repair_page(self, ns: str, cursor: str = "", max_bytes: int = 2 << 20) -> dict:
        """One incremental repair page (reference tree_sync.rs:38-67
        `tree:req:`/`tree:page:` protocol, page cap mesh-v2 §3.1): current
        state-ops for keys > `cursor` in key order, filled until the next
        entry would push the page over `max_bytes`.  Returns
        {entries, next_cursor, done}; applying entries through apply_remote
        is idempotent (HLC-merged), so repair converges without a full join
        snapshot."""
        import json as _json

        ns_map = self._data.get(ns, {})
        keys = sorted(k for k in ns_map if k > cursor)
        entries: List[dict] = []
        used = 0
        last = cursor
        for k in keys:
            d = ns_map[k].to_dict()
            sz = len(_json.dumps(d))
            if entries and used + sz > max_bytes:
                return {"entries": entries, "next_cursor": last, "done": False}
            entries.append(d)
            used += sz
            last = k
        return {"entries": entries, "next_cursor": last, "done": True}

    def compact(self, keep_last: int = 10_000) -> None:
        if len(self._log) > keep_last:
            self._log = self._log[-keep_last:]
