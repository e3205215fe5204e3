"""SWIM-style gossip membership + sync rounds (reference: crates/mesh —
GossipController::event_loop 1 Hz (gossip_controller.rs:1-26): SWIM probe
Ping -> indirect PingReq -> suspect -> dead; round collection drains local
CRDT ops into a RoundBatch shipped to every live peer; partition detection
(partition.rs)).

Transport: HTTP POST between gateways (aiohttp) — the mesh is the low-rate
inter-gateway control plane (stays TCP).
"""
from __future__ import annotations

import asyncio
import logging
import random
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import aiohttp

from .crdt import MeshKV, Op

log = logging.getLogger("smg.mesh")

ALIVE, SUSPECT, DEAD = "alive", "suspect", "dead"


@dataclass
class Member:
    node_id: str
    url: str  # base http url of the peer's mesh endpoint
    state: str = ALIVE
    incarnation: int = 0
    last_ack: float = field(default_factory=time.monotonic)
    suspect_since: Optional[float] = None


class MeshNode:
    def __init__(
        self,
        node_id: str,
        advertise_url: str,
        kv: Optional[MeshKV] = None,
        probe_interval: float = 1.0,
        probe_timeout: float = 0.5,
        suspect_timeout: float = 3.0,
        indirect_k: int = 2,
        mtls_cert: Optional[str] = None,
        mtls_key: Optional[str] = None,
        mtls_ca: Optional[str] = None,
    ):
        self.node_id = node_id
        self.advertise_url = advertise_url.rstrip("/")
        self.kv = kv or MeshKV(node_id)
        self.members: Dict[str, Member] = {}
        self.probe_interval = probe_interval
        self.probe_timeout = probe_timeout
        self.suspect_timeout = suspect_timeout
        self.indirect_k = indirect_k
        self._session: Optional[aiohttp.ClientSession] = None
        self._task: Optional[asyncio.Task] = None
        self._stopped = asyncio.Event()
        self._sent_watermarks: Dict[str, int] = {}  # peer -> last local seq sent
        self.incarnation = 0
        # partition detection (reference mesh/src/partition.rs): when a
        # majority of known members are unreachable at once, this side is
        # (likely) the isolated minority — flagged for metrics/readiness and
        # used to trigger anti-entropy repair on heal
        self.partitioned = False
        self._partition_since: Optional[float] = None
        self.partition_heals = 0
        # incremental page-repair sessions (reference tree_sync.rs:38-67)
        self.max_repair_page_bytes = 2 << 20
        self.repair_max_retries = 3
        self.repairs_completed = 0
        self._repair_tasks: Dict[str, asyncio.Task] = {}
        # chunked sync (reference mesh transport/chunking.rs): cap each sync
        # POST so a post-partition backlog drains in bounded batches
        self.max_sync_ops_per_post = 500
        self.max_sync_posts_per_round = 8
        # mutual TLS (reference mesh/src/mtls.rs): the SAME CA signs every
        # replica; outbound sessions present our cert, the listener requires
        # a peer cert chained to the CA
        self._mtls = (mtls_cert, mtls_key, mtls_ca)

    def client_ssl_context(self):
        """SSLContext for OUTBOUND gossip (client cert + CA pinning), or None
        when mTLS is off."""
        cert, key, ca = self._mtls
        if not (cert and key and ca):
            return None
        import ssl

        ctx = ssl.create_default_context(ssl.Purpose.SERVER_AUTH, cafile=ca)
        ctx.load_cert_chain(cert, key)
        ctx.check_hostname = False  # peers are addressed by IP inside the mesh
        return ctx

    def server_ssl_context(self):
        """SSLContext for the mesh LISTENER: requires a client cert from the
        mesh CA, or None when mTLS is off."""
        cert, key, ca = self._mtls
        if not (cert and key and ca):
            return None
        import ssl

        ctx = ssl.create_default_context(ssl.Purpose.CLIENT_AUTH, cafile=ca)
        ctx.load_cert_chain(cert, key)
        ctx.verify_mode = ssl.CERT_REQUIRED
        return ctx

    # ---- lifecycle ---------------------------------------------------------
    async def start(self, peer_urls: List[str]) -> None:
        self._stopped.clear()
        ssl_ctx = self.client_ssl_context()
        connector = aiohttp.TCPConnector(ssl=ssl_ctx) if ssl_ctx else None
        self._session = aiohttp.ClientSession(
            timeout=aiohttp.ClientTimeout(total=5), connector=connector)
        for url in peer_urls:
            await self.join(url)
        self._task = asyncio.ensure_future(self._event_loop())

    async def stop(self) -> None:
        self._stopped.set()
        if self._task:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
        for t in self._repair_tasks.values():
            if not t.done():
                t.cancel()
        if self._session:
            await self._session.close()

    async def join(self, peer_url: str) -> bool:
        """Contact a seed peer: exchange membership + full CRDT snapshot."""
        peer_url = peer_url.rstrip("/")
        try:
            async with self._session.post(
                peer_url + "/mesh/join",
                json={
                    "node_id": self.node_id,
                    "url": self.advertise_url,
                    "members": self._member_dicts(),
                    "snapshot": [op.to_dict() for op in self.kv.snapshot_ops()],
                },
            ) as resp:
                if resp.status != 200:
                    return False
                data = await resp.json()
        except Exception as exc:
            log.warning("mesh join %s failed: %s", peer_url, exc)
            return False
        self._merge_members(data.get("members", []))
        for opd in data.get("snapshot", []):
            op = Op.from_dict(opd)
            self.kv.apply_remote(op, op.hlc[2])
        return True

    # ---- event loop (1 Hz rounds) ------------------------------------------
    async def _event_loop(self) -> None:
        while not self._stopped.is_set():
            try:
                await self._round()
            except asyncio.CancelledError:
                return
            except Exception as exc:
                log.warning("mesh round error: %s", exc)
            try:
                await asyncio.wait_for(self._stopped.wait(), self.probe_interval)
                return
            except asyncio.TimeoutError:
                pass

    async def _round(self) -> None:
        now = time.monotonic()
        live = [m for m in self.members.values() if m.state != DEAD]
        # SWIM probe: one random member per round
        if live:
            target = random.choice(live)
            was = target.state
            ok = await self._ping(target)
            if not ok:
                ok = await self._indirect_ping(target, live)
            if ok:
                target.state = ALIVE
                target.last_ack = now
                target.suspect_since = None
                if was != ALIVE:
                    # rejoin after suspicion: reconverge via page repair
                    self._schedule_repair(target)
            else:
                if target.state == ALIVE:
                    target.state = SUSPECT
                    target.suspect_since = now
                elif target.state == SUSPECT and now - (target.suspect_since or now) > self.suspect_timeout:
                    target.state = DEAD
                    log.info("mesh member %s declared dead", target.node_id)
        # also probe one dead member occasionally so partitions can HEAL
        dead = [m for m in self.members.values() if m.state == DEAD]
        if dead:
            target = random.choice(dead)
            if await self._ping(target):
                target.state = ALIVE
                target.last_ack = now
                target.suspect_since = None
                log.info("mesh member %s returned from the dead", target.node_id)
                self._schedule_repair(target)
        self._update_partition_state()
        # sync round: ship new ops to every live peer
        await asyncio.gather(*(self._sync_peer(m) for m in self.members.values() if m.state != DEAD))

    # ---- partition detection / heal (reference partition.rs) ---------------
    def _update_partition_state(self) -> None:
        total = len(self.members)
        if total == 0:
            self.partitioned = False
            return
        unreachable = sum(1 for m in self.members.values() if m.state != ALIVE)
        now_partitioned = unreachable * 2 > total
        if now_partitioned and not self.partitioned:
            self.partitioned = True
            self._partition_since = time.monotonic()
            log.warning("mesh partition detected: %d/%d members unreachable", unreachable, total)
        elif not now_partitioned and self.partitioned:
            self.partitioned = False
            dur = time.monotonic() - (self._partition_since or time.monotonic())
            self._partition_since = None
            self.partition_heals += 1
            log.info("mesh partition healed after %.1fs; starting anti-entropy repair", dur)
            for m in self.members.values():
                if m.state == ALIVE:
                    self._schedule_repair(m)

    def partition_state(self) -> dict:
        total = len(self.members)
        unreachable = sum(1 for m in self.members.values() if m.state != ALIVE)
        return {
            "partitioned": self.partitioned,
            "members": total,
            "unreachable": unreachable,
            "since": self._partition_since,
            "heals": self.partition_heals,
        }

    # ---- incremental page repair (reference tree_sync.rs:38-67) -------------
    def _schedule_repair(self, m: Member) -> None:
        existing = self._repair_tasks.get(m.node_id)
        if existing is not None and not existing.done():
            return
        self._repair_tasks[m.node_id] = asyncio.ensure_future(self._repair_peer(m))

    async def _repair_peer(self, m: Member) -> None:
        """Pull the peer's state in byte-capped pages per namespace and merge
        through the CRDT (idempotent) — incremental reconvergence instead of
        a full join snapshot."""
        for ns in self.kv.namespaces():
            cursor = ""
            retries = 0
            while True:
                try:
                    async with self._session.post(
                        m.url + "/mesh/repair",
                        json={"from": self.node_id, "ns": ns, "cursor": cursor,
                              "max_bytes": self.max_repair_page_bytes},
                        timeout=aiohttp.ClientTimeout(total=5),
                    ) as resp:
                        if resp.status != 200:
                            raise RuntimeError(f"repair HTTP {resp.status}")
                        page = await resp.json()
                except Exception as exc:
                    retries += 1
                    if retries > self.repair_max_retries:
                        log.warning("repair of %s ns=%s abandoned: %s", m.node_id, ns, exc)
                        return
                    await asyncio.sleep(0.2 * retries)
                    continue
                retries = 0
                for opd in page.get("entries", []):
                    op = Op.from_dict(opd)
                    self.kv.apply_remote(op, op.hlc[2])
                if page.get("done", True):
                    break
                cursor = page.get("next_cursor", "")
        self.repairs_completed += 1

    def handle_repair(self, payload: dict) -> dict:
        return self.kv.repair_page(
            payload.get("ns", ""), payload.get("cursor", ""),
            int(payload.get("max_bytes") or self.max_repair_page_bytes),
        )

    async def _ping(self, m: Member) -> bool:
        try:
            async with self._session.post(
                m.url + "/mesh/ping",
                json={"from": self.node_id, "members": self._member_dicts()},
                timeout=aiohttp.ClientTimeout(total=self.probe_timeout),
            ) as resp:
                if resp.status == 200:
                    data = await resp.json()
                    self._merge_members(data.get("members", []))
                    return True
        except Exception:
            pass
        return False

    async def _indirect_ping(self, target: Member, live: List[Member]) -> bool:
        helpers = [m for m in live if m.node_id != target.node_id]
        random.shuffle(helpers)
        for h in helpers[: self.indirect_k]:
            try:
                async with self._session.post(
                    h.url + "/mesh/ping_req",
                    json={"from": self.node_id, "target": target.url},
                    timeout=aiohttp.ClientTimeout(total=self.probe_timeout * 2),
                ) as resp:
                    if resp.status == 200 and (await resp.json()).get("ack"):
                        return True
            except Exception:
                continue
        return False

    async def _sync_peer(self, m: Member) -> None:
        """Ship new ops in bounded chunks (reference transport/chunking.rs):
        a post-partition backlog drains over a few capped POSTs per round
        instead of one unbounded body."""
        for _ in range(self.max_sync_posts_per_round):
            since = self._sent_watermarks.get(m.node_id, 0)
            ops = self.kv.ops_since(since)[: self.max_sync_ops_per_post]
            if not ops:
                return
            try:
                async with self._session.post(
                    m.url + "/mesh/sync",
                    json={"from": self.node_id, "ops": [op.to_dict() for op in ops]},
                ) as resp:
                    if resp.status != 200:
                        return
                    self._sent_watermarks[m.node_id] = ops[-1].seq
            except Exception:
                return
            if len(ops) < self.max_sync_ops_per_post:
                return

    # ---- membership helpers -------------------------------------------------
    def _member_dicts(self) -> List[dict]:
        out = [{"node_id": self.node_id, "url": self.advertise_url, "state": ALIVE, "incarnation": self.incarnation}]
        for m in self.members.values():
            out.append({"node_id": m.node_id, "url": m.url, "state": m.state, "incarnation": m.incarnation})
        return out

    def _merge_members(self, dicts: List[dict]) -> None:
        for d in dicts:
            nid = d.get("node_id")
            if not nid or nid == self.node_id:
                continue
            m = self.members.get(nid)
            if m is None:
                self.members[nid] = Member(nid, d["url"], d.get("state", ALIVE), d.get("incarnation", 0))
            else:
                if d.get("incarnation", 0) > m.incarnation:
                    m.incarnation = d["incarnation"]
                    m.state = d.get("state", ALIVE)
                elif d.get("state") == DEAD and m.state != DEAD:
                    pass  # rumors of death need a newer incarnation; keep probing

    # ---- inbound handlers (wired by mesh/server.py) -------------------------
    def handle_ping(self, payload: dict) -> dict:
        self._merge_members(payload.get("members", []))
        return {"ack": True, "members": self._member_dicts()}

    async def handle_ping_req(self, payload: dict) -> dict:
        target_url = payload.get("target", "")
        try:
            async with self._session.post(
                target_url + "/mesh/ping",
                json={"from": self.node_id, "members": []},
                timeout=aiohttp.ClientTimeout(total=self.probe_timeout),
            ) as resp:
                return {"ack": resp.status == 200}
        except Exception:
            return {"ack": False}

    def handle_sync(self, payload: dict) -> dict:
        from_node = payload.get("from", "?")
        applied = 0
        for opd in payload.get("ops", []):
            op = Op.from_dict(opd)
            if self.kv.apply_remote(op, from_node):
                applied += 1
        return {"applied": applied, "watermark": self.kv.watermarks.get(from_node, 0)}

    def handle_join(self, payload: dict) -> dict:
        self._merge_members(
            payload.get("members", []) + [{"node_id": payload["node_id"], "url": payload["url"], "state": ALIVE}]
        )
        for opd in payload.get("snapshot", []):
            op = Op.from_dict(opd)
            self.kv.apply_remote(op, op.hlc[2])
        return {
            "members": self._member_dicts(),
            "snapshot": [op.to_dict() for op in self.kv.snapshot_ops()],
        }

    def live_members(self) -> List[Member]:
        return [m for m in self.members.values() if m.state == ALIVE]
