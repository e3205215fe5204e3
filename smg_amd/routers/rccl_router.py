"""RCCL serving router — the MI355X-native `connection_mode=rccl` transport.

Carries /v1/chat/completions (and /v1/completions, /generate) end-to-end over
the RCCL-over-xGMI lockstep plane (comm/plane.py) to N worker ranks running
TorchEngine, one rank per GPU.  This is the serving integration of the plane:
the same TickGateway core drives bench.py, so the measured benchmark number is
the product's number.

Reference equivalent of this integration: the collapsed same-host engine path
wired in as a first-class router transport —
crates/engine_zmq_client/src/connector.rs:235 (EngineCoreClient submit ->
RequestStream) + model_gateway/src/routers/grpc/zmq_client.rs:1-12 (the ZMQ
adapter presenting the engine surface to the router).  Here the "collapsed
hop" is RCCL p2p over xGMI instead of ZMQ ipc://.

Topology (torchrun, one process per GPU):
  rank 0   — the gateway: HTTP server + policy + plane + its OWN engine
  rank 1.. — worker loop (run_worker_loop): plane tick <-> TorchEngine step

Single-rank mode (world=1) serves from the local engine with no plane.
"""
from __future__ import annotations

import asyncio
import itertools
import json
import statistics
import threading
import time

import numpy as np
from collections import deque
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple

from ..comm.plane import (
    DONE,
    EMB_RECV,
    EMB_SEND,
    KV_RECV,
    KV_SEND,
    PIX_RECV,
    PIX_SEND,
    PREFILLED,
    GatewayPlane,
    PlaneConfig,
    WorkerPlane,
    execute_transfers,
)

PLEN_INFO = 8
from ..policies import SelectWorkerInfo
from .base import RouteRequest, RouteResponse, Router
from ..workers.worker import Worker


class TickGateway:
    """Synchronous rank-0 serving core: one lockstep tick routes pending
    requests (batched policy select — ONE GPU tree kernel per tick), ships
    them over the plane, steps the local engine while remote workers compute,
    then collects token events.

    Thread-contract: submit() may be called from any thread (deque appends
    are atomic); tick() runs on exactly one thread; on_event fires on the
    tick thread.
    """

    def __init__(
        self,
        workers: Sequence[Worker],
        policy,
        plane: Optional[GatewayPlane] = None,
        local_engine=None,
        decode_burst: int = 1,
        model_id: str = "default",
        max_new_arrivals_per_tick: int = 128,
        on_event: Optional[Callable[[int, int, int], None]] = None,
        metrics=None,
        pd_roles: Optional[Dict[int, str]] = None,
        pipeline: bool = True,
    ):
        self.metrics = metrics
        # pipelined ticks defer token reads one tick; PD needs token values
        # inside the tick (handoff instructions), so it stays synchronous.
        # EPD (encode roles only, no prefill split) pipelines fine: its
        # transfer scheduling is tick-phased, not token-event-driven.
        has_prefill = bool(pd_roles) and any(r == "prefill" for r in pd_roles.values())
        self.pipeline = bool(pipeline) and not has_prefill
        self._prev_handle = None
        self._submit_direct = False  # set per-tick by the pipelined path
        # PD/EPD over the plane: rank -> "prefill" | "decode" | "encode".
        # Arrivals route to PREFILL ranks when a prefill fleet exists
        # (prefill_only role on the worker side); PREFILLED events trigger an
        # xGMI KV handoff to a min-load DECODE rank.  ENCODE ranks run the
        # vision tower: pixel-carrying arrivals ship pixels there first, the
        # embeddings hop rank-to-rank next tick, then the request dispatches.
        self.pd_roles = pd_roles if has_prefill else None
        self.epd_roles = pd_roles if (pd_roles and not has_prefill) else None
        roles = pd_roles or {}
        self.encode_idxs = [i for i, w in enumerate(workers)
                            if roles.get(w.rccl_rank) == "encode"]
        if self.pd_roles:
            self.prefill_idxs = [i for i, w in enumerate(workers)
                                 if roles.get(w.rccl_rank) == "prefill"]
        self.decode_idxs = [i for i, w in enumerate(workers)
                            if roles.get(w.rccl_rank, "decode") == "decode"]
        self._pending_mm: deque = deque()  # (rid, tokens, max_new, pixels)
        self._epd_phase2: List[tuple] = []  # (rid, toks, max_new, e_idx, d_idx)
        self._local_encoder = None  # lazy EncodeWorker (world-1 / local encode)
        self._req_meta: Dict[int, Tuple[int, int]] = {}  # rid -> (n_tokens, max_new)
        self._pd_plen: Dict[int, int] = {}  # rid -> prefilled length (PLEN_INFO)
        self._local_transfers: List[Tuple[int, ...]] = []
        self.workers = list(workers)
        self.policy = policy
        self.plane = plane
        self.local_engine = local_engine
        self.decode_burst = decode_burst
        self.model_id = model_id
        self.max_new_arrivals = max_new_arrivals_per_tick
        self.on_event = on_event
        self._pending: deque = deque()  # (rid, tokens, max_new)
        self._local_pending: List[Tuple[int, int, List[int]]] = []
        self.inflight: Dict[int, int] = {}  # rid -> worker index
        self._rid = itertools.count(1)
        self.completed_total = 0
        self.routing_lat: List[float] = []
        self.phase_t = {"route": 0.0, "local": 0.0, "plane": 0.0, "events": 0.0,
                        "launch": 0.0, "finish": 0.0, "ticks": 0}

    # ---- submission (any thread) -----------------------------------------
    def submit(self, tokens: List[int], max_new: int, rid: Optional[int] = None,
               pixels=None) -> int:
        """`pixels` ([3, H, W] uint8) marks an EPD multimodal request: the
        image ships to an encode rank, its [E, d_model] embeddings hop
        rank-to-rank into the chosen decode rank's prefill, then the text
        request dispatches (reference: grpc EncodeStage + engine-side
        Mooncake; here every hop is the xGMI plane)."""
        if rid is None:
            rid = next(self._rid) & 0x7FFFFFFF
        if self.pd_roles:
            self._req_meta[rid] = (len(tokens), max_new)
        toks = np.asarray(tokens, dtype=np.int64)
        if pixels is not None:
            self._pending_mm.append((rid, toks, max_new, pixels))
            return rid
        # numpy end-to-end: no-copy for ndarray submitters (bench/engine), one
        # conversion for list submitters (HTTP path)
        self._pending.append((rid, toks, max_new))
        return rid

    class _LocalMMEndpoint:
        """engine + encoder composite so the gateway's ORDERED local transfer
        list (KV, EMB and PIX entries interleaved) executes as one sequence —
        splitting it per-target could reorder against a peer's single list
        and deadlock the pairing."""

        def __init__(self, engine, encoder):
            self.engine, self.encoder = engine, encoder
            self.cfg = engine.cfg
            self.dtype = engine.dtype
            self.device = engine.device
            self.kv = engine.kv

        def export_kv(self, rid):
            return self.engine.export_kv(rid)

        def import_kv(self, *a, **k):
            return self.engine.import_kv(*a, **k)

        def kv_transfer_shape(self, n):
            return self.engine.kv_transfer_shape(n)

        def accept_embed(self, rid, t):
            self.engine.accept_embed(rid, t)

        def export_embed(self, rid):
            return self.encoder.export_embed(rid)

        def export_pixels(self, rid):
            return self.encoder.export_pixels(rid)

        def accept_pixels(self, rid, t):
            self.encoder.accept_pixels(rid, t)

    def _encoder(self):
        if self._local_encoder is None:
            import torch

            from ..multimodal.encoder import EncodeWorker, ToyVisionEncoder

            eng = self.local_engine
            d_model = eng.cfg.d_model
            self._local_encoder = EncodeWorker(ToyVisionEncoder(
                d_model, image_size=64, patch=16,
                device=str(getattr(eng, "device", "cpu")),
                dtype=getattr(eng, "dtype", torch.float32)))
        return self._local_encoder

    def _process_mm_arrivals(self, cap: int = 8) -> None:
        """Phase 1 of the EPD hop: ship pixels to the encode rank (or encode
        locally when none exists); the embedding transfer + text dispatch
        happen next tick (_process_epd_phase2)."""
        n = 0
        while self._pending_mm and n < cap:
            rid, toks, max_new, pixels = self._pending_mm.popleft()
            n += 1
            # decode target via min-load over the decode pool
            pool = self.decode_idxs or list(range(len(self.workers)))
            d_idx = min(pool, key=lambda i: self.workers[i].active_requests)
            if self.encode_idxs:
                e_idx = min(self.encode_idxs, key=lambda i: self.workers[i].processed_requests)
                e_rank = self.workers[e_idx].rccl_rank
                C, H, W = pixels.shape
                self._encoder()._pixels[rid] = pixels  # outgoing stash
                self._local_transfers.append((rid, e_rank, PIX_SEND, C, H, W))
                self.plane.enqueue_transfer(e_rank, rid, 0, PIX_RECV, C, H, W)
            else:
                e_idx = None
                self._encoder().accept_pixels(rid, pixels)
            self._epd_phase2.append((rid, toks, max_new, e_idx, d_idx))

    def _process_epd_phase2(self) -> None:
        """Phase 2: embeddings hop encode->decode, then the request itself
        dispatches (worker loops execute transfers BEFORE submits, so the
        embedding is present when the request admits)."""
        if not self._epd_phase2:
            return
        batch, self._epd_phase2 = self._epd_phase2, []
        enc = self._encoder()
        E = enc.embed_len()
        for rid, toks, max_new, e_idx, d_idx in batch:
            d_rank = self.workers[d_idx].rccl_rank or 0
            e_rank = self.workers[e_idx].rccl_rank if e_idx is not None else 0
            if e_idx is not None:
                self.plane.enqueue_transfer(e_rank, rid, d_rank, EMB_SEND, E, 0, 0)
                if d_rank == 0:
                    self._local_transfers.append((rid, e_rank, EMB_RECV, E, 0, 0))
                else:
                    self.plane.enqueue_transfer(d_rank, rid, e_rank, EMB_RECV, E, 0, 0)
            else:
                # locally-encoded embedding
                if d_rank == 0:
                    self.local_engine.accept_embed(rid, enc.export_embed(rid))
                else:
                    self._local_transfers.append((rid, d_rank, EMB_SEND, E, 0, 0))
                    self.plane.enqueue_transfer(d_rank, rid, 0, EMB_RECV, E, 0, 0)
            self.workers[d_idx].incr_load()
            self.inflight[rid] = d_idx
            if d_rank == 0:
                # await_embed holds prefill until the local EMB_RECV flush
                # (remote encode); a locally-encoded embedding already pairs
                self.local_engine.submit(toks, max_new, rid=rid,
                                         await_embed=e_idx is not None)
            else:
                self.plane.enqueue(d_rank, rid, max_new, toks)

    @property
    def pending_count(self) -> int:
        return len(self._pending) + len(self._pending_mm)

    # ---- the tick (one thread) -------------------------------------------
    def tick(self) -> int:
        """One lockstep exchange; returns completions this tick.

        Two forms.  The SYNC tick routes, steps, then collects — simple, and
        required for PD (handoff instructions need this tick's token values).
        The PIPELINED tick (default otherwise) launches the engine first and
        routes while the GPU runs, resolving tick N-1's deferred token reads
        afterwards — completions report one tick late, but the gateway's CPU
        phase no longer leaves the GPU idle (measured 56% GPU-busy before)."""
        if (self.pipeline and self.local_engine is not None
                and hasattr(self.local_engine, "step_launch")):
            return self._tick_pipelined()
        return self._tick_sync()

    def _route_arrivals(self) -> None:
        """Batched policy select for this tick's arrivals (one GPU tree
        kernel); local selections land in _local_pending, remote ones in the
        plane's send queue."""
        new_reqs: List[Tuple[int, List[int], int]] = []
        while self._pending and len(new_reqs) < self.max_new_arrivals:
            new_reqs.append(self._pending.popleft())
        if new_reqs:
            # PD: arrivals go to the PREFILL fleet only; EPD: encode ranks
            # never take text requests
            if self.pd_roles:
                route_pool = [self.workers[i] for i in self.prefill_idxs]
                remap = self.prefill_idxs
            elif self.encode_idxs:
                route_pool = [self.workers[i] for i in self.decode_idxs]
                remap = self.decode_idxs
            else:
                route_pool, remap = self.workers, None
            infos = [
                SelectWorkerInfo(
                    request_id=str(rid), model_id=self.model_id, tokens=toks, est_tokens=len(toks)
                )
                for rid, toks, _ in new_reqs
            ]
            t0 = time.perf_counter()
            if hasattr(self.policy, "select_worker_batch"):
                sels = self.policy.select_worker_batch(route_pool, infos)
            else:
                sels = [self.policy.select_worker(route_pool, i) for i in infos]
            dt = time.perf_counter() - t0
            self.routing_lat.extend([dt / len(new_reqs)] * len(new_reqs))
            if len(self.routing_lat) > 8192:  # long-lived server: cap the window
                del self.routing_lat[:-4096]
            self.phase_t["route"] += dt
            for (rid, toks, max_new), sel in zip(new_reqs, sels):
                sel = 0 if sel is None else sel
                if remap is not None:
                    sel = remap[sel]
                self.workers[sel].incr_load()
                self.inflight[rid] = sel
                rank = self.workers[sel].rccl_rank
                if rank in (None, 0):
                    if self._submit_direct:
                        # pipelined mode: routing runs while the GPU executes
                        # this tick, so submit (incl. prefix-key hashing)
                        # happens here, off the launch critical path
                        self.local_engine.submit(toks, max_new, rid=rid)
                    else:
                        self._local_pending.append((rid, max_new, toks))
                else:
                    self.plane.enqueue(rank, rid, max_new, toks)

    def _tick_sync(self) -> int:
        self.phase_t["ticks"] += 1
        # 1) EPD hops (phase2 before phase1: both sides append per-rid
        # entries in the same order, keeping the pairing deadlock-free),
        # then route this tick's arrivals in one batch (one GPU kernel)
        self._process_epd_phase2()
        self._process_mm_arrivals()
        self._route_arrivals()
        # 2) ship remote work first (send AND recv posted together) so the
        # whole plane exchange overlaps the local engine step
        if self.plane is not None:
            tp = time.perf_counter()
            self.plane.tick_send()
            # the gateway's side of this tick's PD handoffs: executed right
            # after tick_send so the peers' instruction lists (sent in that
            # same tensor) pair with ours in order
            self._flush_local_transfers()
            self.phase_t["plane"] += time.perf_counter() - tp
        local_events: List[Tuple[int, int, int]] = []
        if self.local_engine is not None:
            tl = time.perf_counter()
            for rid, max_new, toks in self._local_pending:
                self.local_engine.submit(toks, max_new, rid=rid)
            self._local_pending.clear()
            self.local_engine.step(decode_burst=self.decode_burst)
            local_events = self.local_engine.drain_events()
            self.phase_t["local"] += time.perf_counter() - tl
        remote_arrays = []
        if self.plane is not None:
            tp = time.perf_counter()
            remote_arrays = list(self.plane.tick_recv().values())
            self.phase_t["plane"] += time.perf_counter() - tp
        # 3) completions + event fan-out
        return self._process_events(local_events, remote_arrays)

    def _tick_pipelined(self) -> int:
        """Pipelined tick: launch first, route while the GPU runs, then
        resolve the PREVIOUS tick's deferred token reads (its CUDA event has
        long signalled, so the wait is free).  Arrivals routed this tick are
        submitted at the next launch (+1 tick admission latency); the GPU is
        never drained between ticks."""
        self.phase_t["ticks"] += 1
        eng = self.local_engine
        # 1) submit locals routed LAST tick, launch this tick's GPU work
        tl = time.perf_counter()
        for rid, max_new, toks in self._local_pending:
            eng.submit(toks, max_new, rid=rid)
        self._local_pending.clear()
        handle = eng.step_launch(self.decode_burst)
        dt = time.perf_counter() - tl
        self.phase_t["local"] += dt
        self.phase_t["launch"] += dt
        # 2) plane exchange posted now (remote enqueues from last tick's
        # routing ride out with it, overlapping the local step)
        if self.plane is not None:
            tp = time.perf_counter()
            self.plane.tick_send()
            self._flush_local_transfers()
            self.phase_t["plane"] += time.perf_counter() - tp
        # 3) route this tick's arrivals while the GPU runs (the tree kernel
        # uses its own HIP stream, so it doesn't queue behind the engine);
        # local selections submit straight into the engine here.  EPD hops
        # scheduled here ride the NEXT tick_send, paired with the local
        # flush that follows it.
        self._process_epd_phase2()
        self._process_mm_arrivals()
        self._submit_direct = True
        try:
            self._route_arrivals()
        finally:
            self._submit_direct = False
        # 4) resolve tick N-1
        tl = time.perf_counter()
        prev, self._prev_handle = self._prev_handle, handle
        local_events: List[Tuple[int, int, int]] = []
        if prev is not None:
            eng.step_finish(prev)
            local_events = eng.drain_events()
        dt = time.perf_counter() - tl
        self.phase_t["local"] += dt
        self.phase_t["finish"] += dt
        remote_arrays = []
        if self.plane is not None:
            tp = time.perf_counter()
            remote_arrays = list(self.plane.tick_recv().values())
            self.phase_t["plane"] += time.perf_counter() - tp
        return self._process_events(local_events, remote_arrays)

    def _drain_pipeline(self) -> int:
        """Resolve any outstanding pipelined handle (used by the shutdown /
        barrier paths so the final tick's completions are accounted)."""
        prev, self._prev_handle = self._prev_handle, None
        if prev is None or self.local_engine is None:
            return 0
        self.local_engine.step_finish(prev)
        return self._process_events(self.local_engine.drain_events(), [])

    def _process_events(self, local_events, remote_arrays) -> int:
        tev = time.perf_counter()
        done_now = 0
        if self.pd_roles:
            local_events, remote_arrays = self._handle_pd_events(local_events, remote_arrays)
        cb = self.on_event
        if cb is not None:
            rows = list(local_events)
            for arr in remote_arrays:
                rows.extend(arr.tolist())
            for rid, token, flags in rows:
                cb(rid, token, flags)
                if flags & DONE:
                    done_now += self._complete(rid)
        else:
            # bench fast path: vectorized DONE extraction, no per-token work
            for rid, _tok, flags in local_events:
                if flags & DONE:
                    done_now += self._complete(rid)
            for arr in remote_arrays:
                if len(arr):
                    for rid in arr[(arr[:, 2] & DONE) != 0, 0].tolist():
                        done_now += self._complete(rid)
        self.completed_total += done_now
        self.phase_t["events"] += time.perf_counter() - tev
        m = self.metrics
        if m is not None and not m._null:
            m.plane_ticks.inc()
            n_ev = len(local_events) + sum(len(a) for a in remote_arrays)
            if n_ev:
                m.plane_events_received.inc(n_ev)
        return done_now

    def _handle_pd_events(self, local_events, remote_arrays):
        """Strip PD control events (PLEN_INFO, PREFILLED) out of the streams
        and schedule the KV handoffs.  The PREFILLED event's token is the
        request's FIRST generated token — it re-enters the stream as a plain
        token event so SSE consumers see it."""
        out_local: List[Tuple[int, int, int]] = []

        def handle(rid: int, token: int, flags: int) -> Optional[Tuple[int, int, int]]:
            if flags == PLEN_INFO:
                self._pd_plen[rid] = token
                return None
            if flags == PREFILLED:
                self._schedule_handoff(rid, first_tok=token)
                return (rid, token, 0)  # first token streams normally
            return (rid, token, flags)

        for rid, token, flags in local_events:
            ev = handle(rid, token, flags)
            if ev is not None:
                out_local.append(ev)
        out_remote = []
        for arr in remote_arrays:
            if len(arr) == 0 or not (arr[:, 2] >= PREFILLED).any():
                out_remote.append(arr)
                continue
            kept = []
            for rid, token, flags in arr.tolist():
                ev = handle(rid, token, flags)
                if ev is not None:
                    kept.append(ev)
            out_local.extend(kept)
        return out_local, out_remote

    def _schedule_handoff(self, rid: int, first_tok: int) -> None:
        plen = self._pd_plen.pop(rid, None)
        meta = self._req_meta.pop(rid, None)
        p_idx = self.inflight.get(rid)
        if plen is None or meta is None or p_idx is None:
            return
        max_new = meta[1]
        # min-load decode rank
        d_idx = min(self.decode_idxs, key=lambda i: self.workers[i].active_requests)
        p_rank = self.workers[p_idx].rccl_rank or 0
        d_rank = self.workers[d_idx].rccl_rank or 0
        self.workers[p_idx].decr_load()
        self.workers[d_idx].incr_load()
        self.inflight[rid] = d_idx
        send_instr = (rid, d_rank, KV_SEND, plen, first_tok, max_new)
        recv_instr = (rid, p_rank, KV_RECV, plen, first_tok, max_new)
        if p_rank == 0:
            self._local_transfers.append(send_instr)
        else:
            self.plane.enqueue_transfer(p_rank, *send_instr)
        if d_rank == 0:
            self._local_transfers.append(recv_instr)
        else:
            self.plane.enqueue_transfer(d_rank, *recv_instr)

    def _complete(self, rid: int) -> int:
        wrk = self.inflight.pop(rid, None)
        if wrk is None:
            return 0
        self.workers[wrk].decr_load()  # also bumps processed_requests
        return 1

    # ---- timing / shutdown ------------------------------------------------
    def _flush_local_transfers(self) -> None:
        if self._local_transfers and self.local_engine is not None:
            trs, self._local_transfers = self._local_transfers, []
            target = (self._LocalMMEndpoint(self.local_engine, self._local_encoder)
                      if self._local_encoder is not None else self.local_engine)
            execute_transfers(target, trs)

    def barrier_sync(self) -> None:
        """Plane barrier tick + dist barrier (bench timing bracket).  Any
        handoffs scheduled in the final event pass still ship with this tick,
        so the gateway's local side must execute too — otherwise a remote
        peer blocks in its matching send/recv (shutdown deadlock)."""
        self._drain_pipeline()
        if self.plane is not None:
            import torch.distributed as dist

            self.plane.tick_send(barrier=True)
            self._flush_local_transfers()
            self.plane.tick_recv()
            dist.barrier()

    def stop_workers(self) -> None:
        self._drain_pipeline()
        if self.plane is not None:
            self.plane.tick_send(stop=True)
            self._flush_local_transfers()
            self.plane.tick_recv()  # drain the workers' final event sends

    def p50_routing_ms(self) -> Optional[float]:
        return statistics.median(self.routing_lat) * 1e3 if self.routing_lat else None


def pd_rank_roles(world: int) -> Dict[int, str]:
    """The plane PD topology rule: odd ranks prefill, even ranks decode."""
    return {r: ("prefill" if r % 2 == 1 else "decode") for r in range(world)}


def epd_rank_roles(world: int) -> Dict[int, str]:
    """The plane EPD topology rule: rank 1 runs the vision encoder, every
    other rank decodes (world 1: the gateway encodes locally)."""
    return {r: ("encode" if r == 1 and world > 1 else "decode") for r in range(world)}


def run_worker_loop(engine, plane: WorkerPlane, decode_burst: int = 1,
                    role: str = "regular") -> float:
    """Worker-rank loop (ranks >= 1): lockstep plane ticks against the local
    engine until the gateway sends STOP.  `role="prefill"` makes every
    submission a PD prefill leg (first token sampled, KV parked for the
    handoff the gateway schedules).  Returns this rank's timed-region
    elapsed seconds when the gateway bracketed the run with barrier ticks
    (bench), else 0."""
    import torch
    import torch.distributed as dist

    if role == "encode":
        # EPD encode rank: no text engine — lockstep ticks + PIX/EMB
        # transfers against the EncodeWorker passed as `engine`
        while True:
            reqs, stop = plane.tick([])
            if plane.transfers:
                execute_transfers(engine, plane.transfers)
            if plane.barrier_requested:
                dist.barrier()
                if hasattr(torch.cuda, "is_available") and torch.cuda.is_available():
                    torch.cuda.synchronize()
            if stop:
                break
        return 0.0

    prefill_kw = {"prefill_only": True} if role == "prefill" else {}
    # Pipelined stepping (regular role): launch this tick's GPU work, resolve
    # the PREVIOUS tick's deferred token reads, ship those events next tick —
    # the worker's host phase (plane staging + submits) overlaps its own GPU.
    # PD roles stay synchronous: transfer instructions consume parked token
    # values inside the tick.
    pipelined = role == "regular" and hasattr(engine, "step_launch")
    prev_handle = None
    outbox: List[tuple] = []  # events awaiting the next plane tick
    t0 = t1 = None
    while True:
        reqs, stop = plane.tick(outbox)
        outbox = []
        # PD handoffs pair with their peers FIRST: the gateway (and peer
        # workers) execute their side right after this tick's send, before
        # any barrier — a barrier before the transfer deadlocks the pair
        if plane.transfers:
            execute_transfers(engine, plane.transfers)
        if plane.barrier_requested:
            if prev_handle is not None:
                # settle in-flight work so the timing bracket is honest
                engine.step_finish(prev_handle)
                prev_handle = None
                outbox.extend(engine.drain_events())
            dist.barrier()
            if hasattr(torch.cuda, "is_available") and torch.cuda.is_available():
                torch.cuda.synchronize()
            if t0 is None:
                t0 = time.perf_counter()
            else:
                t1 = time.perf_counter()
        if stop:
            break
        for rid, max_new, prompt in reqs:
            engine.submit(prompt, max_new, rid=rid, **prefill_kw)
        if pipelined:
            launched = engine.step_launch(decode_burst)
            if prev_handle is not None:
                engine.step_finish(prev_handle)
            prev_handle = launched
        else:
            engine.step(decode_burst=decode_burst)
        outbox.extend(engine.drain_events())
    return (t1 - t0) if (t0 is not None and t1 is not None) else 0.0


# ---------------------------------------------------------------------------
# asyncio serving adapter
# ---------------------------------------------------------------------------
class _ReqState:
    __slots__ = ("queue", "tokens", "created")

    def __init__(self):
        self.queue: asyncio.Queue = asyncio.Queue()
        self.tokens: List[int] = []
        self.created = time.time()


class RcclRouter(Router):
    """`connection_mode=rccl` router: OpenAI chat/completions served over the
    lockstep plane.  A dedicated tick thread drives TickGateway; per-request
    events are forwarded onto the asyncio loop for SSE streaming."""

    router_id = "rccl-regular"

    def __init__(self, ctx, config, *, engine=None, plane=None, world: Optional[int] = None):
        import os

        self.ctx = ctx
        self.config = config
        self.model_id = "default"
        world = world if world is not None else int(os.environ.get("WORLD_SIZE", "1"))
        self.world = world
        # fleet: one Worker per rank; rank 0 is the gateway's own engine
        self.workers = []
        for r in range(world):
            w = Worker(f"rccl://rank-{r}", model_id=self.model_id, rccl_rank=r)
            self.workers.append(w)
            try:
                if ctx.worker_registry.get_by_url(w.url) is None:
                    ctx.worker_registry.register(w)
            except Exception:
                pass
        if engine is None:
            engine = self._build_local_engine(config)
        self.engine = engine
        if plane is None and world > 1:
            import torch
            import torch.distributed as dist

            use_gpu = torch.cuda.is_available()
            if not dist.is_initialized():
                dist.init_process_group(backend="nccl" if use_gpu else "gloo")
            rcfg = config.rccl
            plane = GatewayPlane(
                PlaneConfig(
                    max_reqs_per_tick=rcfg.max_batch_requests,
                    max_prompt=rcfg.max_tokens_per_msg,
                    device="cuda:0" if use_gpu else "cpu",
                ),
                list(range(1, world)),
            )
        try:
            self.loop = asyncio.get_running_loop()
        except RuntimeError:
            self.loop = asyncio.get_event_loop()
        self._states: Dict[int, _ReqState] = {}
        policy = ctx.policy_registry.get(self.model_id)
        # PD serving over the plane: in PREFILL_DECODE mode, odd ranks run
        # the prefill leg and even ranks (incl. the gateway's local engine)
        # decode, with the xGMI KV handoff in between (pd_rank_roles mirrors
        # bench --pd and cli.rccl_worker_main's parity rule)
        from ..config import RoutingMode

        if config.mode == RoutingMode.PREFILL_DECODE and world > 1:
            pd_roles = pd_rank_roles(world)
        elif config.mode == RoutingMode.ENCODE_PREFILL_DECODE and world > 1:
            pd_roles = epd_rank_roles(world)
        else:
            pd_roles = None
        self.gw = TickGateway(
            self.workers,
            policy,
            plane=plane,
            local_engine=engine,
            model_id=self.model_id,
            on_event=self._on_event_tick_thread,
            metrics=getattr(ctx, "metrics", None),
            pd_roles=pd_roles,
        )
        self._stop = False
        self.idle_sleep_s = max(1, config.rccl.tick_interval_us) / 1e6
        self._thread = threading.Thread(target=self._tick_loop, name="rccl-tick", daemon=True)
        self._thread.start()

    @staticmethod
    def _build_local_engine(config):
        import torch

        from ..engine.torch_engine import TorchEngine, TorchEngineConfig

        if torch.cuda.is_available():
            cfg = TorchEngineConfig()
            # serving default sized for 288 GB HBM3E: 256 concurrent slots
            # (the 64-slot dataclass default is the CI/tiny-memory shape).
            # graphs=False: the pipelined tick overlaps eager launches fully
            # and hipGraph capture/replay nets negative (steps-60 A/B:
            # 1464.5 no-graphs vs 1395.9 graphs)
            cfg.max_slots = 256
            return TorchEngine(cfg, device="cuda:0", graphs=False)
        return TorchEngine(TorchEngineConfig.tiny(), device="cpu")

    # ---- tick thread ------------------------------------------------------
    def _tick_loop(self):
        gw = self.gw
        while not self._stop:
            busy = gw.inflight or gw.pending_count
            gw.tick()
            if not busy:
                time.sleep(self.idle_sleep_s)
        gw.stop_workers()

    def _on_event_tick_thread(self, rid: int, token: int, flags: int):
        self.loop.call_soon_threadsafe(self._on_event, rid, token, flags)

    def _on_event(self, rid: int, token: int, flags: int):
        st = self._states.get(rid)
        if st is not None:
            st.queue.put_nowait((token, flags))

    # ---- request helpers ---------------------------------------------------
    def _tokenizer(self):
        reg = getattr(self.ctx, "tokenizer_registry", None)
        tok = reg.get(self.model_id) if reg else None
        if tok is None:
            from ..tokenizer.registry import MockTokenizer

            tok = MockTokenizer()
        return tok

    def _encode_chat(self, body: Dict[str, Any], tok) -> List[int]:
        msgs = body.get("messages") or []
        tpl = getattr(self.ctx, "chat_template", None)
        if tpl is not None:
            try:
                text = tpl.render(msgs, add_generation_prompt=True, tools=body.get("tools"))
            except Exception:
                text = None
        else:
            text = None
        if text is None:
            text = "\n".join(f"{m.get('role', 'user')}: {_content_text(m.get('content'))}" for m in msgs)
        return tok.encode(text)

    async def route(self, req: RouteRequest) -> RouteResponse:
        if req.path.endswith("/chat/completions"):
            return await self.route_chat(req)
        if req.path.endswith("/completions"):
            return await self.route_completion(req)
        if req.path.endswith("/generate"):
            return await self.route_generate(req)
        return RouteResponse(status=404, body=b'{"error":"unsupported path for rccl router"}')

    async def _extract_pixels(self, body: Dict[str, Any]):
        """First image_url content part (data:/file URL) -> [3, H, W] u8
        tensor for the EPD pixel path, or None for text-only requests."""
        for m in body.get("messages") or []:
            content = m.get("content")
            if not isinstance(content, list):
                continue
            for part in content:
                if isinstance(part, dict) and part.get("type") == "image_url":
                    url = (part.get("image_url") or {}).get("url") or ""
                    import numpy as np_
                    import torch

                    from ..multimodal.media import decode_image, fetch_image_bytes

                    raw = await fetch_image_bytes(url)
                    arr = decode_image(raw)  # [H, W, 3] u8
                    return torch.from_numpy(np_.ascontiguousarray(
                        arr.transpose(2, 0, 1)))
        return None

    async def route_chat(self, req: RouteRequest) -> RouteResponse:
        body = req.body or {}
        tok = self._tokenizer()
        input_ids = self._encode_chat(body, tok)
        try:
            pixels = await self._extract_pixels(body)
        except Exception as exc:
            return RouteResponse(status=400, body=json.dumps(
                {"error": {"message": f"image fetch/decode failed: {exc}",
                           "type": "invalid_request_error"}}).encode())
        return await self._serve(req, body, tok, input_ids, chat=True, pixels=pixels)

    async def route_completion(self, req: RouteRequest) -> RouteResponse:
        body = req.body or {}
        tok = self._tokenizer()
        prompt = body.get("prompt") or ""
        if isinstance(prompt, list):
            prompt = prompt[0] if prompt else ""
        input_ids = prompt if isinstance(prompt, list) else tok.encode(str(prompt))
        return await self._serve(req, body, tok, input_ids, chat=False)

    async def route_generate(self, req: RouteRequest) -> RouteResponse:
        body = req.body or {}
        tok = self._tokenizer()
        ids = body.get("input_ids") or tok.encode(str(body.get("text") or ""))
        body2 = {"max_tokens": (body.get("sampling_params") or {}).get("max_new_tokens", 16)}
        return await self._serve(req, body2, tok, ids, chat=False, generate=True)

    async def _serve(self, req, body, tok, input_ids, chat: bool, generate: bool = False,
                     pixels=None) -> RouteResponse:
        max_new = int(
            body.get("max_completion_tokens") or body.get("max_tokens") or 16
        )
        st = _ReqState()
        rid = self.gw.submit(input_ids, max_new, pixels=pixels)
        self._states[rid] = st
        model = body.get("model") or self.model_id
        stream = bool(body.get("stream"))
        if stream:
            return RouteResponse(
                status=200,
                headers={"Content-Type": "text/event-stream"},
                stream=self._sse(rid, st, tok, model, chat),
            )
        try:
            timeout = self.config.request_timeout_secs
            deadline = time.monotonic() + timeout
            while True:
                tk, flags = await asyncio.wait_for(st.queue.get(), timeout=max(0.1, deadline - time.monotonic()))
                st.tokens.append(tk)
                if flags & DONE:
                    break
        except asyncio.TimeoutError:
            return RouteResponse(status=504, body=b'{"error":"generation timed out"}')
        finally:
            self._states.pop(rid, None)
        text = tok.decode(st.tokens)
        if generate:
            out = {"text": text, "output_ids": st.tokens, "meta_info": {"completion_tokens": len(st.tokens)}}
        elif chat:
            out = _chat_response(rid, model, text, len(input_ids), len(st.tokens))
        else:
            out = _completion_response(rid, model, text, len(input_ids), len(st.tokens))
        return RouteResponse(status=200, body=json.dumps(out).encode())

    async def _sse(self, rid: int, st: _ReqState, tok, model: str, chat: bool):
        obj = "chat.completion.chunk" if chat else "text_completion"
        created = int(time.time())
        n_out = 0
        try:
            if chat:
                first = {
                    "id": f"chatcmpl-{rid}", "object": obj, "created": created, "model": model,
                    "choices": [{"index": 0, "delta": {"role": "assistant", "content": ""}, "finish_reason": None}],
                }
                yield f"data: {json.dumps(first)}\n\n".encode()
            while True:
                tk, flags = await asyncio.wait_for(st.queue.get(), timeout=self.config.request_timeout_secs)
                st.tokens.append(tk)
                piece = tok.decode_incremental(st.tokens, len(st.tokens) - 1)
                n_out += 1
                done = bool(flags & DONE)
                if chat:
                    chunk = {
                        "id": f"chatcmpl-{rid}", "object": obj, "created": created, "model": model,
                        "choices": [{
                            "index": 0,
                            "delta": {"content": piece},
                            "finish_reason": "stop" if done else None,
                        }],
                    }
                else:
                    chunk = {
                        "id": f"cmpl-{rid}", "object": obj, "created": created, "model": model,
                        "choices": [{"index": 0, "text": piece, "finish_reason": "stop" if done else None}],
                    }
                yield f"data: {json.dumps(chunk)}\n\n".encode()
                if done:
                    break
            yield b"data: [DONE]\n\n"
        finally:
            self._states.pop(rid, None)

    async def get_loads(self) -> Dict[str, Any]:
        return {
            "workers": [
                {"url": w.url, "active": w.active_requests, "processed": w.processed_requests}
                for w in self.workers
            ],
            "p50_routing_ms": self.gw.p50_routing_ms(),
            "completed": self.gw.completed_total,
        }

    async def shutdown(self) -> None:
        self._stop = True
        self._thread.join(timeout=10)


def _content_text(content) -> str:
    if isinstance(content, str):
        return content
    if isinstance(content, list):
        return " ".join(p.get("text", "") for p in content if isinstance(p, dict))
    return ""


def _chat_response(rid, model, text, n_in, n_out):
    return {
        "id": f"chatcmpl-{rid}",
        "object": "chat.completion",
        "created": int(time.time()),
        "model": model,
        "choices": [{
            "index": 0,
            "message": {"role": "assistant", "content": text},
            "finish_reason": "stop",
        }],
        "usage": {"prompt_tokens": n_in, "completion_tokens": n_out, "total_tokens": n_in + n_out},
    }


def _completion_response(rid, model, text, n_in, n_out):
    return {
        "id": f"cmpl-{rid}",
        "object": "text_completion",
        "created": int(time.time()),
        "model": model,
        "choices": [{"index": 0, "text": text, "finish_reason": "stop"}],
        "usage": {"prompt_tokens": n_in, "completion_tokens": n_out, "total_tokens": n_in + n_out},
    }
