"""RCCL-over-xGMI data plane (SURVEY.md §2.5, §5.8 MI355X-native equivalent).

Replaces loopback HTTP/gRPC/ZMQ/SHM between the gateway and co-located
workers with persistent torch.distributed point-to-point channels — RCCL over
xGMI on GPU (backend "nccl" IS RCCL on ROCm), gloo on CPU for tests.

Design: RCCL wants long-lived communicators and a fixed message order per
peer (no tags), so the plane is a fixed-cadence LOCKSTEP TICK:

    every tick, for every worker rank w:
        gateway  -> w : one request tensor  (new requests, padded, int32)
        w -> gateway  : one event tensor    (token/done events, padded, int32)

Exactly one send and one recv per peer per tick, always in the same order —
deadlock-free by construction.  isend/irecv across peers overlap on the tick,
and the gateway posts BOTH directions at tick_send time so the whole exchange
overlaps its local engine step (the world-8 rehearsal showed the serialized
recv+Python-parse path capping scaling at 44% efficiency; see
scripts/world8_rehearsal.py).

Host staging is numpy-vectorized into persistent (pinned on GPU) buffers —
no per-request torch tensor construction, no per-event Python tuple parsing.

Wire layout (int32):
  request tensor [1 + MAX_REQS*(3 + MAX_PROMPT)]:
      [0] = n_new | (STOP_FLAG<<16)
      per request: rid, max_new_tokens, prompt_len, prompt_len tokens (padded)
  event tensor [1 + MAX_EVENTS*3]:
      [0] = n_events
      per event: rid, token, flags (1 = done)
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

STOP_FLAG = 1 << 16
BARRIER_FLAG = 1 << 17  # worker joins a dist.barrier() + device sync after this tick
DONE = 1
ABORT = 2
PREFILLED = 4  # PD prefill leg finished; KV parked for transfer

# PD/EPD handoff instructions carried in the request tensor (both endpoint
# ranks receive the same ordered list, so their send/recv pairs match by
# construction — deadlock-free like the main lockstep exchange)
KV_SEND = 1   # PD: prefill rank ships a parked KV window
KV_RECV = 2
EMB_SEND = 3  # EPD: encode rank ships [n_tokens, d_model] vision embeddings
EMB_RECV = 4
PIX_SEND = 5  # EPD: gateway ships [C, H, W] u8 pixels to the encode rank
PIX_RECV = 6  # (fields: n_tokens=C, first_tok=H, max_new=W)
TR_INTS = 6  # rid, peer_rank, direction, n_tokens, first_tok, max_new


@dataclass
class PlaneConfig:
    max_reqs_per_tick: int = 64
    max_prompt: int = 1024
    max_events_per_tick: int = 4096
    max_transfers_per_tick: int = 32
    device: str = "cpu"  # "cuda:<rank>" on GPU

    @property
    def req_stride(self) -> int:
        return 3 + self.max_prompt

    @property
    def tr_off(self) -> int:
        return 1 + self.max_reqs_per_tick * self.req_stride

    @property
    def req_len(self) -> int:
        return self.tr_off + 1 + self.max_transfers_per_tick * TR_INTS

    @property
    def ev_len(self) -> int:
        return 1 + self.max_events_per_tick * 3


def _host_staging(n: int, gpu: bool) -> Tuple[torch.Tensor, np.ndarray]:
    t = torch.zeros(n, dtype=torch.int32, pin_memory=gpu)
    return t, t.numpy()


class GatewayPlane:
    """Rank-0 side: one staging pair per worker rank."""

    def __init__(self, cfg: PlaneConfig, worker_ranks: List[int]):
        self.cfg = cfg
        self.worker_ranks = worker_ranks
        dev = torch.device(cfg.device)
        self._gpu = dev.type == "cuda"
        self._send = {w: torch.zeros(cfg.req_len, dtype=torch.int32, device=dev) for w in worker_ranks}
        self._recv = {w: torch.zeros(cfg.ev_len, dtype=torch.int32, device=dev) for w in worker_ranks}
        # persistent host staging (pinned when the wire tensors are on-device)
        self._send_h: Dict[int, Tuple[torch.Tensor, np.ndarray]] = {}
        self._recv_h: Dict[int, Tuple[torch.Tensor, np.ndarray]] = {}
        for w in worker_ranks:
            if self._gpu:
                self._send_h[w] = _host_staging(cfg.req_len, True)
                self._recv_h[w] = _host_staging(cfg.ev_len, True)
            else:
                self._send_h[w] = (self._send[w], self._send[w].numpy())
                self._recv_h[w] = (self._recv[w], self._recv[w].numpy())
        self._pending: Dict[int, List[Tuple[int, int, List[int]]]] = {w: [] for w in worker_ranks}
        self._pending_tr: Dict[int, List[Tuple[int, ...]]] = {w: [] for w in worker_ranks}
        self._works = []  # outstanding isend/irecv handles for this tick
        # NCCL Work.wait() only adds a STREAM dependency — before the host
        # rewrites the pinned staging, the previous tick's H2D wire copy must
        # have actually completed (event host-sync; gloo/cpu path skips it)
        self._h2d_ev = torch.cuda.Event() if self._gpu else None

    def enqueue(self, worker_rank: int, rid: int, max_new: int, prompt: List[int]) -> None:
        self._pending[worker_rank].append((rid, max_new, prompt[-self.cfg.max_prompt:]))

    def enqueue_transfer(self, worker_rank: int, rid: int, peer: int, direction: int,
                         n_tokens: int, first_tok: int, max_new: int) -> None:
        """Instruct `worker_rank` to send (KV_SEND) or receive (KV_RECV) the
        KV of `rid` with `peer` next tick (PD handoff).  The gateway enqueues
        the matching instruction on both endpoints in the same tick."""
        self._pending_tr[worker_rank].append((rid, peer, direction, n_tokens, first_tok, max_new))

    def tick_send(self, stop: bool = False, barrier: bool = False) -> None:
        """Phase 1: ship this tick's requests to every worker AND post the
        event recvs, so the whole exchange progresses while the gateway runs
        its local engine step."""
        cfg = self.cfg
        # previous tick's handles must be drained before buffers are reused
        for work in self._works:
            work.wait()
        self._works = []
        if self._h2d_ev is not None:
            self._h2d_ev.synchronize()  # staging safe to rewrite (see __init__)
        for w in self.worker_ranks:
            pend = self._pending[w][: cfg.max_reqs_per_tick]
            self._pending[w] = self._pending[w][len(pend):]
            hdr = len(pend) | (STOP_FLAG if stop else 0) | (BARRIER_FLAG if barrier else 0)
            _, rows = self._send_h[w]
            rows[0] = hdr
            for i, (rid, max_new, prompt) in enumerate(pend):
                base = 1 + i * cfg.req_stride
                rows[base] = rid
                rows[base + 1] = max_new
                rows[base + 2] = len(prompt)
                rows[base + 3: base + 3 + len(prompt)] = prompt  # numpy list assign, one C pass
            trs = self._pending_tr[w][: cfg.max_transfers_per_tick]
            self._pending_tr[w] = self._pending_tr[w][len(trs):]
            rows[cfg.tr_off] = len(trs)
            for i, tr in enumerate(trs):
                base = cfg.tr_off + 1 + i * TR_INTS
                rows[base: base + TR_INTS] = tr
            if self._gpu:
                self._send[w].copy_(self._send_h[w][0], non_blocking=True)
        if self._h2d_ev is not None:
            self._h2d_ev.record()
        ops = []
        for w in self.worker_ranks:
            ops.append(dist.P2POp(dist.isend, self._send[w], w))
            ops.append(dist.P2POp(dist.irecv, self._recv[w], w))
        if ops:
            self._works = dist.batch_isend_irecv(ops)

    def tick_recv(self) -> Dict[int, np.ndarray]:
        """Phase 2: collect every worker's event tensor.  Returns
        {worker_rank: int32 array [n_events, 3] of (rid, token, flags)}."""
        cfg = self.cfg
        for work in self._works:
            work.wait()
        self._works = []
        if self._gpu:
            for w in self.worker_ranks:
                self._recv_h[w][0].copy_(self._recv[w], non_blocking=True)
            torch.cuda.synchronize()
        out: Dict[int, np.ndarray] = {}
        for w in self.worker_ranks:
            ev = self._recv_h[w][1]
            n = min(int(ev[0]), cfg.max_events_per_tick)
            out[w] = ev[1: 1 + n * 3].reshape(n, 3).copy()
        return out

    def tick(self, stop: bool = False, barrier: bool = False) -> Dict[int, np.ndarray]:
        """One lockstep exchange (send + recv back-to-back).  The serving tick
        loop uses tick_send()/tick_recv() split around the gateway's local
        engine step so remote workers compute concurrently."""
        self.tick_send(stop=stop, barrier=barrier)
        if stop:
            for work in self._works:
                work.wait()
            self._works = []
            return {}
        return self.tick_recv()


KV_TAG = 7  # gloo matches p2p by tag-slot: the KV stream must not share the
# tick exchange's slot 0 (observed: a KV send matched the event irecv and
# aborted the pair).  NCCL ignores tags and orders per pair, where our
# program order already pairs correctly.


def execute_transfers(engine, transfers) -> int:
    """Run one tick's ordered PD KV handoffs against the local engine.
    Entries: (rid, peer, direction, n_tokens, first_tok, max_new).  Both
    endpoints iterate the SAME list, so sends and recvs pair in order; fp8
    caches travel as uint8 views (NCCL has no fp8 dtype)."""
    tag = KV_TAG if dist.get_backend() == "gloo" else 0
    done = 0
    for rid, peer, direction, n_tokens, first_tok, max_new in transfers:
        if direction == KV_SEND:
            t, plen, _ft = engine.export_kv(rid)
            payload = t.view(torch.uint8) if t.dtype == torch.float8_e4m3fn else t
            dist.send(payload.contiguous(), dst=peer, tag=tag)
        elif direction == KV_RECV:
            shape = engine.kv_transfer_shape(n_tokens)
            buf = torch.empty(shape, dtype=engine.kv.dtype, device=engine.kv.device)
            payload = buf.view(torch.uint8) if buf.dtype == torch.float8_e4m3fn else buf
            dist.recv(payload, src=peer, tag=tag)
            engine.import_kv(rid, buf, n_tokens, first_tok, max_new)
        elif direction == EMB_SEND:
            emb = engine.export_embed(rid)  # [n_tokens, d_model]
            dist.send(emb.contiguous(), dst=peer, tag=tag)
        elif direction == EMB_RECV:
            buf = torch.empty(n_tokens, engine.cfg.d_model, dtype=engine.dtype,
                              device=engine.device)
            dist.recv(buf, src=peer, tag=tag)
            engine.accept_embed(rid, buf)
        elif direction == PIX_SEND:
            px = engine.export_pixels(rid)  # [C, H, W] u8
            dist.send(px.contiguous(), dst=peer, tag=tag)
        elif direction == PIX_RECV:
            buf = torch.empty(n_tokens, first_tok, max_new, dtype=torch.uint8,
                              device=getattr(engine, "device", "cpu"))
            dist.recv(buf, src=peer, tag=tag)
            engine.accept_pixels(rid, buf)
        done += 1
    return done


class WorkerPlane:
    """Worker-rank side: mirror of the gateway's per-tick exchange."""

    def __init__(self, cfg: PlaneConfig, gateway_rank: int = 0):
        self.cfg = cfg
        self.gateway_rank = gateway_rank
        dev = torch.device(cfg.device)
        self._gpu = dev.type == "cuda"
        self._recv = torch.zeros(cfg.req_len, dtype=torch.int32, device=dev)
        self._send = torch.zeros(cfg.ev_len, dtype=torch.int32, device=dev)
        if self._gpu:
            self._recv_h = _host_staging(cfg.req_len, True)
            self._send_h = _host_staging(cfg.ev_len, True)
        else:
            self._recv_h = (self._recv, self._recv.numpy())
            self._send_h = (self._send, self._send.numpy())
        self._send_work = None
        self.barrier_requested = False
        self.transfers: List[Tuple[int, ...]] = []
        self._h2d_ev = torch.cuda.Event() if self._gpu else None  # see GatewayPlane

    def tick(self, events: List[Tuple[int, int, int]]) -> Tuple[List[Tuple[int, int, List[int]]], bool]:
        """One lockstep exchange: sends `events` [(rid, token, flags)], receives
        new requests.  Returns (new_requests, stop).

        Only the request RECV is awaited here: the event send drains while the
        gateway runs its own local engine step (its recv is posted alongside
        its send), so worker and gateway compute concurrently.  The send
        handle is awaited at the next tick before the buffer is reused."""
        cfg = self.cfg
        if self._send_work is not None:
            self._send_work.wait()
            self._send_work = None
        if self._h2d_ev is not None:
            self._h2d_ev.synchronize()  # last tick's H2D truly done: staging reusable
        _, ev = self._send_h
        n = min(len(events), cfg.max_events_per_tick)
        ev[0] = n
        if n:
            ev[1: 1 + n * 3] = np.asarray(events[:n], dtype=np.int32).reshape(-1)
        if self._gpu:
            self._send.copy_(self._send_h[0], non_blocking=True)
        if self._h2d_ev is not None:
            self._h2d_ev.record()
        ops = [
            dist.P2POp(dist.irecv, self._recv, self.gateway_rank),
            dist.P2POp(dist.isend, self._send, self.gateway_rank),
        ]
        works = dist.batch_isend_irecv(ops)
        works[0].wait()  # requests arrived; step can start
        self._send_work = works[1]
        if self._gpu:
            self._recv_h[0].copy_(self._recv, non_blocking=True)
            torch.cuda.synchronize()
        req = self._recv_h[1]
        hdr = int(req[0])
        stop = bool(hdr & STOP_FLAG)
        self.barrier_requested = bool(hdr & BARRIER_FLAG)
        n_new = hdr & 0xFFFF
        out = []
        for i in range(min(n_new, cfg.max_reqs_per_tick)):
            base = 1 + i * cfg.req_stride
            rid = int(req[base])
            max_new = int(req[base + 1])
            plen = int(req[base + 2])
            prompt = req[base + 3: base + 3 + plen].tolist()
            out.append((rid, max_new, prompt))
        # PD KV-handoff instructions for this tick (ordered; both endpoints
        # hold the same list so their p2p calls pair up)
        n_tr = int(req[cfg.tr_off])
        self.transfers = [
            tuple(int(x) for x in req[cfg.tr_off + 1 + i * TR_INTS:
                                      cfg.tr_off + 1 + (i + 1) * TR_INTS])
            for i in range(min(n_tr, cfg.max_transfers_per_tick))
        ]
        return out, stop
