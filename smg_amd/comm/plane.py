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
deadlock-free by construction.  isend/irecv across peers overlap on the tick.
Tensors live on the GPU: token ids move over xGMI without host staging.

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
from typing import Dict, List, Tuple

import torch
import torch.distributed as dist

STOP_FLAG = 1 << 16
BARRIER_FLAG = 1 << 17  # worker joins a dist.barrier() + device sync after this tick
DONE = 1
ABORT = 2


@dataclass
class PlaneConfig:
    max_reqs_per_tick: int = 64
    max_prompt: int = 1024
    max_events_per_tick: int = 4096
    device: str = "cpu"  # "cuda:<rank>" on GPU

    @property
    def req_stride(self) -> int:
        return 3 + self.max_prompt

    @property
    def req_len(self) -> int:
        return 1 + self.max_reqs_per_tick * self.req_stride

    @property
    def ev_len(self) -> int:
        return 1 + self.max_events_per_tick * 3


class GatewayPlane:
    """Rank-0 side: one staging pair per worker rank."""

    def __init__(self, cfg: PlaneConfig, worker_ranks: List[int]):
        self.cfg = cfg
        self.worker_ranks = worker_ranks
        dev = torch.device(cfg.device)
        self._send = {w: torch.zeros(cfg.req_len, dtype=torch.int32, device=dev) for w in worker_ranks}
        self._recv = {w: torch.zeros(cfg.ev_len, dtype=torch.int32, device=dev) for w in worker_ranks}
        self._pending: Dict[int, List[Tuple[int, int, List[int]]]] = {w: [] for w in worker_ranks}

    def enqueue(self, worker_rank: int, rid: int, max_new: int, prompt: List[int]) -> None:
        self._pending[worker_rank].append((rid, max_new, prompt[-self.cfg.max_prompt:]))

    def tick_send(self, stop: bool = False, barrier: bool = False) -> None:
        """Phase 1: ship this tick's requests to every worker (the workers
        start their engine step as soon as the send lands, overlapping the
        gateway's own local work)."""
        cfg = self.cfg
        for w in self.worker_ranks:
            buf = self._send[w]
            pend = self._pending[w][: cfg.max_reqs_per_tick]
            self._pending[w] = self._pending[w][len(pend):]
            hdr = len(pend) | (STOP_FLAG if stop else 0) | (BARRIER_FLAG if barrier else 0)
            rows = torch.zeros(cfg.req_len, dtype=torch.int32)
            rows[0] = hdr
            for i, (rid, max_new, prompt) in enumerate(pend):
                base = 1 + i * cfg.req_stride
                rows[base] = rid
                rows[base + 1] = max_new
                rows[base + 2] = len(prompt)
                rows[base + 3: base + 3 + len(prompt)] = torch.tensor(prompt, dtype=torch.int32)
            buf.copy_(rows.to(buf.device))
        ops = [dist.P2POp(dist.isend, self._send[w], w) for w in self.worker_ranks]
        if ops:
            for work in dist.batch_isend_irecv(ops):
                work.wait()

    def tick_recv(self) -> Dict[int, List[Tuple[int, int, int]]]:
        """Phase 2: collect every worker's event tensor."""
        cfg = self.cfg
        ops = [dist.P2POp(dist.irecv, self._recv[w], w) for w in self.worker_ranks]
        if ops:
            for work in dist.batch_isend_irecv(ops):
                work.wait()
        out: Dict[int, List[Tuple[int, int, int]]] = {}
        for w in self.worker_ranks:
            ev = self._recv[w].cpu()
            n = int(ev[0])
            events = []
            for i in range(min(n, cfg.max_events_per_tick)):
                base = 1 + i * 3
                events.append((int(ev[base]), int(ev[base + 1]), int(ev[base + 2])))
            out[w] = events
        return out

    def tick(self, stop: bool = False, barrier: bool = False) -> Dict[int, List[Tuple[int, int, int]]]:
        """One lockstep exchange (send + recv back-to-back).  The bench's tick
        loop uses tick_send()/tick_recv() split around the gateway's local
        engine step so remote workers compute concurrently."""
        self.tick_send(stop=stop, barrier=barrier)
        if stop:
            return {}
        return self.tick_recv()


class WorkerPlane:
    """Worker-rank side: mirror of the gateway's per-tick exchange."""

    def __init__(self, cfg: PlaneConfig, gateway_rank: int = 0):
        self.cfg = cfg
        self.gateway_rank = gateway_rank
        dev = torch.device(cfg.device)
        self._recv = torch.zeros(cfg.req_len, dtype=torch.int32, device=dev)
        self._send = torch.zeros(cfg.ev_len, dtype=torch.int32, device=dev)
        self._send_work = None

    def tick(self, events: List[Tuple[int, int, int]]) -> Tuple[List[Tuple[int, int, List[int]]], bool]:
        """One lockstep exchange: sends `events` [(rid, token, flags)], receives
        new requests.  Returns (new_requests, stop).

        Only the request RECV is awaited here: the event send drains while the
        gateway runs its own local engine step (its recv is posted after), so
        worker and gateway compute concurrently.  The send handle is awaited
        at the next tick before the buffer is reused."""
        cfg = self.cfg
        if self._send_work is not None:
            self._send_work.wait()
            self._send_work = None
        ev = torch.zeros(cfg.ev_len, dtype=torch.int32)
        n = min(len(events), cfg.max_events_per_tick)
        ev[0] = n
        for i, (rid, token, flags) in enumerate(events[:n]):
            base = 1 + i * 3
            ev[base], ev[base + 1], ev[base + 2] = rid, token, flags
        self._send.copy_(ev.to(self._send.device))
        ops = [
            dist.P2POp(dist.irecv, self._recv, self.gateway_rank),
            dist.P2POp(dist.isend, self._send, self.gateway_rank),
        ]
        works = dist.batch_isend_irecv(ops)
        works[0].wait()  # requests arrived; step can start
        self._send_work = works[1]
        req = self._recv.cpu()
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
        return out, stop
