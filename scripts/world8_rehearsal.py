#!/usr/bin/env python3
"""World-8 rank-0 bottleneck rehearsal (VERDICT r01 missing #2) — no GPU.

Runs the PRODUCT serving core (routers/rccl_router.py TickGateway +
run_worker_loop) over gloo at world N with a DelayEngine that mimics the
measured MI355X per-tick engine cost (bench profile: local engine step
~8-10 ms/tick at 512 in-flight, decode burst 2).  Measures per-rank tick
occupancy — the fraction of wall time each rank spends computing vs waiting
on the plane — so a rank-0 serialization bottleneck shows up BEFORE the
driver's multi-GPU SCALE run.

Usage:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port 29590 scripts/world8_rehearsal.py \
      --seconds 6 --step-ms 8.0

Rank 0 prints one JSON line with per-rank occupancy and throughput.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch.distributed as dist

from smg_amd.comm.plane import GatewayPlane, PlaneConfig, WorkerPlane
from smg_amd.config import PolicyConfig
from smg_amd.policies import create_policy
from smg_amd.routers.rccl_router import TickGateway
from smg_amd.workers.worker import Worker


class DelayEngine:
    """Engine stand-in with the measured per-tick cost of the GPU engine.

    Mirrors the pipelined TorchEngine API: step_launch busy-costs
    `launch_ms` of HOST time (the measured launch phase: graph replays +
    submits + bookkeeping) and schedules the "GPU" to be done step_ms after
    the device frees up; step_finish busy-waits to that ready time (the
    CUDA-event sync).  The pipelined gateway therefore overlaps its routing
    phase with the simulated device exactly as on hardware."""

    def __init__(self, step_ms: float, launch_ms: float = 0.0, max_new_default: int = 32):
        self.step_ms = step_ms
        self.launch_ms = launch_ms
        self.active = {}  # rid -> [produced, max_new]
        self.events = []
        self.busy_s = 0.0
        self._gpu_free = 0.0

    def submit(self, prompt, max_new, rid=None):
        self.active[rid] = [0, max_new]
        return rid

    def _advance(self, decode_burst: int):
        evs, done = [], []
        for rid, st in self.active.items():
            for _ in range(decode_burst):
                st[0] += 1
                fin = st[0] >= st[1]
                evs.append((rid, 1000 + st[0], 1 if fin else 0))
                if fin:
                    done.append(rid)
                    break
        for rid in done:
            del self.active[rid]
        return evs

    def step_launch(self, decode_burst: int = 1):
        t0 = time.perf_counter()
        target = t0 + self.launch_ms / 1e3
        while time.perf_counter() < target:
            pass  # busy-wait: sleep() oversleeps at ms scale and hides CPU contention
        ready = max(self._gpu_free, t0) + self.step_ms / 1e3
        self._gpu_free = ready
        evs = self._advance(decode_burst)
        self.busy_s += time.perf_counter() - t0
        return (ready, evs)

    def step_finish(self, handle):
        ready, evs = handle
        t0 = time.perf_counter()
        while time.perf_counter() < ready:
            pass  # the CUDA-event sync: waits out the remaining device time
        self.events.extend(evs)
        self.busy_s += time.perf_counter() - t0
        return len(evs)

    def step(self, decode_burst: int = 2):
        return self.step_finish(self.step_launch(decode_burst))

    def drain_events(self):
        out, self.events = self.events, []
        return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=6.0)
    ap.add_argument("--step-ms", type=float, default=13.0,
                    help="simulated GPU device time per tick (r02 profile: ~13 ms)")
    ap.add_argument("--launch-ms", type=float, default=4.0,
                    help="host launch-phase cost per tick (r02 steady state)")
    ap.add_argument("--concurrency", type=int, default=64, help="per-rank in-flight")
    ap.add_argument("--max-new", type=int, default=32)
    ap.add_argument("--decode-burst", type=int, default=1)
    ap.add_argument("--prompt-len", type=int, default=576)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    dist.init_process_group("gloo")
    pcfg = PlaneConfig(max_prompt=args.prompt_len + 8, device="cpu")
    eng = DelayEngine(args.step_ms, launch_ms=args.launch_ms)

    if rank == 0:
        plane = GatewayPlane(pcfg, list(range(1, world))) if world > 1 else None
        workers = [Worker(f"rccl://rank-{r}", rccl_rank=r) for r in range(world)]
        policy = create_policy(PolicyConfig(name="cache_aware", gpu_tree=False))
        gw = TickGateway(workers, policy, plane=plane, local_engine=eng,
                         decode_burst=args.decode_burst)
        rid = 0
        prompt = list(range(args.prompt_len))
        target = args.concurrency * world
        t_start = time.perf_counter()
        t_end = t_start + args.seconds
        while time.perf_counter() < t_end:
            n_new = 0
            while len(gw.inflight) + gw.pending_count < target and n_new < 128:
                rid += 1
                gw.submit(prompt, args.max_new, rid=rid)
                n_new += 1
            gw.tick()
        wall = time.perf_counter() - t_start
        if plane is not None:
            gw.stop_workers()
        # collect per-rank busy fractions
        my = [eng.busy_s / wall]
        gathered = [None] * world
        dist.all_gather_object(gathered, {"rank": rank, "busy_frac": eng.busy_s / wall,
                                          "wall": wall})
        ticks = gw.phase_t["ticks"]
        phases = {k: v for k, v in gw.phase_t.items() if k != "ticks"}
        gw_busy = sum(phases.values())
        result = {
            "world": world,
            "seconds": round(wall, 2),
            "step_ms": args.step_ms,
            "req_per_s": round(gw.completed_total / wall, 1),
            "req_per_s_per_rank": round(gw.completed_total / wall / world, 1),
            "ticks_per_s": round(ticks / wall, 1),
            "rank0_phase_ms_per_tick": {k: round(v * 1e3 / max(1, ticks), 3) for k, v in phases.items()},
            "rank0_occupancy": round(gw_busy / wall, 3),
            "worker_busy_frac": {str(g["rank"]): round(g["busy_frac"], 3) for g in gathered},
            "p50_routing_ms": gw.p50_routing_ms(),
        }
        print("REHEARSAL " + json.dumps(result), flush=True)
    else:
        from smg_amd.routers.rccl_router import run_worker_loop

        plane = WorkerPlane(pcfg)
        t0 = time.perf_counter()
        run_worker_loop(eng, plane, decode_burst=args.decode_burst)
        wall = time.perf_counter() - t0
        gathered = [None] * world
        dist.all_gather_object(gathered, {"rank": rank, "busy_frac": eng.busy_s / wall,
                                          "wall": wall})
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
