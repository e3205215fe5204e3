#!/usr/bin/env python3
"""Gateway scale rig (reference: scripts/scale_test.sh — gateway CPU% +
/health latency at 1000-2000 mock workers, IGW mode).

    python scripts/scale_test.py --workers 1000 --probes 200
"""
from __future__ import annotations

import argparse
import asyncio
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from smg_amd.config import PolicyConfig, RouterConfig  # noqa: E402
from smg_amd.policies import SelectWorkerInfo  # noqa: E402
from smg_amd.server.app_context import AppContext  # noqa: E402
from smg_amd.workers.worker import Worker  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--workers", type=int, default=1000)
    p.add_argument("--probes", type=int, default=200)
    p.add_argument("--selects", type=int, default=2000)
    p.add_argument("--policy", default="cache_aware")
    args = p.parse_args()

    cfg = RouterConfig(policy=PolicyConfig(name=args.policy, gpu_tree=False))
    cfg.health_check.disable = True
    ctx = AppContext(cfg)

    t0 = time.perf_counter()
    for i in range(args.workers):
        ctx.worker_registry.register(Worker(f"http://w{i}:8000", model_id=f"model-{i % 16}"))
    reg_time = time.perf_counter() - t0

    # registry lookup latency (the /health readiness path is O(1) from events)
    lat = []
    for _ in range(args.probes):
        t = time.perf_counter_ns()
        ctx.worker_registry.healthy_count()
        lat.append((time.perf_counter_ns() - t) / 1e3)

    # policy select latency at fleet scale
    policy = ctx.policy_registry.get("model-0")
    workers = ctx.worker_registry.for_model("model-0")
    sel = []
    toks = list(range(256))
    for i in range(args.selects):
        t = time.perf_counter_ns()
        policy.select_worker(workers, SelectWorkerInfo(model_id="model-0", tokens=toks, request_id=str(i)))
        sel.append((time.perf_counter_ns() - t) / 1e3)

    print(json.dumps({
        "workers": args.workers,
        "registration_s": round(reg_time, 3),
        "healthy_count_us": {"p50": round(statistics.median(lat), 2), "max": round(max(lat), 2)},
        "select_worker_us": {
            "p50": round(statistics.median(sel), 2),
            "p99": round(statistics.quantiles(sel, n=100)[98], 2),
        },
        "policy": args.policy,
    }))


if __name__ == "__main__":
    main()
