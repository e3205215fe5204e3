#!/usr/bin/env python3
"""Policy A/B harness (reference: scripts/sim_ab.sh + scripts/sim_load.py —
gateway + N realistic mock workers + Poisson shared-prefix load; TTFT / ITL /
E2E / throughput percentiles per policy).

    python scripts/sim_ab.py --policies cache_aware round_robin --workers 4 \
        --requests 200 --rate 50

Runs fully in-process (no sockets): mock engines behind the sim:// transport,
the same stack the gateway serves over HTTP.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import random
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from smg_amd.config import PolicyConfig, RouterConfig  # noqa: E402
from smg_amd.mock.engine import MockWorkerEngine, SimConfig  # noqa: E402
from smg_amd.routers.base import RouteRequest  # noqa: E402
from smg_amd.routers.factory import RouterManager  # noqa: E402
from smg_amd.server.app_context import AppContext  # noqa: E402
from smg_amd.workers.worker import Worker  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--policies", nargs="+", default=["cache_aware", "round_robin", "least_load"])
    p.add_argument("--workers", type=int, default=4)
    p.add_argument("--requests", type=int, default=200)
    p.add_argument("--rate", type=float, default=50.0, help="Poisson arrival rate req/s")
    p.add_argument("--prefix-pool", type=int, default=8)
    p.add_argument("--prefix-chars", type=int, default=2000)
    p.add_argument("--suffix-chars", type=int, default=200)
    p.add_argument("--max-tokens", type=int, default=24)
    p.add_argument("--speedup", type=float, default=10.0)
    p.add_argument("--prefill-tps", type=float, default=8000.0)
    p.add_argument("--seed", type=int, default=7)
    return p.parse_args()


async def run_policy(policy_name: str, args) -> dict:
    cfg = RouterConfig(policy=PolicyConfig(name=policy_name, gpu_tree=False))
    cfg.health_check.disable = True
    ctx = AppContext(cfg)
    engines = []
    for i in range(args.workers):
        eng = MockWorkerEngine(SimConfig(speedup=args.speedup, model_id="sim-model", prefill_tokens_per_sec=args.prefill_tps))
        await eng.start()
        w = Worker(f"sim://w{i}", model_id="sim-model")
        w.extra["engine"] = eng
        ctx.worker_registry.register(w)
        engines.append(eng)
    ctx.router_manager = RouterManager(ctx, cfg)

    rng = random.Random(args.seed)
    prefixes = ["".join(rng.choice("abcdefgh ") for _ in range(args.prefix_chars))
                for _ in range(args.prefix_pool)]

    lat_e2e, lat_ttft, itls = [], [], []

    async def one_request(i: int):
        prompt = rng.choice(prefixes) + "".join(rng.choice("xyz ") for _ in range(args.suffix_chars))
        body = {"model": "sim-model", "prompt": prompt, "max_tokens": args.max_tokens, "stream": True}
        t0 = time.perf_counter()
        resp = await ctx.router_manager.route(RouteRequest(
            path="/v1/completions", body=body, raw_body=json.dumps(body).encode(), request_id=f"r{i}"))
        first = None
        last = None
        n = 0
        if resp.is_stream:
            async for chunk in resp.stream:
                now = time.perf_counter()
                if first is None:
                    first = now
                else:
                    itls.append(now - last)
                last = now
                n += 1
        t1 = time.perf_counter()
        lat_e2e.append(t1 - t0)
        if first:
            lat_ttft.append(first - t0)

    async def load():
        tasks = []
        for i in range(args.requests):
            tasks.append(asyncio.ensure_future(one_request(i)))
            await asyncio.sleep(rng.expovariate(args.rate))
        await asyncio.gather(*tasks)

    t0 = time.perf_counter()
    await load()
    wall = time.perf_counter() - t0
    for eng in engines:
        await eng.stop()

    def pct(xs, q):
        return statistics.quantiles(xs, n=100)[q - 1] if len(xs) >= 2 else (xs[0] if xs else 0)

    return {
        "policy": policy_name,
        "requests": args.requests,
        "throughput_req_s": round(args.requests / wall, 2),
        "e2e_ms": {"p50": round(pct(lat_e2e, 50) * 1e3, 1), "p99": round(pct(lat_e2e, 99) * 1e3, 1)},
        "ttft_ms": {"p50": round(pct(lat_ttft, 50) * 1e3, 1), "p99": round(pct(lat_ttft, 99) * 1e3, 1)},
        "itl_ms": {"p50": round(pct(itls, 50) * 1e3, 2)} if itls else None,
        "per_worker_processed": [w.processed_requests for w in ctx.worker_registry.all()],
    }


async def main():
    args = parse_args()
    results = []
    for pol in args.policies:
        r = await run_policy(pol, args)
        results.append(r)
        print(json.dumps(r), flush=True)
    best = max(results, key=lambda r: r["throughput_req_s"])
    print(f"\n# best throughput: {best['policy']} at {best['throughput_req_s']} req/s", file=sys.stderr)


if __name__ == "__main__":
    asyncio.run(main())
