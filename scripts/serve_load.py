#!/usr/bin/env python3
"""HTTP-in-the-loop load test of the PRODUCT serving path:
`smg launch --connection-mode rccl` (aiohttp server -> RcclRouter ->
TickGateway -> TorchEngine on cuda:0), measured from an OpenAI-client's
view — req/s, completion latency, and streaming TTFT through real SSE.

The bench.py headline drives TickGateway directly; this script closes the
loop over the full route table (reference rig: scripts/sim_load.py against
the gateway binary).

Usage: python scripts/serve_load.py [--seconds 20] [--concurrency 48]
       [--max-tokens 32] [--port 8099] [--stream-frac 0.25]
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import statistics
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def percentile(xs, p):
    if not xs:
        return None
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(p / 100 * len(xs)))]


async def one_request(session, url, body, stream):
    t0 = time.perf_counter()
    ttft = None
    if stream:
        body = dict(body, stream=True)
        async with session.post(url, json=body) as resp:
            assert resp.status == 200, await resp.text()
            async for line in resp.content:
                if line.startswith(b"data:") and ttft is None:
                    ttft = time.perf_counter() - t0
        return time.perf_counter() - t0, ttft
    async with session.post(url, json=body) as resp:
        assert resp.status == 200, await resp.text()
        payload = await resp.json()
        assert payload.get("choices"), payload
    return time.perf_counter() - t0, None


async def load(args):
    import aiohttp

    url = f"http://127.0.0.1:{args.port}/v1/chat/completions"
    rng = random.Random(7)
    prefixes = ["".join(rng.choice("abcdefgh ") for _ in range(args.prompt_chars))
                for _ in range(8)]
    lat, ttfts, done = [], [], [0]
    t_end = time.perf_counter() + args.seconds

    async with aiohttp.ClientSession() as session:
        async def worker(i):
            while time.perf_counter() < t_end:
                body = {
                    "model": "default",
                    "messages": [{"role": "user", "content": rng.choice(prefixes) + f" q{rng.random()}"}],
                    "max_tokens": args.max_tokens,
                }
                stream = rng.random() < args.stream_frac
                el, ttft = await one_request(session, url, body, stream)
                lat.append(el)
                if ttft is not None:
                    ttfts.append(ttft)
                done[0] += 1

        t0 = time.perf_counter()
        await asyncio.gather(*(worker(i) for i in range(args.concurrency)))
        wall = time.perf_counter() - t0
    return {
        "metric": "HTTP serving load (product path: aiohttp + RcclRouter + SSE)",
        "req_per_s": round(done[0] / wall, 1),
        "requests": done[0],
        "seconds": round(wall, 1),
        "concurrency": args.concurrency,
        "max_tokens": args.max_tokens,
        "latency_ms": {"p50": round(1e3 * percentile(lat, 50), 1),
                       "p95": round(1e3 * percentile(lat, 95), 1)},
        "stream_ttft_ms": {"p50": round(1e3 * percentile(ttfts, 50), 1) if ttfts else None,
                           "p95": round(1e3 * percentile(ttfts, 95), 1) if ttfts else None,
                           "n": len(ttfts)},
    }


async def wait_ready(port, timeout=180):
    import aiohttp

    t0 = time.perf_counter()
    async with aiohttp.ClientSession() as s:
        while time.perf_counter() - t0 < timeout:
            try:
                async with s.get(f"http://127.0.0.1:{port}/health") as r:
                    if r.status == 200:
                        return True
            except Exception:
                pass
            await asyncio.sleep(1.0)
    return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=20.0)
    ap.add_argument("--concurrency", type=int, default=48)
    ap.add_argument("--max-tokens", type=int, default=32)
    ap.add_argument("--prompt-chars", type=int, default=512)
    ap.add_argument("--stream-frac", type=float, default=0.25)
    ap.add_argument("--port", type=int, default=8099)
    ap.add_argument("--external", action="store_true",
                    help="server already running; just drive load")
    args = ap.parse_args()

    proc = None
    if not args.external:
        proc = subprocess.Popen(
            [sys.executable, "-m", "smg_amd.cli", "launch", "--connection-mode", "rccl",
             "--host", "127.0.0.1", "--port", str(args.port)],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        )
    try:
        ok = asyncio.run(wait_ready(args.port))
        if not ok:
            print(json.dumps({"error": "server never became ready"}))
            return 1
        result = asyncio.run(load(args))
        print("SERVE_LOAD " + json.dumps(result), flush=True)
        return 0
    finally:
        if proc is not None:
            proc.terminate()
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()


if __name__ == "__main__":
    raise SystemExit(main())
