#!/usr/bin/env python3
"""TorchEngine phase micro-benchmark: prefill throughput (miss + prefix-hit)
and decode step latency by batch size.  Run on MI355X."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig  # noqa: E402


def sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def main():
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    cfg = TorchEngineConfig.bench_1b() if dev.startswith("cuda") else TorchEngineConfig.tiny()
    cfg.max_slots = 56
    cfg.max_seq = 700
    eng = TorchEngine(cfg, device=dev, graphs=dev.startswith("cuda"))
    plen = 576 if dev.startswith("cuda") else 64

    # ---- prefill, unique prompts (miss path) ----
    n = 16
    for i in range(n):
        eng.submit([(i * 7919 + j) % cfg.vocab_size for j in range(plen)], max_new_tokens=10_000)
    sync()
    t0 = time.time()
    while eng.waiting or any(r.prefilled < len(r.tokens) for r in eng.running.values()):
        eng.step(decode_burst=1)
    sync()
    dt = time.time() - t0
    print(f"prefill {n}x{plen} unique: {dt*1e3:.1f} ms -> {n*plen/dt:.0f} tok/s")

    # ---- decode latency by active batch ----
    for _ in range(3):
        eng.step()  # warm graphs
    sync()
    t0 = time.time()
    iters = 50
    for _ in range(iters):
        eng.step(decode_burst=1)
    sync()
    dt = time.time() - t0
    b = len(eng.running)
    print(f"decode batch={b}: {dt*1e3/iters:.2f} ms/step -> {b*iters/dt:.0f} tok/s")

    # ---- prefill with prefix-cache hits ----
    shared = [(9999 + j) % cfg.vocab_size for j in range(plen)]
    eng2 = TorchEngine(cfg, device=dev, graphs=dev.startswith("cuda"))
    eng2.submit(shared + [1, 2, 3, 4], max_new_tokens=10_000)
    while eng2.waiting or any(r.prefilled < len(r.tokens) for r in eng2.running.values()):
        eng2.step()
    sync()
    t0 = time.time()
    for i in range(n):
        eng2.submit(shared + [i, i + 1, i + 2, i + 3], max_new_tokens=10_000)
    while eng2.waiting or any(r.prefilled < len(r.tokens) for r in eng2.running.values()):
        eng2.step(decode_burst=1)
    sync()
    dt = time.time() - t0
    print(f"prefill {n}x{plen+4} shared-prefix: {dt*1e3:.1f} ms (hits={eng2.prefix_cache_hits})")


if __name__ == "__main__":
    main()
