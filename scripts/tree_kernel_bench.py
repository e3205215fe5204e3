#!/usr/bin/env python3
"""GPU radix-index kernel latency: batched match+decide+insert at several
batch sizes and prompt lengths (run on MI355X)."""
import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402  (HIP runtime ordering)

from smg_amd.kvindex.gpu_tree import GpuTokenTree  # noqa: E402


def main():
    tree = GpuTokenTree(page_size=16, capacity=1 << 22)
    rng = random.Random(1)
    urls = [f"rccl://rank-{i}" for i in range(8)]
    prefixes = [[rng.randrange(30000) for _ in range(512)] for _ in range(8)]

    def batch(n, plen=576):
        out = []
        for _ in range(n):
            p = rng.choice(prefixes)
            out.append(p + [rng.randrange(30000) for _ in range(plen - len(p))])
        return out

    # warm + populate
    tree.match_and_insert_batch(batch(256), urls=urls, candidates=list(range(8)),
                                loads=[0] * 8, processed=[0] * 8, cache_threshold=0.3,
                                min_load_idx=0, imbalanced=False)
    for n in (1, 16, 64, 256, 1024):
        reqs = batch(n)
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            tree.match_and_insert_batch(reqs, urls=urls, candidates=list(range(8)),
                                        loads=[0] * 8, processed=[0] * 8, cache_threshold=0.3,
                                        min_load_idx=0, imbalanced=False)
        dt = (time.perf_counter() - t0) / iters
        print(f"batch={n:5d} plen=576: {dt*1e6:8.1f} us/launch  {dt*1e6/n:7.2f} us/req")
    print("stats:", {k: v for k, v in tree.stats().items() if k != 'tenant_nodes'})


if __name__ == "__main__":
    main()
