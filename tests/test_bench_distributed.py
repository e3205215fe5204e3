"""Multi-process data-plane tests on CPU (gloo, world_size 2): the RCCL tick
protocol and bench.py's gateway/worker lockstep must run correct-by-construction
without a GPU (the driver's 8-GPU scaling run uses the same code path)."""
import json
import os
import subprocess
import sys

import pytest


def test_bench_two_rank_gloo():
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", "29581",
        "bench.py", "--gpus", "2", "--tiny",
        "--steps", "2", "--warmup", "1", "--reqs-per-step", "4",
        "--concurrency", "8", "--prefix-len", "32", "--suffix-len", "8", "--max-new", "4",
    ]
    out = subprocess.run(
        cmd, capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))), env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    result = json.loads(line)
    assert result["n_gpus"] == 2
    assert result["value"] > 0
    assert result["scaling"] == "weak"
    assert result["config"]["p50_routing_latency_ms"] is not None


def test_bench_single_process():
    cmd = [
        sys.executable, "bench.py", "--tiny",
        "--steps", "2", "--warmup", "1", "--reqs-per-step", "4",
        "--concurrency", "8", "--prefix-len", "32", "--suffix-len", "8", "--max-new", "4",
    ]
    out = subprocess.run(
        cmd, capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    result = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert result["n_gpus"] == 1
    assert result["unit"] == "req/s"
