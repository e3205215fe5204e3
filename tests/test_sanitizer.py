"""Sanitizer discipline: the host C++ tree runs its randomized
mutation workload under ASan+UBSan in CPU CI.  The reference gets data-race /
memory safety from Rust's type system; the C++/HIP rebuild proves the host
side with sanitizers instead (device kernels are covered by differential
tests + the stream-serialized maintenance design)."""
import subprocess

import pytest


def test_host_tree_under_asan_ubsan():
    from smg_amd.csrc.build import build_sanitized

    try:
        path = build_sanitized(verbose=False)
    except (FileNotFoundError, subprocess.CalledProcessError) as exc:
        pytest.skip(f"sanitizer toolchain unavailable: {exc}")
    out = subprocess.run([str(path)], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    assert "OK" in out.stdout
