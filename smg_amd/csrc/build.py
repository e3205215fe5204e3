"""In-tree build of smg_amd._core (hipcc, gfx950).

hipcc cross-compiles device code without a GPU, so this runs in CPU-only CI;
the built .so travels to the GPU box with the repo snapshot.  Invoked by
__graft_entry__.build() and `python -m smg_amd.csrc.build`.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

CSRC = Path(__file__).resolve().parent
PKG = CSRC.parent
ARCH = os.environ.get("SMG_GFX_ARCH", "gfx950")


def _newer(out: Path, srcs) -> bool:
    if not out.exists():
        return False
    ts = out.stat().st_mtime
    return all(s.stat().st_mtime < ts for s in srcs)


def build(verbose: bool = True, force: bool = False) -> Path:
    import pybind11

    out = PKG / "_core.so"
    srcs = [CSRC / "gpu_tree.hip", CSRC / "bpe.hip", CSRC / "image.hip", CSRC / "attn_decode.hip", CSRC / "fused_decode.hip", CSRC / "rms_gemm.hip", CSRC / "bindings.cpp", CSRC / "host_tree.cpp"]
    if not force and _newer(out, srcs):
        return out
    py_inc = sysconfig.get_paths()["include"]
    mode = os.environ.get("SMG_BUILD_MODE", "release")
    opt_flags = ["-O3"] if mode != "debug" else ["-O1", "-g", "-UNDEBUG"]
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        *opt_flags,
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{pybind11.get_include()}",
        f"-I{py_inc}",
        f"-I{CSRC}",
        str(CSRC / "gpu_tree.hip"),
        str(CSRC / "bpe.hip"),
        str(CSRC / "image.hip"),
        str(CSRC / "attn_decode.hip"),
        str(CSRC / "fused_decode.hip"),
        str(CSRC / "rms_gemm.hip"),
        str(CSRC / "bindings.cpp"),
        "-o",
        str(out),
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return out


def build_sanitized(verbose: bool = False, force: bool = False) -> Path:
    """ASan+UBSan build of the host C++ self-test (SURVEY §5.2 sanitizer
    discipline): pure host code, so plain g++ compiles it — no GPU needed.
    Run by tests/test_sanitizer.py in CPU CI."""
    out = CSRC / "host_tree_selftest"
    srcs = [CSRC / "host_tree_selftest.cpp", CSRC / "host_tree.cpp"]
    if not force and _newer(out, srcs):
        return out
    cmd = [
        "g++", "-std=c++17", "-g", "-O1",
        "-fsanitize=address,undefined", "-fno-omit-frame-pointer",
        "-fno-sanitize-recover=all",
        str(CSRC / "host_tree_selftest.cpp"),
        "-o", str(out),
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return out


def build_debug(verbose: bool = True, force: bool = False) -> Path:
    """Device-debug variant of the extension (-g -O1, assertions on) for
    kernel debugging sessions; select with SMG_BUILD_MODE=debug."""
    os.environ["SMG_BUILD_MODE"] = "debug"
    return build(verbose=verbose, force=True)


if __name__ == "__main__":
    if "--sanitize" in sys.argv:
        path = build_sanitized(verbose=True, force="--force" in sys.argv)
        subprocess.run([str(path)], check=True)
    else:
        build(force="--force" in sys.argv)
