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
    srcs = [CSRC / "gpu_tree.hip", CSRC / "bpe.hip", CSRC / "image.hip", CSRC / "attn_decode.hip", CSRC / "fused_decode.hip", CSRC / "bindings.cpp", CSRC / "host_tree.cpp"]
    if not force and _newer(out, srcs):
        return out
    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
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
        str(CSRC / "bindings.cpp"),
        "-o",
        str(out),
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
