"""KV prefix index: paged radix trees + positional KV-event indexer.

Three interchangeable tree backends with identical semantics:
  * PagedRadixTree (pytree.py)     — pure-Python reference, used in CPU tests;
  * _core host tree (C++)          — fast host fallback;
  * _core GPU tree (gfx950 HIP)    — device-resident, batched kernels, the
                                     production backend on MI355X.
`make_token_tree(gpu=True)` returns the best available backend.
"""
from __future__ import annotations

from .pytree import MatchResult, PagedRadixTree, StringTree, TokenTree

# torch must load its libamdhip64 BEFORE _core.so links the system one (same
# soname, first load wins): a second runtime would leave torch.cuda
# unavailable on a live GPU box.  Imported at module load so the first
# routing decision never pays the ~2 s torch import.
try:
    import torch  # noqa: F401
except ImportError:
    pass


def _core():
    try:
        from .. import _core as core  # built in-tree by __graft_entry__.build()

        return core
    except ImportError:
        return None


def gpu_available() -> bool:
    core = _core()
    if core is None:
        return False
    try:
        return bool(core.hip_device_count() > 0)
    except Exception:
        return False


def make_token_tree(page_size: int = 16, gpu: bool = True, device: int = 0, capacity: int = 1 << 22):
    if gpu and gpu_available():
        from .gpu_tree import GpuTokenTree

        return GpuTokenTree(page_size=page_size, device=device, capacity=capacity)
    core = _core()
    if core is not None and hasattr(core, "HostTokenTree"):
        from .host_tree import HostTokenTree

        return HostTokenTree(page_size=page_size)
    return TokenTree(page_size=page_size)


def make_text_tree(gpu: bool = True, device: int = 0, page_size: int = 8):
    if gpu and gpu_available():
        from .gpu_tree import GpuTextTree

        return GpuTextTree(page_size=page_size, device=device)
    core = _core()
    if core is not None and hasattr(core, "HostTokenTree"):
        from .host_tree import HostTextTree

        return HostTextTree(page_size=page_size)
    return StringTree(page_size=page_size)


__all__ = [
    "MatchResult",
    "PagedRadixTree",
    "StringTree",
    "TokenTree",
    "gpu_available",
    "make_text_tree",
    "make_token_tree",
]
