"""Engine-side bridge seams (reference grpc_servicer/: vllm/servicer.py,
sglang/servicer.py, tokenspeed/servicer.py, mlx/servicer.py).

The engine runtimes are not installed in this image, so each bridge is built
against the ENGINE'S PUBLIC API SHAPE (duck-typed) and unit-tested with
fakes that mimic it: the translation logic — sampling-param mapping,
delta-from-cumulative token streaming, finish-reason normalization, load
snapshot extraction — is the real content of the reference servicers and is
fully exercised without the runtime.  Instantiating a bridge against a
missing runtime raises a clear driver-absent error."""
from .vllm import VllmBridge, translate_sampling_params_vllm
from .sglang import SglangBridge, build_sglang_generate_payload

__all__ = [
    "VllmBridge", "translate_sampling_params_vllm",
    "SglangBridge", "build_sglang_generate_payload", "make_bridge",
]


def make_bridge(kind: str, engine=None, **kw):
    if kind == "vllm":
        return VllmBridge(engine, **kw)
    if kind == "sglang":
        return SglangBridge(engine, **kw)
    raise ValueError(f"unknown engine bridge {kind!r} (available: vllm, sglang)")
