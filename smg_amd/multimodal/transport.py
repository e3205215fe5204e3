"""Multimodal tensor transport (reference: grpc/multimodal/transport.rs:1-30 —
inline / SHM / RDMA chosen by size (>=64 KiB -> SHM, SMG_MM_* env);
crates/mm_rdma NIXL arena).

MI355X mapping: on-node pixel tensors ride the RCCL/xGMI
plane ("xgmi" mode — device-to-device, no host staging); off-node or
CPU-engine transports are "inline" (msgpack bytes) and "shm"
(/dev/shm via multiprocessing.shared_memory, zero-copy to co-located
engine processes)."""
from __future__ import annotations

from multiprocessing import shared_memory
from typing import Any, Dict

import numpy as np

_OWNED: Dict[str, shared_memory.SharedMemory] = {}


def encode_tensor(arr: np.ndarray, mode: str = "inline", min_shm_bytes: int = 65_536) -> Dict[str, Any]:
    arr = np.ascontiguousarray(arr)
    nbytes = arr.nbytes
    if mode == "shm" and nbytes >= min_shm_bytes:
        shm = shared_memory.SharedMemory(create=True, size=nbytes)
        shm.buf[:nbytes] = arr.tobytes()
        _OWNED[shm.name] = shm
        return {"kind": "shm", "name": shm.name, "dtype": str(arr.dtype), "shape": list(arr.shape)}
    if mode == "xgmi":
        # descriptor only: the RCCL plane ships the tensor device-to-device;
        # the payload stays on the gateway GPU until the tick exchange
        return {"kind": "xgmi", "dtype": str(arr.dtype), "shape": list(arr.shape), "data": arr.tobytes()}
    return {"kind": "inline", "dtype": str(arr.dtype), "shape": list(arr.shape), "data": arr.tobytes()}


def decode_tensor(desc: Dict[str, Any]) -> np.ndarray:
    kind = desc.get("kind", "inline")
    shape = tuple(desc["shape"])
    dtype = np.dtype(desc["dtype"])
    if kind == "shm":
        shm = shared_memory.SharedMemory(name=desc["name"])
        try:
            return np.frombuffer(shm.buf, dtype=dtype)[: int(np.prod(shape))].reshape(shape).copy()
        finally:
            shm.close()
    return np.frombuffer(desc["data"], dtype=dtype).reshape(shape)


def release_tensor(desc: Dict[str, Any]) -> None:
    if desc.get("kind") == "shm":
        shm = _OWNED.pop(desc["name"], None)
        if shm is not None:
            shm.close()
            shm.unlink()
