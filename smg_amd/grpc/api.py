"""Engine gRPC API (reference: crates/grpc_client/proto/sglang_scheduler.proto —
Generate server-stream :11, Embed, HealthCheck, Abort, GetModelInfo,
GetServerInfo, GetLoads, FlushCache, SubscribeKvEvents :50-59).

This image has grpcio but no protoc/grpcio-tools, so messages are msgpack
maps over gRPC's generic (bytes) method handlers — same RPC surface and
streaming semantics, self-describing wire format.  Both ends (gateway client,
engine servicer) live in this repo, exactly like the reference's
crates/grpc_client + grpc_servicer pair.
"""
from __future__ import annotations

from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List, Optional

import msgpack

SERVICE = "smg.Scheduler"


def method(name: str) -> str:
    return f"/{SERVICE}/{name}"


def dumps(obj: Any) -> bytes:
    if hasattr(obj, "to_dict"):
        obj = obj.to_dict()
    return msgpack.packb(obj, use_bin_type=True)


def loads(data: bytes) -> Dict[str, Any]:
    return msgpack.unpackb(data, raw=False)


@dataclass
class SamplingParams:
    max_new_tokens: int = 128
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    stop: List[str] = field(default_factory=list)
    stop_token_ids: List[int] = field(default_factory=list)
    ignore_eos: bool = False
    skip_special_tokens: bool = True

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "SamplingParams":
        return cls(**{k: v for k, v in (d or {}).items() if k in cls.__dataclass_fields__})


@dataclass
class GenerateRequest:
    request_id: str
    input_ids: List[int] = field(default_factory=list)
    text: Optional[str] = None
    sampling: SamplingParams = field(default_factory=SamplingParams)
    stream: bool = True
    multimodal: Optional[Dict[str, Any]] = None  # pixel refs / embeddings
    bootstrap_host: Optional[str] = None
    bootstrap_port: Optional[int] = None
    bootstrap_room: Optional[int] = None
    dp_rank: Optional[int] = None
    lora_id: Optional[str] = None  # adapter id minted at LoadLoraAdapter time

    def to_dict(self):
        d = asdict(self)
        d["sampling"] = self.sampling.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "GenerateRequest":
        d = dict(d)
        d["sampling"] = SamplingParams.from_dict(d.get("sampling") or {})
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


@dataclass
class GenerateChunk:
    """One streamed chunk: new token ids (and optionally text)."""

    request_id: str
    token_ids: List[int] = field(default_factory=list)
    finished: bool = False
    finish_reason: Optional[str] = None  # stop | length | abort
    prompt_tokens: int = 0
    completion_tokens: int = 0
    cached_tokens: int = 0

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d) -> "GenerateChunk":
        return cls(**{k: v for k, v in d.items() if k in cls.__dataclass_fields__})


@dataclass
class EmbedRequest:
    request_id: str
    input_ids: List[int] = field(default_factory=list)
    text: Optional[str] = None

    def to_dict(self):
        return asdict(self)


METHODS = {
    "Generate": "server_stream",
    "Embed": "unary",
    "EncodeImage": "unary",
    "Rerank": "unary",
    "Classify": "unary",
    "HealthCheck": "unary",
    "Abort": "unary",
    "GetModelInfo": "unary",
    "GetServerInfo": "unary",
    "GetLoads": "unary",
    "FlushCache": "unary",
    "SubscribeKvEvents": "server_stream",
    "StartProfile": "unary",
    "StopProfile": "unary",
    # LoRA adapter management (reference sglang_scheduler.proto:385-420)
    "LoadLoraAdapter": "unary",
    "UnloadLoraAdapter": "unary",
    "ListLoraAdapters": "unary",
}
