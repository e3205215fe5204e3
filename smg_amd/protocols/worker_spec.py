"""Worker/admin protocol types (reference: crates/protocols/src/worker.rs —
WorkerSpec :604, TransportMode :773, SchedulerLoadSnapshot :1214,
WorkerLoadResponse :1250; model_card.rs:47 ModelCard; transcription.rs:19).

These are the typed wire shapes of /workers CRUD, /get_loads and /v1/models;
parse helpers validate and default like the reference's serde(default)."""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

WORKER_TYPES = ("regular", "prefill", "decode", "encode")
CONNECTION_MODES = ("http", "grpc", "rccl")
RUNTIME_TYPES = ("sglang", "vllm", "trtllm", "tokenspeed", "mlx", "external", "smg")
TRANSPORT_MODES = ("inline", "shm", "rdma", "xgmi")


class WorkerSpecError(ValueError):
    pass


@dataclass
class WorkerSpec:
    """POST /workers body (worker.rs:604).  api_key accepted on input, never
    serialized back (to_dict drops it, like serde(skip_serializing))."""

    url: str
    models: List[str] = field(default_factory=list)  # empty = wildcard
    worker_type: str = "regular"
    connection_mode: str = "http"
    runtime_type: str = "smg"
    provider: Optional[str] = None
    labels: Dict[str, str] = field(default_factory=dict)
    priority: int = 100
    cost: float = 1.0
    api_key: Optional[str] = None
    bootstrap_port: Optional[int] = None
    dp_base_url: Optional[str] = None
    dp_rank: Optional[int] = None
    dp_size: Optional[int] = None
    kv_connector: Optional[str] = None
    kv_role: Optional[str] = None
    kv_engine_id: Optional[str] = None
    kv_block_size: Optional[int] = None
    tokenizer_path: Optional[str] = None
    transport_mode: str = "inline"

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "WorkerSpec":
        if not d.get("url"):
            raise WorkerSpecError("worker spec requires `url`")
        known = {f.name for f in dataclasses.fields(cls)}
        kw = {k: v for k, v in d.items() if k in known}
        if "runtime" in d and "runtime_type" not in d:  # serde alias
            kw["runtime_type"] = d["runtime"]
        models = kw.get("models")
        if isinstance(models, str):
            kw["models"] = [models]
        spec = cls(**kw)
        if spec.worker_type not in WORKER_TYPES:
            raise WorkerSpecError(f"unknown worker_type {spec.worker_type!r}")
        if spec.connection_mode not in CONNECTION_MODES:
            raise WorkerSpecError(f"unknown connection_mode {spec.connection_mode!r}")
        if spec.transport_mode not in TRANSPORT_MODES:
            raise WorkerSpecError(f"unknown transport_mode {spec.transport_mode!r}")
        return spec

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d.pop("api_key", None)  # never serialize credentials back
        return {k: v for k, v in d.items() if v not in (None, [], {})}


@dataclass
class SchedulerLoadSnapshot:
    """Per-DP-rank engine load (worker.rs:1214) — the canonical schema every
    engine's GetLoads maps into."""

    dp_rank: int = 0
    num_running_reqs: int = 0
    num_waiting_reqs: int = 0
    num_waiting_uncached_tokens: int = 0
    num_total_reqs: int = 0
    num_used_tokens: int = 0
    max_total_num_tokens: int = 0
    token_usage: float = 0.0
    gen_throughput: float = 0.0
    cache_hit_rate: float = 0.0
    utilization: float = 0.0
    max_running_requests: int = 0
    kv_transfer_latency_ms: Optional[float] = None
    kv_transfer_speed_gb_s: Optional[float] = None
    prefill_queue_reqs: Optional[int] = None
    decode_queue_reqs: Optional[int] = None
    disagg_mode: Optional[str] = None

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "SchedulerLoadSnapshot":
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in (d or {}).items() if k in known and v is not None})


@dataclass
class WorkerLoadResponse:
    """All DP ranks of one worker (worker.rs:1250)."""

    timestamp: str = ""
    dp_rank_count: int = 0
    loads: List[SchedulerLoadSnapshot] = field(default_factory=list)

    def effective_token_usage(self) -> float:
        if not self.loads:
            return 0.0
        return sum(s.token_usage for s in self.loads) / len(self.loads)

    def total_queued_tokens(self) -> int:
        return sum(s.num_waiting_uncached_tokens for s in self.loads)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "timestamp": self.timestamp,
            "dp_rank_count": self.dp_rank_count,
            "loads": [
                {k: v for k, v in dataclasses.asdict(s).items() if v is not None}
                for s in self.loads
            ],
        }


@dataclass
class ModelCard:
    """/v1/models entry (model_card.rs:47): identity + capability surface."""

    id: str
    display_name: Optional[str] = None
    aliases: List[str] = field(default_factory=list)
    model_type: List[str] = field(default_factory=lambda: ["chat"])  # endpoint kinds
    hf_model_type: Optional[str] = None
    architectures: List[str] = field(default_factory=list)
    provider: Optional[str] = None
    context_length: Optional[int] = None
    supports_vision: bool = False
    supports_tools: bool = False
    created: int = 0
    owned_by: str = "smg"

    def to_openai(self) -> Dict[str, Any]:
        out = {"id": self.id, "object": "model", "created": self.created,
               "owned_by": self.owned_by}
        if self.display_name:
            out["display_name"] = self.display_name
        if self.context_length:
            out["context_length"] = self.context_length
        if self.aliases:
            out["aliases"] = list(self.aliases)
        return out


@dataclass
class TranscriptionRequest:
    """Multipart /v1/audio/transcriptions fields (transcription.rs:19)."""

    model: str = ""
    language: Optional[str] = None
    prompt: Optional[str] = None
    response_format: str = "json"
    temperature: float = 0.0
    timestamp_granularities: List[str] = field(default_factory=list)
    stream: bool = False

    @classmethod
    def from_form(cls, form: Dict[str, Any]) -> "TranscriptionRequest":
        known = {f.name for f in dataclasses.fields(cls)}
        kw = {k: v for k, v in form.items() if k in known}
        if "temperature" in kw:
            kw["temperature"] = float(kw["temperature"])
        if "stream" in kw and isinstance(kw["stream"], str):
            kw["stream"] = kw["stream"].lower() == "true"
        if isinstance(kw.get("timestamp_granularities"), str):
            kw["timestamp_granularities"] = [kw["timestamp_granularities"]]
        return cls(**kw)
