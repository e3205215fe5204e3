"""Router configuration types.

Mirrors the reference's layered config surface (RouterConfig at
model_gateway/src/config/types.rs:19, PolicyConfig at types.rs:443,
RoutingMode types.rs:298) as plain dataclasses so YAML/JSON/CLI all
construct the same object.  Not a translation: only the externally
observable knobs are kept; MI355X-specific knobs (GPU kv-index sizing,
RCCL fan-out) are added here.
"""
from __future__ import annotations

import dataclasses
import enum
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple


class ConnectionMode(str, enum.Enum):
    HTTP = "http"
    GRPC = "grpc"
    # MI355X-native on-node data plane: RCCL p2p over xGMI instead of loopback
    RCCL = "rccl"


class RoutingMode(str, enum.Enum):
    REGULAR = "regular"
    PREFILL_DECODE = "prefill_decode"
    ENCODE_PREFILL_DECODE = "encode_prefill_decode"
    OPENAI = "openai"


POLICY_NAMES = (
    "random",
    "round_robin",
    "passthrough",
    "cache_aware",
    "power_of_two",
    "least_load",
    "prefix_hash",
    "consistent_hashing",
    "manual",
    "bucket",
)


@dataclass
class PolicyConfig:
    """Per-policy tuning knobs (reference: config/types.rs:443 PolicyConfig enum)."""

    name: str = "cache_aware"
    # cache_aware
    cache_threshold: float = 0.3
    balance_abs_threshold: int = 64
    balance_rel_threshold: float = 1.5
    balance_token_usage_threshold: float = 1.0
    overload_token_usage_threshold: float = 1.0
    eviction_interval_secs: int = 120
    max_tree_size: int = 67_108_864
    block_size: int = 16
    # manual
    max_idle_secs: int = 14_400
    assignment_mode: str = "random"  # random | min_load | min_group
    # prefix_hash
    prefix_token_count: int = 256
    prefix_hash_load_factor: float = 1.25
    # least_load
    least_load_kv_pressure_weight: float = 0.15
    least_load_default_throughput: float = 2000.0
    least_load_mean_prefill_tokens: int = 1024
    # MI355X: run the prefix tree on-GPU when a device is present
    gpu_tree: bool = True
    gpu_tree_device: int = 0

    def validate(self) -> None:
        if self.name not in POLICY_NAMES:
            raise ValueError(f"unknown policy {self.name!r}; valid: {POLICY_NAMES}")
        if not (0.0 <= self.cache_threshold <= 1.0):
            raise ValueError("cache_threshold must be in [0,1]")
        if self.assignment_mode not in ("random", "min_load", "min_group"):
            raise ValueError("assignment_mode must be random|min_load|min_group")
        if self.block_size <= 0 or self.block_size & (self.block_size - 1):
            raise ValueError("block_size must be a positive power of two")


@dataclass
class RetryConfig:
    """reference: config/types.rs:673."""

    max_retries: int = 5
    initial_backoff_ms: int = 50
    max_backoff_ms: int = 30_000
    backoff_multiplier: float = 1.5
    jitter_factor: float = 0.2
    disable: bool = False


@dataclass
class CircuitBreakerConfig:
    failure_threshold: int = 10
    success_threshold: int = 3
    timeout_duration_secs: int = 60
    window_duration_secs: int = 120
    disable: bool = False


@dataclass
class HealthCheckConfig:
    """reference: config/types.rs:701."""

    failure_threshold: int = 3
    success_threshold: int = 2
    timeout_secs: int = 5
    check_interval_secs: int = 60
    endpoint: str = "/health"
    disable: bool = False
    remove_unhealthy_workers: bool = False
    drain_settle_secs: int = 0


@dataclass
class RateLimitConfig:
    max_concurrent_requests: int = -1  # -1 = unlimited
    queue_size: int = 100
    queue_timeout_secs: int = 60
    tokens_per_second: Optional[int] = None


@dataclass
class PrioritySchedulerConfig:
    enabled: bool = False
    default_max_class: str = "default"  # system|interactive|default|bulk
    config_path: Optional[str] = None
    tenant_metric_top_n: int = 0


@dataclass
class TenantRateLimitConfig:
    enabled: bool = False
    config_path: Optional[str] = None


@dataclass
class TokenizerCacheConfig:
    enable_l0: bool = True
    l0_max_entries: int = 8192
    enable_l1: bool = False
    l1_max_memory: int = 64 << 20


@dataclass
class MeshConfig:
    enabled: bool = False
    server_name: Optional[str] = None
    host: str = "0.0.0.0"
    advertise_host: Optional[str] = None
    port: int = 32300
    peer_urls: List[str] = field(default_factory=list)
    # mutual TLS between gateway replicas (reference crates/mesh/src/mtls.rs):
    # when cert+key+ca are set, the mesh listener REQUIRES client certs from
    # the same CA and outbound gossip presents this node's cert
    mtls_cert: Optional[str] = None
    mtls_key: Optional[str] = None
    mtls_ca: Optional[str] = None


@dataclass
class DiscoveryConfig:
    enabled: bool = False
    selector: Dict[str, str] = field(default_factory=dict)
    port: int = 8000
    namespace: Optional[str] = None
    prefill_selector: Dict[str, str] = field(default_factory=dict)
    decode_selector: Dict[str, str] = field(default_factory=dict)
    encode_selector: Dict[str, str] = field(default_factory=dict)
    router_selector: Dict[str, str] = field(default_factory=dict)
    model_id_from: str = "label"  # label | annotation | namespace


@dataclass
class TraceConfig:
    """reference: config/types.rs:788."""

    enabled: bool = False
    otlp_endpoint: Optional[str] = None


@dataclass
class AuthConfig:
    api_key: Optional[str] = None
    tenant_api_keys: Dict[str, str] = field(default_factory=dict)  # key -> tenant id
    jwt_issuer: Optional[str] = None
    jwt_audience: Optional[str] = None
    jwt_jwks_uri: Optional[str] = None  # http(s) URI or local file path
    jwt_jwks_inline: Optional[Dict] = None  # inline {"keys": [...]} set
    jwt_leeway_secs: float = 60.0
    jwt_jwks_cache_ttl_secs: float = 3600.0
    jwt_enable_jti_check: bool = False
    jwt_role_claim: str = "roles"
    jwt_role_mapping: Dict[str, str] = field(default_factory=dict)
    jwt_require_exp: bool = True  # reference jsonwebtoken Validation::default required_spec_claims={"exp"}
    admin_role: str = "admin"  # JWT role allowed to hit control-plane mutations (Role::is_admin)
    control_plane_api_keys: List[str] = field(default_factory=list)
    disable_audit_logging: bool = False


@dataclass
class StorageConfig:
    backend: str = "memory"  # memory | none | postgres | redis | oracle
    history_backend: Optional[str] = None
    postgres_db_url: Optional[str] = None
    postgres_pool_max_size: int = 16
    redis_url: Optional[str] = None
    redis_pool_max_size: int = 16
    redis_retention_days: int = 30
    oracle_dsn: Optional[str] = None


@dataclass
class RcclPlaneConfig:
    """MI355X on-node data plane: persistent RCCL p2p channels over xGMI.

    Replaces loopback HTTP/gRPC/ZMQ/SHM for co-located
    workers.  One rank per GPU, gateway on rank 0; fixed-cadence tick
    exchange with double-buffered staging tensors.
    """

    enabled: bool = False
    world_size: int = 1
    tick_interval_us: int = 200
    max_batch_requests: int = 64
    max_tokens_per_msg: int = 8192
    staging_dtype: str = "int32"


@dataclass
class RouterConfig:
    """Top-level gateway configuration (reference RouterConfig config/types.rs:19)."""

    host: str = "0.0.0.0"
    port: int = 30000
    health_check_port: Optional[int] = None
    worker_urls: List[str] = field(default_factory=list)
    prefill_urls: List[Tuple[str, Optional[int]]] = field(default_factory=list)
    decode_urls: List[str] = field(default_factory=list)
    encode_urls: List[Tuple[str, Optional[int]]] = field(default_factory=list)

    mode: RoutingMode = RoutingMode.REGULAR
    connection_mode: ConnectionMode = ConnectionMode.HTTP
    enable_igw: bool = False

    policy: PolicyConfig = field(default_factory=PolicyConfig)
    prefill_policy: Optional[PolicyConfig] = None
    decode_policy: Optional[PolicyConfig] = None
    encode_policy: Optional[PolicyConfig] = None

    dp_aware: bool = False
    dp_minimum_tokens_scheduler: bool = False
    routing_key_override: Optional[str] = None

    retry: RetryConfig = field(default_factory=RetryConfig)
    circuit_breaker: CircuitBreakerConfig = field(default_factory=CircuitBreakerConfig)
    health_check: HealthCheckConfig = field(default_factory=HealthCheckConfig)
    rate_limit: RateLimitConfig = field(default_factory=RateLimitConfig)
    priority_scheduler: PrioritySchedulerConfig = field(default_factory=PrioritySchedulerConfig)
    tenant_rate_limit: TenantRateLimitConfig = field(default_factory=TenantRateLimitConfig)
    auth: AuthConfig = field(default_factory=AuthConfig)
    storage: StorageConfig = field(default_factory=StorageConfig)
    mesh: MeshConfig = field(default_factory=MeshConfig)
    discovery: DiscoveryConfig = field(default_factory=DiscoveryConfig)
    trace: TraceConfig = field(default_factory=TraceConfig)
    tokenizer_cache: TokenizerCacheConfig = field(default_factory=TokenizerCacheConfig)
    rccl: RcclPlaneConfig = field(default_factory=RcclPlaneConfig)

    # worker lifecycle
    worker_startup_timeout_secs: int = 600
    worker_startup_delay: float = 0.0
    worker_startup_check_interval: int = 30
    load_monitor_interval: int = 5
    engine_metrics: bool = False

    # multimodal transport
    multimodal_tensor_transport: str = "inline"  # inline | shm | rdma | xgmi
    multimodal_shm_min_bytes: int = 65_536

    # tokenizer / parsers
    model_path: Optional[str] = None
    tokenizer_path: Optional[str] = None
    chat_template: Optional[str] = None
    disable_tokenizer_autoload: bool = False
    reasoning_parser: Optional[str] = None
    tool_call_parser: Optional[str] = None

    # misc server knobs
    model_aliases: Dict[str, str] = field(default_factory=dict)
    request_timeout_secs: int = 1800
    shutdown_grace_period_secs: int = 10
    max_payload_size: int = 512 << 20
    cors_allowed_origins: List[str] = field(default_factory=list)
    request_id_headers: List[str] = field(
        default_factory=lambda: ["x-request-id", "x-correlation-id", "x-trace-id", "request-id"]
    )
    storage_context_headers: List[str] = field(default_factory=list)
    trust_tenant_header: bool = False
    tenant_header_name: str = "x-smg-tenant"
    log_dir: Optional[str] = None
    log_level: str = "info"
    log_json: bool = False
    prometheus_port: Optional[int] = 29000
    prometheus_host: str = "0.0.0.0"
    mcp_config_path: Optional[str] = None
    tls_cert_path: Optional[str] = None
    tls_key_path: Optional[str] = None
    # plugin modules may only be loaded from inside this directory; when unset,
    # POST /wasm is refused (plugins load native code — operator opt-in only)
    plugin_dir: Optional[str] = None

    def validate(self) -> None:
        from . import validation

        validation.validate_config(self)

    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    @property
    def is_pd(self) -> bool:
        return self.mode in (RoutingMode.PREFILL_DECODE, RoutingMode.ENCODE_PREFILL_DECODE)
