from .circuit_breaker import CircuitBreaker, CircuitState
from .hash_ring import HashRing
from .registry import WorkerRegistry
from .worker import HealthState, Worker, WorkerType

__all__ = [
    "CircuitBreaker",
    "CircuitState",
    "HashRing",
    "HealthState",
    "Worker",
    "WorkerRegistry",
    "WorkerType",
]
