"""Per-worker circuit breaker (reference: model_gateway/src/worker/circuit_breaker.rs:103).

Closed -> Open after `failure_threshold` consecutive failures; Open -> HalfOpen after
`timeout_duration_secs`; HalfOpen -> Closed after `success_threshold` consecutive
successes, HalfOpen -> Open on any failure.  Lock-free in the reference; here a plain
object mutated from the event loop (and guarded for thread use by the GIL — all
transitions are single attribute writes).
"""
from __future__ import annotations

import enum
import time

from ..config import CircuitBreakerConfig


class CircuitState(str, enum.Enum):
    CLOSED = "closed"
    OPEN = "open"
    HALF_OPEN = "half_open"


class CircuitBreaker:
    __slots__ = (
        "config",
        "state",
        "_consecutive_failures",
        "_consecutive_successes",
        "_opened_at",
        "_clock",
    )

    def __init__(self, config: CircuitBreakerConfig | None = None, clock=time.monotonic):
        self.config = config or CircuitBreakerConfig()
        self.state = CircuitState.CLOSED
        self._consecutive_failures = 0
        self._consecutive_successes = 0
        self._opened_at = 0.0
        self._clock = clock

    def can_execute(self) -> bool:
        if self.config.disable:
            return True
        if self.state == CircuitState.CLOSED:
            return True
        if self.state == CircuitState.OPEN:
            if self._clock() - self._opened_at >= self.config.timeout_duration_secs:
                self.state = CircuitState.HALF_OPEN
                self._consecutive_successes = 0
                return True
            return False
        return True  # HALF_OPEN: allow probes

    def record_success(self) -> None:
        self._consecutive_failures = 0
        if self.state == CircuitState.HALF_OPEN:
            self._consecutive_successes += 1
            if self._consecutive_successes >= self.config.success_threshold:
                self.state = CircuitState.CLOSED
        elif self.state == CircuitState.OPEN:
            # a success observed while open (e.g. in-flight before trip) does not close
            pass

    def record_failure(self) -> None:
        self._consecutive_successes = 0
        if self.state == CircuitState.HALF_OPEN:
            self._trip()
            return
        self._consecutive_failures += 1
        if self.state == CircuitState.CLOSED and self._consecutive_failures >= self.config.failure_threshold:
            self._trip()

    def record_outcome(self, success: bool) -> None:
        if success:
            self.record_success()
        else:
            self.record_failure()

    def _trip(self) -> None:
        self.state = CircuitState.OPEN
        self._opened_at = self._clock()
        self._consecutive_failures = 0

    def reset(self) -> None:
        self.state = CircuitState.CLOSED
        self._consecutive_failures = 0
        self._consecutive_successes = 0
