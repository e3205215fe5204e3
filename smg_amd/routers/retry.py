"""Retry executor with exponential backoff + jitter (reference:
model_gateway/src/routers/common/retry.rs `RetryExecutor`; config types.rs:673)."""
from __future__ import annotations

import asyncio
import random
from typing import Awaitable, Callable, Optional, TypeVar

from ..config import RetryConfig

T = TypeVar("T")

RETRYABLE_STATUS = {408, 429, 500, 502, 503, 504}


def is_retryable_status(status: int) -> bool:
    return status in RETRYABLE_STATUS


class RetryExecutor:
    def __init__(self, config: Optional[RetryConfig] = None, rng: Optional[random.Random] = None):
        self.config = config or RetryConfig()
        self._rng = rng or random.Random()

    def backoff_secs(self, attempt: int) -> float:
        cfg = self.config
        base = min(
            cfg.initial_backoff_ms * (cfg.backoff_multiplier**attempt),
            cfg.max_backoff_ms,
        )
        jitter = 1.0 + cfg.jitter_factor * (2.0 * self._rng.random() - 1.0)
        return max(0.0, base * jitter / 1000.0)

    async def execute(
        self,
        attempt_fn: Callable[[int], Awaitable[T]],
        should_retry: Callable[[T], bool],
        on_retry: Optional[Callable[[int, T], None]] = None,
    ) -> T:
        """Run attempt_fn until should_retry(result) is False or retries are
        exhausted.  Exceptions from attempt_fn are retried like failures and
        re-raised on the final attempt."""
        max_attempts = 1 if self.config.disable else self.config.max_retries + 1
        last_exc: Optional[BaseException] = None
        result: Optional[T] = None
        for attempt in range(max_attempts):
            try:
                result = await attempt_fn(attempt)
                last_exc = None
            except asyncio.CancelledError:
                raise
            except Exception as exc:  # noqa: BLE001 — retry any transport error
                last_exc = exc
                result = None
            if last_exc is None and result is not None and not should_retry(result):
                return result
            if attempt + 1 < max_attempts:
                if on_retry is not None:
                    on_retry(attempt, result)
                await asyncio.sleep(self.backoff_secs(attempt))
        if last_exc is not None:
            raise last_exc
        return result  # type: ignore[return-value]
