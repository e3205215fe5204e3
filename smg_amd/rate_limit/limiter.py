"""Concurrency limiter + bounded wait queue (reference:
model_gateway/src/middleware/{concurrency,token_bucket}.rs)."""
from __future__ import annotations

import asyncio
import time
from aiohttp import web

from ..config import RateLimitConfig
from ..protocols.openai import error_body


class ConcurrencyLimiter:
    def __init__(self, config: RateLimitConfig):
        self.config = config
        self._sem = asyncio.Semaphore(max(1, config.max_concurrent_requests))
        self._queued = 0
        self._bucket_tokens = float(config.tokens_per_second or 0)
        self._bucket_last = time.monotonic()

    def _take_token(self) -> bool:
        rate = self.config.tokens_per_second
        if not rate:
            return True
        now = time.monotonic()
        self._bucket_tokens = min(rate, self._bucket_tokens + (now - self._bucket_last) * rate)
        self._bucket_last = now
        if self._bucket_tokens >= 1.0:
            self._bucket_tokens -= 1.0
            return True
        return False

    async def admit(self, request: web.Request, handler):
        if not self._take_token():
            return web.Response(
                status=429, body=error_body("rate limit exceeded", 429, "rate_limit_error"),
                content_type="application/json",
            )
        if self._sem.locked() and self._queued >= self.config.queue_size:
            return web.Response(
                status=429, body=error_body("concurrency queue full", 429, "rate_limit_error"),
                content_type="application/json",
            )
        self._queued += 1
        try:
            try:
                await asyncio.wait_for(self._sem.acquire(), timeout=self.config.queue_timeout_secs)
            except asyncio.TimeoutError:
                return web.Response(
                    status=429, body=error_body("queued too long", 429, "rate_limit_error"),
                    content_type="application/json",
                )
        finally:
            self._queued -= 1
        try:
            return await handler(request)
        finally:
            self._sem.release()
