"""Router trait (reference: model_gateway/src/routers/mod.rs:54 `trait RouterTrait`,
route_* methods mod.rs:87-288).  Routers receive the parsed request + raw body
and return an HTTP-shaped RouteResponse (status, headers, body or async stream).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Dict, Optional


@dataclass
class RouteRequest:
    path: str
    method: str = "POST"
    body: Optional[Dict[str, Any]] = None
    raw_body: bytes = b""
    headers: Dict[str, str] = field(default_factory=dict)
    request_id: str = ""
    tenant_id: Optional[str] = None
    routing_key: Optional[str] = None
    model_override: Optional[str] = None


@dataclass
class RouteResponse:
    status: int = 200
    headers: Dict[str, str] = field(default_factory=dict)
    body: bytes = b""
    stream: Optional[AsyncIterator[bytes]] = None  # set for SSE / chunked responses

    @property
    def is_stream(self) -> bool:
        return self.stream is not None


class Router:
    """Per-protocol router base.  Methods mirror RouterTrait's route_* surface."""

    router_id = "base"

    async def route(self, req: RouteRequest) -> RouteResponse:
        raise NotImplementedError

    async def route_chat(self, req: RouteRequest) -> RouteResponse:
        return await self.route(req)

    async def route_completion(self, req: RouteRequest) -> RouteResponse:
        return await self.route(req)

    async def route_embeddings(self, req: RouteRequest) -> RouteResponse:
        return await self.route(req)

    async def route_generate(self, req: RouteRequest) -> RouteResponse:
        return await self.route(req)

    async def route_rerank(self, req: RouteRequest) -> RouteResponse:
        return await self.route(req)

    async def flush_cache(self) -> RouteResponse:
        return RouteResponse(status=200, body=b'{"status":"ok"}')

    async def get_loads(self) -> Dict[str, Any]:
        return {}

    async def shutdown(self) -> None:
        pass
