"""RouterFactory + RouterManager (reference: routers/factory.rs:51 keyed on
ConnectionMode x RoutingMode, RouterIds factory.rs:37-48; RouterManager
router_manager.rs — per-model multiplexing in IGW mode)."""
from __future__ import annotations

from typing import Dict, Optional

from ..config import ConnectionMode, RouterConfig, RoutingMode
from .base import RouteRequest, RouteResponse, Router
from .http_router import HttpRouter
from .pd_router import PDRouter


ROUTER_IDS = {
    (ConnectionMode.HTTP, RoutingMode.REGULAR): "http-regular",
    (ConnectionMode.HTTP, RoutingMode.PREFILL_DECODE): "http-pd",
    (ConnectionMode.HTTP, RoutingMode.OPENAI): "http-openai",
    (ConnectionMode.GRPC, RoutingMode.REGULAR): "grpc-regular",
    (ConnectionMode.GRPC, RoutingMode.PREFILL_DECODE): "grpc-pd",
    (ConnectionMode.GRPC, RoutingMode.ENCODE_PREFILL_DECODE): "grpc-epd",
    (ConnectionMode.RCCL, RoutingMode.REGULAR): "rccl-regular",
    (ConnectionMode.RCCL, RoutingMode.PREFILL_DECODE): "rccl-pd",
}


def create_router(ctx, config: Optional[RouterConfig] = None) -> Router:
    config = config or ctx.config
    key = (config.connection_mode, config.mode)
    router_id = ROUTER_IDS.get(key)
    if router_id is None:
        raise ValueError(f"unsupported router combination {key}")
    if router_id == "http-regular" or router_id == "http-openai":
        return HttpRouter(ctx.worker_registry, ctx.policy_registry, config, metrics=ctx.metrics)
    if router_id == "http-pd":
        return PDRouter(ctx.worker_registry, ctx.policy_registry, config, metrics=ctx.metrics)
    if router_id == "grpc-regular":
        from .grpc.router import GrpcRouter

        return GrpcRouter(ctx, config)
    if router_id in ("rccl-regular", "rccl-pd"):
        # RcclRouter reads config.mode itself: PREFILL_DECODE turns on the
        # plane PD topology (odd ranks prefill, KV handoff over xGMI)
        from .rccl_router import RcclRouter

        return RcclRouter(ctx, config)
    raise ValueError(f"router {router_id} not yet wired")


class RouterManager:
    """model_id -> router multiplexing (IGW); single shared router otherwise."""

    def __init__(self, ctx, config: RouterConfig):
        self.ctx = ctx
        self.config = config
        self.default_router = create_router(ctx, config)
        self.model_routers: Dict[str, Router] = {}

    def router_for(self, model_id: Optional[str]) -> Router:
        if model_id is not None and model_id in self.model_routers:
            return self.model_routers[model_id]
        return self.default_router

    def add_model_router(self, model_id: str, router: Router) -> None:
        self.model_routers[model_id] = router

    async def route(self, req: RouteRequest) -> RouteResponse:
        model = req.model_override or (req.body or {}).get("model") if req.body else None
        return await self.router_for(model if isinstance(model, str) else None).route(req)

    async def shutdown(self) -> None:
        await self.default_router.shutdown()
        for r in self.model_routers.values():
            await r.shutdown()
