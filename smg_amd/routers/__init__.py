from .base import RouteRequest, RouteResponse, Router
from .factory import RouterManager, create_router
from .http_router import HttpRouter
from .pd_router import PDRouter

__all__ = ["HttpRouter", "PDRouter", "RouteRequest", "RouteResponse", "Router", "RouterManager", "create_router"]
