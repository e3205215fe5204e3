"""Mesh HTTP endpoints (reference: crates/mesh gossip gRPC service
gossip_service.rs / proto/gossip.proto — here aiohttp routes mounted on a
dedicated mesh port or the main app)."""
from __future__ import annotations

from aiohttp import web

from .swim import MeshNode

MESH_KEY = web.AppKey("smg_mesh", MeshNode)


def add_mesh_routes(app: web.Application, mesh: MeshNode) -> None:
    app[MESH_KEY] = mesh

    async def ping(request: web.Request):
        return web.json_response(mesh.handle_ping(await request.json()))

    async def ping_req(request: web.Request):
        return web.json_response(await mesh.handle_ping_req(await request.json()))

    async def sync(request: web.Request):
        return web.json_response(mesh.handle_sync(await request.json()))

    async def join(request: web.Request):
        return web.json_response(mesh.handle_join(await request.json()))

    async def members(request: web.Request):
        return web.json_response({"node_id": mesh.node_id, "members": mesh._member_dicts()})

    async def repair(request: web.Request):
        return web.json_response(mesh.handle_repair(await request.json()))

    async def partition(request: web.Request):
        return web.json_response(mesh.partition_state())

    app.router.add_post("/mesh/ping", ping)
    app.router.add_post("/mesh/ping_req", ping_req)
    app.router.add_post("/mesh/sync", sync)
    app.router.add_post("/mesh/join", join)
    app.router.add_post("/mesh/repair", repair)
    app.router.add_get("/mesh/members", members)
    app.router.add_get("/mesh/partition", partition)


async def start_mesh_server(mesh: MeshNode, host: str, port: int):
    app = web.Application()
    add_mesh_routes(app, mesh)
    runner = web.AppRunner(app, access_log=None)
    await runner.setup()
    # mTLS listener (reference mtls.rs): requires a peer cert from the mesh CA
    site = web.TCPSite(runner, host, port, ssl_context=mesh.server_ssl_context())
    await site.start()
    return runner
