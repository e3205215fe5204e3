"""Mock worker HTTP server (reference: crates/mock_worker/src/http.rs and the
minimal scripts/mock_worker.py).  Serves a MockWorkerEngine over aiohttp so the
gateway can be exercised end-to-end on CPU."""
from __future__ import annotations

import argparse
import asyncio
import json
from typing import Optional

from aiohttp import web

from .engine import MockWorkerEngine, SimConfig


def build_mock_app(engine: MockWorkerEngine) -> web.Application:
    async def catch_all(request: web.Request):
        body = None
        if request.can_read_body:
            try:
                body = json.loads(await request.read() or b"{}")
            except json.JSONDecodeError:
                body = {}
        status, headers, payload = await engine.handle(request.path, body, dict(request.headers))
        if hasattr(payload, "__aiter__"):
            resp = web.StreamResponse(status=status, headers={"content-type": "text/event-stream", **headers})
            resp.enable_chunked_encoding()
            await resp.prepare(request)
            try:
                async for chunk in payload:
                    await resp.write(chunk)
            except (ConnectionResetError, asyncio.CancelledError):
                pass
            await resp.write_eof()
            return resp
        ctype = headers.get("content-type", "application/json")
        return web.Response(status=status, body=payload, content_type=ctype.split(";")[0])

    app = web.Application()
    app.router.add_route("*", "/{tail:.*}", catch_all)
    return app


async def serve_mock_worker(
    host: str = "127.0.0.1", port: int = 8001, config: Optional[SimConfig] = None
):
    engine = MockWorkerEngine(config)
    await engine.start()
    app = build_mock_app(engine)
    runner = web.AppRunner(app, access_log=None)
    await runner.setup()
    site = web.TCPSite(runner, host, port)
    await site.start()
    return engine, runner


def main() -> None:
    p = argparse.ArgumentParser(description="smg mock worker (engine simulator)")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8001)
    p.add_argument("--model-id", default="mock-model")
    p.add_argument("--prefill-tps", type=float, default=8000.0)
    p.add_argument("--decode-base-ms", type=float, default=6.0)
    p.add_argument("--decode-slope-ms", type=float, default=0.35)
    p.add_argument("--kv-tokens", type=int, default=524_288)
    p.add_argument("--speedup", type=float, default=1.0)
    args = p.parse_args()
    cfg = SimConfig(
        prefill_tokens_per_sec=args.prefill_tps,
        decode_base_secs=args.decode_base_ms / 1e3,
        decode_per_request_secs=args.decode_slope_ms / 1e3,
        kv_capacity_tokens=args.kv_tokens,
        model_id=args.model_id,
        speedup=args.speedup,
    )

    async def _run():
        await serve_mock_worker(args.host, args.port, cfg)
        await asyncio.Event().wait()

    try:
        asyncio.run(_run())
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
