"""torchrun helper for the rccl serving CI test (gloo, world 2):
rank 0 boots the FULL gateway (`smg launch --connection-mode rccl` path:
startup() -> RouterManager -> RcclRouter) and fires OpenAI chat requests at
it over HTTP; rank 1 runs the worker loop (cli.rccl_worker_main).  Rank 0
prints one JSON line with the observed results."""
import asyncio
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def build_cfg(port: int):
    from smg_amd.cli import to_router_config

    argv = [
        "launch",
        "--host", "127.0.0.1", "--port", str(port),
        "--connection-mode", "rccl",
        "--policy", "round_robin",
    ]
    if os.environ.get("SMG_TEST_PD"):
        argv += ["--pd-disaggregation"]
    cfg = to_router_config(argv)
    cfg.health_check.disable = True
    cfg.prometheus_port = None
    return cfg


async def rank0(cfg, port: int):
    import aiohttp

    from smg_amd.server.app import startup

    ctx = await startup(cfg)
    results = {"completions": [], "stream_chunks": 0, "worker_processed": {}}
    async with aiohttp.ClientSession() as s:
        for i in range(6):
            body = {
                "model": "default",
                "messages": [{"role": "user", "content": f"hello number {i} " * 8}],
                "max_tokens": 4,
            }
            async with s.post(f"http://127.0.0.1:{port}/v1/chat/completions", json=body) as r:
                assert r.status == 200, await r.text()
                j = await r.json()
                results["completions"].append(j["usage"]["completion_tokens"])
        # one streaming request
        body = {
            "model": "default",
            "messages": [{"role": "user", "content": "stream me"}],
            "max_tokens": 3,
            "stream": True,
        }
        async with s.post(f"http://127.0.0.1:{port}/v1/chat/completions", json=body) as r:
            assert r.status == 200
            async for raw in r.content:
                if raw.startswith(b"data: ") and b"[DONE]" not in raw:
                    results["stream_chunks"] += 1
    router = ctx.router_manager.default_router
    for w in router.workers:
        results["worker_processed"][w.url] = w.processed_requests
    results["p50_routing_ms"] = router.gw.p50_routing_ms()
    await ctx.shutdown()
    print("RESULT " + json.dumps(results), flush=True)


def main():
    rank = int(os.environ.get("RANK", "0"))
    port = int(os.environ.get("SMG_TEST_PORT", "31890"))
    cfg = build_cfg(port)
    if rank == 0:
        asyncio.run(rank0(cfg, port))
    else:
        from smg_amd.cli import rccl_worker_main

        rccl_worker_main(cfg)


if __name__ == "__main__":
    sys.exit(main())
