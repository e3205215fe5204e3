"""Kubernetes watch discovery against a mock API server (reference
service_discovery.rs kube-rs watchers; the raw-HTTP watch makes this
testable in-image, which the old client-package poll was not)."""
import asyncio
import json

import pytest
from aiohttp import web
from aiohttp.test_utils import TestServer

from smg_amd.discovery.source import KubernetesDiscovery, start_discovery
from smg_amd.workers.registry import WorkerRegistry


def pod(name, ip, phase="Running", rv="1", model="m1"):
    return {
        "metadata": {"name": name, "resourceVersion": rv,
                     "labels": {"app": "worker", "smg.ai/model-id": model},
                     "annotations": {}},
        "status": {"phase": phase, "podIP": ip},
    }


class MockK8s:
    def __init__(self):
        self.pods = {}
        self.events = asyncio.Queue()
        self.watch_conns = 0
        self.fail_next_watch = False
        self.gone_next_watch = False

    def app(self):
        app = web.Application()
        app.router.add_get("/api/v1/namespaces/{ns}/pods", self.handler)
        return app

    async def handler(self, request):
        if request.query.get("watch") != "true":
            return web.json_response({
                "items": list(self.pods.values()),
                "metadata": {"resourceVersion": "100"},
            })
        self.watch_conns += 1
        if self.fail_next_watch:
            self.fail_next_watch = False
            return web.Response(status=500)
        if self.gone_next_watch:
            self.gone_next_watch = False
            return web.Response(status=410)
        resp = web.StreamResponse()
        await resp.prepare(request)
        try:
            while True:
                ev = await self.events.get()
                if ev is None:
                    break
                await resp.write(json.dumps(ev).encode() + b"\n")
        except (ConnectionResetError, asyncio.CancelledError):
            pass
        return resp


def test_watch_applies_events(runner):
    async def run():
        mock = MockK8s()
        mock.pods["w1"] = pod("w1", "10.0.0.1")
        server = TestServer(mock.app())
        await server.start_server()
        reg = WorkerRegistry()
        src = KubernetesDiscovery({"app": "worker"}, 8000, namespace="ns",
                                  api_base=f"http://127.0.0.1:{server.port}", token="t")
        task = await start_discovery(reg, src)
        try:
            for _ in range(50):
                if reg.get_by_url("http://10.0.0.1:8000"):
                    break
                await asyncio.sleep(0.05)
            w = reg.get_by_url("http://10.0.0.1:8000")
            assert w is not None and w.model_id == "m1"
            # ADDED event -> second worker appears without any polling
            await mock.events.put({"type": "ADDED", "object": pod("w2", "10.0.0.2", rv="2")})
            for _ in range(50):
                if reg.get_by_url("http://10.0.0.2:8000"):
                    break
                await asyncio.sleep(0.05)
            assert reg.get_by_url("http://10.0.0.2:8000") is not None
            # DELETED event -> removed
            await mock.events.put({"type": "DELETED", "object": pod("w1", "10.0.0.1", rv="3")})
            for _ in range(50):
                if reg.get_by_url("http://10.0.0.1:8000") is None:
                    break
                await asyncio.sleep(0.05)
            assert reg.get_by_url("http://10.0.0.1:8000") is None
        finally:
            task.cancel()
            await asyncio.gather(task, return_exceptions=True)
            await server.close()

    runner(run())


def test_watch_reconnects_with_backoff(runner):
    async def run():
        mock = MockK8s()
        mock.pods["w1"] = pod("w1", "10.0.0.1")
        mock.fail_next_watch = True  # first watch attempt 500s
        server = TestServer(mock.app())
        await server.start_server()
        reg = WorkerRegistry()
        src = KubernetesDiscovery({"app": "worker"}, 8000, namespace="ns",
                                  api_base=f"http://127.0.0.1:{server.port}", token="t")
        task = await start_discovery(reg, src)
        try:
            for _ in range(100):
                if mock.watch_conns >= 2:
                    break
                await asyncio.sleep(0.05)
            assert mock.watch_conns >= 2  # reconnected after the 500
            assert src.reconnects >= 1
            assert reg.get_by_url("http://10.0.0.1:8000") is not None
        finally:
            task.cancel()
            await asyncio.gather(task, return_exceptions=True)
            await server.close()

    runner(run())


def test_watch_410_relists(runner):
    async def run():
        mock = MockK8s()
        mock.gone_next_watch = True
        mock.pods["w1"] = pod("w1", "10.0.0.1")
        server = TestServer(mock.app())
        await server.start_server()
        src = KubernetesDiscovery({"app": "worker"}, 8000, namespace="ns",
                                  api_base=f"http://127.0.0.1:{server.port}", token="t")
        reg = WorkerRegistry()
        task = await start_discovery(reg, src)
        try:
            for _ in range(100):
                if mock.watch_conns >= 2 and reg.get_by_url("http://10.0.0.1:8000"):
                    break
                await asyncio.sleep(0.05)
            assert src._resource_version is not None  # relisted after Gone
            assert reg.get_by_url("http://10.0.0.1:8000") is not None
        finally:
            task.cancel()
            await asyncio.gather(task, return_exceptions=True)
            await server.close()

    runner(run())


def test_non_running_pod_excluded(runner):
    async def run():
        mock = MockK8s()
        mock.pods["w1"] = pod("w1", "10.0.0.1", phase="Pending")
        server = TestServer(mock.app())
        await server.start_server()
        src = KubernetesDiscovery({"app": "worker"}, 8000, namespace="ns",
                                  api_base=f"http://127.0.0.1:{server.port}", token="t")
        try:
            workers = await src.poll()
            assert workers == []
        finally:
            await server.close()

    runner(run())
