"""Discovery sources + registry reconciliation."""
from __future__ import annotations

import asyncio
import json
import logging
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..workers.worker import Worker, WorkerType

log = logging.getLogger("smg.discovery")


@dataclass
class DiscoveredWorker:
    url: str
    model_id: str = "default"
    worker_type: str = "regular"
    labels: Dict[str, str] = field(default_factory=dict)
    bootstrap_port: Optional[int] = None


class DiscoverySource:
    async def poll(self) -> List[DiscoveredWorker]:
        raise NotImplementedError


class FileDiscovery(DiscoverySource):
    """JSON file: [{"url": ..., "model_id": ..., "worker_type": ..., "labels": {}}]"""

    def __init__(self, path: str):
        self.path = path

    async def poll(self) -> List[DiscoveredWorker]:
        if not os.path.exists(self.path):
            return []
        try:
            with open(self.path) as f:
                data = json.load(f)
        except (json.JSONDecodeError, OSError) as exc:
            log.warning("discovery file unreadable: %s", exc)
            return []
        out = []
        for d in data if isinstance(data, list) else data.get("workers", []):
            if isinstance(d, dict) and d.get("url"):
                out.append(
                    DiscoveredWorker(
                        url=d["url"],
                        model_id=d.get("model_id", "default"),
                        worker_type=d.get("worker_type", "regular"),
                        labels=d.get("labels") or {},
                        bootstrap_port=d.get("bootstrap_port"),
                    )
                )
        return out


class KubernetesDiscovery(DiscoverySource):
    """Pod-WATCH discovery against the raw Kubernetes API over HTTP
    (reference service_discovery.rs:8-24 kube-rs watchers): a streaming
    `?watch=true` GET delivers ADDED/MODIFIED/DELETED pod events with
    resourceVersion resume, reconnect with exponential backoff, and a
    relist on 410 Gone.  No client package needed — in-cluster auth reads
    the serviceaccount token; `api_base` injection makes the watch testable
    against a local mock API server.

    Selector-matched Running pods become workers on `port` (or the
    smg.ai/worker-ports annotation); model id from the smg.ai/model-id
    label."""

    SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

    def __init__(self, selector: Dict[str, str], port: int, namespace: Optional[str] = None,
                 model_id_from: str = "label", api_base: Optional[str] = None,
                 token: Optional[str] = None):
        self.selector = selector
        self.port = port
        self.namespace = namespace or "default"
        self.model_id_from = model_id_from
        self._api_base = api_base
        self._token = token
        self._pods: Dict[str, DiscoveredWorker] = {}  # pod name -> worker
        self._resource_version: Optional[str] = None
        self.reconnects = 0

    # ---- in-cluster config -------------------------------------------------
    def api_base(self) -> str:
        if self._api_base:
            return self._api_base
        host = os.environ.get("KUBERNETES_SERVICE_HOST")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        if not host:
            raise RuntimeError("not in a Kubernetes cluster (KUBERNETES_SERVICE_HOST unset)")
        return f"https://{host}:{port}"

    def _headers(self) -> Dict[str, str]:
        token = self._token
        if token is None and os.path.exists(f"{self.SA_DIR}/token"):
            with open(f"{self.SA_DIR}/token") as f:
                token = f.read().strip()
        return {"Authorization": f"Bearer {token}"} if token else {}

    def _selector_qs(self) -> str:
        return ",".join(f"{k}={v}" for k, v in self.selector.items())

    def _pod_to_worker(self, pod: dict) -> Optional[DiscoveredWorker]:
        status = pod.get("status") or {}
        meta = pod.get("metadata") or {}
        if status.get("phase") != "Running" or not status.get("podIP"):
            return None
        ann = meta.get("annotations") or {}
        labels = meta.get("labels") or {}
        port = ann.get("smg.ai/worker-ports", self.port)
        model_id = labels.get("smg.ai/model-id", "default")
        return DiscoveredWorker(url=f"http://{status['podIP']}:{port}", model_id=model_id)

    def _apply_event(self, etype: str, pod: dict) -> None:
        name = (pod.get("metadata") or {}).get("name", "")
        if not name:
            return
        if etype == "DELETED":
            self._pods.pop(name, None)
            return
        w = self._pod_to_worker(pod)
        if w is None:
            self._pods.pop(name, None)  # e.g. pod left Running
        else:
            self._pods[name] = w

    # ---- list (poll fallback + watch bootstrap) -----------------------------
    async def poll(self) -> List[DiscoveredWorker]:
        import aiohttp

        url = f"{self.api_base()}/api/v1/namespaces/{self.namespace}/pods"
        async with aiohttp.ClientSession() as session:
            async with session.get(
                url, params={"labelSelector": self._selector_qs()},
                headers=self._headers(), ssl=False,
                timeout=aiohttp.ClientTimeout(total=15),
            ) as resp:
                if resp.status != 200:
                    raise RuntimeError(f"k8s list pods HTTP {resp.status}")
                data = await resp.json()
        self._pods.clear()
        for pod in data.get("items", []):
            self._apply_event("ADDED", pod)
        self._resource_version = (data.get("metadata") or {}).get("resourceVersion")
        return list(self._pods.values())

    # ---- the watch ----------------------------------------------------------
    async def watch(self, on_change) -> None:
        """Run forever: relist, then stream watch events, invoking
        `on_change(workers)` after every membership change.  Reconnects with
        100ms -> 30s backoff; a 410 Gone restarts from a fresh list."""
        import aiohttp

        backoff = 0.1
        while True:
            try:
                await self.poll()  # bootstrap list + resourceVersion
                await on_change(list(self._pods.values()))
                url = f"{self.api_base()}/api/v1/namespaces/{self.namespace}/pods"
                params = {"labelSelector": self._selector_qs(), "watch": "true"}
                if self._resource_version:
                    params["resourceVersion"] = self._resource_version
                async with aiohttp.ClientSession() as session:
                    async with session.get(
                        url, params=params, headers=self._headers(), ssl=False,
                        timeout=aiohttp.ClientTimeout(total=None, sock_read=300),
                    ) as resp:
                        if resp.status == 410:
                            self._resource_version = None
                            raise RuntimeError("k8s watch 410 Gone (relist)")
                        if resp.status != 200:
                            raise RuntimeError(f"k8s watch HTTP {resp.status}")
                        backoff = 0.1  # stream established
                        async for line in resp.content:
                            line = line.strip()
                            if not line:
                                continue
                            try:
                                ev = json.loads(line)
                            except json.JSONDecodeError:
                                continue
                            obj = ev.get("object") or {}
                            rv = (obj.get("metadata") or {}).get("resourceVersion")
                            if rv:
                                self._resource_version = rv
                            if ev.get("type") == "ERROR":
                                if (obj.get("code") == 410):
                                    self._resource_version = None
                                raise RuntimeError(f"k8s watch ERROR event: {obj}")
                            self._apply_event(ev.get("type", ""), obj)
                            await on_change(list(self._pods.values()))
                raise RuntimeError("k8s watch stream ended")
            except asyncio.CancelledError:
                return
            except Exception as exc:
                log.warning("k8s watch reconnecting in %.1fs: %s", backoff, exc)
                self.reconnects += 1
                await asyncio.sleep(backoff)
                backoff = min(backoff * 2, 30.0)


async def reconcile(registry, source: DiscoverySource, circuit_breaker_config=None) -> None:
    """One reconciliation pass: add new workers, remove vanished discovered ones."""
    discovered = await source.poll()
    want = {d.url.rstrip("/"): d for d in discovered}
    have = {w.url: w for w in registry.all() if w.labels.get("discovered") == "true"}
    for url, d in want.items():
        if registry.get_by_url(url) is None:
            labels = dict(d.labels)
            labels["discovered"] = "true"
            registry.register(
                Worker(
                    d.url,
                    model_id=d.model_id,
                    worker_type=WorkerType(d.worker_type),
                    labels=labels,
                    bootstrap_port=d.bootstrap_port,
                    circuit_breaker_config=circuit_breaker_config,
                )
            )
    for url, w in have.items():
        if url not in want:
            registry.remove(w.worker_id)


def _apply_discovered(registry, discovered: List[DiscoveredWorker], circuit_breaker_config=None) -> None:
    want = {d.url.rstrip("/"): d for d in discovered}
    have = {w.url: w for w in registry.all() if w.labels.get("discovered") == "true"}
    for url, d in want.items():
        if registry.get_by_url(url) is None:
            labels = dict(d.labels)
            labels["discovered"] = "true"
            registry.register(
                Worker(
                    d.url,
                    model_id=d.model_id,
                    worker_type=WorkerType(d.worker_type),
                    labels=labels,
                    bootstrap_port=d.bootstrap_port,
                    circuit_breaker_config=circuit_breaker_config,
                )
            )
    for url, w in have.items():
        if url not in want:
            registry.remove(w.worker_id)


async def start_discovery(registry, source: DiscoverySource, interval: float = 10.0,
                          circuit_breaker_config=None) -> asyncio.Task:
    """Watch-capable sources stream events into the registry (one long-lived
    connection, incremental updates); poll-only sources reconcile on a
    timer."""
    if hasattr(source, "watch"):
        async def on_change(workers):
            _apply_discovered(registry, workers, circuit_breaker_config)

        return asyncio.ensure_future(source.watch(on_change))

    async def loop():
        while True:
            try:
                await reconcile(registry, source, circuit_breaker_config)
            except asyncio.CancelledError:
                return
            except Exception as exc:
                log.warning("discovery pass failed: %s", exc)
            await asyncio.sleep(interval)

    return asyncio.ensure_future(loop())
