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
    """Pod-watch discovery: selector-matched pods become workers on
    `port` (or the smg.ai/worker-ports annotation); model id from label /
    annotation / namespace per ModelIdSource."""

    def __init__(self, selector: Dict[str, str], port: int, namespace: Optional[str] = None,
                 model_id_from: str = "label"):
        try:
            import kubernetes  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "KubernetesDiscovery requires the `kubernetes` client package "
                "(not installed in this image); use FileDiscovery or static --worker-urls"
            ) from e
        self.selector = selector
        self.port = port
        self.namespace = namespace
        self.model_id_from = model_id_from

    async def poll(self) -> List[DiscoveredWorker]:
        from kubernetes import client, config

        config.load_incluster_config()
        v1 = client.CoreV1Api()
        sel = ",".join(f"{k}={v}" for k, v in self.selector.items())
        pods = v1.list_namespaced_pod(self.namespace or "default", label_selector=sel)
        out = []
        for pod in pods.items:
            if pod.status.phase != "Running" or not pod.status.pod_ip:
                continue
            port = pod.metadata.annotations.get("smg.ai/worker-ports", self.port)
            model_id = pod.metadata.labels.get("smg.ai/model-id", "default")
            out.append(DiscoveredWorker(url=f"http://{pod.status.pod_ip}:{port}", model_id=model_id))
        return out


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


async def start_discovery(registry, source: DiscoverySource, interval: float = 10.0,
                          circuit_breaker_config=None) -> asyncio.Task:
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
