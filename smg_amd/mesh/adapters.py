"""Mesh <-> gateway adapters (reference: model_gateway/src/mesh/adapters/ —
worker_sync.rs (`worker:` namespace, mesh-imported workers), tree_sync.rs
(`td:` tree deltas + repair), rate_limit_sync.rs (`rl:` EpochMaxWins shards),
wiring.rs)."""
from __future__ import annotations

import logging
from ..workers.worker import Worker, WorkerType
from .crdt import Op, epoch_max_wins_merge
from .swim import MeshNode

log = logging.getLogger("smg.mesh.adapters")

NS_WORKER = "worker"
NS_TREE = "td"
NS_RATELIMIT = "rl"


class WorkerSyncAdapter:
    """Workers registered on one gateway appear on every replica
    (mesh-imported; removed when the origin removes them)."""

    def __init__(self, mesh: MeshNode, registry):
        self.mesh = mesh
        self.registry = registry
        self._importing = False
        mesh.kv.register_namespace(NS_WORKER)
        mesh.kv.watch(NS_WORKER, self._on_remote_op)
        registry.subscribe(self._on_local_event)
        # publish pre-existing local workers
        for w in registry.all():
            self._publish(w)

    def _on_local_event(self, kind: str, worker: Worker) -> None:
        if self._importing or worker.labels.get("mesh_origin"):
            return
        if kind == "add":
            self._publish(worker)
        elif kind == "remove":
            self.mesh.kv.delete(NS_WORKER, worker.url)

    def _publish(self, worker: Worker) -> None:
        if worker.labels.get("mesh_origin"):
            return
        self.mesh.kv.put(
            NS_WORKER,
            worker.url,
            {
                "url": worker.url,
                "model_id": worker.model_id,
                "worker_type": worker.worker_type.value,
                "labels": worker.labels,
                "origin": self.mesh.node_id,
            },
        )

    def _on_remote_op(self, op: Op) -> None:
        self._importing = True
        try:
            if op.value is None:
                w = self.registry.get_by_url(op.key)
                if w is not None and w.labels.get("mesh_origin"):
                    self.registry.remove(w.worker_id)
                return
            v = op.value
            if v.get("origin") == self.mesh.node_id:
                return
            if self.registry.get_by_url(v["url"]) is None:
                labels = dict(v.get("labels") or {})
                labels["mesh_origin"] = v.get("origin", "?")
                self.registry.register(
                    Worker(
                        v["url"],
                        model_id=v.get("model_id", "default"),
                        worker_type=WorkerType(v.get("worker_type", "regular")),
                        labels=labels,
                    )
                )
        finally:
            self._importing = False


class TreeSyncAdapter:
    """Cache-tree delta sync (reference tree_sync.rs:1-20): local cache-aware
    inserts publish the routed prefix's TOKENS + tenant, and replicas insert
    them into their own tree — a request landing on any gateway replica routes
    to the worker that already holds the prefix.  The join snapshot doubles
    as the repair protocol (:38-67)."""

    MAX_SYNC_TOKENS = 1024

    def __init__(self, mesh: MeshNode, policy_registry):
        self.mesh = mesh
        self.policies = policy_registry
        self._importing = False
        mesh.kv.register_namespace(NS_TREE)
        mesh.kv.watch(NS_TREE, self._on_remote_op)
        # hook every current + future cache_aware policy
        self._wire_existing()

    def _wire_existing(self) -> None:
        for policy in self.policies.all_policies():
            self._wire(policy)
        orig_get = self.policies.get
        adapter = self

        def hooked_get(model_id=None, role="regular"):
            policy = orig_get(model_id, role)
            adapter._wire(policy)
            return policy

        self.policies.get = hooked_get

    def _wire(self, policy) -> None:
        if hasattr(policy, "mesh_hook") and policy.mesh_hook is None:
            policy.mesh_hook = self.publish_insert

    def publish_insert(self, model_id: str, tokens, tenant_url: str) -> None:
        if self._importing:
            return
        tokens = list(tokens)[: self.MAX_SYNC_TOKENS]
        import hashlib

        key = hashlib.blake2b(bytes(str((model_id, tenant_url, tokens[:64]))[:512], "utf8"),
                              digest_size=12).hexdigest()
        self.mesh.kv.put(
            NS_TREE, f"{model_id}:{key}",
            {"model": model_id, "tenant": tenant_url, "tokens": tokens},
        )

    def _on_remote_op(self, op: Op) -> None:
        if op.value is None:
            return
        v = op.value
        tokens = v.get("tokens")
        if not tokens:
            return
        self._importing = True
        try:
            policy = self.policies.get(v.get("model"))
            if hasattr(policy, "token_trees"):
                policy._token_tree(v.get("model") or "default").insert(tokens, v.get("tenant"))
        except Exception as exc:
            log.debug("tree sync apply failed: %s", exc)
        finally:
            self._importing = False


class RateLimitSyncAdapter:
    """Distributed token-bucket shards: each gateway publishes its shard usage
    per (tenant, epoch); EpochMaxWins merge keeps the cluster-wide max
    (reference rate_limit_sync.rs + epoch_max_wins.rs)."""

    def __init__(self, mesh: MeshNode, rate_limit_manager=None):
        self.mesh = mesh
        self.manager = rate_limit_manager
        mesh.kv.register_namespace(NS_RATELIMIT, epoch_max_wins_merge)
        mesh.kv.watch(NS_RATELIMIT, self._on_remote_op)

    def publish_usage(self, tenant: str, epoch: int, used: int) -> None:
        self.mesh.kv.put(NS_RATELIMIT, f"{tenant}:{self.mesh.node_id}", {"epoch": epoch, "used": used})

    def cluster_usage(self, tenant: str, epoch: int) -> int:
        total = 0
        for key, v in self.mesh.kv.items(NS_RATELIMIT).items():
            if key.startswith(tenant + ":") and v.get("epoch") == epoch:
                total += v.get("used", 0)
        return total

    def _on_remote_op(self, op: Op) -> None:
        if self.manager is not None and op.value is not None:
            tenant = op.key.split(":", 1)[0]
            self.manager.observe_remote_usage(tenant, op.value.get("epoch", 0), op.value.get("used", 0))


class MeshAdapters:
    def __init__(self, mesh: MeshNode, ctx):
        self.worker_sync = WorkerSyncAdapter(mesh, ctx.worker_registry)
        self.tree_sync = TreeSyncAdapter(mesh, ctx.policy_registry)
        self.rate_limit_sync = RateLimitSyncAdapter(mesh, getattr(ctx, "tenant_rate_limiter", None))
