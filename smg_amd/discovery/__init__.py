"""Service discovery (reference: model_gateway/src/service_discovery.rs —
kube-rs pod watchers with label selectors :8-24, per-role selectors,
`smg.ai/worker-ports` annotation, ModelIdSource :35).

Pluggable sources:
  * FileDiscovery — polls a JSON file describing the worker fleet (the
    testable, cluster-free source; also useful for static fleets);
  * KubernetesDiscovery — pod watch via the `kubernetes` client package,
    raising a clear error when the package/cluster is absent from this image.
"""
from .source import DiscoveredWorker, FileDiscovery, KubernetesDiscovery, start_discovery

__all__ = ["DiscoveredWorker", "FileDiscovery", "KubernetesDiscovery", "start_discovery"]
