"""Per-tenant rate limiting (reference: model_gateway/src/rate_limit/ —
config.rs YAML limits, RateLimitManager::reserve -> settle (manager.rs:73),
pluggable backend (local_backend.rs), distributed shards via the mesh `rl:`
namespace (adapters/rate_limit_sync.rs))."""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, Optional


@dataclass
class TenantLimit:
    requests_per_minute: Optional[int] = None
    tokens_per_minute: Optional[int] = None
    max_concurrent: Optional[int] = None


@dataclass
class TenantRateLimitSettings:
    default: TenantLimit = field(default_factory=TenantLimit)
    tenants: Dict[str, TenantLimit] = field(default_factory=dict)

    @classmethod
    def from_yaml(cls, path: Optional[str]) -> "TenantRateLimitSettings":
        s = cls()
        if not path:
            return s
        import yaml

        with open(path) as f:
            data = yaml.safe_load(f) or {}

        def mk(d):
            return TenantLimit(
                requests_per_minute=d.get("requests_per_minute"),
                tokens_per_minute=d.get("tokens_per_minute"),
                max_concurrent=d.get("max_concurrent"),
            )

        if "default" in data:
            s.default = mk(data["default"])
        for t, d in (data.get("tenants") or {}).items():
            s.tenants[t] = mk(d)
        return s


class Reservation:
    __slots__ = ("tenant", "tokens", "settled")

    def __init__(self, tenant: str, tokens: int):
        self.tenant = tenant
        self.tokens = tokens
        self.settled = False


class RateLimitManager:
    """Epoch-based accounting (1-minute epochs align with the mesh's
    EpochMaxWins shard merge)."""

    def __init__(self, settings: Optional[TenantRateLimitSettings] = None, mesh_adapter=None, clock=time.time):
        self.settings = settings or TenantRateLimitSettings()
        self.mesh_adapter = mesh_adapter
        self.clock = clock
        self._requests: Dict[str, Dict[int, int]] = {}
        self._tokens: Dict[str, Dict[int, int]] = {}
        self._concurrent: Dict[str, int] = {}
        self._remote_usage: Dict[str, Dict[int, int]] = {}  # tenant -> epoch -> tokens

    def limit_for(self, tenant: str) -> TenantLimit:
        return self.settings.tenants.get(tenant, self.settings.default)

    def _epoch(self) -> int:
        return int(self.clock() // 60)

    def reserve(self, tenant: str, est_tokens: int = 0) -> Optional[Reservation]:
        """None = rejected (429)."""
        lim = self.limit_for(tenant)
        epoch = self._epoch()
        if lim.max_concurrent is not None and self._concurrent.get(tenant, 0) >= lim.max_concurrent:
            return None
        reqs = self._requests.setdefault(tenant, {}).get(epoch, 0)
        if lim.requests_per_minute is not None and reqs >= lim.requests_per_minute:
            return None
        if lim.tokens_per_minute is not None:
            used = self._tokens.setdefault(tenant, {}).get(epoch, 0)
            used += self._remote_usage.get(tenant, {}).get(epoch, 0)
            if used + est_tokens > lim.tokens_per_minute:
                return None
        self._requests[tenant][epoch] = reqs + 1
        if est_tokens:
            self._tokens.setdefault(tenant, {}).setdefault(epoch, 0)
            self._tokens[tenant][epoch] += est_tokens
        self._concurrent[tenant] = self._concurrent.get(tenant, 0) + 1
        self._publish(tenant, epoch)
        return Reservation(tenant, est_tokens)

    def settle(self, res: Reservation, actual_tokens: Optional[int] = None) -> None:
        if res.settled:
            return
        res.settled = True
        self._concurrent[res.tenant] = max(0, self._concurrent.get(res.tenant, 0) - 1)
        if actual_tokens is not None and actual_tokens != res.tokens:
            epoch = self._epoch()
            tok = self._tokens.setdefault(res.tenant, {})
            tok[epoch] = max(0, tok.get(epoch, 0) + (actual_tokens - res.tokens))
            self._publish(res.tenant, epoch)

    def observe_remote_usage(self, tenant: str, epoch: int, used: int) -> None:
        cur = self._remote_usage.setdefault(tenant, {})
        cur[epoch] = max(cur.get(epoch, 0), used)

    def _publish(self, tenant: str, epoch: int) -> None:
        if self.mesh_adapter is not None:
            self.mesh_adapter.publish_usage(tenant, epoch, self._tokens.get(tenant, {}).get(epoch, 0))

    def gc(self) -> None:
        cutoff = self._epoch() - 2
        for store in (self._requests, self._tokens, self._remote_usage):
            for tenant in list(store):
                store[tenant] = {e: v for e, v in store[tenant].items() if e >= cutoff}
