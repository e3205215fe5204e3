"""Plugin manager: load/unload modules, run OnRequest/OnResponse phases."""
from __future__ import annotations

import importlib.util
import logging
import uuid
from typing import Any, Dict, List, Optional

log = logging.getLogger("smg.plugins")


class ShortCircuit(Exception):
    """Raised by a plugin to answer the request immediately."""

    def __init__(self, status: int, body: bytes, headers: Optional[Dict[str, str]] = None):
        super().__init__(f"short-circuit {status}")
        self.status = status
        self.body = body
        self.headers = headers or {}


class PluginManager:
    def __init__(self, max_body_size: int = 10 << 20):
        self._modules: Dict[str, dict] = {}  # uuid -> {name, module, path}
        self.max_body_size = max_body_size

    def add_module(self, path: str, name: Optional[str] = None) -> str:
        mod_id = uuid.uuid4().hex
        spec = importlib.util.spec_from_file_location(f"smg_plugin_{mod_id}", path)
        if spec is None or spec.loader is None:
            raise ValueError(f"cannot load plugin from {path}")
        module = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(module)
        if not hasattr(module, "on_request") and not hasattr(module, "on_response"):
            raise ValueError("plugin must export on_request and/or on_response")
        self._modules[mod_id] = {"id": mod_id, "name": name or path, "module": module, "path": path}
        return mod_id

    def remove_module(self, mod_id: str) -> bool:
        return self._modules.pop(mod_id, None) is not None

    def list_modules(self) -> List[dict]:
        return [{"id": m["id"], "name": m["name"], "path": m["path"]} for m in self._modules.values()]

    def __len__(self) -> int:
        return len(self._modules)

    def run_phase(self, phase: str, ctx: Dict[str, Any]) -> Dict[str, Any]:
        """phase: on_request | on_response.  May raise ShortCircuit."""
        for m in list(self._modules.values()):
            fn = getattr(m["module"], phase, None)
            if fn is None:
                continue
            try:
                out = fn(ctx)
            except ShortCircuit:
                raise
            except Exception as exc:
                log.warning("plugin %s %s failed: %s", m["name"], phase, exc)
                continue
            if isinstance(out, dict):
                ctx.update(out)
        return ctx
