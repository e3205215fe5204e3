"""Request/response plugin hooks (reference: crates/wasm — wasmtime
component-model middleware with OnRequest/OnResponse attach points
(module.rs), live add/remove over REST, storage hooks (hooked.rs)).

This image has no wasmtime, so plugins are sandbox-free Python modules
loaded from file paths (documented deviation): each module exports
`on_request(ctx) -> ctx|None` and/or `on_response(ctx) -> ctx|None`.
A returned dict replaces the phase's mutable fields; raising ShortCircuit
answers the request immediately — the same contract the WASM modules have.
"""
from .manager import PluginManager, ShortCircuit

__all__ = ["PluginManager", "ShortCircuit"]
