"""Worker-registration workflow (reference: model_gateway/src/workflow/
steps/mod.rs:91-260 — classify → detect connection mode → discover metadata
→ discover DP info → create worker, with per-step retry/timeout/failure
actions over the wfaas engine).

MI355X mapping: `sim://` and `rccl://` workers are the in-process /
RCCL-data-plane engines (local branch — no HTTP probing, DP info from the
engine object); `http(s)://` workers are external and get probed like the
reference's external branch.
"""
from __future__ import annotations

from typing import Any, Dict

from ..workflow import (
    BackoffStrategy,
    FailureAction,
    RetryPolicy,
    StepDefinition,
    WorkflowDefinition,
    WorkflowEngine,
    WorkflowError,
)


def build_worker_registration(ctx) -> WorkflowDefinition:
    """data = {"payload": <job payload>, "worker": out, ...discovered}."""

    async def classify(data: Dict[str, Any]):
        url = data["payload"]["url"]
        if url.startswith(("sim://", "rccl://")):
            data["kind"] = "local"
        elif url.startswith(("http://", "https://", "grpc://")):
            data["kind"] = "external"
        else:
            raise WorkflowError(f"unsupported worker URL scheme: {url}")

    async def detect_connection_mode(data: Dict[str, Any]):
        url = data["payload"]["url"]
        if data["kind"] == "local":
            data["connection"] = "data_plane"
            return
        data["connection"] = "grpc" if url.startswith("grpc://") else "http"
        session = getattr(ctx, "client_session", None)
        if data["connection"] == "http" and session is not None:
            async with session.get(url.rstrip("/") + "/health") as resp:
                if resp.status >= 500:
                    raise WorkflowError(f"worker {url} unhealthy: HTTP {resp.status}")

    async def discover_metadata(data: Dict[str, Any]):
        url = data["payload"]["url"]
        session = getattr(ctx, "client_session", None)
        if data["kind"] == "external" and data["connection"] == "http" and session is not None:
            async with session.get(url.rstrip("/") + "/get_model_info") as resp:
                if resp.status == 200:
                    data["metadata"] = await resp.json()

    async def discover_dp_info(data: Dict[str, Any]):
        meta = data.get("metadata") or {}
        data["dp_size"] = int(meta.get("dp_size") or data["payload"].get("dp_size") or 1)

    async def create_worker(data: Dict[str, Any]):
        from ..workers.worker import Worker, WorkerType

        p = data["payload"]
        w = Worker(
            p["url"],
            model_id=p.get("model_id") or (data.get("metadata") or {}).get("model_id", "default"),
            worker_type=WorkerType(p.get("worker_type", "regular")),
            labels=p.get("labels") or {},
            api_key=p.get("api_key"),
            bootstrap_port=p.get("bootstrap_port"),
            model_aliases=p.get("model_aliases") or [],
            circuit_breaker_config=ctx.config.circuit_breaker,
        )
        if data.get("dp_size", 1) > 1:
            w.extra["dp_size"] = data["dp_size"]
        data["worker"] = ctx.worker_registry.register(w)

    probe_retry = RetryPolicy(max_attempts=3, backoff=BackoffStrategy("fixed", base=0.2))
    return (
        WorkflowDefinition(id="worker_registration", name="Worker Registration", default_timeout=10.0)
        .add_step(StepDefinition("classify_worker_type", classify, timeout=5.0))
        .add_step(
            StepDefinition(
                "detect_connection_mode",
                detect_connection_mode,
                depends_on=["classify_worker_type"],
                retry=probe_retry,
            )
        )
        .add_step(
            StepDefinition(
                "discover_metadata",
                discover_metadata,
                depends_on=["detect_connection_mode"],
                retry=probe_retry,
                failure_action=FailureAction.CONTINUE_NEXT_STEP,
            )
        )
        .add_step(
            StepDefinition(
                "discover_dp_info",
                discover_dp_info,
                # runs even when metadata discovery failed (any-of on the
                # connection step, matching ContinueNextStep semantics)
                depends_on=["detect_connection_mode"],
            )
        )
        .add_step(
            StepDefinition(
                "create_worker",
                create_worker,
                depends_on=["discover_dp_info"],
                timeout=5.0,
            )
        )
    )


def make_engine(ctx) -> WorkflowEngine:
    engine = WorkflowEngine()
    engine.register_workflow(build_worker_registration(ctx))
    return engine


async def register_worker_via_workflow(engine: WorkflowEngine, payload: Dict[str, Any]):
    data: Dict[str, Any] = {"payload": payload}
    iid = await engine.start_workflow("worker_registration", data)
    state = await engine.wait_for_completion(iid, timeout=60.0)
    if state.status.value != "completed":
        raise WorkflowError(state.error or f"worker registration {state.status.value}")
    return data["worker"]
