"""Typed workflow engine (reference: crates/workflow — wfaas).

Deterministic control-plane orchestration: DAG of StepDefinitions with
retry policies, failure actions, conditional execution, an event bus and a
pluggable state store.  The gateway's JobQueue uses these for worker
registration / tokenizer load / MCP init flows (reference src/workflow/).
"""
from .engine import (
    BackoffStrategy,
    FailureAction,
    InMemoryStore,
    RetryPolicy,
    StepDefinition,
    StepResult,
    StepState,
    StepStatus,
    WorkflowDefinition,
    WorkflowEngine,
    WorkflowError,
    WorkflowEvent,
    WorkflowState,
    WorkflowStatus,
)

__all__ = [
    "BackoffStrategy",
    "FailureAction",
    "InMemoryStore",
    "RetryPolicy",
    "StepDefinition",
    "StepResult",
    "StepState",
    "StepStatus",
    "WorkflowDefinition",
    "WorkflowEngine",
    "WorkflowError",
    "WorkflowEvent",
    "WorkflowState",
    "WorkflowStatus",
]
