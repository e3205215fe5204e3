"""Typed workflow engine (reference: crates/workflow/src/{definition,engine,
event,executor,state,types}.rs — WorkflowDefinition/StepDefinition DAG with
all-of + any-of dependencies (definition.rs:108,127), run_if conditions
(:167), RetryPolicy with fixed/exponential/linear backoff (types.rs:90-116),
FailureAction fail/continue/retry-indefinitely (types.rs:120), WorkflowEngine
with start/cancel/status/wait (engine.rs:414,1113-1150), EventBus
(event.rs:16-59) and pluggable StateStore (state.rs)).

asyncio-native: each workflow instance is one task; ready steps run
concurrently as their dependencies resolve.
"""
from __future__ import annotations

import asyncio
import enum
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Optional


class WorkflowError(RuntimeError):
    pass


class WorkflowStatus(str, enum.Enum):
    PENDING = "pending"
    RUNNING = "running"
    COMPLETED = "completed"
    FAILED = "failed"
    CANCELLED = "cancelled"


class StepStatus(str, enum.Enum):
    PENDING = "pending"
    RUNNING = "running"
    SUCCEEDED = "succeeded"
    FAILED = "failed"
    RETRYING = "retrying"
    SKIPPED = "skipped"


class StepResult(str, enum.Enum):
    SUCCESS = "success"
    FAILURE = "failure"
    SKIP = "skip"


class FailureAction(str, enum.Enum):
    FAIL_WORKFLOW = "fail_workflow"
    CONTINUE_NEXT_STEP = "continue_next_step"
    RETRY_INDEFINITELY = "retry_indefinitely"


@dataclass
class BackoffStrategy:
    kind: str = "exponential"  # fixed | exponential | linear
    base: float = 1.0
    max: float = 30.0
    increment: float = 1.0

    def delay(self, attempt: int) -> float:
        if self.kind == "fixed":
            return self.base
        if self.kind == "linear":
            return min(self.max, self.base + self.increment * (attempt - 1))
        return min(self.max, self.base * (2 ** (attempt - 1)))


@dataclass
class RetryPolicy:
    max_attempts: int = 3
    backoff: BackoffStrategy = field(default_factory=BackoffStrategy)


@dataclass
class StepDefinition:
    """One DAG node.  `execute(data) -> StepResult | None` (None = success);
    raising marks the attempt failed (executor.rs:9 StepExecutor)."""

    id: str
    execute: Callable[[Any], Awaitable[Any]]
    depends_on: List[str] = field(default_factory=list)  # all must succeed
    depends_on_any: List[str] = field(default_factory=list)  # one must succeed
    retry: Optional[RetryPolicy] = None
    timeout: Optional[float] = None
    failure_action: FailureAction = FailureAction.FAIL_WORKFLOW
    run_if: Optional[Callable[[Any], bool]] = None
    delay: float = 0.0

    def all_dependencies(self) -> List[str]:
        return [*self.depends_on, *self.depends_on_any]


@dataclass
class WorkflowDefinition:
    id: str
    name: str = ""
    steps: List[StepDefinition] = field(default_factory=list)
    default_retry: RetryPolicy = field(default_factory=RetryPolicy)
    default_timeout: float = 60.0

    def add_step(self, step: StepDefinition) -> "WorkflowDefinition":
        self.steps.append(step)
        return self

    def validate(self) -> None:
        """definition.rs:253 — unique ids, known deps, no cycles."""
        ids = [s.id for s in self.steps]
        if len(set(ids)) != len(ids):
            raise WorkflowError(f"duplicate step ids in workflow {self.id!r}")
        known = set(ids)
        for s in self.steps:
            for d in s.all_dependencies():
                if d not in known:
                    raise WorkflowError(f"step {s.id!r} depends on unknown step {d!r}")
        # cycle check (Kahn)
        indeg = {s.id: len(set(s.all_dependencies())) for s in self.steps}
        dependents: Dict[str, List[str]] = {s.id: [] for s in self.steps}
        for s in self.steps:
            for d in set(s.all_dependencies()):
                dependents[d].append(s.id)
        queue = [i for i, n in indeg.items() if n == 0]
        seen = 0
        while queue:
            cur = queue.pop()
            seen += 1
            for nxt in dependents[cur]:
                indeg[nxt] -= 1
                if indeg[nxt] == 0:
                    queue.append(nxt)
        if seen != len(self.steps):
            raise WorkflowError(f"dependency cycle in workflow {self.id!r}")


@dataclass
class StepState:
    status: StepStatus = StepStatus.PENDING
    attempt: int = 0
    last_error: Optional[str] = None
    started_at: Optional[float] = None
    completed_at: Optional[float] = None


@dataclass
class WorkflowState:
    instance_id: str
    definition_id: str
    status: WorkflowStatus = WorkflowStatus.PENDING
    data: Any = None
    steps: Dict[str, StepState] = field(default_factory=dict)
    started_at: float = 0.0
    completed_at: Optional[float] = None
    error: Optional[str] = None


@dataclass
class WorkflowEvent:
    """event.rs:16 — kind in {workflow_started, step_started, step_succeeded,
    step_failed, step_retrying, step_skipped, workflow_completed,
    workflow_failed, workflow_cancelled}."""

    kind: str
    instance_id: str
    step_id: Optional[str] = None
    attempt: int = 0
    error: Optional[str] = None
    duration: float = 0.0


class InMemoryStore:
    """state.rs InMemoryStore: instance_id -> WorkflowState."""

    def __init__(self):
        self._states: Dict[str, WorkflowState] = {}

    def save(self, state: WorkflowState) -> None:
        self._states[state.instance_id] = state

    def load(self, instance_id: str) -> Optional[WorkflowState]:
        return self._states.get(instance_id)

    def remove(self, instance_id: str) -> None:
        self._states.pop(instance_id, None)

    def all(self) -> List[WorkflowState]:
        return list(self._states.values())


class WorkflowEngine:
    def __init__(self, store: Optional[InMemoryStore] = None):
        self.store = store or InMemoryStore()
        self._definitions: Dict[str, WorkflowDefinition] = {}
        self._tasks: Dict[str, asyncio.Task] = {}
        self._subscribers: List[Callable[[WorkflowEvent], None]] = []
        self._shutting_down = False

    # ---- registry / events --------------------------------------------------
    def register_workflow(self, definition: WorkflowDefinition) -> None:
        definition.validate()
        self._definitions[definition.id] = definition

    def subscribe(self, fn: Callable[[WorkflowEvent], None]) -> None:
        self._subscribers.append(fn)

    def _emit(self, ev: WorkflowEvent) -> None:
        for fn in self._subscribers:
            try:
                fn(ev)
            except Exception:
                pass

    # ---- lifecycle ----------------------------------------------------------
    def active_workflow_count(self) -> int:
        return sum(1 for t in self._tasks.values() if not t.done())

    def is_shutting_down(self) -> bool:
        return self._shutting_down

    def shutdown(self) -> None:
        self._shutting_down = True

    async def force_cancel_all(self) -> int:
        n = 0
        for iid, task in list(self._tasks.items()):
            if not task.done():
                task.cancel()
                n += 1
        await asyncio.gather(*self._tasks.values(), return_exceptions=True)
        return n

    async def start_workflow(self, definition_id: str, data: Any) -> str:
        if self._shutting_down:
            raise WorkflowError("engine is shutting down")
        definition = self._definitions.get(definition_id)
        if definition is None:
            raise WorkflowError(f"unknown workflow {definition_id!r}")
        instance_id = f"wf_{uuid.uuid4().hex[:16]}"
        state = WorkflowState(
            instance_id=instance_id,
            definition_id=definition_id,
            status=WorkflowStatus.RUNNING,
            data=data,
            steps={s.id: StepState() for s in definition.steps},
            started_at=time.monotonic(),
        )
        self.store.save(state)
        self._tasks[instance_id] = asyncio.get_event_loop().create_task(self._run(definition, state))
        self._emit(WorkflowEvent("workflow_started", instance_id))
        return instance_id

    async def cancel_workflow(self, instance_id: str) -> None:
        task = self._tasks.get(instance_id)
        if task is None:
            raise WorkflowError(f"workflow not found: {instance_id}")
        task.cancel()
        try:
            await task
        except asyncio.CancelledError:
            pass

    async def get_status(self, instance_id: str) -> WorkflowState:
        state = self.store.load(instance_id)
        if state is None:
            raise WorkflowError(f"workflow not found: {instance_id}")
        return state

    async def wait_for_completion(self, instance_id: str, timeout: Optional[float] = None) -> WorkflowState:
        task = self._tasks.get(instance_id)
        if task is not None:
            try:
                await asyncio.wait_for(asyncio.shield(task), timeout)
            except asyncio.TimeoutError:
                raise WorkflowError(f"workflow {instance_id} did not complete in {timeout}s")
            except asyncio.CancelledError:
                pass
        return await self.get_status(instance_id)

    # ---- execution ----------------------------------------------------------
    async def _run(self, definition: WorkflowDefinition, state: WorkflowState) -> None:
        try:
            pending = {s.id: s for s in definition.steps}
            running: Dict[str, asyncio.Task] = {}
            while pending or running:
                # launch every ready step (engine.rs dependency scheduling)
                for sid in list(pending):
                    step = pending[sid]
                    ss = state.steps[sid]
                    all_ok = all(
                        state.steps[d].status == StepStatus.SUCCEEDED for d in step.depends_on
                    )
                    any_ok = not step.depends_on_any or any(
                        state.steps[d].status == StepStatus.SUCCEEDED for d in step.depends_on_any
                    )
                    blocked = any(
                        state.steps[d].status in (StepStatus.FAILED, StepStatus.SKIPPED)
                        for d in step.depends_on
                    ) or (
                        step.depends_on_any
                        and all(
                            state.steps[d].status in (StepStatus.FAILED, StepStatus.SKIPPED)
                            for d in step.depends_on_any
                        )
                    )
                    if blocked:
                        ss.status = StepStatus.SKIPPED
                        self._emit(WorkflowEvent("step_skipped", state.instance_id, sid))
                        del pending[sid]
                    elif all_ok and any_ok:
                        del pending[sid]
                        running[sid] = asyncio.get_event_loop().create_task(
                            self._run_step(definition, step, state)
                        )
                if not running:
                    break  # everything left is unreachable
                done, _ = await asyncio.wait(running.values(), return_when=asyncio.FIRST_COMPLETED)
                for sid in [k for k, t in running.items() if t in done]:
                    task = running.pop(sid)
                    exc = task.exception()
                    if exc is not None:
                        # FailureAction.FAIL_WORKFLOW propagates here
                        state.status = WorkflowStatus.FAILED
                        state.error = str(exc)
                        state.completed_at = time.monotonic()
                        self.store.save(state)
                        for t in running.values():
                            t.cancel()
                        self._emit(
                            WorkflowEvent("workflow_failed", state.instance_id, sid, error=str(exc))
                        )
                        return
            state.status = WorkflowStatus.COMPLETED
            state.completed_at = time.monotonic()
            self.store.save(state)
            self._emit(
                WorkflowEvent(
                    "workflow_completed", state.instance_id, duration=state.completed_at - state.started_at
                )
            )
        except asyncio.CancelledError:
            state.status = WorkflowStatus.CANCELLED
            state.completed_at = time.monotonic()
            self.store.save(state)
            self._emit(WorkflowEvent("workflow_cancelled", state.instance_id))
            raise

    async def _run_step(
        self, definition: WorkflowDefinition, step: StepDefinition, state: WorkflowState
    ) -> None:
        ss = state.steps[step.id]
        if step.run_if is not None and not step.run_if(state.data):
            ss.status = StepStatus.SKIPPED
            self._emit(WorkflowEvent("step_skipped", state.instance_id, step.id))
            return
        if step.delay > 0:
            await asyncio.sleep(step.delay)
        retry = step.retry or definition.default_retry
        timeout = step.timeout if step.timeout is not None else definition.default_timeout
        attempt = 0
        while True:
            attempt += 1
            ss.status = StepStatus.RUNNING
            ss.attempt = attempt
            ss.started_at = time.monotonic()
            self._emit(WorkflowEvent("step_started", state.instance_id, step.id, attempt))
            try:
                result = await asyncio.wait_for(step.execute(state.data), timeout)
                if result == StepResult.SKIP:
                    ss.status = StepStatus.SKIPPED
                    self._emit(WorkflowEvent("step_skipped", state.instance_id, step.id))
                    return
                if result == StepResult.FAILURE:
                    raise WorkflowError(f"step {step.id} returned failure")
                ss.status = StepStatus.SUCCEEDED
                ss.completed_at = time.monotonic()
                self._emit(
                    WorkflowEvent(
                        "step_succeeded", state.instance_id, step.id,
                        duration=ss.completed_at - ss.started_at,
                    )
                )
                return
            except asyncio.CancelledError:
                raise
            except Exception as exc:
                ss.last_error = str(exc)
                indefinite = step.failure_action == FailureAction.RETRY_INDEFINITELY
                will_retry = indefinite or attempt < retry.max_attempts
                self._emit(
                    WorkflowEvent(
                        "step_failed", state.instance_id, step.id, attempt, error=str(exc)
                    )
                )
                if not will_retry:
                    ss.status = StepStatus.FAILED
                    ss.completed_at = time.monotonic()
                    if step.failure_action == FailureAction.CONTINUE_NEXT_STEP:
                        return  # dependents see FAILED and get skipped; siblings go on
                    raise
                delay = retry.backoff.delay(attempt)
                ss.status = StepStatus.RETRYING
                self._emit(
                    WorkflowEvent("step_retrying", state.instance_id, step.id, attempt + 1)
                )
                await asyncio.sleep(delay)
