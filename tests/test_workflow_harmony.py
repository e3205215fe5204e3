"""Typed workflow engine (reference crates/workflow) and Harmony gpt-oss
format (reference grpc/harmony/) tests."""
import asyncio

import pytest

from smg_amd.workflow import (
    BackoffStrategy,
    FailureAction,
    RetryPolicy,
    StepDefinition,
    StepResult,
    StepStatus,
    WorkflowDefinition,
    WorkflowEngine,
    WorkflowError,
    WorkflowStatus,
)


def _run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_workflow_dag_order_and_events():
    order = []
    events = []

    async def step(name):
        order.append(name)

    wf = (
        WorkflowDefinition(id="t", default_timeout=5.0)
        .add_step(StepDefinition("a", lambda d: step("a")))
        .add_step(StepDefinition("b", lambda d: step("b"), depends_on=["a"]))
        .add_step(StepDefinition("c", lambda d: step("c"), depends_on=["a"]))
        .add_step(StepDefinition("d", lambda d: step("d"), depends_on=["b", "c"]))
    )

    async def run():
        eng = WorkflowEngine()
        eng.register_workflow(wf)
        eng.subscribe(lambda ev: events.append(ev.kind))
        iid = await eng.start_workflow("t", {})
        state = await eng.wait_for_completion(iid, timeout=5.0)
        assert state.status == WorkflowStatus.COMPLETED
        assert all(s.status == StepStatus.SUCCEEDED for s in state.steps.values())

    _run(run())
    assert order[0] == "a" and order[-1] == "d" and set(order) == {"a", "b", "c", "d"}
    assert "workflow_completed" in events and events.count("step_succeeded") == 4


def test_workflow_retry_then_succeed():
    attempts = []

    async def flaky(d):
        attempts.append(1)
        if len(attempts) < 3:
            raise RuntimeError("transient")

    wf = WorkflowDefinition(id="r", default_timeout=5.0).add_step(
        StepDefinition("flaky", flaky, retry=RetryPolicy(max_attempts=5, backoff=BackoffStrategy("fixed", base=0.01)))
    )

    async def run():
        eng = WorkflowEngine()
        eng.register_workflow(wf)
        iid = await eng.start_workflow("r", {})
        state = await eng.wait_for_completion(iid, timeout=5.0)
        assert state.status == WorkflowStatus.COMPLETED
        assert state.steps["flaky"].attempt == 3

    _run(run())


def test_workflow_failure_actions():
    async def boom(d):
        raise RuntimeError("dead")

    async def ok(d):
        d["ran"] = d.get("ran", 0) + 1

    # ContinueNextStep: sibling still runs, dependent of the failed step skips
    wf = (
        WorkflowDefinition(id="f", default_timeout=5.0, default_retry=RetryPolicy(1, BackoffStrategy("fixed", base=0.01)))
        .add_step(StepDefinition("boom", boom, failure_action=FailureAction.CONTINUE_NEXT_STEP))
        .add_step(StepDefinition("dependent", ok, depends_on=["boom"]))
        .add_step(StepDefinition("sibling", ok))
    )

    async def run():
        eng = WorkflowEngine()
        eng.register_workflow(wf)
        data = {}
        iid = await eng.start_workflow("f", data)
        state = await eng.wait_for_completion(iid, timeout=5.0)
        assert state.status == WorkflowStatus.COMPLETED
        assert state.steps["boom"].status == StepStatus.FAILED
        assert state.steps["dependent"].status == StepStatus.SKIPPED
        assert state.steps["sibling"].status == StepStatus.SUCCEEDED
        assert data["ran"] == 1

        # FAIL_WORKFLOW: whole workflow fails
        wf2 = WorkflowDefinition(id="f2", default_timeout=5.0, default_retry=RetryPolicy(1, BackoffStrategy("fixed", base=0.01))).add_step(
            StepDefinition("boom", boom)
        )
        eng.register_workflow(wf2)
        iid = await eng.start_workflow("f2", {})
        state = await eng.wait_for_completion(iid, timeout=5.0)
        assert state.status == WorkflowStatus.FAILED

    _run(run())


def test_workflow_run_if_and_skip_result():
    async def skipper(d):
        return StepResult.SKIP

    async def ok(d):
        d["ok"] = True

    wf = (
        WorkflowDefinition(id="s", default_timeout=5.0)
        .add_step(StepDefinition("cond", ok, run_if=lambda d: d.get("go", False)))
        .add_step(StepDefinition("sk", skipper))
    )

    async def run():
        eng = WorkflowEngine()
        eng.register_workflow(wf)
        data = {}
        iid = await eng.start_workflow("s", data)
        state = await eng.wait_for_completion(iid, timeout=5.0)
        assert state.steps["cond"].status == StepStatus.SKIPPED
        assert state.steps["sk"].status == StepStatus.SKIPPED
        assert "ok" not in data

    _run(run())


def test_workflow_validation():
    async def ok(d):
        pass

    wf = WorkflowDefinition(id="bad").add_step(StepDefinition("a", ok, depends_on=["zzz"]))
    eng = WorkflowEngine()
    with pytest.raises(WorkflowError, match="unknown step"):
        eng.register_workflow(wf)
    cyc = (
        WorkflowDefinition(id="cyc")
        .add_step(StepDefinition("a", ok, depends_on=["b"]))
        .add_step(StepDefinition("b", ok, depends_on=["a"]))
    )
    with pytest.raises(WorkflowError, match="cycle"):
        eng.register_workflow(cyc)


def test_worker_registration_workflow():
    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.server.app_context import AppContext
    from smg_amd.server.worker_workflow import make_engine, register_worker_via_workflow

    async def run():
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        ctx = AppContext(cfg)
        eng = make_engine(ctx)
        w = await register_worker_via_workflow(eng, {"url": "sim://wf-worker", "model_id": "m1"})
        assert w.url == "sim://wf-worker"
        assert ctx.worker_registry.get_by_url("sim://wf-worker") is not None
        with pytest.raises(WorkflowError):
            await register_worker_via_workflow(eng, {"url": "bogus://x"})

    _run(run())


# ---- harmony ----------------------------------------------------------------
def test_harmony_builder_and_complete_parse():
    from smg_amd.protocols.harmony import build_harmony_prompt, parse_complete

    prompt = build_harmony_prompt(
        {
            "messages": [
                {"role": "system", "content": "Be brief."},
                {"role": "user", "content": "weather in SF?"},
            ],
            "tools": [{"type": "function", "function": {"name": "get_weather", "description": "wx", "parameters": {"type": "object"}}}],
        }
    )
    assert prompt.endswith("<|start|>assistant")
    assert "# Valid channels: analysis, commentary, final" in prompt
    assert "get_weather" in prompt and "Be brief." in prompt

    out = (
        "<|channel|>analysis<|message|>User wants weather; call the tool.<|end|>"
        "<|start|>assistant<|channel|>commentary to=functions.get_weather<|message|>{\"city\": \"SF\"}<|call|>"
        "<|start|>assistant<|channel|>final<|message|>It is sunny.<|return|>"
    )
    res = parse_complete(out)
    assert res.reasoning_content == "User wants weather; call the tool."
    assert res.content == "It is sunny."
    assert res.tool_calls[0]["function"]["name"] == "get_weather"
    assert res.tool_calls[0]["function"]["arguments"] == '{"city": "SF"}'


def test_harmony_stream_parser_events():
    from smg_amd.protocols.harmony import HarmonyStreamParser

    sp = HarmonyStreamParser()
    full = (
        "<|channel|>analysis<|message|>thinking<|end|>"
        "<|start|>assistant<|channel|>commentary to=functions.f<|message|>{\"a\":1}<|call|>"
        "<|start|>assistant<|channel|>final<|message|>done<|return|>"
    )
    events = []
    for i in range(0, len(full), 7):  # ragged chunks crossing token boundaries
        events.extend(sp.feed(full[i:i + 7]))
    events.extend(sp.finalize())
    kinds = [e["type"] for e in events]
    assert "".join(e["text"] for e in events if e["type"] == "reasoning") == "thinking"
    assert "".join(e["text"] for e in events if e["type"] == "content") == "done"
    start = [e for e in events if e["type"] == "tool_call_start"][0]
    assert start["name"] == "f"
    assert "".join(e["arguments"] for e in events if e["type"] == "tool_call_args") == '{"a":1}'
    assert "tool_call_end" in kinds


def test_harmony_reasoning_and_tool_parser_pipeline():
    """The pipeline order: reasoning parser first, tool parser on its normal
    output — harmony parsers compose across that boundary."""
    from smg_amd.parsers.reasoning import get_reasoning_parser
    from smg_amd.parsers.tool.factory import get_parser

    out = (
        "<|channel|>analysis<|message|>let me check<|end|>"
        "<|start|>assistant<|channel|>commentary to=functions.add<|message|>{\"x\": 2}<|call|>"
        "<|start|>assistant<|channel|>final<|message|>sum is 4<|return|>"
    )
    rp = get_reasoning_parser("gpt-oss-120b")
    reasoning, normal = rp.parse(out)
    assert reasoning == "let me check"
    tp = get_parser("gpt-oss-120b")
    content, calls = tp.parse(normal)
    assert content == "sum is 4"
    assert calls == [{"name": "add", "arguments": '{"x": 2}', "index": 0}]


def test_harmony_streaming_through_reasoning_parser():
    from smg_amd.parsers.reasoning import get_reasoning_parser

    rp = get_reasoning_parser("harmony")
    out = (
        "<|channel|>analysis<|message|>hmm<|end|>"
        "<|start|>assistant<|channel|>final<|message|>hello world<|return|>"
    )
    reasoning, normal = "", ""
    for i in range(0, len(out), 5):
        r, n = rp.parse_streaming(out[i:i + 5])
        reasoning += r
        normal += n
    assert reasoning == "hmm"
    assert normal == "hello world"
