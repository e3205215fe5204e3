"""Engine bridge seams vs fake engines mimicking the PUBLIC API shapes
(reference grpc_servicer vllm/servicer.py + sglang/servicer.py +
request_manager.py — the translation logic is what those files are)."""
import asyncio

import pytest

from smg_amd.grpc import api
from smg_amd.grpc.bridges import (
    SglangBridge,
    VllmBridge,
    build_sglang_generate_payload,
    make_bridge,
    translate_sampling_params_vllm,
)
from smg_amd.grpc.bridges.sglang import normalize_finish_reason


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


# ---- fakes mimicking the engines' public API shapes ------------------------
class FakeSeqOut:
    def __init__(self, ids, finish_reason=None):
        self.token_ids = ids
        self.finish_reason = finish_reason


class FakeRequestOutput:
    def __init__(self, ids, finished, prompt_ids, reason=None, cached=0):
        self.outputs = [FakeSeqOut(ids, reason)]
        self.finished = finished
        self.prompt_token_ids = prompt_ids
        self.num_cached_tokens = cached


class FakeAsyncLLM:
    """vLLM AsyncLLM shape: generate() yields CUMULATIVE RequestOutputs."""

    def __init__(self):
        self.aborted = []

    async def generate(self, prompt, sampling_params, request_id):
        self.last_params = sampling_params
        ids = []
        prompt_ids = prompt["prompt_token_ids"]
        for t in [101, 102, 103]:
            ids.append(t)
            yield FakeRequestOutput(list(ids), False, prompt_ids, cached=16)
        yield FakeRequestOutput(list(ids) + [104], True, prompt_ids, reason="length")

    async def abort(self, request_id):
        self.aborted.append(request_id)


class FakeSglangScheduler:
    """SGLang scheduler-client shape: submit() + per-step batch outputs with
    CUMULATIVE output_ids and dict finish reasons."""

    def __init__(self):
        self.payloads = []

    async def submit(self, payload):
        self.payloads.append(payload)

    async def outputs(self):
        rid = self.payloads[-1]["rid"]
        yield {rid: {"output_ids": [7], "finished": False, "prompt_tokens": 5}}
        yield {"other": {"output_ids": [1], "finished": False}}  # other request's step
        yield {rid: {"output_ids": [7, 8, 9], "finished": True,
                     "finish_reason": {"type": "FINISH_LENGTH", "length": 3},
                     "prompt_tokens": 5, "cached_tokens": 4}}


def test_vllm_param_translation():
    sp = api.SamplingParams(max_new_tokens=32, temperature=0.0, top_p=0.0, top_k=0,
                            stop=["</s>"], ignore_eos=True)
    out = translate_sampling_params_vllm(sp)
    # proto zero-defaults map back to library semantic defaults (servicer.py:754)
    assert out["top_p"] == 1.0 and out["top_k"] == -1
    assert out["temperature"] == 0.0  # explicit zero temperature is greedy, kept
    assert out["max_tokens"] == 32 and out["stop"] == ["</s>"]
    assert out["detokenize"] is True  # stop strings force detokenize
    assert out["ignore_eos"] is True


def test_vllm_bridge_delta_streaming():
    eng = FakeAsyncLLM()
    bridge = VllmBridge(eng)
    req = api.GenerateRequest(request_id="r1", input_ids=[1, 2, 3, 4, 5],
                              sampling=api.SamplingParams(max_new_tokens=8))

    async def go():
        return [c async for c in bridge.generate(req)]

    chunks = run(go())
    # cumulative outputs became per-chunk deltas
    assert [c.token_ids for c in chunks] == [[101], [102], [103], [104]]
    assert chunks[-1].finished and chunks[-1].finish_reason == "length"
    assert chunks[-1].completion_tokens == 4
    assert chunks[0].prompt_tokens == 5 and chunks[0].cached_tokens == 16
    run(bridge.abort("r1"))
    assert eng.aborted == ["r1"]


def test_sglang_payload_and_stream():
    req = api.GenerateRequest(
        request_id="s1", input_ids=[9, 9, 9],
        sampling=api.SamplingParams(max_new_tokens=3, stop=["x"]),
        bootstrap_host="10.0.0.1", bootstrap_port=7001, bootstrap_room=42,
        dp_rank=2, lora_id="lora-a")
    payload = build_sglang_generate_payload(req)
    assert payload["rid"] == "s1" and payload["input_ids"] == [9, 9, 9]
    assert payload["sampling_params"]["max_new_tokens"] == 3
    assert payload["bootstrap_room"] == 42 and payload["data_parallel_rank"] == 2
    assert payload["lora_id"] == "lora-a"

    sched = FakeSglangScheduler()
    bridge = SglangBridge(sched)

    async def go():
        return [c async for c in bridge.generate(req)]

    chunks = run(go())
    assert [c.token_ids for c in chunks] == [[7], [8, 9]]
    assert chunks[-1].finished and chunks[-1].finish_reason == "length"
    assert chunks[-1].cached_tokens == 4


def test_finish_reason_normalization():
    assert normalize_finish_reason(None) is None
    assert normalize_finish_reason({"type": "length"}) == "length"
    assert normalize_finish_reason("FINISH_ABORT") == "abort"
    assert normalize_finish_reason({"type": "stop_token"}) == "stop"


def test_missing_runtime_raises_clearly():
    with pytest.raises(RuntimeError, match="vllm"):
        VllmBridge(None)
    with pytest.raises(RuntimeError, match="sglang"):
        SglangBridge(None)
    with pytest.raises(ValueError, match="unknown engine bridge"):
        make_bridge("mlx9000")
