"""vLLM bridge (reference grpc_servicer/smg_grpc_servicer/vllm/servicer.py —
wraps AsyncLLM, translates sampling params :712-784, streams cumulative
RequestOutput into incremental chunks :878-1035)."""
from __future__ import annotations

from typing import Any, AsyncIterator, Dict, Optional

from .. import api


def translate_sampling_params_vllm(sp: api.SamplingParams, *, has_stop_strings: bool = False) -> Dict[str, Any]:
    """Our SamplingParams -> vllm.SamplingParams kwargs (reference
    servicer.py:754-784 semantics: proto zero-defaults mapped back to the
    library's semantic defaults; detokenize forced on when stop strings are
    used)."""
    out: Dict[str, Any] = {
        "temperature": sp.temperature if sp.temperature is not None else 1.0,
        "top_p": sp.top_p if sp.top_p else 1.0,
        "top_k": sp.top_k if sp.top_k else -1,
        "max_tokens": sp.max_new_tokens,
        "stop": list(sp.stop) or None,
        "stop_token_ids": list(sp.stop_token_ids) or None,
        "skip_special_tokens": sp.skip_special_tokens,
        "ignore_eos": sp.ignore_eos,
        "n": 1,
    }
    if has_stop_strings or sp.stop:
        out["detokenize"] = True
    return {k: v for k, v in out.items() if v is not None}


class VllmBridge:
    """Duck-typed over vLLM's AsyncLLM: needs .generate(prompt, sampling_params,
    request_id) -> async iterator of RequestOutput-shaped objects
    (outputs[0].token_ids CUMULATIVE, .finished, outputs[0].finish_reason)."""

    def __init__(self, async_llm=None, model_id: str = "vllm-model"):
        if async_llm is None:
            try:
                import vllm  # noqa: F401
            except ImportError as e:
                raise RuntimeError(
                    "VllmBridge requires the vllm package (not installed in this "
                    "image) or an engine object implementing AsyncLLM.generate") from e
            raise RuntimeError("pass the constructed AsyncLLM engine explicitly")
        self.engine = async_llm
        self.model_id = model_id

    async def generate(self, req: api.GenerateRequest) -> AsyncIterator[api.GenerateChunk]:
        """Cumulative RequestOutput stream -> incremental GenerateChunk deltas
        (reference _chunk_response/_complete_response: ship only the tokens
        appended since the previous output)."""
        try:
            from vllm import SamplingParams as VSP  # type: ignore

            params = VSP(**translate_sampling_params_vllm(req.sampling))
        except ImportError:
            params = translate_sampling_params_vllm(req.sampling)  # fakes take the dict
        prompt = {"prompt_token_ids": list(req.input_ids)} if req.input_ids else req.text
        sent = 0
        async for out in self.engine.generate(prompt, params, req.request_id):
            seq = out.outputs[0]
            ids = list(seq.token_ids)
            delta = ids[sent:]
            sent = len(ids)
            finished = bool(getattr(out, "finished", False))
            if not delta and not finished:
                continue
            yield api.GenerateChunk(
                request_id=req.request_id,
                token_ids=delta,
                finished=finished,
                finish_reason=(getattr(seq, "finish_reason", None) or "stop") if finished else None,
                prompt_tokens=len(getattr(out, "prompt_token_ids", None) or req.input_ids),
                completion_tokens=sent,
                cached_tokens=int(getattr(out, "num_cached_tokens", 0) or 0),
            )
            if finished:
                return

    async def abort(self, request_id: str) -> None:
        abort = getattr(self.engine, "abort", None)
        if abort is not None:
            await abort(request_id)

    def load_snapshot(self) -> Dict[str, Any]:
        """Scheduler-stats shape -> our GetLoads fields (reference
        _latest_scheduler_stats :92-126)."""
        stats = getattr(self.engine, "scheduler_stats", None)
        if stats is None:
            return {}
        return {
            "num_running_reqs": int(getattr(stats, "num_running_reqs", 0)),
            "num_queue_reqs": int(getattr(stats, "num_waiting_reqs", 0)),
            "token_usage": float(getattr(stats, "kv_cache_usage", 0.0)),
            "gen_throughput": None,
        }
