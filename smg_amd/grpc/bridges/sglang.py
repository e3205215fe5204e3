"""SGLang bridge (reference grpc_servicer/smg_grpc_servicer/sglang/
servicer.py + request_manager.py — talks to the SGLang scheduler with
tokenized requests and folds per-step batch outputs back into per-request
streams)."""
from __future__ import annotations

from typing import Any, AsyncIterator, Dict, List, Optional

from .. import api


def build_sglang_generate_payload(req: api.GenerateRequest) -> Dict[str, Any]:
    """Our GenerateRequest -> the SGLang tokenized generate payload
    (scheduler-side TokenizedGenerateReqInput shape: rid, input_ids,
    sampling_params dict, stream)."""
    sp = req.sampling
    return {
        "rid": req.request_id,
        "input_ids": list(req.input_ids),
        "sampling_params": {
            "max_new_tokens": sp.max_new_tokens,
            "temperature": sp.temperature,
            "top_p": sp.top_p,
            "top_k": sp.top_k,
            "stop": list(sp.stop),
            "stop_token_ids": list(sp.stop_token_ids),
            "ignore_eos": sp.ignore_eos,
            "skip_special_tokens": sp.skip_special_tokens,
        },
        "stream": True,
        **({"lora_id": req.lora_id} if req.lora_id else {}),
        **({"data_parallel_rank": req.dp_rank} if req.dp_rank is not None else {}),
        **({
            "bootstrap_host": req.bootstrap_host,
            "bootstrap_port": req.bootstrap_port,
            "bootstrap_room": req.bootstrap_room,
        } if req.bootstrap_host or req.bootstrap_room is not None else {}),
    }


def normalize_finish_reason(reason) -> Optional[str]:
    """SGLang finish reasons arrive as dicts ({'type': 'length', ...}) or
    FINISH_* objects; normalize to the OpenAI strings (reference
    request_manager.py)."""
    if reason is None:
        return None
    if isinstance(reason, dict):
        t = reason.get("type")
    else:
        t = str(reason)
    t = (t or "").lower()
    if "length" in t:
        return "length"
    if "abort" in t:
        return "abort"
    return "stop"


class SglangBridge:
    """Duck-typed over an SGLang scheduler client: needs .submit(payload) and
    an output stream of per-step BATCH dicts {rid: {output_ids cumulative,
    finished, finish_reason, prompt_tokens, cached_tokens}} (the
    BatchTokenIDOut fan-in the reference request_manager implements)."""

    def __init__(self, scheduler=None, model_id: str = "sglang-model"):
        if scheduler is None:
            try:
                import sglang  # noqa: F401
            except ImportError as e:
                raise RuntimeError(
                    "SglangBridge requires the sglang package (not installed in "
                    "this image) or a scheduler client object") from e
            raise RuntimeError("pass the constructed scheduler client explicitly")
        self.scheduler = scheduler
        self.model_id = model_id

    async def generate(self, req: api.GenerateRequest) -> AsyncIterator[api.GenerateChunk]:
        payload = build_sglang_generate_payload(req)
        await self.scheduler.submit(payload)
        sent = 0
        async for batch in self.scheduler.outputs():
            entry = batch.get(req.request_id)
            if entry is None:
                continue
            ids = list(entry.get("output_ids") or [])
            delta = ids[sent:]
            sent = len(ids)
            finished = bool(entry.get("finished"))
            if not delta and not finished:
                continue
            yield api.GenerateChunk(
                request_id=req.request_id,
                token_ids=delta,
                finished=finished,
                finish_reason=normalize_finish_reason(entry.get("finish_reason")) if finished else None,
                prompt_tokens=int(entry.get("prompt_tokens") or len(req.input_ids)),
                completion_tokens=sent,
                cached_tokens=int(entry.get("cached_tokens") or 0),
            )
            if finished:
                return

    async def abort(self, request_id: str) -> None:
        abort = getattr(self.scheduler, "abort", None)
        if abort is not None:
            await abort(request_id)
