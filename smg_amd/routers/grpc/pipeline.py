"""gRPC router pipeline (reference: model_gateway/src/routers/grpc/pipeline.rs:192
— stage lists :258: Preparation -> WorkerSelection -> ClientAcquisition ->
[Encode (EPD)] -> RequestBuilding -> DispatchMetadata -> RequestExecution ->
ResponseProcessing; endpoints pipeline.rs:64-71).

Each stage is an async callable over a shared PipelineContext; the response
processing stage runs the streaming tail: incremental detok ->
StopSequenceDecoder -> streaming reasoning parser -> streaming tool parser ->
SSE encode (reference regular/streaming.rs).
"""
from __future__ import annotations

import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ...grpc import api
from ...grpc.client import ClientPool
from ...parsers.reasoning import get_reasoning_parser
from ...parsers.tool import get_parser
from ...parsers.tool.stream import StreamingToolParser
from ...policies import SelectWorkerInfo
from ...protocols.openai import error_body
from ...tokenizer.chat_template import ChatTemplate
from ...tokenizer.stop import DecodeStream, StopSequenceDecoder
from ..base import RouteRequest, RouteResponse


@dataclass
class PipelineContext:
    req: RouteRequest
    endpoint: str  # chat | completion | generate | embedding
    body: Dict[str, Any] = field(default_factory=dict)
    model_id: Optional[str] = None
    text: str = ""
    input_ids: List[int] = field(default_factory=list)
    sampling: api.SamplingParams = field(default_factory=api.SamplingParams)
    worker: Any = None
    prefill_worker: Any = None  # PD/EPD: ctx.worker is the decode leg
    encode_worker: Any = None
    dp_rank: Optional[int] = None
    client: Any = None
    gen_request: Optional[api.GenerateRequest] = None
    request_id: str = ""
    stream: bool = False
    tokenizer: Any = None
    multimodal: Optional[Dict[str, Any]] = None
    error: Optional[RouteResponse] = None


class Stage:
    name = "stage"

    async def run(self, ctx: PipelineContext) -> bool:
        """Returns False to short-circuit (ctx.error set)."""
        raise NotImplementedError


class PreparationStage(Stage):
    """Tokenize + chat-template render + sampling extraction
    (reference regular/stages prep + utils/chat_utils.rs)."""

    name = "preparation"

    def __init__(self, app_ctx):
        self.app = app_ctx
        self._template: Optional[ChatTemplate] = None

    def template(self) -> ChatTemplate:
        if self._template is None:
            from ...tokenizer.chat_template import load_chat_template

            self._template = load_chat_template(self.app.config.chat_template)
        return self._template

    async def _process_images(self, ctx: PipelineContext, urls) -> bool:
        from ...multimodal.media import MediaError, decode_image, fetch_image_bytes
        from ...multimodal.processors import processor_for_model
        from ...multimodal.transport import encode_tensor

        proc = processor_for_model(ctx.model_id)
        mode = self.app.config.multimodal_tensor_transport
        images = []
        for url in urls:
            try:
                data = await fetch_image_bytes(url)
                arr = decode_image(data)
            except MediaError as e:
                ctx.error = RouteResponse(status=400, body=error_body(str(e)))
                return False
            out = proc.process(arr)
            desc = encode_tensor(
                out["pixel_values"], mode=mode, min_shm_bytes=self.app.config.multimodal_shm_min_bytes
            )
            desc["height"] = out["height"]
            desc["width"] = out["width"]
            if "grid_thw" in out:
                desc["grid_thw"] = list(out["grid_thw"])
            images.append(desc)
        ctx.multimodal = {"images": images}
        return True

    async def _process_audio(self, ctx: PipelineContext, parts) -> bool:
        """input_audio content parts -> whisper-style log-mel feature tensors
        (reference audio/processors/qwen3_audio.rs preprocess)."""
        import base64

        from ...multimodal.audio import AudioError, preprocess_audio
        from ...multimodal.transport import encode_tensor

        mode = self.app.config.multimodal_tensor_transport
        audios = []
        for part in parts:
            try:
                data = base64.b64decode(part.get("data") or "")
                out = preprocess_audio(data)
            except (AudioError, ValueError) as e:
                ctx.error = RouteResponse(status=400, body=error_body(f"audio decode failed: {e}"))
                return False
            desc = encode_tensor(
                out["features"], mode=mode, min_shm_bytes=self.app.config.multimodal_shm_min_bytes
            )
            desc["feature_length"] = out["feature_length"]
            desc["sample_rate"] = out["sample_rate"]
            audios.append(desc)
        if ctx.multimodal is None:
            ctx.multimodal = {}
        ctx.multimodal["audios"] = audios
        return True

    async def run(self, ctx: PipelineContext) -> bool:
        body = ctx.req.body or {}
        ctx.body = body
        ctx.model_id = self.app.worker_registry.resolve_model(
            ctx.req.model_override or (body.get("model") if isinstance(body.get("model"), str) else None)
        )
        ctx.stream = bool(body.get("stream", False))
        ctx.request_id = ctx.req.request_id or uuid.uuid4().hex

        # sampling params (OpenAI surface -> engine sampling)
        sp = api.SamplingParams()
        sp.max_new_tokens = int(
            body.get("max_completion_tokens") or body.get("max_tokens") or body.get("max_new_tokens") or 128
        )
        sp.temperature = float(body.get("temperature", 1.0))
        sp.top_p = float(body.get("top_p", 1.0))
        stop = body.get("stop")
        if isinstance(stop, str):
            sp.stop = [stop]
        elif isinstance(stop, list):
            sp.stop = [s for s in stop if isinstance(s, str)]
        ctx.sampling = sp

        # text assembly
        if ctx.endpoint == "chat":
            msgs = body.get("messages")
            if not isinstance(msgs, list) or not msgs:
                ctx.error = RouteResponse(status=400, body=error_body("'messages' must be a non-empty array"))
                return False
            ctx.text = self.template().render(msgs, tools=body.get("tools"))
            # multimodal: fetch + preprocess image parts on the gfx950 kernel
            # (reference grpc/multimodal/process.rs + assemble.rs)
            image_urls = [
                p.get("image_url", {}).get("url") if isinstance(p.get("image_url"), dict) else p.get("image_url")
                for m in msgs
                if isinstance(m.get("content"), list)
                for p in m["content"]
                if isinstance(p, dict) and p.get("type") == "image_url"
            ]
            if image_urls:
                ok = await self._process_images(ctx, [u for u in image_urls if u])
                if not ok:
                    return False
            # audio parts -> log-mel features (reference audio/ +
            # processors/qwen3_audio.rs; OpenAI input_audio content part)
            audio_parts = [
                p.get("input_audio")
                for m in msgs
                if isinstance(m.get("content"), list)
                for p in m["content"]
                if isinstance(p, dict) and p.get("type") == "input_audio" and isinstance(p.get("input_audio"), dict)
            ]
            if audio_parts:
                ok = await self._process_audio(ctx, audio_parts)
                if not ok:
                    return False
        elif ctx.endpoint == "completion":
            p = body.get("prompt")
            if isinstance(p, list) and p and all(isinstance(x, int) for x in p):
                ctx.input_ids = p
            else:
                ctx.text = p if isinstance(p, str) else "\n".join(p or [])
        elif ctx.endpoint == "generate":
            ids = body.get("input_ids")
            if isinstance(ids, list) and ids:
                ctx.input_ids = ids
            else:
                ctx.text = body.get("text") or ""
        elif ctx.endpoint == "embedding":
            i = body.get("input")
            ctx.text = i if isinstance(i, str) else "\n".join(x for x in (i or []) if isinstance(x, str))
        elif ctx.endpoint == "rerank":
            ctx.text = str(body.get("query") or "")
        elif ctx.endpoint == "classify":
            i = body.get("input") or body.get("text") or ""
            ctx.text = i if isinstance(i, str) else ""

        # tokenize
        if not ctx.input_ids:
            reg = self.app.tokenizer_registry
            tok = reg.get(ctx.model_id) if reg else None
            if tok is None:
                from ...tokenizer.registry import MockTokenizer

                tok = MockTokenizer()
            ctx.tokenizer = tok
            ctx.input_ids = tok.encode(ctx.text)
        else:
            reg = self.app.tokenizer_registry
            ctx.tokenizer = reg.get(ctx.model_id) if reg else None
            if ctx.tokenizer is None:
                from ...tokenizer.registry import MockTokenizer

                ctx.tokenizer = MockTokenizer()
        return True


class WorkerSelectionStage(Stage):
    """Policy select on token ids (reference common/stages/worker_selection.rs:41)."""

    name = "worker_selection"

    def __init__(self, app_ctx, metrics=None):
        self.app = app_ctx

    async def run(self, ctx: PipelineContext) -> bool:
        from ...config import RoutingMode
        from ...workers.worker import WorkerType

        info = SelectWorkerInfo(
            request_id=ctx.request_id,
            model_id=ctx.model_id,
            tokens=ctx.input_ids,
            routing_key=ctx.req.routing_key,
            tenant_id=ctx.req.tenant_id,
            est_tokens=len(ctx.input_ids),
        )
        mode = self.app.config.mode
        t0 = time.perf_counter_ns()
        if mode in (RoutingMode.PREFILL_DECODE, RoutingMode.ENCODE_PREFILL_DECODE):
            # per-leg policies (reference worker_selection.rs:48-56 modes)
            prefill_pool = self.app.worker_registry.for_model(ctx.model_id, worker_type=WorkerType.PREFILL)
            decode_pool = self.app.worker_registry.for_model(ctx.model_id, worker_type=WorkerType.DECODE)
            if not prefill_pool or not decode_pool:
                ctx.error = RouteResponse(status=503, body=error_body("no prefill/decode workers", 503))
                return False
            p_idx = self.app.policy_registry.get(ctx.model_id, "prefill").select_worker(prefill_pool, info)
            d_idx = self.app.policy_registry.get(ctx.model_id, "decode").select_worker(decode_pool, info)
            if p_idx is None or d_idx is None:
                ctx.error = RouteResponse(status=503, body=error_body("selection failed", 503))
                return False
            ctx.prefill_worker = prefill_pool[p_idx]
            ctx.worker = decode_pool[d_idx]
            if mode == RoutingMode.ENCODE_PREFILL_DECODE and ctx.multimodal:
                encode_pool = self.app.worker_registry.for_model(ctx.model_id, worker_type=WorkerType.ENCODE)
                if not encode_pool:
                    ctx.error = RouteResponse(status=503, body=error_body("no encode workers", 503))
                    return False
                e_idx = self.app.policy_registry.get(ctx.model_id, "encode").select_worker(encode_pool, info)
                ctx.encode_worker = encode_pool[e_idx] if e_idx is not None else encode_pool[0]
        else:
            workers = self.app.worker_registry.for_model(ctx.model_id)
            if not workers:
                ctx.error = RouteResponse(
                    status=503, body=error_body(f"no available worker for model {ctx.model_id or 'any'}", 503)
                )
                return False
            policy = self.app.policy_registry.get(ctx.model_id)
            idx = policy.select_worker(workers, info)
            if idx is None:
                ctx.error = RouteResponse(status=503, body=error_body("selection failed", 503))
                return False
            ctx.worker = workers[idx]
        self.app.metrics.observe_routing_latency((time.perf_counter_ns() - t0) / 1e9)
        # DP-aware rank routing (reference dp_min_token.rs via monitor.rs:164)
        if self.app.config.dp_aware and ctx.worker.dp_size:
            ctx.dp_rank = self.app.policy_registry.dp_policy.select_dp_rank(ctx.worker)
        return True


class ClientAcquisitionStage(Stage):
    name = "client_acquisition"

    def __init__(self, pool: ClientPool):
        self.pool = pool

    async def run(self, ctx: PipelineContext) -> bool:
        ctx.client = self.pool.get(ctx.worker.url)
        return True


class EncodeStage(Stage):
    """EPD only: dispatch the vision encode to the encode fleet and replace
    raw pixels with embeddings (reference common/stages/encode.rs; pixel
    transport inline/SHM/RDMA -> here inline/shm/xgmi descriptors)."""

    name = "encode"

    def __init__(self, pool: ClientPool):
        self.pool = pool

    async def run(self, ctx: PipelineContext) -> bool:
        if ctx.encode_worker is None or not ctx.multimodal:
            return True
        client = self.pool.get(ctx.encode_worker.url)
        ctx.encode_worker.incr_load()
        try:
            result = await client.encode_image(ctx.request_id, ctx.multimodal)
            ctx.encode_worker.record_outcome(True)
        except Exception as exc:
            ctx.encode_worker.record_outcome(False)
            ctx.error = RouteResponse(status=502, body=error_body(f"encode failed: {exc}", 502))
            return False
        finally:
            ctx.encode_worker.decr_load()
        ctx.multimodal = {"embeddings": result.get("embeddings", [])}
        return True


class RequestBuildingStage(Stage):
    name = "request_building"

    async def run(self, ctx: PipelineContext) -> bool:
        ctx.gen_request = api.GenerateRequest(
            request_id=ctx.request_id,
            input_ids=ctx.input_ids,
            sampling=ctx.sampling,
            stream=True,
            multimodal=ctx.multimodal,
            dp_rank=ctx.dp_rank,
        )
        if ctx.prefill_worker is not None:
            # bootstrap metadata for the engine-side KV handoff
            # (reference request_execution.rs:253 + pd_types.rs:15)
            import random as _random
            from urllib.parse import urlparse

            ctx.gen_request.bootstrap_host = (
                ctx.prefill_worker.bootstrap_host
                or urlparse("http://" + ctx.prefill_worker.url.split("://", 1)[-1]).hostname
            )
            ctx.gen_request.bootstrap_port = ctx.prefill_worker.bootstrap_port
            ctx.gen_request.bootstrap_room = _random.getrandbits(63)
        return True


class DispatchMetadataStage(Stage):
    """Request-id stamping incl. the `_dp{rank}` suffix
    (reference request_execution.rs:110)."""

    name = "dispatch_metadata"

    async def run(self, ctx: PipelineContext) -> bool:
        if ctx.dp_rank is not None:
            ctx.gen_request.request_id = f"{ctx.request_id}_dp{ctx.dp_rank}"
        return True


class RequestExecutionStage(Stage):
    """Single dispatch (reference request_execution.rs:410; PD dual :253)."""

    name = "request_execution"

    async def run(self, ctx: PipelineContext) -> bool:
        ctx.worker.incr_load(len(ctx.input_ids))
        return True


class ResponseProcessingStage(Stage):
    """The per-token streaming tail (reference regular/streaming.rs)."""

    name = "response_processing"

    def __init__(self, app_ctx):
        self.app = app_ctx

    async def run(self, ctx: PipelineContext) -> bool:
        return True

    # ---- helpers shared by GrpcRouter -------------------------------------
    def make_processors(self, ctx: PipelineContext):
        stop = StopSequenceDecoder(ctx.sampling.stop, ctx.sampling.stop_token_ids)
        detok = DecodeStream(ctx.tokenizer)
        rp_name = self.app.config.reasoning_parser
        reasoning = get_reasoning_parser(rp_name) if rp_name else None
        tp_name = self.app.config.tool_call_parser
        tool_stream = StreamingToolParser(get_parser(tp_name)) if tp_name and (ctx.body or {}).get("tools") else None
        return detok, stop, reasoning, tool_stream
