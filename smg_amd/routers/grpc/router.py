"""gRPC-mode router (reference: model_gateway/src/routers/grpc/ — the deep
pipeline).  Drives the stage pipeline, executes the engine
Generate stream, and runs the per-token tail: incremental detok -> stop scan
-> reasoning parser -> tool parser -> OpenAI SSE chunks.
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import AsyncIterator, Dict, List, Optional

from ...grpc import api
from ...grpc.client import ClientPool
from ...protocols.openai import error_body
from ..base import RouteRequest, RouteResponse, Router
from .pipeline import (
    ClientAcquisitionStage,
    DispatchMetadataStage,
    EncodeStage,
    PipelineContext,
    PreparationStage,
    RequestBuildingStage,
    RequestExecutionStage,
    ResponseProcessingStage,
    WorkerSelectionStage,
)

log = logging.getLogger("smg.router.grpc")

ENDPOINT_BY_PATH = {
    "/v1/chat/completions": "chat",
    "/v1/completions": "completion",
    "/generate": "generate",
    "/v1/embeddings": "embedding",
    "/v1/rerank": "rerank",
    "/rerank": "rerank",
    "/v1/classify": "classify",
}


class GrpcRouter(Router):
    router_id = "grpc-regular"

    def __init__(self, app_ctx, config=None):
        self.app = app_ctx
        self.config = config or app_ctx.config
        self.pool = ClientPool()
        self.prep = PreparationStage(app_ctx)
        self.select = WorkerSelectionStage(app_ctx)
        self.acquire = ClientAcquisitionStage(self.pool)
        self.encode = EncodeStage(self.pool)
        self.build = RequestBuildingStage()
        self.meta = DispatchMetadataStage()
        self.execute = RequestExecutionStage()
        self.process = ResponseProcessingStage(app_ctx)
        self.stages = [self.prep, self.select, self.acquire, self.encode, self.build, self.meta, self.execute]

    async def shutdown(self) -> None:
        await self.pool.close()

    async def route(self, req: RouteRequest) -> RouteResponse:
        endpoint = ENDPOINT_BY_PATH.get(req.path)
        if endpoint is None:
            return RouteResponse(status=404, body=error_body(f"unsupported gRPC-mode path {req.path}", 404))
        ctx = PipelineContext(req=req, endpoint=endpoint)
        for stage in self.stages:
            ok = await stage.run(ctx)
            if not ok:
                return ctx.error
        if endpoint == "embedding":
            return await self._embed(ctx)
        if endpoint == "rerank":
            return await self._rerank(ctx)
        if endpoint == "classify":
            return await self._classify(ctx)
        if ctx.stream:
            return RouteResponse(
                status=200,
                headers={"content-type": "text/event-stream"},
                stream=self._stream_response(ctx),
            )
        return await self._unary_response(ctx)

    # ---- embeddings -------------------------------------------------------
    async def _embed(self, ctx: PipelineContext) -> RouteResponse:
        try:
            d = await ctx.client.embed(api.EmbedRequest(ctx.request_id, ctx.input_ids))
            ctx.worker.decr_load(len(ctx.input_ids))
            ctx.worker.record_outcome(True)
        except Exception as exc:
            ctx.worker.decr_load(len(ctx.input_ids))
            ctx.worker.record_outcome(False)
            return RouteResponse(status=502, body=error_body(f"engine error: {exc}", 502))
        body = {
            "object": "list",
            "data": [{"object": "embedding", "index": 0, "embedding": d.get("embedding", [])}],
            "model": ctx.model_id,
            "usage": {"prompt_tokens": len(ctx.input_ids), "total_tokens": len(ctx.input_ids)},
        }
        return RouteResponse(status=200, body=json.dumps(body).encode())

    async def _rerank(self, ctx: PipelineContext) -> RouteResponse:
        body = ctx.body or {}
        try:
            d = await ctx.client.rerank(body.get("query", ""), body.get("documents") or [])
            ctx.worker.record_outcome(True)
        except Exception as exc:
            ctx.worker.record_outcome(False)
            return RouteResponse(status=502, body=error_body(f"engine error: {exc}", 502))
        finally:
            ctx.worker.decr_load(len(ctx.input_ids))
        return RouteResponse(status=200, body=json.dumps(
            {"results": d.get("results", []), "model": ctx.model_id}).encode())

    async def _classify(self, ctx: PipelineContext) -> RouteResponse:
        body = ctx.body or {}
        try:
            d = await ctx.client.classify(body.get("input") or body.get("text") or "")
            ctx.worker.record_outcome(True)
        except Exception as exc:
            ctx.worker.record_outcome(False)
            return RouteResponse(status=502, body=error_body(f"engine error: {exc}", 502))
        finally:
            ctx.worker.decr_load(len(ctx.input_ids))
        return RouteResponse(status=200, body=json.dumps(
            {"object": "classification", "model": ctx.model_id, "data": d.get("data", [])}).encode())

    def _fire_prefill(self, ctx: PipelineContext):
        """PD dual dispatch: run the prefill leg concurrently and drain its
        stream (reference request_execution.rs:253); KV moves engine-side via
        the bootstrap metadata."""
        if ctx.prefill_worker is None:
            return None
        pclient = self.pool.get(ctx.prefill_worker.url)
        ctx.prefill_worker.incr_load(len(ctx.input_ids))

        async def drain():
            ok = True
            try:
                async for chunk in pclient.generate(ctx.gen_request):
                    if chunk.finished:
                        break
            except Exception as exc:
                ok = False
                log.debug("prefill leg failed on %s: %s", ctx.prefill_worker.url, exc)
            finally:
                ctx.prefill_worker.decr_load(len(ctx.input_ids))
                ctx.prefill_worker.record_outcome(ok)

        return asyncio.ensure_future(drain())

    # ---- unary ------------------------------------------------------------
    async def _unary_response(self, ctx: PipelineContext) -> RouteResponse:
        detok, stop, reasoning, tool_stream = self.process.make_processors(ctx)
        prefill_task = self._fire_prefill(ctx)
        text_parts: List[str] = []
        finish_reason = "stop"
        usage = {"prompt_tokens": len(ctx.input_ids), "completion_tokens": 0}
        ok = True
        try:
            async for chunk in ctx.client.generate(ctx.gen_request):
                for tid in chunk.token_ids:
                    usage["completion_tokens"] += 1
                    piece = detok.push(tid)
                    emitted, outcome = stop.process_token(tid, piece)
                    if emitted:
                        text_parts.append(emitted)
                    if stop.stopped:
                        await ctx.client.abort(ctx.gen_request.request_id)
                        finish_reason = "stop"
                        break
                if stop.stopped:
                    break
                if chunk.finished:
                    finish_reason = chunk.finish_reason or "stop"
                    break
            if not stop.stopped:
                text_parts.append(stop.flush())
        except Exception as exc:
            ok = False
            log.warning("generate failed on %s: %s", ctx.worker.url, exc)
            return RouteResponse(status=502, body=error_body(f"engine error: {exc}", 502))
        finally:
            ctx.worker.decr_load(len(ctx.input_ids))
            ctx.worker.record_outcome(ok)
            if ctx.dp_rank is not None and ctx.worker.dp_loads:
                ctx.worker.dp_loads[ctx.dp_rank] = max(0, ctx.worker.dp_loads[ctx.dp_rank] - 1)

        text = "".join(text_parts)
        reasoning_text = None
        if reasoning is not None:
            reasoning_text, text = reasoning.parse(text)
        tool_calls = None
        if (ctx.body or {}).get("tools") and self.config.tool_call_parser:
            from ...parsers.tool import parse_complete

            text, calls = parse_complete(self.config.tool_call_parser, text)
            if calls:
                tool_calls = [
                    {
                        "id": f"call_{ctx.request_id[:8]}_{i}",
                        "type": "function",
                        "function": {"name": c["name"], "arguments": c["arguments"]},
                    }
                    for i, c in enumerate(calls)
                ]
        usage["total_tokens"] = usage["prompt_tokens"] + usage["completion_tokens"]
        body = self._final_body(ctx, text, reasoning_text, tool_calls, finish_reason, usage)
        return RouteResponse(status=200, body=json.dumps(body).encode())

    def _final_body(self, ctx, text, reasoning_text, tool_calls, finish_reason, usage):
        created = int(time.time())
        rid = f"chatcmpl-{ctx.request_id[:24]}"
        if ctx.endpoint == "chat":
            msg: Dict = {"role": "assistant", "content": text}
            if reasoning_text:
                msg["reasoning_content"] = reasoning_text
            if tool_calls:
                msg["tool_calls"] = tool_calls
                msg["content"] = text or None
                finish_reason = "tool_calls"
            return {
                "id": rid,
                "object": "chat.completion",
                "created": created,
                "model": ctx.model_id,
                "choices": [{"index": 0, "message": msg, "finish_reason": finish_reason}],
                "usage": usage,
            }
        if ctx.endpoint == "completion":
            return {
                "id": rid,
                "object": "text_completion",
                "created": created,
                "model": ctx.model_id,
                "choices": [{"index": 0, "text": text, "finish_reason": finish_reason}],
                "usage": usage,
            }
        return {"text": text, "meta_info": {"id": ctx.request_id, "finish_reason": finish_reason, "usage": usage}}

    # ---- streaming ---------------------------------------------------------
    async def _stream_response(self, ctx: PipelineContext) -> AsyncIterator[bytes]:
        detok, stop, reasoning, tool_stream = self.process.make_processors(ctx)
        prefill_task = self._fire_prefill(ctx)
        created = int(time.time())
        rid = f"chatcmpl-{ctx.request_id[:24]}"
        first = True
        ok = True
        completion_tokens = 0
        ttft_t0 = time.perf_counter()

        def sse(payload: dict) -> bytes:
            return b"data: " + json.dumps(payload).encode() + b"\n\n"

        def chat_chunk(delta: dict, finish: Optional[str] = None) -> dict:
            return {
                "id": rid,
                "object": "chat.completion.chunk",
                "created": created,
                "model": ctx.model_id,
                "choices": [{"index": 0, "delta": delta, "finish_reason": finish}],
            }

        def emit_text(piece: str) -> List[bytes]:
            nonlocal first
            out = []
            if not piece:
                return out
            deltas: List[Dict] = []
            if reasoning is not None:
                r, n = reasoning.parse_streaming(piece)
                if r:
                    deltas.append({"reasoning_content": r})
                piece = n
            if tool_stream is not None and piece:
                for ev in tool_stream.feed(piece):
                    if ev[0] == "text":
                        deltas.append({"content": ev[1]})
                    elif ev[0] == "tool_name":
                        deltas.append(
                            {
                                "tool_calls": [
                                    {
                                        "index": ev[1],
                                        "id": f"call_{ctx.request_id[:8]}_{ev[1]}",
                                        "type": "function",
                                        "function": {"name": ev[2], "arguments": ""},
                                    }
                                ]
                            }
                        )
                    elif ev[0] == "tool_args":
                        deltas.append(
                            {"tool_calls": [{"index": ev[1], "function": {"arguments": ev[2]}}]}
                        )
            elif piece:
                deltas.append({"content": piece})
            for d in deltas:
                if first and ctx.endpoint == "chat":
                    d = {"role": "assistant", **d}
                    first = False
                if ctx.endpoint == "chat":
                    out.append(sse(chat_chunk(d)))
                elif ctx.endpoint == "completion":
                    out.append(
                        sse({"id": rid, "object": "text_completion", "model": ctx.model_id,
                             "choices": [{"index": 0, "text": d.get("content", ""), "finish_reason": None}]})
                    )
                else:
                    out.append(sse({"text": d.get("content", ""), "meta_info": {"id": ctx.request_id}}))
            return out

        finish_reason = "stop"
        try:
            async for chunk in ctx.client.generate(ctx.gen_request):
                if completion_tokens == 0 and chunk.token_ids:
                    self.app.metrics.observe_ttft(time.perf_counter() - ttft_t0)
                for tid in chunk.token_ids:
                    completion_tokens += 1
                    piece = detok.push(tid)
                    emitted, outcome = stop.process_token(tid, piece)
                    for b in emit_text(emitted):
                        yield b
                    if stop.stopped:
                        await ctx.client.abort(ctx.gen_request.request_id)
                        break
                if stop.stopped:
                    break
                if chunk.finished:
                    finish_reason = chunk.finish_reason or "stop"
                    break
            if not stop.stopped:
                for b in emit_text(stop.flush()):
                    yield b
            if tool_stream is not None:
                fin, _, calls = tool_stream.finish()
                for ev in fin:
                    if ev[0] == "tool_args":
                        yield sse(chat_chunk({"tool_calls": [{"index": ev[1], "function": {"arguments": ev[2]}}]}))
                    elif ev[0] == "tool_name":
                        yield sse(chat_chunk({"tool_calls": [{"index": ev[1], "id": f"call_{ctx.request_id[:8]}_{ev[1]}", "type": "function", "function": {"name": ev[2], "arguments": ""}}]}))
                if calls:
                    finish_reason = "tool_calls"
            if ctx.endpoint == "chat":
                yield sse(chat_chunk({}, finish_reason))
                if (ctx.body or {}).get("stream_options", {}).get("include_usage"):
                    yield sse({
                        "id": rid, "object": "chat.completion.chunk", "created": created,
                        "model": ctx.model_id, "choices": [],
                        "usage": {"prompt_tokens": len(ctx.input_ids), "completion_tokens": completion_tokens,
                                  "total_tokens": len(ctx.input_ids) + completion_tokens},
                    })
            yield b"data: [DONE]\n\n"
        except Exception as exc:
            ok = False
            log.warning("stream failed on %s: %s", ctx.worker.url, exc)
            yield sse({"error": {"message": str(exc), "type": "upstream_error"}})
        finally:
            self.app.metrics.generate_tokens.inc(completion_tokens) if not self.app.metrics._null else None
            ctx.worker.decr_load(len(ctx.input_ids))
            ctx.worker.record_outcome(ok)
            if ctx.dp_rank is not None and ctx.worker.dp_loads:
                ctx.worker.dp_loads[ctx.dp_rank] = max(0, ctx.worker.dp_loads[ctx.dp_rank] - 1)

    async def get_loads(self) -> Dict:
        out = {}
        for w in self.app.worker_registry.all():
            if w.url.startswith("grpc"):
                try:
                    out[w.url] = await self.pool.get(w.url).get_loads()
                except Exception as exc:
                    out[w.url] = {"error": str(exc)}
        return {"workers": out}

    async def flush_cache(self) -> RouteResponse:
        for w in self.app.worker_registry.all():
            if w.url.startswith("grpc"):
                try:
                    await self.pool.get(w.url).flush_cache()
                except Exception:
                    pass
        for p in self.app.policy_registry.all_policies():
            p.reset()
        return RouteResponse(status=200, body=b'{"status":"cache flushed"}')
