"""Gateway-side gRPC engine client (reference: crates/grpc_client — tonic
clients, channel pooling channel.rs, abort-on-drop streams abort_on_drop.rs).

grpc.aio channels with per-URL pooling; Generate returns an async iterator of
GenerateChunk; cancellation of the iterator aborts the RPC (abort-on-drop).
"""
from __future__ import annotations

import asyncio
import logging
from typing import AsyncIterator, Dict, Optional

import grpc
import grpc.aio

from . import api

log = logging.getLogger("smg.grpc.client")

_BYTES = lambda b: b  # identity serializers: we speak msgpack bytes


class EngineClient:
    def __init__(self, target: str):
        # accept grpc://host:port or host:port
        self.target = target.split("://", 1)[-1]
        self._channel: Optional[grpc.aio.Channel] = None

    def channel(self) -> grpc.aio.Channel:
        if self._channel is None:
            self._channel = grpc.aio.insecure_channel(
                self.target,
                options=[
                    ("grpc.max_receive_message_length", 256 << 20),
                    ("grpc.max_send_message_length", 256 << 20),
                    ("grpc.keepalive_time_ms", 30_000),
                ],
            )
        return self._channel

    async def close(self):
        if self._channel is not None:
            await self._channel.close()
            self._channel = None

    # ---- RPCs -------------------------------------------------------------
    async def generate(self, req: api.GenerateRequest) -> AsyncIterator[api.GenerateChunk]:
        call = self.channel().unary_stream(
            api.method("Generate"), request_serializer=_BYTES, response_deserializer=_BYTES
        )(api.dumps(req))
        try:
            async for raw in call:
                yield api.GenerateChunk.from_dict(api.loads(raw))
        finally:
            call.cancel()  # abort-on-drop

    async def _unary(self, name: str, payload: dict, timeout: Optional[float] = 10.0) -> dict:
        call = self.channel().unary_unary(
            api.method(name), request_serializer=_BYTES, response_deserializer=_BYTES
        )
        raw = await call(api.dumps(payload), timeout=timeout)
        return api.loads(raw)

    async def health_check(self) -> bool:
        try:
            d = await self._unary("HealthCheck", {}, timeout=5.0)
            return bool(d.get("healthy"))
        except Exception:
            return False

    async def abort(self, request_id: str) -> None:
        try:
            await self._unary("Abort", {"request_id": request_id})
        except Exception:
            pass

    async def get_loads(self) -> dict:
        return await self._unary("GetLoads", {})

    async def get_model_info(self) -> dict:
        return await self._unary("GetModelInfo", {})

    async def flush_cache(self) -> dict:
        return await self._unary("FlushCache", {})

    async def embed(self, req: api.EmbedRequest) -> dict:
        return await self._unary("Embed", req.to_dict())

    async def encode_image(self, request_id: str, multimodal: dict) -> dict:
        return await self._unary("EncodeImage", {"request_id": request_id, "multimodal": multimodal}, timeout=60.0)

    async def rerank(self, query: str, documents) -> dict:
        return await self._unary("Rerank", {"query": query, "documents": list(documents)}, timeout=30.0)

    async def load_lora_adapter(self, lora_name: str, lora_path: str, lora_id: str, pinned: bool = False) -> dict:
        return await self._unary(
            "LoadLoraAdapter",
            {"lora_name": lora_name, "lora_path": lora_path, "lora_id": lora_id, "pinned": pinned},
        )

    async def unload_lora_adapter(self, lora_name: str, lora_id: str) -> dict:
        return await self._unary("UnloadLoraAdapter", {"lora_name": lora_name, "lora_id": lora_id})

    async def list_lora_adapters(self) -> dict:
        return await self._unary("ListLoraAdapters", {})

    async def classify(self, text: str) -> dict:
        return await self._unary("Classify", {"input": text}, timeout=30.0)

    async def subscribe_kv_events(self) -> AsyncIterator[dict]:
        call = self.channel().unary_stream(
            api.method("SubscribeKvEvents"), request_serializer=_BYTES, response_deserializer=_BYTES
        )(api.dumps({}))
        try:
            async for raw in call:
                yield api.loads(raw)
        finally:
            call.cancel()


class ProtoEngineClient(EngineClient):
    """Reference-wire client: speaks `sglang.grpc.scheduler.SglangScheduler`
    with the protobuf messages of grpc/proto_wire.py, so this gateway can
    front engines that implement the reference's engine protocol (tonic or
    grpcio peers).  Selected per worker with url scheme grpc+proto://."""

    def _stub(self, name: str):
        from . import proto_wire as pw

        req_cls, resp_cls, streaming = pw.METHODS[name]
        factory = self.channel().unary_stream if streaming else self.channel().unary_unary
        return factory(
            pw.method_path(name),
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString,
        )

    async def generate(self, req: api.GenerateRequest) -> AsyncIterator[api.GenerateChunk]:
        from . import proto_wire as pw

        m = pw.GenerateRequest()
        m.request_id = req.request_id
        if req.text:
            m.tokenized.original_text = req.text
        m.tokenized.input_ids.extend(req.input_ids)
        sp = m.sampling_params
        s = req.sampling
        sp.temperature = s.temperature
        sp.top_p = s.top_p
        sp.top_k = s.top_k
        sp.max_new_tokens = s.max_new_tokens
        sp.stop.extend(s.stop)
        sp.stop_token_ids.extend(s.stop_token_ids)
        sp.ignore_eos = s.ignore_eos
        sp.skip_special_tokens = s.skip_special_tokens
        m.stream = True
        if req.lora_id:
            m.lora_id = req.lora_id
        if req.dp_rank is not None:
            m.data_parallel_rank = req.dp_rank
        if req.bootstrap_host or req.bootstrap_room is not None:
            m.disaggregated_params.bootstrap_host = req.bootstrap_host or ""
            m.disaggregated_params.bootstrap_port = req.bootstrap_port or 0
            m.disaggregated_params.bootstrap_room = req.bootstrap_room or 0
        call = self._stub("Generate")(m)
        sent = 0
        try:
            async for resp in call:
                which = resp.WhichOneof("response")
                if which == "chunk":
                    c = resp.chunk
                    sent += len(c.token_ids)
                    yield api.GenerateChunk(
                        request_id=resp.request_id, token_ids=list(c.token_ids),
                        prompt_tokens=c.prompt_tokens, completion_tokens=c.completion_tokens,
                        cached_tokens=c.cached_tokens)
                elif which == "complete":
                    c = resp.complete
                    yield api.GenerateChunk(
                        request_id=resp.request_id,
                        token_ids=list(c.output_ids)[sent:],
                        finished=True, finish_reason=c.finish_reason or "stop",
                        prompt_tokens=c.prompt_tokens, completion_tokens=c.completion_tokens,
                        cached_tokens=c.cached_tokens)
                    break
        finally:
            call.cancel()  # abort-on-drop

    async def health_check(self) -> bool:
        from . import proto_wire as pw

        try:
            r = await self._stub("HealthCheck")(pw.HealthCheckRequest(), timeout=5.0)
            return bool(r.healthy)
        except Exception:
            return False

    async def abort(self, request_id: str) -> None:
        from . import proto_wire as pw

        try:
            m = pw.AbortRequest()
            m.request_id = request_id
            await self._stub("Abort")(m, timeout=10.0)
        except Exception:
            pass

    async def get_loads(self) -> dict:
        from . import proto_wire as pw

        r = await self._stub("GetLoads")(pw.GetLoadsRequest(), timeout=10.0)
        loads = r.loads[0] if r.loads else None
        return {
            "loads": {
                "num_running_reqs": loads.num_running_reqs if loads else 0,
                "num_queue_reqs": loads.num_waiting_reqs if loads else 0,
                "num_inflight_tokens": loads.num_used_tokens if loads else 0,
                "num_queue_tokens": loads.num_waiting_uncached_tokens if loads else 0,
                "token_usage": loads.token_usage if loads else 0.0,
                "gen_throughput": loads.gen_throughput if loads else 0.0,
            }
        }

    async def get_model_info(self) -> dict:
        from . import proto_wire as pw

        r = await self._stub("GetModelInfo")(pw.GetModelInfoRequest(), timeout=10.0)
        return {"model_path": r.model_path, "is_generation": r.is_generation,
                "max_context_length": r.max_context_length}

    async def flush_cache(self) -> dict:
        from . import proto_wire as pw

        r = await self._stub("FlushCache")(pw.FlushCacheRequest(), timeout=10.0)
        return {"status": "ok" if r.success else "failed"}


class ClientPool:
    """Per-URL client cache (reference channel.rs pooling).  URLs with the
    grpc+proto:// scheme get the reference-wire ProtoEngineClient."""

    def __init__(self):
        self._clients: Dict[str, EngineClient] = {}

    def get(self, url: str) -> EngineClient:
        c = self._clients.get(url)
        if c is None:
            c = ProtoEngineClient(url) if url.startswith("grpc+proto") else EngineClient(url)
            self._clients[url] = c
        return c

    async def close(self):
        await asyncio.gather(*(c.close() for c in self._clients.values()), return_exceptions=True)
        self._clients.clear()
