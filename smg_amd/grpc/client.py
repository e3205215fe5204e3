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


class ClientPool:
    """Per-URL client cache (reference channel.rs pooling)."""

    def __init__(self):
        self._clients: Dict[str, EngineClient] = {}

    def get(self, url: str) -> EngineClient:
        c = self._clients.get(url)
        if c is None:
            c = EngineClient(url)
            self._clients[url] = c
        return c

    async def close(self):
        await asyncio.gather(*(c.close() for c in self._clients.values()), return_exceptions=True)
        self._clients.clear()
