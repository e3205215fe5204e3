"""Engine-side gRPC servicer (reference: grpc_servicer/ Python package —
vllm/servicer.py, sglang/servicer.py).

Bridges the smg.Scheduler RPC surface onto an engine: the CPU mock simulator
(tests / CPU fleets) or the GPU TorchEngine (MI355X workers).  grpc.aio with
generic bytes handlers + msgpack (see api.py).
"""
from __future__ import annotations

import asyncio
import logging
from typing import AsyncIterator, Optional

import grpc

from ..mock.engine import MockWorkerEngine, SimConfig
from . import api

log = logging.getLogger("smg.grpc.servicer")


class EngineAdapter:
    """Uniform async facade over MockWorkerEngine / TorchEngine."""

    def __init__(self, engine):
        self.engine = engine
        self.is_mock = isinstance(engine, MockWorkerEngine)
        self._torch_task: Optional[asyncio.Task] = None
        self._streams = {}
        self._kv_event_subs = []
        # lora_id -> {"lora_name", "lora_path", "pinned"} (reference
        # sglang_scheduler.proto:385-420 — the engine owns GPU residency;
        # this registry is the management surface the gateway drives)
        self.lora_adapters: dict = {}

    # ---- LoRA management ---------------------------------------------------
    def load_lora(self, lora_name: str, lora_path: str, lora_id: str, pinned: bool = False):
        import os

        if not lora_id:
            raise ValueError("lora_id is required (caller-minted)")
        if not self.is_mock and lora_path and not os.path.isdir(lora_path):
            raise FileNotFoundError(f"adapter path not readable: {lora_path}")
        self.lora_adapters[lora_id] = {"lora_name": lora_name, "lora_path": lora_path, "pinned": pinned}
        return sorted(self.lora_adapters)

    def unload_lora(self, lora_name: str, lora_id: str):
        entry = self.lora_adapters.get(lora_id)
        if entry is None or (lora_name and entry["lora_name"] != lora_name):
            raise KeyError(f"adapter not loaded: name={lora_name!r} id={lora_id!r}")
        if entry["pinned"]:
            raise PermissionError(f"adapter {lora_id!r} is pinned")
        del self.lora_adapters[lora_id]
        return sorted(self.lora_adapters)

    async def start(self):
        if self.is_mock:
            await self.engine.start()
        else:
            self._torch_task = asyncio.ensure_future(self._torch_loop())

    async def stop(self):
        if self.is_mock:
            await self.engine.stop()
        elif self._torch_task:
            self._torch_task.cancel()

    async def _torch_loop(self):
        """Drive the GPU engine; fan out per-request events."""
        loop = asyncio.get_event_loop()
        while True:
            if self.engine.n_active() == 0:
                await asyncio.sleep(0.002)
                continue
            await loop.run_in_executor(None, self.engine.step)
            for rid, token, done in self.engine.drain_events():
                q = self._streams.get(rid)
                if q is not None:
                    q.put_nowait((token, bool(done)))
            await asyncio.sleep(0)

    async def generate(self, req: api.GenerateRequest) -> AsyncIterator[api.GenerateChunk]:
        if self.is_mock:
            sim_req = self.engine.sim.submit(req.input_ids, req.sampling.max_new_tokens, rid=req.request_id)
            n = 0
            while True:
                kind, tok = await sim_req.stream_q.get()
                if kind == "token":
                    n += 1
                    yield api.GenerateChunk(req.request_id, [tok], False, None, len(req.input_ids), n)
                elif kind == "done":
                    yield api.GenerateChunk(req.request_id, [], True, "length", len(req.input_ids), n)
                    return
                elif kind == "aborted":
                    yield api.GenerateChunk(req.request_id, [], True, "abort", len(req.input_ids), n)
                    return
        else:
            q: asyncio.Queue = asyncio.Queue()
            self._streams[req.request_id] = q
            self.engine.submit(req.input_ids, req.sampling.max_new_tokens, rid=req.request_id)
            n = 0
            try:
                while True:
                    token, done = await q.get()
                    n += 1
                    yield api.GenerateChunk(req.request_id, [token], False, None, len(req.input_ids), n)
                    if done:
                        yield api.GenerateChunk(req.request_id, [], True, "length", len(req.input_ids), n)
                        return
            finally:
                self._streams.pop(req.request_id, None)

    def abort(self, request_id: str) -> None:
        if self.is_mock:
            self.engine.sim.abort(request_id)

    def loads(self) -> dict:
        if self.is_mock:
            return self.engine.sim.load_snapshot()
        return self.engine.load_snapshot()

    def drain_kv_events(self):
        if self.is_mock:
            return self.engine.sim.drain_kv_events()
        return []


class SchedulerServicer(grpc.GenericRpcHandler):
    """Generic bytes-level handler for the smg.Scheduler service."""

    def __init__(self, adapter: EngineAdapter, model_id: str = "mock-model"):
        self.adapter = adapter
        self.model_id = model_id

    def service(self, handler_call_details):
        name = handler_call_details.method.rsplit("/", 1)[-1]
        if handler_call_details.method != api.method(name) or name not in api.METHODS:
            return None
        kind = api.METHODS[name]
        fn = getattr(self, f"_h_{name.lower()}", None)
        if fn is None:
            return None
        if kind == "server_stream":
            return grpc.unary_stream_rpc_method_handler(fn)
        return grpc.unary_unary_rpc_method_handler(fn)

    # ---- handlers (sync grpc server; executed in thread pool) ------------
    def _h_generate(self, request: bytes, context):
        req = api.GenerateRequest.from_dict(api.loads(request))
        loop = self._loop()
        agen = self.adapter.generate(req)
        try:
            while True:
                chunk = asyncio.run_coroutine_threadsafe(agen.__anext__(), loop).result()
                yield api.dumps(chunk)
                if chunk.finished:
                    break
        except StopAsyncIteration:
            pass
        except Exception as exc:  # abort on client cancel / engine error
            log.warning("generate stream error: %s", exc)
            context.abort(grpc.StatusCode.INTERNAL, str(exc))

    def _h_embed(self, request: bytes, context):
        d = api.loads(request)
        import hashlib

        ids = d.get("input_ids") or []
        vec = [int.from_bytes(hashlib.blake2b(bytes(str(i), "utf8"), digest_size=4).digest(), "little") % 1000 / 1000.0 for i in ids[:16]]
        vec += [0.0] * (16 - len(vec))
        return api.dumps({"request_id": d.get("request_id"), "embedding": vec})

    def _h_healthcheck(self, request: bytes, context):
        return api.dumps({"healthy": True})

    def _h_rerank(self, request: bytes, context):
        d = api.loads(request)
        query = d.get("query", "")
        docs = d.get("documents") or []
        results = [
            {"index": i, "relevance_score": 1.0 / (1 + abs(hash(query) - hash(doc)) % 100)}
            for i, doc in enumerate(docs)
        ]
        results.sort(key=lambda r: -r["relevance_score"])
        return api.dumps({"results": results})

    def _h_classify(self, request: bytes, context):
        d = api.loads(request)
        text = d.get("input") or d.get("text") or ""
        label = "positive" if hash(text) % 2 == 0 else "negative"
        return api.dumps({"data": [{"index": 0, "label": label, "score": 0.9}]})

    def _h_encodeimage(self, request: bytes, context):
        """EPD encode leg: vision-tower forward for the pixel tensor (mock:
        deterministic pooled embedding; the GPU engine runs its tower).
        Returns an embedding descriptor the prefill leg consumes
        (reference grpc_servicer encoder_servicer.py)."""
        import hashlib

        d = api.loads(request)
        images = (d.get("multimodal") or {}).get("images") or []
        embeddings = []
        for im in images:
            h = hashlib.blake2b(im.get("data") or b"", digest_size=32).digest()
            vec = [b / 255.0 for b in h]
            embeddings.append({"embedding": vec, "height": im.get("height"), "width": im.get("width")})
        return api.dumps({"request_id": d.get("request_id"), "embeddings": embeddings})

    def _h_abort(self, request: bytes, context):
        d = api.loads(request)
        self.adapter.abort(d.get("request_id", ""))
        return api.dumps({"status": "aborted"})

    def _h_getmodelinfo(self, request: bytes, context):
        return api.dumps({"model_path": self.model_id, "is_generation": True, "max_context_length": 131072})

    def _h_getserverinfo(self, request: bytes, context):
        return api.dumps({"version": "0.1.0", "engine": "mock" if self.adapter.is_mock else "torch"})

    def _h_getloads(self, request: bytes, context):
        return api.dumps({"loads": self.adapter.loads()})

    def _h_flushcache(self, request: bytes, context):
        if self.adapter.is_mock:
            self.adapter.engine.sim.prefix_cache.clear()
        return api.dumps({"status": "ok"})

    def _h_subscribekvevents(self, request: bytes, context):
        import time

        while context.is_active():
            events = self.adapter.drain_kv_events()
            if events:
                yield api.dumps({"events": events, "block_size": getattr(self.adapter.engine.config, "block_size", 16)})
            time.sleep(0.05)

    def _h_startprofile(self, request: bytes, context):
        """Engine profiling passthrough (reference /start_profile ->
        StartProfile RPC, sglang_scheduler.proto:37-40).  On the GPU engine
        this starts torch.profiler, which on ROCm records through
        roctracer/kineto — the rocprofv3-family tooling."""
        if self.adapter.is_mock:
            return api.dumps({"status": "profiling not active on the mock engine"})
        if getattr(self, "_profiler", None) is not None:
            return api.dumps({"status": "already profiling"})
        try:
            import torch
            from torch.profiler import ProfilerActivity, profile

            self._profiler = profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA])
            self._profiler.__enter__()
            return api.dumps({"status": "profiling started"})
        except Exception as exc:
            self._profiler = None
            return api.dumps({"status": f"profiler unavailable: {exc}"})

    def _h_stopprofile(self, request: bytes, context):
        prof = getattr(self, "_profiler", None)
        if prof is None:
            return api.dumps({"status": "not profiling"})
        self._profiler = None
        try:
            prof.__exit__(None, None, None)
            import os
            import tempfile

            d = api.loads(request) if request else {}
            out = (d.get("output_dir") or tempfile.gettempdir())
            os.makedirs(out, exist_ok=True)
            path = os.path.join(out, "smg_engine_trace.json")
            prof.export_chrome_trace(path)
            return api.dumps({"status": "profiling stopped", "trace": path})
        except Exception as exc:
            return api.dumps({"status": f"profiler stop failed: {exc}"})

    def _h_loadloraadapter(self, request: bytes, context):
        d = api.loads(request)
        try:
            ids = self.adapter.load_lora(
                d.get("lora_name", ""), d.get("lora_path", ""), d.get("lora_id", ""), bool(d.get("pinned"))
            )
            return api.dumps({"success": True, "message": "", "loaded_lora_ids": ids})
        except (ValueError, FileNotFoundError) as e:
            return api.dumps({"success": False, "message": str(e), "loaded_lora_ids": sorted(self.adapter.lora_adapters)})

    def _h_unloadloraadapter(self, request: bytes, context):
        d = api.loads(request)
        try:
            ids = self.adapter.unload_lora(d.get("lora_name", ""), d.get("lora_id", ""))
            return api.dumps({"success": True, "message": "", "loaded_lora_ids": ids})
        except (KeyError, PermissionError) as e:
            return api.dumps({"success": False, "message": str(e), "loaded_lora_ids": sorted(self.adapter.lora_adapters)})

    def _h_listloraadapters(self, request: bytes, context):
        return api.dumps(
            {
                "adapters": [
                    {"lora_id": lid, **entry} for lid, entry in sorted(self.adapter.lora_adapters.items())
                ]
            }
        )

    def _loop(self):
        return self._event_loop

    def bind_loop(self, loop):
        self._event_loop = loop


class SglangProtoServicer(grpc.GenericRpcHandler):
    """Wire-compatible `sglang.grpc.scheduler.SglangScheduler` service
    (reference sglang_scheduler.proto:11-59) over the same EngineAdapter —
    protobuf messages from grpc/proto_wire.py, so a reference-protocol router
    (tonic) can front THIS repo's engines directly.  Registered alongside the
    msgpack SchedulerServicer; the method path picks the dialect."""

    def __init__(self, adapter: EngineAdapter, model_id: str = "mock-model"):
        from . import proto_wire as pw

        self.pw = pw
        self.adapter = adapter
        self.model_id = model_id
        self._event_loop = None

    def bind_loop(self, loop):
        self._event_loop = loop

    def service(self, handler_call_details):
        pw = self.pw
        prefix = f"/{pw.SERVICE_NAME}/"
        if not handler_call_details.method.startswith(prefix):
            return None
        name = handler_call_details.method[len(prefix):]
        spec = pw.METHODS.get(name)
        fn = getattr(self, f"_p_{name}", None)
        if spec is None or fn is None:
            return None
        req_cls, resp_cls, streaming = spec
        if streaming:
            return grpc.unary_stream_rpc_method_handler(
                fn, request_deserializer=req_cls.FromString,
                response_serializer=lambda m: m.SerializeToString())
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=req_cls.FromString,
            response_serializer=lambda m: m.SerializeToString())

    # ---- handlers ---------------------------------------------------------
    def _p_Generate(self, req, context):
        pw = self.pw
        sp = req.sampling_params
        internal = api.GenerateRequest(
            request_id=req.request_id,
            input_ids=list(req.tokenized.input_ids),
            text=req.tokenized.original_text or None,
            sampling=api.SamplingParams(
                max_new_tokens=sp.max_new_tokens if sp.HasField("max_new_tokens") else 128,
                temperature=sp.temperature or 1.0,
                top_p=sp.top_p or 1.0,
                top_k=sp.top_k or -1,
                stop=list(sp.stop),
                stop_token_ids=list(sp.stop_token_ids),
                ignore_eos=sp.ignore_eos,
                skip_special_tokens=sp.skip_special_tokens,
            ),
            stream=True,
            lora_id=req.lora_id or None,
            dp_rank=req.data_parallel_rank or None,
        )
        if req.HasField("disaggregated_params"):
            internal.bootstrap_host = req.disaggregated_params.bootstrap_host or None
            internal.bootstrap_port = req.disaggregated_params.bootstrap_port or None
            internal.bootstrap_room = req.disaggregated_params.bootstrap_room or None
        loop = self._event_loop
        agen = self.adapter.generate(internal)
        all_ids = []
        try:
            while True:
                chunk = asyncio.run_coroutine_threadsafe(agen.__anext__(), loop).result()
                all_ids.extend(chunk.token_ids)
                resp = pw.GenerateResponse()
                resp.request_id = req.request_id
                if chunk.finished:
                    resp.complete.output_ids.extend(all_ids)
                    resp.complete.finish_reason = chunk.finish_reason or "stop"
                    resp.complete.prompt_tokens = chunk.prompt_tokens
                    resp.complete.completion_tokens = chunk.completion_tokens
                    resp.complete.cached_tokens = chunk.cached_tokens
                    yield resp
                    break
                resp.chunk.token_ids.extend(chunk.token_ids)
                resp.chunk.prompt_tokens = chunk.prompt_tokens
                resp.chunk.completion_tokens = chunk.completion_tokens
                resp.chunk.cached_tokens = chunk.cached_tokens
                yield resp
        except StopAsyncIteration:
            pass
        except Exception as exc:
            log.warning("proto generate stream error: %s", exc)
            context.abort(grpc.StatusCode.INTERNAL, str(exc))

    def _p_HealthCheck(self, req, context):
        r = self.pw.HealthCheckResponse()
        r.healthy = True
        return r

    def _p_Abort(self, req, context):
        self.adapter.abort(req.request_id)
        r = self.pw.AbortResponse()
        r.success = True
        return r

    def _p_GetModelInfo(self, req, context):
        r = self.pw.GetModelInfoResponse()
        r.model_path = self.model_id
        r.served_model_name = self.model_id
        r.is_generation = True
        r.max_context_length = 131072
        return r

    def _p_GetServerInfo(self, req, context):
        r = self.pw.GetServerInfoResponse()
        r.server_type = "grpc"
        r.sglang_version = "smg-amd-0.1.0"
        return r

    def _p_GetLoads(self, req, context):
        snap = self.adapter.loads()
        r = self.pw.GetLoadsResponse()
        r.dp_rank_count = 1
        ld = r.loads.add()
        ld.num_running_reqs = int(snap.get("num_running_reqs") or 0)
        ld.num_waiting_reqs = int(snap.get("num_queue_reqs") or 0)
        ld.num_used_tokens = int(snap.get("num_inflight_tokens") or 0)
        ld.num_waiting_uncached_tokens = int(snap.get("num_queue_tokens") or 0)
        ld.token_usage = float(snap.get("token_usage") or 0.0)
        ld.gen_throughput = float(snap.get("gen_throughput") or 0.0)
        r.aggregate.total_running_reqs = ld.num_running_reqs
        r.aggregate.total_waiting_reqs = ld.num_waiting_reqs
        return r

    def _p_FlushCache(self, req, context):
        if self.adapter.is_mock:
            self.adapter.engine.sim.prefix_cache.clear()
        r = self.pw.FlushCacheResponse()
        r.success = True
        return r

    def _p_Embed(self, req, context):
        import hashlib

        ids = list(req.tokenized.input_ids)
        vec = [int.from_bytes(hashlib.blake2b(bytes(str(i), "utf8"), digest_size=4).digest(),
                              "little") % 1000 / 1000.0 for i in ids[:16]]
        vec += [0.0] * (16 - len(vec))
        r = self.pw.EmbedResponse()
        r.embedding_dim = len(vec)
        r.embedding.extend(vec)
        r.prompt_tokens = len(ids)
        return r

    def _p_SubscribeKvEvents(self, req, context):
        import time

        seq = int(req.start_sequence_number)
        while context.is_active():
            events = self.adapter.drain_kv_events()
            if events:
                seq += 1
                b = self.pw.KvEventBatch()
                b.sequence_number = seq
                b.timestamp = time.time()
                for ev in events:
                    e = b.events.add()
                    e.event_id = seq
                    if ev.get("type") == "stored":
                        for blk in ev.get("blocks", []):
                            kb = e.stored.blocks.add()
                            kb.block_hash = int(blk.get("block_hash", 0))
                            kb.token_ids.extend(blk.get("token_ids", []))
                            kb.block_size = int(blk.get("block_size", 16))
                    elif ev.get("type") == "removed":
                        e.removed.block_hashes.extend(ev.get("block_hashes", []))
                    else:
                        e.cleared.SetInParent()
                yield b
            time.sleep(0.05)


async def serve_grpc_worker(
    host: str = "127.0.0.1",
    port: int = 50051,
    engine=None,
    model_id: str = "mock-model",
    sim_config: Optional[SimConfig] = None,
):
    """Start a gRPC engine worker; returns (server, adapter, bound_port).
    Serves BOTH dialects on one port: the in-repo msgpack service
    (smg.Scheduler) and the reference-wire proto service
    (sglang.grpc.scheduler.SglangScheduler)."""
    from concurrent.futures import ThreadPoolExecutor

    if engine is None:
        engine = MockWorkerEngine(sim_config or SimConfig(model_id=model_id))
    adapter = EngineAdapter(engine)
    await adapter.start()
    servicer = SchedulerServicer(adapter, model_id)
    servicer.bind_loop(asyncio.get_event_loop())
    proto_servicer = SglangProtoServicer(adapter, model_id)
    proto_servicer.bind_loop(asyncio.get_event_loop())
    server = grpc.server(ThreadPoolExecutor(max_workers=32))
    server.add_generic_rpc_handlers((servicer, proto_servicer))
    bound = server.add_insecure_port(f"{host}:{port}")
    server.start()
    return server, adapter, bound
