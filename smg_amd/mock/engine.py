"""Continuous-batching engine simulator (reference: crates/mock_worker/src/engine.rs:1-23,
defaults engine.rs:77-92).

Models a real LLM engine on CPU for gateway testing without GPUs:
  * prefill latency proportional to the UNCACHED prompt length (chunked),
  * decode latency = base + slope * batch_size per step, one token per running
    request per step,
  * finite KV with admission queueing,
  * radix prefix cache over prompt tokens (the same PagedRadixTree the
    cache-aware policy uses) emitting KvBlocksStored/Removed events,
  * load snapshot for /get_loads (queued/inflight tokens, token_usage,
    gen_throughput).

`SimEngine.step(now)` is pure/synchronous for deterministic tests; `run()` is
the asyncio pacing loop used when serving; `MockWorkerServer` exposes the
engine over HTTP (aiohttp) and `handle()` is the in-process sim:// transport.
"""
from __future__ import annotations

import asyncio
import itertools
import json
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Dict, List, Optional, Tuple

from ..kvindex.pytree import PagedRadixTree


@dataclass
class SimConfig:
    prefill_tokens_per_sec: float = 8000.0
    decode_base_secs: float = 0.006
    decode_per_request_secs: float = 0.00035
    kv_capacity_tokens: int = 524_288
    block_size: int = 16
    max_batch: int = 256
    model_id: str = "mock-model"
    dp_size: int = 0
    speedup: float = 1.0  # >1 accelerates simulated time (tests)
    failure_rate: float = 0.0  # fault injection: fraction of requests erroring
    fail_health: bool = False  # fault injection: health endpoint down


@dataclass
class _SimRequest:
    rid: str
    prompt_tokens: List[int]
    max_new_tokens: int
    stream_q: asyncio.Queue = field(default_factory=asyncio.Queue)
    uncached_tokens: int = 0
    prefilled: float = 0.0
    generated: int = 0
    done: bool = False
    ttft: Optional[float] = None
    submitted_at: float = 0.0


class SimEngine:
    def __init__(self, config: Optional[SimConfig] = None):
        self.config = config or SimConfig()
        self.waiting: List[_SimRequest] = []
        self.running: List[_SimRequest] = []
        self.prefix_cache = PagedRadixTree(page_size=self.config.block_size)
        self.kv_used = 0
        self.total_generated = 0
        self._recent_tokens: List[Tuple[float, int]] = []
        self.kv_events: List[Dict[str, Any]] = []  # pending KvBlocksStored/Removed
        self._counter = itertools.count()
        self.aborted: set = set()

    # ---- submission ------------------------------------------------------
    def submit(self, prompt_tokens: List[int], max_new_tokens: int, rid: Optional[str] = None) -> _SimRequest:
        req = _SimRequest(
            rid=rid or uuid.uuid4().hex,
            prompt_tokens=list(prompt_tokens),
            max_new_tokens=max(1, max_new_tokens),
            submitted_at=time.monotonic(),
        )
        match = self.prefix_cache.match(req.prompt_tokens)
        req.uncached_tokens = len(req.prompt_tokens) - match.matched_token_count
        self.waiting.append(req)
        return req

    def abort(self, rid: str) -> None:
        self.aborted.add(rid)

    # ---- pure step (deterministic) ---------------------------------------
    def step(self, now: float) -> float:
        """Advance one engine step at time `now`; returns the step duration in
        simulated seconds."""
        cfg = self.config
        # admission: prompt + full output must fit in KV
        while self.waiting and len(self.running) < cfg.max_batch:
            nxt = self.waiting[0]
            need = len(nxt.prompt_tokens) + nxt.max_new_tokens
            if self.kv_used + need > cfg.kv_capacity_tokens:
                break
            self.waiting.pop(0)
            self.kv_used += need
            self.running.append(nxt)

        batch = len(self.running)
        if batch == 0:
            return cfg.decode_base_secs

        # prefill phase: requests with un-prefilled prompt consume prefill budget
        step_dur = cfg.decode_base_secs + cfg.decode_per_request_secs * batch
        prefill_budget = cfg.prefill_tokens_per_sec * step_dur
        finished: List[_SimRequest] = []
        for req in self.running:
            if req.rid in self.aborted:
                req.done = True
                finished.append(req)
                req.stream_q.put_nowait(("aborted", None))
                continue
            if req.prefilled < req.uncached_tokens:
                take = min(prefill_budget, req.uncached_tokens - req.prefilled)
                req.prefilled += take
                prefill_budget -= take
                if req.prefilled < req.uncached_tokens or take > 0 and prefill_budget <= 0:
                    if req.prefilled < req.uncached_tokens:
                        continue
            # decode one token
            tok = 1000 + (next(self._counter) % 30000)
            req.generated += 1
            self.total_generated += 1
            if req.ttft is None:
                req.ttft = now - req.submitted_at
            req.stream_q.put_nowait(("token", tok))
            if req.generated >= req.max_new_tokens:
                req.done = True
                finished.append(req)
                req.stream_q.put_nowait(("done", None))

        for req in finished:
            self.running.remove(req)
            self.kv_used -= len(req.prompt_tokens) + req.max_new_tokens
            # store the prompt prefix in the cache (engine keeps KV blocks)
            self.prefix_cache.insert(req.prompt_tokens, "self")
            n_blocks = len(req.prompt_tokens) // self.config.block_size
            if n_blocks:
                self.kv_events.append(
                    {"type": "stored", "rid": req.rid, "num_blocks": n_blocks, "tokens": req.prompt_tokens}
                )
        self._recent_tokens.append((now, batch))
        cutoff = now - 10.0
        while self._recent_tokens and self._recent_tokens[0][0] < cutoff:
            self._recent_tokens.pop(0)
        return step_dur

    # ---- load snapshot ----------------------------------------------------
    def load_snapshot(self) -> Dict[str, Any]:
        now = time.monotonic()
        window = [c for t, c in self._recent_tokens if t > now - 2.0]
        tput = sum(window) / 2.0 if window else 0.0
        snap = {
            "num_queue_tokens": sum(len(r.prompt_tokens) for r in self.waiting),
            "num_inflight_tokens": sum(
                len(r.prompt_tokens) + r.generated for r in self.running
            ),
            "num_queue_reqs": len(self.waiting),
            "num_running_reqs": len(self.running),
            "token_usage": self.kv_used / self.config.kv_capacity_tokens,
            "gen_throughput": tput,
        }
        if self.config.dp_size:
            per = max(1, len(self.running) // self.config.dp_size)
            snap["dp_loads"] = [per] * self.config.dp_size
        return snap

    def drain_kv_events(self) -> List[Dict[str, Any]]:
        out, self.kv_events = self.kv_events, []
        return out


class MockWorkerEngine:
    """Async serving wrapper: pacing loop + OpenAI-compatible `handle()`.

    Used directly as the sim:// in-process transport and by MockWorkerServer
    for real HTTP serving (reference mock_worker http.rs).
    """

    def __init__(self, config: Optional[SimConfig] = None):
        self.sim = SimEngine(config)
        self.config = self.sim.config
        self._loop_task: Optional[asyncio.Task] = None
        self._stopped = asyncio.Event()

    async def start(self) -> None:
        self._stopped.clear()
        if self._loop_task is None:
            self._loop_task = asyncio.ensure_future(self._run())

    async def stop(self) -> None:
        self._stopped.set()
        if self._loop_task is not None:
            self._loop_task.cancel()
            try:
                await self._loop_task
            except (asyncio.CancelledError, Exception):
                pass
            self._loop_task = None

    async def _run(self) -> None:
        while not self._stopped.is_set():
            dur = self.sim.step(time.monotonic())
            await asyncio.sleep(dur / max(self.config.speedup, 1e-6))

    # ---- request handling -------------------------------------------------
    @staticmethod
    def _tokens_from_body(body: Dict[str, Any]) -> List[int]:
        ids = body.get("input_ids")
        if isinstance(ids, list) and ids and all(isinstance(x, int) for x in ids):
            return ids
        prompt = body.get("prompt")
        if isinstance(prompt, list) and prompt and all(isinstance(x, int) for x in prompt):
            return prompt
        text = ""
        if isinstance(prompt, str):
            text = prompt
        elif isinstance(body.get("text"), str):
            text = body["text"]
        elif isinstance(body.get("messages"), list):
            text = "\n".join(
                m.get("content", "") for m in body["messages"] if isinstance(m.get("content"), str)
            )
        # deterministic pseudo-tokenization: 1 token per 4 chars
        return [(hash(text[i : i + 4]) & 0x7FFF) for i in range(0, len(text), 4)]

    async def handle(self, path: str, body: Optional[Dict[str, Any]], headers: Dict[str, str]):
        """In-process transport entry: returns (status, headers, bytes|async-iter)."""
        if path in ("/health", "/health_generate"):
            if self.config.fail_health:
                return 503, {}, b'{"status":"unhealthy"}'
            return 200, {}, b'{"status":"ok"}'
        if path == "/get_loads":
            return 200, {}, json.dumps({"loads": self.sim.load_snapshot()}).encode()
        if path == "/get_model_info":
            return 200, {}, json.dumps({"model_path": self.config.model_id, "is_generation": True}).encode()
        if path == "/v1/models":
            return 200, {}, json.dumps(
                {"object": "list", "data": [{"id": self.config.model_id, "object": "model"}]}
            ).encode()
        if path == "/flush_cache":
            self.sim.prefix_cache.clear()
            return 200, {}, b'{"status":"ok"}'
        if path in ("/v1/chat/completions", "/v1/completions", "/generate"):
            return await self._generate(path, body or {}, headers)
        if path in ("/v1/rerank", "/rerank"):
            body = body or {}
            docs = body.get("documents") or []
            query = body.get("query", "")
            results = [
                {"index": i, "relevance_score": 1.0 / (1 + abs(hash(query) - hash(d)) % 100)}
                for i, d in enumerate(docs)
            ]
            results.sort(key=lambda r: -r["relevance_score"])
            return 200, {}, json.dumps({"results": results, "model": self.config.model_id}).encode()
        if path == "/v1/classify":
            body = body or {}
            text = body.get("input") or body.get("text") or ""
            label = "positive" if hash(text) % 2 == 0 else "negative"
            return 200, {}, json.dumps(
                {"object": "classification", "model": self.config.model_id,
                 "data": [{"index": 0, "label": label, "score": 0.9}]}).encode()
        return 404, {}, b'{"error":"not found"}'

    async def _generate(self, path: str, body: Dict[str, Any], headers: Dict[str, str]):
        if self.config.failure_rate > 0.0:
            import random

            if random.random() < self.config.failure_rate:
                return 500, {}, b'{"error":{"message":"injected failure","type":"server_error"}}'
        tokens = self._tokens_from_body(body)
        max_new = body.get("max_tokens") or body.get("max_new_tokens") or (
            (body.get("sampling_params") or {}).get("max_new_tokens") if isinstance(body.get("sampling_params"), dict) else None
        ) or 16
        stream = bool(body.get("stream", False))
        req = self.sim.submit(tokens, int(max_new))
        model = body.get("model") or self.config.model_id
        rid = f"cmpl-{req.rid[:16]}"

        if stream:
            return 200, {"content-type": "text/event-stream"}, self._sse_stream(path, req, rid, model)
        # unary: wait for completion
        toks: List[int] = []
        while True:
            kind, tok = await req.stream_q.get()
            if kind == "token":
                toks.append(tok)
            elif kind in ("done", "aborted"):
                break
        text = "".join(f" tok{t}" for t in toks)
        resp = self._final_body(path, rid, model, text, toks, len(tokens))
        return 200, {"content-type": "application/json"}, json.dumps(resp).encode()

    def _final_body(self, path, rid, model, text, toks, prompt_len):
        usage = {
            "prompt_tokens": prompt_len,
            "completion_tokens": len(toks),
            "total_tokens": prompt_len + len(toks),
        }
        if path == "/v1/chat/completions":
            return {
                "id": rid,
                "object": "chat.completion",
                "created": int(time.time()),
                "model": model,
                "choices": [
                    {
                        "index": 0,
                        "message": {"role": "assistant", "content": text},
                        "finish_reason": "stop",
                    }
                ],
                "usage": usage,
            }
        if path == "/v1/completions":
            return {
                "id": rid,
                "object": "text_completion",
                "created": int(time.time()),
                "model": model,
                "choices": [{"index": 0, "text": text, "finish_reason": "stop"}],
                "usage": usage,
            }
        return {"text": text, "output_ids": toks, "meta_info": {"id": rid, "usage": usage}}

    async def _sse_stream(self, path: str, req: _SimRequest, rid: str, model: str) -> AsyncIterator[bytes]:
        first = True
        while True:
            kind, tok = await req.stream_q.get()
            if kind == "token":
                text = f" tok{tok}"
                if path == "/v1/chat/completions":
                    delta = {"role": "assistant", "content": text} if first else {"content": text}
                    chunk = {
                        "id": rid,
                        "object": "chat.completion.chunk",
                        "created": int(time.time()),
                        "model": model,
                        "choices": [{"index": 0, "delta": delta, "finish_reason": None}],
                    }
                elif path == "/v1/completions":
                    chunk = {
                        "id": rid,
                        "object": "text_completion",
                        "model": model,
                        "choices": [{"index": 0, "text": text, "finish_reason": None}],
                    }
                else:
                    chunk = {"text": text, "token_ids": [tok], "meta_info": {"id": rid}}
                first = False
                yield b"data: " + json.dumps(chunk).encode() + b"\n\n"
            elif kind in ("done", "aborted"):
                if path == "/v1/chat/completions":
                    chunk = {
                        "id": rid,
                        "object": "chat.completion.chunk",
                        "model": model,
                        "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}],
                    }
                    yield b"data: " + json.dumps(chunk).encode() + b"\n\n"
                yield b"data: [DONE]\n\n"
                return
