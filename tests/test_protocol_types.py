"""Protocol type breadth (reference crates/protocols — realtime_events.rs,
worker.rs WorkerSpec/SchedulerLoadSnapshot, model_card.rs, transcription.rs)."""
import pytest

from smg_amd.protocols.realtime_events import (
    CLIENT_EVENTS,
    SERVER_EVENTS,
    RealtimeEventError,
    audio_bearing,
    make_event,
    parse_event,
)
from smg_amd.protocols.worker_spec import (
    ModelCard,
    SchedulerLoadSnapshot,
    TranscriptionRequest,
    WorkerLoadResponse,
    WorkerSpec,
    WorkerSpecError,
)


class TestRealtimeEvents:
    def test_registry_covers_reference_surface(self):
        assert len(CLIENT_EVENTS) == 11
        assert len(SERVER_EVENTS) >= 45
        assert "response.function_call_arguments.delta" in SERVER_EVENTS
        assert "input_audio_buffer.speech_started" in SERVER_EVENTS

    def test_parse_valid_and_invalid(self):
        assert parse_event({"type": "input_audio_buffer.append", "audio": "QQ=="}) \
            == "input_audio_buffer.append"
        with pytest.raises(RealtimeEventError, match="missing"):
            parse_event({"type": "conversation.item.create"})
        with pytest.raises(RealtimeEventError, match="unknown"):
            parse_event({"type": "not.a.thing"})
        with pytest.raises(RealtimeEventError, match="no `type`"):
            parse_event({"audio": "x"})
        # server-only type is not a client event
        with pytest.raises(RealtimeEventError, match="unknown client"):
            parse_event({"type": "session.created", "session": {}})

    def test_make_event_validates_and_stamps_id(self):
        ev = make_event("response.output_text.delta", content_index=0, delta="hi",
                        item_id="i1", output_index=0, response_id="r1")
        assert ev["event_id"].startswith("event_")
        with pytest.raises(RealtimeEventError):
            make_event("response.output_text.delta", delta="missing the rest")

    def test_audio_bearing_flags(self):
        assert audio_bearing("input_audio_buffer.append")
        assert not audio_bearing("session.update")


class TestWorkerSpec:
    def test_parse_defaults_and_alias(self):
        s = WorkerSpec.from_dict({"url": "http://w:1", "runtime": "vllm",
                                  "models": "m1", "api_key": "sek"})
        assert s.runtime_type == "vllm" and s.models == ["m1"]
        assert s.worker_type == "regular" and s.cost == 1.0
        d = s.to_dict()
        assert "api_key" not in d  # credentials never serialize back
        assert d["url"] == "http://w:1"

    def test_rejects_bad_enums(self):
        with pytest.raises(WorkerSpecError, match="worker_type"):
            WorkerSpec.from_dict({"url": "http://w", "worker_type": "gpu"})
        with pytest.raises(WorkerSpecError, match="connection_mode"):
            WorkerSpec.from_dict({"url": "http://w", "connection_mode": "zmq"})
        with pytest.raises(WorkerSpecError, match="url"):
            WorkerSpec.from_dict({})

    def test_load_response_aggregates(self):
        r = WorkerLoadResponse(dp_rank_count=2, loads=[
            SchedulerLoadSnapshot(dp_rank=0, token_usage=0.2, num_waiting_uncached_tokens=100),
            SchedulerLoadSnapshot(dp_rank=1, token_usage=0.6, num_waiting_uncached_tokens=50),
        ])
        assert abs(r.effective_token_usage() - 0.4) < 1e-9
        assert r.total_queued_tokens() == 150
        d = r.to_dict()
        assert len(d["loads"]) == 2 and "kv_transfer_latency_ms" not in d["loads"][0]

    def test_snapshot_from_engine_dict(self):
        s = SchedulerLoadSnapshot.from_dict({"num_running_reqs": 3, "token_usage": 0.5,
                                             "bogus": 1, "gen_throughput": None})
        assert s.num_running_reqs == 3 and s.gen_throughput == 0.0


class TestModelCardTranscription:
    def test_model_card_openai_shape(self):
        c = ModelCard(id="org/m", display_name="M", context_length=8192, aliases=["m"])
        o = c.to_openai()
        assert o["object"] == "model" and o["context_length"] == 8192

    def test_transcription_form(self):
        t = TranscriptionRequest.from_form({"model": "whisper", "temperature": "0.3",
                                            "stream": "true",
                                            "timestamp_granularities": "word"})
        assert t.temperature == 0.3 and t.stream is True
        assert t.timestamp_granularities == ["word"]


def test_ws_relay_rejects_malformed_events(runner):
    """The WS relay validates client events against the typed registry."""
    import json as _json

    from aiohttp import WSMsgType, web
    from aiohttp.test_utils import TestServer

    from smg_amd.workers.worker import Worker
    from tests.test_gateway_e2e import make_ctx, start_client, stop_all

    async def run():
        seen = []

        async def ws_handler(request):
            ws = web.WebSocketResponse()
            await ws.prepare(request)
            async for msg in ws:
                if msg.type == WSMsgType.TEXT:
                    seen.append(_json.loads(msg.data)["type"])
                    await ws.send_str(_json.dumps({"type": "session.created",
                                                   "event_id": "e1", "session": {}}))
            return ws

        worker_app = web.Application()
        worker_app.router.add_get("/v1/realtime", ws_handler)
        worker_srv = TestServer(worker_app)
        await worker_srv.start_server()
        ctx, engines = make_ctx(n_workers=0)
        ctx.worker_registry.register(Worker(f"http://127.0.0.1:{worker_srv.port}",
                                            model_id="mock-model"))
        client = await start_client(ctx, engines)
        try:
            ws = await client.ws_connect("/v1/realtime?model=mock-model")
            # malformed: missing required field -> error event, NOT relayed
            await ws.send_str(_json.dumps({"type": "conversation.item.create"}))
            msg = await ws.receive(timeout=5)
            err = _json.loads(msg.data)
            assert err["type"] == "error" and "missing" in err["error"]["message"]
            # valid event relays through and the worker answers
            await ws.send_str(_json.dumps({"type": "input_audio_buffer.clear"}))
            msg = await ws.receive(timeout=5)
            assert _json.loads(msg.data)["type"] == "session.created"
            assert seen == ["input_audio_buffer.clear"]
            await ws.close()
        finally:
            await stop_all(client, engines)
            await worker_srv.close()

    runner(run())
