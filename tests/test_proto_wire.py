"""Wire-level golden tests for the descriptor-built engine protocol
(smg_amd/grpc/proto_wire.py vs reference
crates/grpc_client/proto/sglang_scheduler.proto + common.proto).

The expected bytes are produced by a first-principles protobuf wire-format
encoder written here from the spec (varint/zigzag-free proto3 encoding),
independent of the protobuf runtime — so a passing test means our field
numbers/types serialize exactly as the reference's tonic/prost peers expect.
"""
import struct

import pytest

pw = pytest.importorskip("smg_amd.grpc.proto_wire")


# ---- first-principles wire encoder (proto3 spec) ---------------------------
def varint(n: int) -> bytes:
    out = b""
    n &= (1 << 64) - 1
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            out += bytes([b])
            return out


def tag(field_no: int, wire_type: int) -> bytes:
    return varint((field_no << 3) | wire_type)


def f_varint(no, val) -> bytes:
    return tag(no, 0) + varint(val)


def f_len(no, payload: bytes) -> bytes:
    return tag(no, 2) + varint(len(payload)) + payload


def f_str(no, s: str) -> bytes:
    return f_len(no, s.encode())


def f_float(no, v) -> bytes:
    return tag(no, 5) + struct.pack("<f", v)


def f_double(no, v) -> bytes:
    return tag(no, 1) + struct.pack("<d", v)


def packed_u32(no, vals) -> bytes:
    return f_len(no, b"".join(varint(v) for v in vals))


# ---- golden: GenerateRequest ----------------------------------------------
def test_generate_request_wire_bytes():
    m = pw.GenerateRequest()
    m.request_id = "req-1"
    m.tokenized.original_text = "hi"
    m.tokenized.input_ids.extend([1, 2, 3])
    m.sampling_params.temperature = 0.7
    m.sampling_params.top_k = -1
    m.sampling_params.max_new_tokens = 8
    m.sampling_params.stop.append("</s>")
    m.stream = True
    got = m.SerializeToString()

    sp = (
        f_float(1, 0.7)            # temperature
        + f_varint(3, -1)          # top_k (int32 negative -> 10-byte varint)
        + f_varint(8, 8)           # max_new_tokens (optional uint32)
        + f_str(9, "</s>")         # stop[0]
    )
    tok = f_str(1, "hi") + packed_u32(2, [1, 2, 3])
    expected = (
        f_str(1, "req-1")
        + f_len(2, tok)            # tokenized
        + f_len(4, sp)             # sampling_params
        + f_varint(17, 1)          # stream = true
    )
    assert got == expected, (got.hex(), expected.hex())
    # round trip
    m2 = pw.GenerateRequest.FromString(expected)
    assert m2.request_id == "req-1"
    assert list(m2.tokenized.input_ids) == [1, 2, 3]
    assert m2.sampling_params.max_new_tokens == 8
    assert abs(m2.sampling_params.temperature - 0.7) < 1e-6
    assert m2.stream is True


def test_generate_response_oneof_wire():
    r = pw.GenerateResponse()
    r.request_id = "r9"
    r.chunk.token_ids.extend([42, 7])
    r.chunk.completion_tokens = 2
    got = r.SerializeToString()
    chunk = packed_u32(1, [42, 7]) + f_varint(3, 2)
    assert got == f_str(1, "r9") + f_len(2, chunk)
    # switching the oneof replaces the chunk
    r.complete.output_ids.extend([42, 7, 9])
    r.complete.finish_reason = "stop"
    r.complete.prompt_tokens = 5
    got = r.SerializeToString()
    comp = packed_u32(1, [42, 7, 9]) + f_str(2, "stop") + f_varint(3, 5)
    assert got == f_str(1, "r9") + f_len(3, comp)
    assert pw.GenerateResponse.FromString(got).WhichOneof("response") == "complete"


def test_sampling_constraint_oneof_and_logit_bias_map():
    sp = pw.SamplingParams()
    sp.json_schema = '{"type":"object"}'
    assert sp.WhichOneof("constraint") == "json_schema"
    sp.regex = "a+"
    assert sp.WhichOneof("constraint") == "regex"
    sp.logit_bias["50256"] = -100.0
    data = sp.SerializeToString()
    sp2 = pw.SamplingParams.FromString(data)
    assert sp2.regex == "a+" and sp2.logit_bias["50256"] == -100.0
    # map entry wire shape: field 22, submessage {key=1: str, value=2: float}
    entry = f_str(1, "50256") + f_float(2, -100.0)
    assert f_len(22, entry) in data


def test_kv_event_batch_wire():
    b = pw.KvEventBatch()
    b.sequence_number = 5
    b.timestamp = 123.5
    ev = b.events.add()
    ev.event_id = 1
    blk = ev.stored.blocks.add()
    blk.block_hash = -7
    blk.token_ids.extend([11, 12])
    blk.block_size = 16
    ev.stored.parent_block_hash = 99
    b.dp_rank = 3
    got = b.SerializeToString()
    block = (
        f_varint(1, -7)           # int64 block_hash (negative -> 10-byte varint)
        + packed_u32(2, [11, 12])
        + f_varint(3, 16)
    )
    stored = f_len(1, block) + f_varint(2, 99)
    event = f_varint(1, 1) + f_len(2, stored)
    expected = (
        f_varint(1, 5)
        + f_double(2, 123.5)
        + f_len(3, event)
        + f_varint(4, 3)
    )
    assert got == expected, (got.hex(), expected.hex())


def test_proto3_optional_presence():
    sp = pw.SamplingParams()
    assert not sp.HasField("max_new_tokens")
    sp.max_new_tokens = 0  # explicit zero must serialize (presence-tracked)
    assert sp.HasField("max_new_tokens")
    assert f_varint(8, 0) in sp.SerializeToString()
    load = pw.SchedulerLoad()
    assert not load.HasField("memory")
    load.memory.weight_gb = 1.5
    assert load.HasField("memory")


def test_get_loads_response_roundtrip():
    g = pw.GetLoadsResponse()
    g.dp_rank_count = 2
    ld = g.loads.add()
    ld.dp_rank = 0
    ld.num_running_reqs = 7
    ld.token_usage = 0.25
    ld.num_waiting_uncached_tokens = 640
    g.aggregate.total_running_reqs = 7
    out = pw.GetLoadsResponse.FromString(g.SerializeToString())
    assert out.loads[0].num_waiting_uncached_tokens == 640
    assert out.aggregate.total_running_reqs == 7


def test_service_method_table():
    assert pw.method_path("Generate") == "/sglang.grpc.scheduler.SglangScheduler/Generate"
    req_cls, resp_cls, streaming = pw.METHODS["Generate"]
    assert req_cls is pw.GenerateRequest and streaming
    assert not pw.METHODS["Embed"][2]
    assert pw.METHODS["SubscribeKvEvents"][1] is pw.KvEventBatch


def test_timestamp_and_struct_wkt_fields():
    m = pw.GenerateRequest()
    m.timestamp.seconds = 1700000000
    m.sampling_params.custom_params["foo"] = "bar"
    out = pw.GenerateRequest.FromString(m.SerializeToString())
    assert out.timestamp.seconds == 1700000000
    assert out.sampling_params.custom_params["foo"] == "bar"


# ---- loopback: proto client <-> proto servicer over real grpcio ------------
def test_proto_loopback_generate():
    """The reference-wire service end to end: ProtoEngineClient speaks
    sglang.grpc.scheduler.SglangScheduler to the engine servicer (mock
    engine), streaming chunks + final complete."""
    import asyncio

    from smg_amd.grpc import api
    from smg_amd.grpc.client import ProtoEngineClient
    from smg_amd.grpc.servicer import serve_grpc_worker
    from smg_amd.mock.engine import SimConfig

    async def run():
        server, adapter, port = await serve_grpc_worker(
            port=0, sim_config=SimConfig(speedup=50.0, model_id="mock-model")
        )
        client = ProtoEngineClient(f"grpc+proto://127.0.0.1:{port}")
        try:
            assert await client.health_check()
            req = api.GenerateRequest(
                request_id="p1",
                input_ids=list(range(48)),
                sampling=api.SamplingParams(max_new_tokens=6),
            )
            chunks = []
            async for c in client.generate(req):
                chunks.append(c)
            assert chunks[-1].finished
            total = sum(len(c.token_ids) for c in chunks)
            assert total == 6
            assert chunks[-1].finish_reason in ("stop", "length")
            loads = await client.get_loads()
            assert "loads" in loads
            info = await client.get_model_info()
            assert info["model_path"] == "mock-model"
            fl = await client.flush_cache()
            assert fl["status"] == "ok"
        finally:
            await client.close()
            await adapter.stop()
            server.stop(grace=None)

    asyncio.new_event_loop().run_until_complete(run())
