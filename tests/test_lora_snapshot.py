"""LoRA adapter management over the msgpack-gRPC plane (reference
sglang_scheduler.proto:385-420) and tree snapshot/replay (reference
kv_index/src/snapshot.rs)."""
import asyncio

import pytest

from smg_amd.kvindex.pytree import StringTree, TokenTree


# ---- lora -------------------------------------------------------------------
def test_lora_grpc_roundtrip():
    from smg_amd.grpc.client import EngineClient
    from smg_amd.grpc.servicer import serve_grpc_worker
    from smg_amd.mock.engine import MockWorkerEngine, SimConfig

    async def run():
        engine = MockWorkerEngine(SimConfig(speedup=50.0, model_id="m"))
        server, adapter, port = await serve_grpc_worker(port=0, engine=engine)
        client = EngineClient(f"127.0.0.1:{port}")
        try:
            r = await client.load_lora_adapter("sql-expert", "/tmp/adapters/sql", "lora-1")
            assert r["success"] and r["loaded_lora_ids"] == ["lora-1"]
            r = await client.load_lora_adapter("chat-tune", "/tmp/adapters/chat", "lora-2", pinned=True)
            assert r["loaded_lora_ids"] == ["lora-1", "lora-2"]
            r = await client.list_lora_adapters()
            assert [a["lora_id"] for a in r["adapters"]] == ["lora-1", "lora-2"]
            assert r["adapters"][1]["pinned"] is True
            # pinned adapters refuse unload
            r = await client.unload_lora_adapter("chat-tune", "lora-2")
            assert not r["success"] and "pinned" in r["message"]
            r = await client.unload_lora_adapter("sql-expert", "lora-1")
            assert r["success"] and r["loaded_lora_ids"] == ["lora-2"]
            # wrong name or unknown id errors
            r = await client.unload_lora_adapter("nope", "lora-2")
            assert not r["success"]
            # missing caller-minted id is rejected (proto:395-397)
            r = await client.load_lora_adapter("x", "/p", "")
            assert not r["success"] and "lora_id" in r["message"]
        finally:
            await client.close()
            await adapter.stop()
            server.stop(grace=None)

    asyncio.new_event_loop().run_until_complete(run())


def test_generate_request_carries_lora_id():
    from smg_amd.grpc import api

    req = api.GenerateRequest(request_id="r1", input_ids=[1, 2], lora_id="lora-9")
    d = req.to_dict()
    assert d["lora_id"] == "lora-9"
    back = api.GenerateRequest.from_dict(d)
    assert back.lora_id == "lora-9"


# ---- snapshot ---------------------------------------------------------------
def test_token_tree_snapshot_roundtrip():
    tree = TokenTree(page_size=4)
    tree.insert(list(range(16)), "w0")
    tree.insert(list(range(8)) + [99, 98, 97, 96], "w1")
    tree.insert(list(range(12)), "w1")
    blob = tree.snapshot()
    assert isinstance(blob, bytes)
    restored = TokenTree.from_snapshot(blob)
    assert len(restored) == len(tree)
    assert restored.tenant_token_count == tree.tenant_token_count
    # identical match behavior (tenant attribution + depth)
    for toks in (list(range(16)), list(range(8)) + [99, 98, 97, 96], [5, 5, 5, 5]):
        a = tree.match(toks)
        b = restored.match(toks)
        assert (a.tenant, a.matched_token_count) == (b.tenant, b.matched_token_count)


def test_snapshot_preserves_lru_stamps():
    tree = TokenTree(page_size=2)
    tree.insert([1, 2, 3, 4], "old")
    tree.insert([1, 2, 9, 9], "new")  # shares first page; fresher stamp
    restored = TokenTree.from_snapshot(tree.snapshot())
    # MRU tenant on the shared first page must still be "new"
    assert restored.match([1, 2]).tenant == "new"
    # eviction order survives: evict to 1 node keeps the most recent path's leaf
    removed = restored.evict(max_nodes=2)
    assert removed == 1
    assert restored.match([1, 2, 9, 9]).matched_token_count == 4


def test_string_tree_snapshot():
    tree = StringTree(page_size=4)
    tree.insert_text("hello world this is a prefix", "a")
    tree.insert_text("hello world but different tail", "b")
    restored = StringTree.from_snapshot(tree.snapshot())
    m = restored.match_text("hello world this is a prefix")
    assert m.tenant == "a" and m.matched_token_count >= 24


def test_empty_tree_snapshot():
    tree = TokenTree(page_size=16)
    restored = TokenTree.from_snapshot(tree.snapshot())
    assert len(restored) == 0
    assert restored.match([1] * 16).matched_token_count == 0
