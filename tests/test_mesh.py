"""HA mesh tests: SWIM membership, CRDT sync, worker/rate-limit adapters
(reference: crates/mesh + src/mesh/adapters e2e behavior)."""
import asyncio

import pytest
from aiohttp import web
from aiohttp.test_utils import TestServer

from smg_amd.config import PolicyConfig, RouterConfig
from smg_amd.mesh.adapters import MeshAdapters, RateLimitSyncAdapter, WorkerSyncAdapter
from smg_amd.mesh.crdt import MeshKV, Op, epoch_max_wins_merge, lww_merge
from smg_amd.mesh.server import add_mesh_routes
from smg_amd.mesh.swim import ALIVE, DEAD, MeshNode
from smg_amd.server.app_context import AppContext
from smg_amd.workers.worker import Worker


class TestCrdt:
    def test_lww(self):
        kv = MeshKV("a")
        kv.register_namespace("x")
        op0 = kv.put("x", "k", 1)
        newer = Op("x", "k", 2, (op0.hlc[0] + 10, 0, "b"), 1)
        assert kv.apply_remote(newer, "b")
        assert kv.get("x", "k") == 2
        older = Op("x", "k", 3, (1, 0, "c"), 1)
        assert not kv.apply_remote(older, "c")
        assert kv.get("x", "k") == 2

    def test_epoch_max_wins(self):
        kv = MeshKV("a")
        kv.register_namespace("rl", epoch_max_wins_merge)
        kv.put("rl", "t1:a", {"epoch": 1, "used": 5})
        assert kv.apply_remote(Op("rl", "t1:a", {"epoch": 1, "used": 9}, (0, 0, "b"), 1), "b")
        assert kv.get("rl", "t1:a")["used"] == 9
        assert not kv.apply_remote(Op("rl", "t1:a", {"epoch": 1, "used": 3}, (99, 0, "c"), 2), "c")
        assert kv.apply_remote(Op("rl", "t1:a", {"epoch": 2, "used": 1}, (0, 0, "d"), 3), "d")
        assert kv.get("rl", "t1:a")["epoch"] == 2

    def test_ops_since_watermark(self):
        kv = MeshKV("a")
        kv.register_namespace("x")
        kv.put("x", "k1", 1)
        kv.put("x", "k2", 2)
        ops = kv.ops_since(0)
        assert len(ops) == 2
        assert kv.ops_since(ops[-1].seq) == []


async def make_node(name, peers=()):
    mesh = MeshNode(name, "placeholder", probe_interval=0.1, probe_timeout=0.3, suspect_timeout=0.5)
    app = web.Application()
    add_mesh_routes(app, mesh)
    server = TestServer(app)
    await server.start_server()
    mesh.advertise_url = f"http://127.0.0.1:{server.port}"
    await mesh.start([p.advertise_url for p in peers])
    return mesh, server


def test_membership_and_sync(runner):
    async def run():
        a, sa = await make_node("a")
        b, sb = await make_node("b", peers=[a])
        try:
            await asyncio.sleep(0.3)
            assert "b" in a.members and a.members["b"].state == ALIVE
            assert "a" in b.members
            # CRDT propagation a -> b
            a.kv.register_namespace("x")
            b.kv.register_namespace("x")
            a.kv.put("x", "hello", {"v": 42})
            for _ in range(30):
                if b.kv.get("x", "hello"):
                    break
                await asyncio.sleep(0.1)
            assert b.kv.get("x", "hello") == {"v": 42}
        finally:
            await a.stop()
            await b.stop()
            await sa.close()
            await sb.close()

    runner(run())


def test_failure_detection(runner):
    async def run():
        a, sa = await make_node("a")
        b, sb = await make_node("b", peers=[a])
        try:
            await asyncio.sleep(0.3)
            await b.stop()
            await sb.close()
            for _ in range(60):
                if a.members.get("b") and a.members["b"].state == DEAD:
                    break
                await asyncio.sleep(0.1)
            assert a.members["b"].state == DEAD
        finally:
            await a.stop()
            await sa.close()

    runner(run())


def test_worker_sync_adapter(runner):
    async def run():
        a, sa = await make_node("a")
        b, sb = await make_node("b", peers=[a])
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        ctx_a, ctx_b = AppContext(cfg), AppContext(cfg)
        MeshAdapters(a, ctx_a)
        MeshAdapters(b, ctx_b)
        try:
            ctx_a.worker_registry.register(Worker("http://wX:9000", model_id="m"))
            for _ in range(30):
                if ctx_b.worker_registry.get_by_url("http://wX:9000"):
                    break
                await asyncio.sleep(0.1)
            imported = ctx_b.worker_registry.get_by_url("http://wX:9000")
            assert imported is not None
            assert imported.labels.get("mesh_origin") == "a"
            # removal propagates
            ctx_a.worker_registry.remove_by_url("http://wX:9000")
            for _ in range(30):
                if not ctx_b.worker_registry.get_by_url("http://wX:9000"):
                    break
                await asyncio.sleep(0.1)
            assert ctx_b.worker_registry.get_by_url("http://wX:9000") is None
        finally:
            await a.stop()
            await b.stop()
            await sa.close()
            await sb.close()

    runner(run())


def test_join_snapshot_bootstraps_late_node(runner):
    async def run():
        a, sa = await make_node("a")
        a.kv.register_namespace("worker")
        a.kv.put("worker", "http://w1", {"url": "http://w1", "model_id": "m", "worker_type": "regular", "origin": "a"})
        b, sb = await make_node("b", peers=[a])
        try:
            assert b.kv.get("worker", "http://w1") is not None  # via join snapshot
        finally:
            await a.stop()
            await b.stop()
            await sa.close()
            await sb.close()

    runner(run())


def test_rate_limit_shards(runner):
    async def run():
        a, sa = await make_node("a")
        b, sb = await make_node("b", peers=[a])
        ra = RateLimitSyncAdapter(a)
        rb = RateLimitSyncAdapter(b)
        try:
            ra.publish_usage("acme", epoch=7, used=40)
            rb.publish_usage("acme", epoch=7, used=25)
            for _ in range(40):
                if ra.cluster_usage("acme", 7) >= 65 and rb.cluster_usage("acme", 7) >= 65:
                    break
                await asyncio.sleep(0.1)
            assert ra.cluster_usage("acme", 7) == 65
            assert rb.cluster_usage("acme", 7) == 65
        finally:
            await a.stop()
            await b.stop()
            await sa.close()
            await sb.close()

    runner(run())


def test_tree_delta_sync_prewarms_replica(runner):
    """A prefix routed on gateway A routes to the SAME worker on gateway B
    (reference tree_sync.rs behavior)."""
    import asyncio

    from smg_amd.policies import SelectWorkerInfo

    async def run():
        a, sa = await make_node("a")
        b, sb = await make_node("b", peers=[a])
        cfg = RouterConfig(policy=PolicyConfig(name="cache_aware", gpu_tree=False, block_size=4))
        ctx_a, ctx_b = AppContext(cfg), AppContext(cfg)
        MeshAdapters(a, ctx_a)
        MeshAdapters(b, ctx_b)
        try:
            for ctx in (ctx_a, ctx_b):
                ctx.worker_registry.register(Worker("http://w0:1", model_id="m"))
                ctx.worker_registry.register(Worker("http://w1:1", model_id="m"))
            toks = list(range(64))
            pol_a = ctx_a.policy_registry.get("m")
            workers_a = ctx_a.worker_registry.for_model("m")
            sel_a = pol_a.select_worker(workers_a, SelectWorkerInfo(model_id="m", tokens=toks))
            chosen_url = workers_a[sel_a].url
            # wait for the delta to reach B
            pol_b = ctx_b.policy_registry.get("m")
            for _ in range(40):
                tree = pol_b.token_trees.get("m")
                if tree is not None and len(tree) > 0:
                    break
                await asyncio.sleep(0.1)
            workers_b = ctx_b.worker_registry.for_model("m")
            sel_b = pol_b.select_worker(workers_b, SelectWorkerInfo(model_id="m", tokens=toks))
            assert workers_b[sel_b].url == chosen_url
        finally:
            await a.stop()
            await b.stop()
            await sa.close()
            await sb.close()

    runner(run())


class TestPartitionAndRepair:
    """Mesh hardening (reference mesh/src/partition.rs + tree_sync.rs:38-67
    repair protocol): partition detection, incremental page repair replacing
    join-snapshot healing, chunked sync batches."""

    def test_repair_page_paging(self):
        kv = MeshKV("a")
        kv.register_namespace("td")
        for i in range(50):
            kv.put("td", f"k{i:04d}", {"tokens": list(range(20)), "tenant": f"w{i}"})
        pages, cursor, total = 0, "", 0
        while True:
            page = kv.repair_page("td", cursor, max_bytes=800)
            total += len(page["entries"])
            pages += 1
            if page["done"]:
                break
            cursor = page["next_cursor"]
        assert total == 50
        assert pages > 3  # byte cap forced multiple pages
        # idempotent application on a fresh node
        kv2 = MeshKV("b")
        kv2.register_namespace("td")
        cursor = ""
        while True:
            page = kv.repair_page("td", cursor, max_bytes=800)
            for opd in page["entries"]:
                op = Op.from_dict(opd)
                kv2.apply_remote(op, op.hlc[2])
            if page["done"]:
                break
            cursor = page["next_cursor"]
        assert kv2.items("td") == kv.items("td")

    def test_partition_detection_and_heal_counter(self, runner):
        async def run():
            a, sa = await make_node("a")
            b, sb = await make_node("b", peers=[a])
            c, sc = await make_node("c", peers=[a])
            try:
                await asyncio.sleep(0.4)
                assert len(a.members) == 2
                # kill both peers' inbound servers -> a sees a majority gone
                await b.stop()
                await sb.close()
                await c.stop()
                await sc.close()
                for _ in range(80):
                    if a.partitioned:
                        break
                    await asyncio.sleep(0.1)
                assert a.partitioned
                st = a.partition_state()
                assert st["unreachable"] == 2 and st["members"] == 2
            finally:
                await a.stop()
                await sa.close()

        runner(run())

    def test_kill_and_heal_reconverges_without_snapshot(self, runner):
        """A peer that missed ops it can never receive via the op-log
        (compacted during the outage) reconverges through DEAD->ALIVE page
        repair — the reference's tree:req:/tree:page: behavior."""

        async def run():
            a, sa = await make_node("a")
            b, sb = await make_node("b", peers=[a])
            a.kv.register_namespace("td")
            b.kv.register_namespace("td")
            try:
                await asyncio.sleep(0.3)
                a.kv.put("td", "seed", {"v": 1})
                for _ in range(30):
                    if b.kv.get("td", "seed"):
                        break
                    await asyncio.sleep(0.1)
                assert b.kv.get("td", "seed") == {"v": 1}
                # ---- outage window: a writes 80 ops b will never see via
                # the log (compacted away + watermark already advanced)
                for i in range(80):
                    a.kv.put("td", f"missed{i:03d}", {"v": i})
                a.kv._log.clear()  # compaction during the outage
                a._sent_watermarks["b"] = a.kv.local_seq()
                await asyncio.sleep(0.4)  # sync rounds run; nothing can ship
                assert b.kv.get("td", "missed000") is None
                # ---- heal: b observed a as DEAD, then the dead-probe
                # succeeds -> DEAD->ALIVE transition schedules page repair
                b.members["a"].state = DEAD
                for _ in range(100):
                    if b.kv.get("td", "missed079") is not None:
                        break
                    await asyncio.sleep(0.1)
                assert b.kv.get("td", "missed079") == {"v": 79}
                assert all(b.kv.get("td", f"missed{i:03d}") == {"v": i} for i in range(80))
                assert b.repairs_completed >= 1
            finally:
                await a.stop()
                await b.stop()
                await sa.close()
                await sb.close()

        runner(run())

    def test_chunked_sync_drains_backlog(self, runner):
        async def run():
            a, sa = await make_node("a")
            a.max_sync_ops_per_post = 20
            b, sb = await make_node("b", peers=[a])
            a.kv.register_namespace("x")
            b.kv.register_namespace("x")
            try:
                await asyncio.sleep(0.2)
                for i in range(150):  # backlog >> per-POST cap
                    a.kv.put("x", f"bk{i:04d}", i)
                for _ in range(80):
                    if b.kv.get("x", "bk0149") is not None:
                        break
                    await asyncio.sleep(0.1)
                assert b.kv.get("x", "bk0149") == 149
            finally:
                await a.stop()
                await b.stop()
                await sa.close()
                await sb.close()

        runner(run())


class TestMeshMtls:
    """Mutual TLS between replicas (reference crates/mesh/src/mtls.rs): the
    listener requires a client cert from the mesh CA; outbound gossip
    presents this node's cert; a cert from a DIFFERENT CA is rejected."""

    @staticmethod
    def make_ca_and_cert(tmp_path, name, ca_name="mesh-ca"):
        import subprocess

        ca_key = tmp_path / f"{ca_name}.key"
        ca_crt = tmp_path / f"{ca_name}.crt"
        if not ca_crt.exists():
            subprocess.run(["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
                            "-keyout", str(ca_key), "-out", str(ca_crt), "-days", "1",
                            "-subj", f"/CN={ca_name}"], check=True, capture_output=True)
        key = tmp_path / f"{name}.key"
        csr = tmp_path / f"{name}.csr"
        crt = tmp_path / f"{name}.crt"
        ext = tmp_path / f"{name}.ext"
        ext.write_text("subjectAltName=IP:127.0.0.1,DNS:localhost\n")
        subprocess.run(["openssl", "req", "-newkey", "rsa:2048", "-nodes",
                        "-keyout", str(key), "-out", str(csr), "-subj", f"/CN={name}"],
                       check=True, capture_output=True)
        subprocess.run(["openssl", "x509", "-req", "-in", str(csr), "-CA", str(ca_crt),
                        "-CAkey", str(ca_key), "-CAcreateserial", "-out", str(crt),
                        "-days", "1", "-extfile", str(ext)], check=True, capture_output=True)
        return str(crt), str(key), str(ca_crt)

    def test_mtls_gossip_and_rejection(self, runner, tmp_path):
        from smg_amd.mesh.server import start_mesh_server

        async def run():
            a_crt, a_key, ca = self.make_ca_and_cert(tmp_path, "node-a")
            b_crt, b_key, _ = self.make_ca_and_cert(tmp_path, "node-b")
            import socket

            s = socket.socket(); s.bind(("127.0.0.1", 0)); port_a = s.getsockname()[1]; s.close()
            s = socket.socket(); s.bind(("127.0.0.1", 0)); port_b = s.getsockname()[1]; s.close()
            a = MeshNode("a", f"https://127.0.0.1:{port_a}", probe_interval=0.1,
                         probe_timeout=0.5, suspect_timeout=1.0,
                         mtls_cert=a_crt, mtls_key=a_key, mtls_ca=ca)
            b = MeshNode("b", f"https://127.0.0.1:{port_b}", probe_interval=0.1,
                         probe_timeout=0.5, suspect_timeout=1.0,
                         mtls_cert=b_crt, mtls_key=b_key, mtls_ca=ca)
            ra = await start_mesh_server(a, "127.0.0.1", port_a)
            rb = await start_mesh_server(b, "127.0.0.1", port_b)
            try:
                await a.start([])
                await b.start([a.advertise_url])
                assert "a" in b.members
                a.kv.register_namespace("x"); b.kv.register_namespace("x")
                a.kv.put("x", "secure", 1)
                for _ in range(40):
                    if b.kv.get("x", "secure"):
                        break
                    await asyncio.sleep(0.1)
                assert b.kv.get("x", "secure") == 1  # gossip over mTLS works

                # a client WITHOUT a cert is rejected by the handshake
                import aiohttp
                import ssl as _ssl

                anon_ctx = _ssl.create_default_context(cafile=ca)
                anon_ctx.check_hostname = False
                async with aiohttp.ClientSession(
                        connector=aiohttp.TCPConnector(ssl=anon_ctx)) as sess:
                    with pytest.raises(Exception):
                        async with sess.post(a.advertise_url + "/mesh/ping",
                                             json={"from": "evil"},
                                             timeout=aiohttp.ClientTimeout(total=3)) as r:
                            await r.read()

                # a cert from a DIFFERENT CA is rejected too
                evil_crt, evil_key, _evil_ca = self.make_ca_and_cert(
                    tmp_path, "evil", ca_name="other-ca")
                evil_ctx = _ssl.create_default_context(cafile=ca)
                evil_ctx.check_hostname = False
                evil_ctx.load_cert_chain(evil_crt, evil_key)
                async with aiohttp.ClientSession(
                        connector=aiohttp.TCPConnector(ssl=evil_ctx)) as sess:
                    with pytest.raises(Exception):
                        async with sess.post(a.advertise_url + "/mesh/ping",
                                             json={"from": "evil"},
                                             timeout=aiohttp.ClientTimeout(total=3)) as r:
                            await r.read()
            finally:
                await a.stop()
                await b.stop()
                await ra.cleanup()
                await rb.cleanup()

        runner(run())
