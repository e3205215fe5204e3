"""Prometheus exposition aggregation (reference worker/metrics_aggregator.rs:
parse, per-worker labels, family merge with label padding)."""
import asyncio

from smg_amd.observability.aggregate import Family, merge_expositions, parse_prometheus


W1 = """\
# HELP vllm_num_requests_running Number of running requests
# TYPE vllm_num_requests_running gauge
vllm_num_requests_running{model="m1"} 3
# TYPE vllm_tokens_total counter
vllm_tokens_total 1234
# TYPE request_latency histogram
request_latency_bucket{le="0.1"} 5
request_latency_bucket{le="+Inf"} 9
request_latency_sum 0.81
request_latency_count 9
"""

W2 = """\
# TYPE vllm_num_requests_running gauge
vllm_num_requests_running{model="m1",dp_rank="0"} 7
# TYPE vllm_tokens_total counter
vllm_tokens_total 99
"""


def test_parse_prometheus_families():
    fams = parse_prometheus(W1)
    assert fams["vllm_num_requests_running"].kind == "gauge"
    assert fams["vllm_num_requests_running"].help.startswith("Number of")
    assert fams["vllm_tokens_total"].samples[0][2] == "1234"
    # histogram samples attach to the base family
    rl = fams["request_latency"]
    names = [s[0] for s in rl.samples]
    assert "request_latency_bucket" in names and "request_latency_count" in names


def test_parse_replaces_colons():
    fams = parse_prometheus("# TYPE a:b counter\na:b 1\n")
    assert "a_b" in fams


def test_merge_adds_worker_label_and_pads():
    merged = merge_expositions([
        ("http://w1:8000", parse_prometheus(W1)),
        ("http://w2:8000", parse_prometheus(W2)),
    ])
    # both workers' samples present, each labeled
    assert 'vllm_num_requests_running{dp_rank="",model="m1",worker="http://w1:8000"} 3' in merged
    assert 'vllm_num_requests_running{dp_rank="0",model="m1",worker="http://w2:8000"} 7' in merged
    assert 'vllm_tokens_total{worker="http://w1:8000"} 1234' in merged
    assert 'vllm_tokens_total{worker="http://w2:8000"} 99' in merged
    # one TYPE line per family
    assert merged.count("# TYPE vllm_tokens_total counter") == 1


def test_engine_metrics_endpoint(runner):
    from tests.test_gateway_e2e import make_ctx, start_client, stop_all

    async def run():
        ctx, engines = make_ctx(n_workers=2)
        client = await start_client(ctx, engines)
        try:
            r = await client.get("/engine_metrics")
            assert r.status == 200
            text = await r.text()
            assert "smg_worker_active_requests" in text
            assert 'worker="sim://worker-0"' in text and 'worker="sim://worker-1"' in text
            r = await client.get("/engine_metrics?format=json")
            data = await r.json()
            assert "sim://worker-0" in data
        finally:
            await stop_all(client, engines)

    runner(run())


def test_extended_metric_families_registered():
    from smg_amd.observability.metrics import GatewayMetrics

    m = GatewayMetrics()
    if m._null:
        return  # prometheus_client absent (not this image)
    text = m.export().decode()
    for family in (
        "smg_router_tpot_seconds",
        "smg_pd_prefill_duration_seconds",
        "smg_worker_circuit_breaker_state",
        "smg_scheduler_admitted_total",
        "smg_mesh_gossip_rounds_total",
        "smg_tokenizer_l0_hits_total",
        "smg_kv_events_applied_total",
        "smg_event_loop_lag_seconds",
        "smg_inflight_request_age_bucket",
    ):
        assert family in text, family


def test_event_loop_canary_observes(runner):
    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.server.app import startup

    async def run():
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        cfg.health_check.disable = True
        ctx = await startup(cfg, serve=False)
        try:
            if ctx.metrics._null:
                return
            await asyncio.sleep(1.2)  # > 2 canary intervals
            text = ctx.metrics.export().decode()
            import re

            m = re.search(r"smg_event_loop_lag_seconds_count (\d+)", text)
            assert m and int(m.group(1)) >= 1
        finally:
            await ctx.shutdown()

    runner(run())
