"""Engine-side prefix KV cache: correctness (identical decode with/without a
cache hit) and hit accounting (reference: engine prefix caching that
cache-aware routing exploits)."""
import pytest

from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig


def cfg(page=8, slots=4):
    c = TorchEngineConfig.tiny()
    c.prefix_cache_page = page
    c.prefix_cache_slots = slots
    c.prefix_cache_max = 128
    return c


def run_one(eng, prompt, n=4):
    rid = eng.submit(prompt, max_new_tokens=n)
    while not eng.finished(rid):
        eng.step()
    return eng.collect(rid)


class TestPrefixCache:
    def test_hit_after_identical_prompt(self):
        eng = TorchEngine(cfg(), device="cpu")
        prompt = list(range(40))
        out1 = run_one(eng, prompt)
        assert eng.prefix_cache_hits == 0
        out2 = run_one(eng, prompt)
        assert eng.prefix_cache_hits == 1
        assert out1 == out2  # restored KV must reproduce the same decode

    def test_shared_prefix_divergent_suffix(self):
        eng = TorchEngine(cfg(), device="cpu")
        shared = list(range(32))
        out_a = run_one(eng, shared + [100, 101, 102, 103])
        out_b = run_one(eng, shared + [200, 201, 202, 203])
        assert eng.prefix_cache_hits == 1  # second request reused the 32-token prefix
        # reference: no-cache engine must produce identical tokens
        ref = TorchEngine(cfg(slots=0), device="cpu")
        assert run_one(ref, shared + [200, 201, 202, 203]) == out_b

    def test_disabled_when_zero_slots(self):
        eng = TorchEngine(cfg(slots=0), device="cpu")
        p = list(range(40))
        run_one(eng, p)
        run_one(eng, p)
        assert eng.prefix_cache_hits == 0

    def test_lru_eviction(self):
        eng = TorchEngine(cfg(slots=2), device="cpu")
        prompts = [[i * 1000 + j for j in range(24)] for i in range(3)]
        for p in prompts:
            run_one(eng, p, n=2)
        # prompt 0 was evicted by 2-slot LRU
        run_one(eng, prompts[0], n=2)
        assert eng.prefix_cache_miss >= 4

    def test_short_prompt_not_cached(self):
        eng = TorchEngine(cfg(page=16), device="cpu")
        run_one(eng, [1, 2, 3], n=2)
        run_one(eng, [1, 2, 3], n=2)
        assert eng.prefix_cache_hits == 0


@pytest.mark.gpu
class TestPrefixCacheGpu:
    def test_gpu_hit_reproduces_decode(self):
        eng = TorchEngine(cfg(), device="cuda:0")
        prompt = list(range(48))
        out1 = run_one(eng, prompt, n=6)
        out2 = run_one(eng, prompt, n=6)
        assert eng.prefix_cache_hits == 1
        assert out1 == out2
