"""GPU positional indexer vs the host PositionalIndexer (differential, gpu)."""
import pytest

pytest.importorskip("torch")
pytest.importorskip("smg_amd._core")

from smg_amd.kvindex.event_index import PositionalIndexer, compute_content_hashes


def toks(n, base=0):
    return [base + i for i in range(n)]


@pytest.mark.gpu
class TestGpuEventIndex:
    def make(self):
        from smg_amd.kvindex.gpu_event_index import GpuPositionalIndexer

        return GpuPositionalIndexer(block_size=16, capacity=1 << 16)

    def test_matches_host_indexer(self):
        gpu = self.make()
        host = PositionalIndexer(block_size=16)
        t = toks(64)
        h = compute_content_hashes(t, 16)
        for ix in (gpu, host):
            ix.apply_stored("m", "http://w0", h)
            ix.apply_stored("m", "http://w1", h[:2])
        assert gpu.find_matches("m", t) == host.find_matches("m", t)

    def test_removed(self):
        gpu = self.make()
        t = toks(64)
        h = compute_content_hashes(t, 16)
        gpu.apply_stored("m", "http://w0", h)
        gpu.apply_removed("m", "http://w0", h[2:])
        # removal clears the attribution along the tail of the path
        assert gpu.find_matches("m", t).get("http://w0", 0) <= 32

    def test_batch_scoring(self):
        gpu = self.make()
        a, b = toks(64), toks(64, base=9000)
        gpu.apply_stored("m", "http://w0", compute_content_hashes(a, 16))
        out = gpu.find_matches_batch("m", [a, b])
        assert out[0] == {"http://w0": 64}
        assert out[1] == {}

    def test_remove_worker(self):
        gpu = self.make()
        t = toks(32)
        gpu.apply_stored("m", "http://w0", compute_content_hashes(t, 16))
        gpu.remove_worker("m", "http://w0")
        assert gpu.find_matches("m", t) == {}
