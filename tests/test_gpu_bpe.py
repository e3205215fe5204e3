"""Batch BPE tests: the _core merge loop (host path on CPU, GPU kernel on
MI355X) must produce token-identical output to the HuggingFace tokenizer for
byte-level BPE vocabularies."""
import json
import os

import pytest

pytest.importorskip("torch")
core = pytest.importorskip("smg_amd._core")
tokenizers = pytest.importorskip("tokenizers")

from smg_amd.tokenizer.gpu_bpe import GpuBpeTokenizer

SAMPLES = [
    "hello world",
    "Hello, World! It's a test.",
    "  leading spaces and   runs   of spaces",
    "numbers 12345 and punct!!! #@$",
    "CamelCaseWords and snake_case_words",
    "they're we've I'll don't it's",
    "unicode héllo wörld ünïcode — em😀ji",
    "\nnewlines\n\nand\ttabs\n",
    "a" * 200,
    "The quick brown fox jumps over the lazy dog. " * 5,
    "",
]


@pytest.fixture(scope="module")
def tok_file(tmp_path_factory):
    from tokenizers import Tokenizer
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel
    from tokenizers.decoders import ByteLevel as DecBL
    from tokenizers.trainers import BpeTrainer

    tk = Tokenizer(BPE())
    tk.pre_tokenizer = ByteLevel(add_prefix_space=False, use_regex=True)
    tk.decoder = DecBL()
    corpus = SAMPLES * 50 + ["common words appear often in corpora " * 10] * 100
    tk.train_from_iterator(corpus, BpeTrainer(vocab_size=800, initial_alphabet=ByteLevel.alphabet()))
    path = tmp_path_factory.mktemp("tok") / "tokenizer.json"
    tk.save(str(path))
    return str(path)


class TestHostBpe:
    def test_matches_hf_exactly(self, tok_file):
        from tokenizers import Tokenizer

        hf = Tokenizer.from_file(tok_file)
        gb = GpuBpeTokenizer(tok_file, use_gpu=False)
        assert not gb.on_gpu
        for s in SAMPLES:
            expected = hf.encode(s, add_special_tokens=False).ids
            got = gb.encode(s)
            assert got == expected, f"{s!r}: {got} != {expected}"

    def test_batch(self, tok_file):
        from tokenizers import Tokenizer

        hf = Tokenizer.from_file(tok_file)
        gb = GpuBpeTokenizer(tok_file, use_gpu=False)
        outs = gb.encode_batch(SAMPLES)
        for s, got in zip(SAMPLES, outs):
            assert got == hf.encode(s, add_special_tokens=False).ids

    def test_decode_roundtrip(self, tok_file):
        gb = GpuBpeTokenizer(tok_file, use_gpu=False)
        s = "hello world, it's a test"
        assert gb.decode(gb.encode(s)) == s

    def test_cxx_pretokenizer_sane(self, tok_file):
        # the fallback C++ scanner must produce a valid partition of the bytes
        gb = GpuBpeTokenizer(tok_file, use_gpu=False)
        for s in SAMPLES:
            raw = s.encode("utf-8")
            offs = gb._bpe.pretokenize(raw).tolist()
            assert offs[0] == 0
            if raw:
                assert offs[-1] == len(raw)
            assert all(b > a for a, b in zip(offs, offs[1:]))


@pytest.mark.gpu
class TestGpuBpe:
    def test_gpu_matches_hf(self, tok_file):
        from tokenizers import Tokenizer

        hf = Tokenizer.from_file(tok_file)
        gb = GpuBpeTokenizer(tok_file, use_gpu=True)
        assert gb.on_gpu, "GPU BPE must run the kernel on a GPU box"
        for s in SAMPLES:
            assert gb.encode(s) == hf.encode(s, add_special_tokens=False).ids

    def test_gpu_large_batch(self, tok_file):
        gb = GpuBpeTokenizer(tok_file, use_gpu=True)
        batch = [f"request {i}: the quick brown fox {i} " * 8 for i in range(512)]
        host = GpuBpeTokenizer(tok_file, use_gpu=False)
        assert gb.encode_batch(batch) == host.encode_batch(batch)
