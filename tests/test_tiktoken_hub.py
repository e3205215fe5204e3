"""Tiktoken-format tokenizer + HF-hub resolution (reference:
crates/tokenizer/src/tiktoken.rs, kimi_k2_tokenizer.rs, hub.rs)."""
import base64
import json
import os

import pytest

from smg_amd.tokenizer.tiktoken_bpe import (
    CL100K_BASE_PATTERN,
    KIMI_K2_PATTERN,
    CoreBPE,
    TiktokenTokenizer,
    is_tiktoken_dir,
    load_tiktoken_bpe,
)
from smg_amd.tokenizer.hub import load_tokenizer, resolve_model_dir


def write_tiktoken(path, vocab):
    with open(path, "wb") as f:
        for tok, rank in vocab.items():
            f.write(base64.b64encode(tok) + b" " + str(rank).encode() + b"\n")


def full_byte_vocab():
    """All 256 bytes + a few merges — enough for lossless round trips."""
    v = {bytes([i]): i for i in range(256)}
    nxt = 256
    for merge in [b"he", b"ll", b"hell", b"hello", b" w", b" wo", b"or", b"ld"]:
        v[merge] = nxt
        nxt += 1
    return v


class TestCoreBpe:
    def test_greedy_lowest_rank_merge(self):
        enc = {b"a": 0, b"b": 1, b"c": 2, b"ab": 3, b"bc": 4, b"abc": 5}
        core = CoreBPE(enc, {})
        assert core._bpe(b"abc") == [5]  # direct vocab hit
        # 'abcb': merge ab(3) first, then abc(5): -> [abc, b]
        assert core._bpe(b"abcb") == [5, 1]
        # 'bcb': bc(4) merges, cb unknown -> [bc, b]
        assert core._bpe(b"bcb") == [4, 1]
        # no merges possible
        assert core._bpe(b"cba") == [2, 1, 0]

    def test_roundtrip_full_vocab(self, tmp_path):
        p = tmp_path / "v.tiktoken"
        write_tiktoken(p, full_byte_vocab())
        enc = load_tiktoken_bpe(str(p))
        assert enc[b"hello"] == 259
        core = CoreBPE(enc, {})
        for text in ["hello world", "héllo wörld!", "  spaces\n\nand\tmore ", "数字123"]:
            ids = core.encode_ordinary(text)
            assert core.decode(ids) == text
        # the merge actually fires: 'hello' is one token
        assert core.encode_ordinary("hello")[0] == 259

    def test_cl100k_pattern_splits(self):
        import regex

        pat = regex.compile(CL100K_BASE_PATTERN, regex.V1)
        pieces = pat.findall("Hello world's test123 done")
        assert pieces == ["Hello", " world", "'s", " test", "123", " done"]
        # numbers chunk in <=3 digit groups
        assert pat.findall("12345") == ["123", "45"]

    def test_kimi_pattern_han_runs(self):
        import regex

        pat = regex.compile(KIMI_K2_PATTERN, regex.V1)
        pieces = pat.findall("汉字处理abc then 中文")
        assert pieces[0] == "汉字处理"  # Han run kept together
        assert "abc" in pieces
        assert pieces[-1] == "中文"

    def test_special_tokens(self):
        enc = {bytes([i]): i for i in range(256)}
        core = CoreBPE(enc, {"<|im_end|>": 1000, "<|im_start|>": 1001})
        ids = core.encode("<|im_start|>hi<|im_end|>")
        assert ids[0] == 1001 and ids[-1] == 1000
        assert ids[1:3] == [ord("h"), ord("i")]
        assert core.decode(ids) == "<|im_start|>hi<|im_end|>"


class TestTiktokenTokenizer:
    def make_model_dir(self, tmp_path, kimi=False):
        d = tmp_path / "model"
        d.mkdir()
        write_tiktoken(d / "tiktoken.model", full_byte_vocab())
        cfg = {
            "added_tokens_decoder": {
                "100000": {"content": "<|endoftext|>", "special": True},
                "100001": {"content": "<|pad|>", "special": True},
            },
            "eos_token": "<|endoftext|>",
        }
        if kimi:
            cfg["auto_map"] = {"AutoTokenizer": ["tokenization_kimi.TikTokenTokenizer", None]}
        (d / "tokenizer_config.json").write_text(json.dumps(cfg))
        return str(d)

    def test_from_dir_specials_and_eos(self, tmp_path):
        d = self.make_model_dir(tmp_path)
        assert is_tiktoken_dir(d)
        tok = TiktokenTokenizer.from_dir(d)
        assert tok.eos_token_ids == [100000]
        assert tok.vocab_size >= 100002
        ids = tok.encode("hi<|endoftext|>")
        assert ids[-1] == 100000
        assert tok.decode(ids) == "hi<|endoftext|>"
        assert tok.decode_incremental(ids, len(ids) - 1) == "<|endoftext|>"

    def test_kimi_pattern_detection(self, tmp_path):
        d = self.make_model_dir(tmp_path, kimi=True)
        tok = TiktokenTokenizer.from_dir(d)
        assert tok._core.pat.pattern == KIMI_K2_PATTERN

    def test_batch(self, tmp_path):
        tok = TiktokenTokenizer.from_dir(self.make_model_dir(tmp_path))
        outs = tok.encode_batch(["one", "two"])
        assert len(outs) == 2 and all(isinstance(o, list) for o in outs)


class TestHub:
    def test_resolve_local_path(self, tmp_path):
        assert resolve_model_dir(str(tmp_path)) == str(tmp_path)

    def test_resolve_hf_cache_layout(self, tmp_path, monkeypatch):
        cache = tmp_path / "hub"
        snap = cache / "models--org--mymodel" / "snapshots" / "abc123"
        snap.mkdir(parents=True)
        (snap / "tokenizer_config.json").write_text("{}")
        (snap / "tokenizer.json").write_text("{}")
        monkeypatch.setenv("HF_HUB_CACHE", str(cache))
        assert resolve_model_dir("org/mymodel") == str(snap)

    def test_resolve_missing_returns_none(self, tmp_path, monkeypatch):
        monkeypatch.setenv("HF_HUB_CACHE", str(tmp_path / "empty"))
        monkeypatch.setenv("HF_HUB_OFFLINE", "1")
        assert resolve_model_dir("org/definitely-not-cached") is None

    def test_load_tokenizer_picks_tiktoken(self, tmp_path, monkeypatch):
        cache = tmp_path / "hub"
        snap = cache / "models--moonshotai--Kimi-K2" / "snapshots" / "r1"
        snap.mkdir(parents=True)
        write_tiktoken(snap / "tiktoken.model", full_byte_vocab())
        (snap / "tokenizer_config.json").write_text(json.dumps(
            {"auto_map": {"AutoTokenizer": ["tokenization_kimi.TikTokenTokenizer", None]}}))
        monkeypatch.setenv("HF_HUB_CACHE", str(cache))
        tok = load_tokenizer("moonshotai/Kimi-K2")
        assert isinstance(tok, TiktokenTokenizer)
        assert tok._core.pat.pattern == KIMI_K2_PATTERN
        ids = tok.encode("hello 世界")
        assert tok.decode(ids) == "hello 世界"
