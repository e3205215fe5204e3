"""Streaming tail tests: stop decoder, incremental detok, streaming tool parser
(reference stop.rs / stream.rs / parser streaming tests)."""
import json

import pytest

from smg_amd.parsers.tool.stream import StreamingToolParser
from smg_amd.tokenizer.registry import MockTokenizer
from smg_amd.tokenizer.stop import DecodeStream, StopOutcome, StopSequenceDecoder


class TestStopDecoder:
    def test_plain_passthrough(self):
        d = StopSequenceDecoder(["END"])
        out, oc = d.process_text("hello world")
        assert out == "hello world"
        assert oc == StopOutcome.NONE

    def test_full_stop_hidden(self):
        d = StopSequenceDecoder(["STOP"])
        out, oc = d.process_text("before STOP after")
        assert out == "before "
        assert oc == StopOutcome.STOPPED_WITH_TEXT
        assert d.stopped

    def test_stop_included(self):
        d = StopSequenceDecoder(["STOP"], include_stop=True)
        out, _ = d.process_text("xSTOP")
        assert out == "xSTOP"

    def test_partial_withheld_then_released(self):
        d = StopSequenceDecoder(["</s>"])
        out1, _ = d.process_text("abc</")
        assert out1 == "abc"
        out2, _ = d.process_text("xyz")  # "</xyz" is not a stop: release
        assert out2 == "</xyz"
        assert not d.stopped

    def test_partial_withheld_then_matched(self):
        d = StopSequenceDecoder(["</s>"])
        out1, _ = d.process_text("abc</")
        out2, oc = d.process_text("s> tail")
        assert out1 == "abc"
        assert out2 == ""
        assert oc == StopOutcome.STOPPED
        assert d.matched == "</s>"

    def test_stop_token_id(self):
        d = StopSequenceDecoder([], stop_token_ids={7})
        out, oc = d.process_token(3, "a")
        assert (out, oc) == ("a", StopOutcome.NONE)
        out, oc = d.process_token(7, "<eos>")
        assert oc == StopOutcome.STOPPED

    def test_flush_releases_held(self):
        d = StopSequenceDecoder(["</s>"])
        d.process_text("abc</")
        assert d.flush() == "</"

    def test_multiple_stops_earliest_wins(self):
        d = StopSequenceDecoder(["YY", "XX"])
        out, _ = d.process_text("aXXbYY")
        assert out == "a"
        assert d.matched == "XX"


class TestDecodeStream:
    def test_mock_tokenizer_roundtrip(self):
        tok = MockTokenizer()
        ds = DecodeStream(tok)
        ids = [5, 6, 7]
        text = "".join(ds.push(i) for i in ids)
        assert text == tok.decode(ids)

    def test_hf_tokenizer_multibyte(self):
        pytest.importorskip("tokenizers")
        import os

        # build a tiny BPE on the fly
        from tokenizers import Tokenizer
        from tokenizers.models import BPE
        from tokenizers.pre_tokenizers import ByteLevel as PreBL
        from tokenizers.decoders import ByteLevel as DecBL
        from tokenizers.trainers import BpeTrainer

        tk = Tokenizer(BPE())
        tk.pre_tokenizer = PreBL()
        tk.decoder = DecBL()
        tk.train_from_iterator(["héllo wörld ünïcode — em😀ji"] * 50, BpeTrainer(vocab_size=300))

        class W:
            def decode(self, ids):
                return tk.decode(ids)

        ids = tk.encode("héllo — em😀ji").ids
        ds = DecodeStream(W())
        text = "".join(ds.push(i) for i in ids)
        assert text == tk.decode(ids)


class TestStreamingToolParser:
    def chunked(self, text, n=7):
        return [text[i: i + n] for i in range(0, len(text), n)]

    def collect(self, parser_name, text, chunk=7):
        sp = StreamingToolParser(parser_name)
        events = []
        for c in self.chunked(text, chunk):
            events.extend(sp.feed(c))
        fin, normal, calls = sp.finish()
        events.extend(fin)
        return events, normal, calls

    def test_qwen_streaming(self):
        text = 'hello <tool_call>\n{"name": "get_weather", "arguments": {"city": "Paris"}}\n</tool_call>'
        events, normal, calls = self.collect("qwen", text)
        texts = "".join(e[1] for e in events if e[0] == "text")
        assert texts.startswith("hello")
        names = [e for e in events if e[0] == "tool_name"]
        assert names and names[0][2] == "get_weather"
        args = "".join(e[2] for e in events if e[0] == "tool_args")
        assert json.loads(args) == {"city": "Paris"}
        assert calls[0]["name"] == "get_weather"

    def test_plain_text_no_marker(self):
        events, normal, calls = self.collect("qwen", "just a plain answer, no tools")
        assert calls == []
        assert "".join(e[1] for e in events if e[0] == "text") == "just a plain answer, no tools"

    def test_partial_marker_withheld(self):
        sp = StreamingToolParser("qwen")
        ev1 = sp.feed("answer <tool")
        text1 = "".join(e[1] for e in ev1 if e[0] == "text")
        assert "<tool" not in text1
        ev2 = sp.feed("box>")  # not a marker after all
        fin, normal, calls = sp.finish()
        all_text = text1 + "".join(e[1] for e in list(ev2) + fin if e[0] == "text")
        assert all_text == "answer <toolbox>"

    def test_mistral_streaming(self):
        text = 'Sure. [TOOL_CALLS] [{"name": "search", "arguments": {"q": "rust"}}]'
        events, normal, calls = self.collect("mistral", text, chunk=5)
        assert calls[0]["name"] == "search"
        args = "".join(e[2] for e in events if e[0] == "tool_args")
        assert json.loads(args) == {"q": "rust"}

    def test_json_streaming(self):
        text = '{"name": "calc", "arguments": {"expr": "1+2*3"}}'
        events, normal, calls = self.collect("json", text, chunk=9)
        assert calls and calls[0]["name"] == "calc"
        args = "".join(e[2] for e in events if e[0] == "tool_args")
        assert json.loads(args) == {"expr": "1+2*3"}
