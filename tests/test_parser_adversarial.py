"""Adversarial streaming-parser cases (r01 review: "tests exercise happy
paths").  Every format is driven one CHARACTER at a time, with malformed /
truncated / marker-split inputs, and checked against the complete-parse
result — the invariants the reference's per-model streaming machines
guarantee (crates/tool_parser streaming tests)."""
import json

import pytest

from smg_amd.parsers.tool import parse_complete
from smg_amd.parsers.tool.stream import StreamingToolParser

SAMPLES = {
    "qwen": '<tool_call>\n{"name": "get_weather", "arguments": {"city": "Paris, France", "days": 3}}\n</tool_call>',
    "mistral": '[TOOL_CALLS][{"name": "get_weather", "arguments": {"city": "Paris"}}]',
    "llama": '<|python_tag|>{"name": "get_weather", "parameters": {"city": "Paris"}}',
    "kimik2": '<|tool_calls_section_begin|><|tool_call_begin|>functions.get_weather:0'
              '<|tool_call_argument_begin|>{"city": "Paris"}<|tool_call_end|><|tool_calls_section_end|>',
    "cohere": '<|START_ACTION|>[{"tool_name": "get_weather", "parameters": {"city": "Paris"}}]<|END_ACTION|>',
    "qwen_xml": '<tool_call>\n<function=get_weather>\n<parameter=city>\nParis\n</parameter>\n</function>\n</tool_call>',
}


def drip(name: str, text: str, chunk: int = 1):
    sp = StreamingToolParser(name)
    events = []
    for i in range(0, len(text), chunk):
        events.extend(sp.feed(text[i: i + chunk]))
    fin, normal, calls = sp.finish()
    events.extend(fin)
    return events, normal, calls


class TestCharAtATime:
    @pytest.mark.parametrize("name", sorted(SAMPLES))
    def test_stream_equals_complete(self, name):
        text = SAMPLES[name]
        events, _, calls = drip(name, text)
        _, ref_calls = parse_complete(name, text)
        assert [c["name"] for c in calls] == [c["name"] for c in ref_calls]
        assert [c["name"] for c in ref_calls] == ["get_weather"]
        # streamed arg deltas concatenate to the final arguments
        streamed = {}
        for ev in events:
            if ev[0] == "tool_args":
                streamed[ev[1]] = streamed.get(ev[1], "") + ev[2]
        if streamed:
            assert streamed[0] == calls[0]["arguments"]

    @pytest.mark.parametrize("name", sorted(SAMPLES))
    @pytest.mark.parametrize("chunk", [3, 7])
    def test_odd_chunk_sizes(self, name, chunk):
        events, _, calls = drip(name, SAMPLES[name], chunk=chunk)
        assert calls and calls[0]["name"] == "get_weather"

    @pytest.mark.parametrize("name", sorted(SAMPLES))
    def test_leading_text_preserved(self, name):
        prefix = "Sure, let me check the weather. "
        text = prefix + SAMPLES[name]
        events, _, calls = drip(name, text)
        emitted = "".join(e[1] for e in events if e[0] == "text")
        assert emitted.startswith(prefix.rstrip()) or prefix.rstrip().startswith(emitted.rstrip())
        assert calls and calls[0]["name"] == "get_weather"


class TestMalformed:
    @pytest.mark.parametrize("name", sorted(SAMPLES))
    def test_truncated_midway_no_crash(self, name):
        text = SAMPLES[name]
        for cut in (len(text) // 3, len(text) // 2, len(text) - 3):
            sp = StreamingToolParser(name)
            for ch in text[:cut]:
                sp.feed(ch)
            sp.finish()  # must not raise

    def test_marker_lookalike_text(self):
        """Text containing an unterminated marker prefix stays text."""
        events, normal, calls = drip("qwen", "compare a<b and c<tool for me")
        assert not calls
        assert "".join(e[1] for e in events if e[0] == "text") == "compare a<b and c<tool for me"

    def test_unclosed_tool_call_recovers_args(self):
        text = '<tool_call>\n{"name": "f", "arguments": {"x": 1'
        sp = StreamingToolParser("qwen")
        for ch in text:
            sp.feed(ch)
        _, _, calls = sp.finish()
        assert calls and calls[0]["name"] == "f"

    def test_garbage_json_degrades_gracefully(self):
        text = '<tool_call>\nnot json at all\n</tool_call>'
        sp = StreamingToolParser("qwen")
        for ch in text:
            sp.feed(ch)
        sp.finish()  # no exception is the contract


class TestFormatEdges:
    def test_qwen_xml_value_with_markup(self):
        text = ('<tool_call>\n<function=post>\n<parameter=body>\n'
                'a <b> c & "quotes" </parameter-ish>\n</parameter>\n</function>\n</tool_call>')
        _, calls = parse_complete("qwen_xml", text)
        assert calls[0]["name"] == "post"
        assert "<b> c" in json.loads(calls[0]["arguments"])["body"]

    def test_pythonic_nested_commas_and_strings(self):
        text = '[search(q="a, b(c), d", k=3)]'
        tools = [{"type": "function", "function": {"name": "search", "parameters": {
            "properties": {"q": {"type": "string"}, "k": {"type": "integer"}},
            "required": ["q"]}}}]
        _, calls = parse_complete("pythonic", text, tools)
        args = json.loads(calls[0]["arguments"])
        assert args["q"] == "a, b(c), d"
        assert args["k"] == 3

    def test_mistral_multiple_calls(self):
        text = ('[TOOL_CALLS][{"name": "a", "arguments": {"i": 1}}, '
                '{"name": "b", "arguments": {"i": 2}}]')
        _, calls = parse_complete("mistral", text)
        assert [c["name"] for c in calls] == ["a", "b"]

    def test_unicode_split_inside_marker(self):
        """deepseek's markers are multi-byte; byte-ish splits inside the
        marker must still resolve."""
        text = ('<｜tool▁calls▁begin｜><｜tool▁call▁begin｜>function<｜tool▁sep｜>get_weather\n'
                '```json\n{"city": "Paris"}\n```<｜tool▁call▁end｜><｜tool▁calls▁end｜>')
        events, _, calls = drip("deepseek", text)
        assert calls and calls[0]["name"] == "get_weather"
