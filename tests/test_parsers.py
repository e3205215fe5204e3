"""Tool + reasoning parser tests (model: reference crates/tool_parser/src/tests.rs
and reasoning_parser colocated tests)."""
import json

import pytest

from smg_amd.parsers.reasoning import get_reasoning_parser, parse_reasoning_complete
from smg_amd.parsers.tool import PARSERS, get_parser, parse_complete
from smg_amd.parsers.tool.partial_json import is_complete_json, parse_partial


class TestPartialJson:
    def test_complete(self):
        v, n = parse_partial('{"a": 1}')
        assert v == {"a": 1}

    def test_truncated_object(self):
        v, _ = parse_partial('{"name": "get_weather", "arguments": {"city": "Par')
        assert v["name"] == "get_weather"
        assert v["arguments"]["city"] == "Par"

    def test_truncated_array(self):
        v, _ = parse_partial('[{"a": 1}, {"b": 2')
        assert v[0] == {"a": 1}

    def test_dangling_key(self):
        v, _ = parse_partial('{"name": "f", "arguments":')
        assert v["name"] == "f"


class TestJsonParser:
    def test_single(self):
        normal, calls = parse_complete("json", '{"name": "f", "arguments": {"x": 1}}')
        assert calls[0]["name"] == "f"
        assert json.loads(calls[0]["arguments"]) == {"x": 1}

    def test_array(self):
        _, calls = parse_complete("json", '[{"name": "a", "arguments": {}}, {"name": "b", "arguments": {}}]')
        assert [c["name"] for c in calls] == ["a", "b"]

    def test_plain_text(self):
        normal, calls = parse_complete("json", "just words, no tools")
        assert calls == []
        assert normal == "just words, no tools"


class TestQwen:
    def test_tagged(self):
        text = 'reply text\n<tool_call>\n{"name": "get_weather", "arguments": {"city": "Tokyo"}}\n</tool_call>'
        normal, calls = parse_complete("qwen", text)
        assert normal == "reply text"
        assert calls[0]["name"] == "get_weather"

    def test_multiple(self):
        text = (
            '<tool_call>\n{"name": "a", "arguments": {}}\n</tool_call>'
            '<tool_call>\n{"name": "b", "arguments": {}}\n</tool_call>'
        )
        _, calls = parse_complete("qwen", text)
        assert [c["name"] for c in calls] == ["a", "b"]


class TestMistral:
    def test_bot_token(self):
        text = 'Sure. [TOOL_CALLS] [{"name": "search", "arguments": {"q": "x[1]"}}]'
        normal, calls = parse_complete("mistral", text)
        assert calls[0]["name"] == "search"
        assert json.loads(calls[0]["arguments"])["q"] == "x[1]"
        assert normal == "Sure."


class TestLlama:
    def test_python_tag(self):
        text = '<|python_tag|>{"name": "f", "parameters": {"a": 2}}'
        _, calls = parse_complete("llama", text)
        assert calls[0]["name"] == "f"
        assert json.loads(calls[0]["arguments"]) == {"a": 2}

    def test_semicolon_multiple(self):
        text = '<|python_tag|>{"name": "f", "parameters": {}};{"name": "g", "parameters": {}}'
        _, calls = parse_complete("llama", text)
        assert len(calls) == 2


class TestPythonic:
    def test_calls(self):
        text = '[get_weather(city="Paris", days=3), ping()]'
        _, calls = parse_complete("pythonic", text)
        assert calls[0]["name"] == "get_weather"
        assert json.loads(calls[0]["arguments"]) == {"city": "Paris", "days": 3}
        assert calls[1]["name"] == "ping"


class TestDeepSeek:
    def test_fenced(self):
        text = (
            "<｜tool▁calls▁begin｜><｜tool▁call▁begin｜>function<｜tool▁sep｜>get_weather\n"
            '```json\n{"city": "SF"}\n```<｜tool▁call▁end｜><｜tool▁calls▁end｜>'
        )
        _, calls = parse_complete("deepseek", text)
        assert calls[0]["name"] == "get_weather"
        assert json.loads(calls[0]["arguments"]) == {"city": "SF"}


class TestKimiK2:
    def test_id_format(self):
        text = (
            "<|tool_calls_section_begin|><|tool_call_begin|>functions.search:0"
            '<|tool_call_argument_begin|>{"q": "news"}<|tool_call_end|><|tool_calls_section_end|>'
        )
        _, calls = parse_complete("kimik2", text)
        assert calls[0]["name"] == "search"
        assert json.loads(calls[0]["arguments"]) == {"q": "news"}


class TestQwenXml:
    def test_xml(self):
        text = "<tool_call><function=get_weather><parameter=city>Berlin</parameter></function></tool_call>"
        _, calls = parse_complete("qwen_xml", text)
        assert calls[0]["name"] == "get_weather"
        assert json.loads(calls[0]["arguments"])["city"] == "Berlin"


class TestMinimax:
    def test_invoke(self):
        text = (
            '<minimax:tool_call><invoke name="calc"><parameter name="expr">1+1</parameter>'
            "</invoke></minimax:tool_call>"
        )
        _, calls = parse_complete("minimax_m2", text)
        assert calls[0]["name"] == "calc"


class TestCohere:
    def test_action(self):
        text = '<|START_ACTION|>[{"tool_name": "ping", "parameters": {}}]<|END_ACTION|>'
        _, calls = parse_complete("cohere", text)
        assert calls[0]["name"] == "ping"

    def test_response_tags_stripped(self):
        normal, calls = parse_complete("cohere", "<|START_RESPONSE|>Hello<|END_RESPONSE|>")
        assert normal == "Hello"
        assert calls == []


class TestSarashina:
    def test_python_literal(self):
        text = "<|tool_calls|>[{'name': 'get_weather', 'arguments': {'city': 'Tokyo'}}]"
        _, calls = parse_complete("sarashina", text)
        assert calls[0]["name"] == "get_weather"


class TestStep3:
    def test_steptml(self):
        text = (
            "<｜tool_calls_begin｜><｜tool_call_begin｜>function<｜tool_sep｜>"
            '<steptml:invoke name="get_weather"><steptml:parameter name="city">Oslo</steptml:parameter>'
            "</steptml:invoke><｜tool_call_end｜><｜tool_calls_end｜>"
        )
        _, calls = parse_complete("step3", text)
        assert calls[0]["name"] == "get_weather"
        assert json.loads(calls[0]["arguments"])["city"] == "Oslo"


def test_all_registered_names_parse_plain_text():
    for name, parser in PARSERS.items():
        normal, calls = parser.parse("plain text answer")
        assert calls == [], name


def test_model_mapping():
    assert get_parser("Qwen3-32B-Instruct").name in ("qwen", "qwen_coder")
    assert get_parser("deepseek-chat").name == "deepseek"
    assert get_parser("mistral-large").name == "mistral"


class TestReasoning:
    def test_deepseek_r1_always_in_reasoning(self):
        reasoning, normal = parse_reasoning_complete("deepseek_r1", "thinking hard</think>the answer")
        assert reasoning == "thinking hard"
        assert normal == "the answer"

    def test_qwen3_explicit_start(self):
        reasoning, normal = parse_reasoning_complete("qwen3", "<think>step 1</think>result")
        assert reasoning == "step 1"
        assert normal == "result"
        reasoning, normal = parse_reasoning_complete("qwen3", "no thinking here")
        assert reasoning == ""
        assert normal == "no thinking here"

    def test_kimi_unicode_tokens(self):
        reasoning, normal = parse_reasoning_complete("kimi", "◁think▷deep◁/think▷out")
        assert reasoning == "deep"
        assert normal == "out"

    def test_cohere_markers(self):
        reasoning, normal = parse_reasoning_complete(
            "cohere_cmd", "<|START_THINKING|>analyze<|END_THINKING|>42"
        )
        assert reasoning == "analyze"
        assert normal == "42"

    def test_streaming_split_marker(self):
        p = get_reasoning_parser("qwen3")
        r1, n1 = p.parse_streaming("<thi")
        assert (r1, n1) == ("", "")
        r2, n2 = p.parse_streaming("nk>reason")
        assert r2 == "reason"
        r3, n3 = p.parse_streaming("ing</think> done")
        assert r3 == "ing"
        assert n3 == " done"

    def test_streaming_truncated(self):
        p = get_reasoning_parser("deepseek_r1")
        r, n = p.parse_streaming("all reasoning no end")
        assert r == "all reasoning no end"
        assert n == ""


class TestQwenXmlDepth:
    """qwen_xml edge formats beyond happy path:
    schema-driven typing, nested JSON values, multi-block, trailing text."""

    def tools(self):
        return [{"type": "function", "function": {
            "name": "write_file",
            "parameters": {"type": "object", "properties": {
                "path": {"type": "string"},
                "content": {"type": "string"},
                "meta": {"type": "object"},
                "lines": {"type": "integer"},
                "ratio": {"type": "number"},
                "append": {"type": "boolean"},
            }}}}]

    def parser(self):
        from smg_amd.parsers.tool.factory import get_parser

        return get_parser("qwen_xml")

    def test_schema_typed_values(self):
        text = (
            "<tool_call><function=write_file>\n"
            "<parameter=path>\n/tmp/a.txt\n</parameter>\n"
            "<parameter=content>\nline1\nline2 with <brackets> & \"quotes\"\n</parameter>\n"
            "<parameter=meta>\n{\"k\": [1, 2], \"nested\": {\"x\": true}}\n</parameter>\n"
            "<parameter=lines>\n2\n</parameter>\n"
            "<parameter=ratio>\n0.5\n</parameter>\n"
            "<parameter=append>\ntrue\n</parameter>\n"
            "</function></tool_call>"
        )
        normal, calls = self.parser().parse(text, tools=self.tools())
        assert len(calls) == 1
        import json as _json

        args = _json.loads(calls[0]["arguments"])
        assert args["path"] == "/tmp/a.txt"
        # string params keep internal newlines/brackets verbatim
        assert args["content"] == 'line1\nline2 with <brackets> & "quotes"'
        assert args["meta"] == {"k": [1, 2], "nested": {"x": True}}
        assert args["lines"] == 2 and args["ratio"] == 0.5 and args["append"] is True

    def test_string_param_containing_closing_tag_lookalike(self):
        # a string value containing a nested <parameter=...> open does not
        # truncate at the INNER close (rfind takes the last close before the
        # next opening)
        text = (
            "<tool_call><function=write_file>"
            "<parameter=content>uses <parameter=inner>x</parameter> style</parameter>"
            "</function></tool_call>"
        )
        normal, calls = self.parser().parse(text)
        import json as _json

        args = _json.loads(calls[0]["arguments"])
        assert "content" in args

    def test_multiple_blocks_and_trailing_text(self):
        text = (
            "thinking first. "
            "<tool_call><function=a><parameter=x>1</parameter></function></tool_call>"
            "<tool_call><function=b><parameter=y>2</parameter></function></tool_call>"
            " and a closing remark"
        )
        normal, calls = self.parser().parse(text)
        assert [c["name"] for c in calls] == ["a", "b"]
        assert "thinking first." in normal and "closing remark" in normal


class TestPythonicDepth:
    def parser(self):
        from smg_amd.parsers.tool.factory import get_parser

        return get_parser("pythonic")

    def test_positional_args_map_by_schema_order(self):
        tools = [{"type": "function", "function": {
            "name": "get_weather",
            "parameters": {"type": "object", "properties": {
                "city": {"type": "string"}, "unit": {"type": "string"}}}}}]
        normal, calls = self.parser().parse(
            '[get_weather("Paris", unit="C")]', tools=tools)
        import json as _json

        args = _json.loads(calls[0]["arguments"])
        assert args == {"city": "Paris", "unit": "C"}

    def test_nested_structures(self):
        normal, calls = self.parser().parse(
            "[configure(opts={'a': [1, 2], 'b': {'c': None}}, flags=[True, False])]")
        import json as _json

        args = _json.loads(calls[0]["arguments"])
        assert args["opts"] == {"a": [1, 2], "b": {"c": None}}
        assert args["flags"] == [True, False]

    def test_multiple_calls_with_leading_text(self):
        normal, calls = self.parser().parse(
            "Let me check both. [first(x=1), second(y='two')]")
        assert [c["name"] for c in calls] == ["first", "second"]
        assert normal == "Let me check both."
