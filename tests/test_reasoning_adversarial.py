"""Adversarial streaming-reasoning cases: char-at-a-time equivalence with
complete parsing, marker splits, nested/repeated blocks, unclosed blocks
(reference: crates/reasoning_parser streaming tests)."""
import pytest

from smg_amd.parsers.reasoning import get_reasoning_parser, parse_reasoning_complete

CASES = {
    "deepseek_r1": "I should check.</think>The answer is 4.",  # always-in-reasoning
    "qwen3": "<think>step one</think>The answer is 4.",
    "kimi": "◁think▷hmm◁/think▷done",
    "cohere_cmd": "<|START_THINKING|>alpha<|END_THINKING|>omega",
    "kimi_k3": "<|open|>think<|sep|>deep<|close|>think<|sep|>out",
}


def drip(name, text, chunk=1):
    p = get_reasoning_parser(name)
    r_parts, n_parts = [], []
    for i in range(0, len(text), chunk):
        r, n = p.parse_streaming(text[i: i + chunk])
        r_parts.append(r)
        n_parts.append(n)
    return "".join(r_parts), "".join(n_parts)


class TestStreamingEquivalence:
    @pytest.mark.parametrize("name", sorted(CASES))
    @pytest.mark.parametrize("chunk", [1, 3])
    def test_char_at_a_time(self, name, chunk):
        text = CASES[name]
        r_s, n_s = drip(name, text, chunk)
        r_c, n_c = parse_reasoning_complete(name, text)
        assert r_s.strip() == r_c
        assert n_s.strip() == n_c

    def test_repeated_blocks(self):
        text = "<think>a</think>mid<think>b</think>end"
        r, n = parse_reasoning_complete("qwen3", text)
        assert r == "ab"
        assert n == "midend"
        r_s, n_s = drip("qwen3", text)
        assert r_s.strip() == "ab" and n_s.strip() == "midend"

    def test_unclosed_block_streams_as_reasoning(self):
        r_s, n_s = drip("qwen3", "<think>never ends, model ran out")
        assert "never ends" in r_s
        assert n_s == ""

    def test_marker_lookalike_stays_text(self):
        text = "a < b and <thin ice> here"
        r_s, n_s = drip("qwen3", text)
        assert r_s == ""
        assert n_s == text

    def test_always_in_reasoning_without_end(self):
        # deepseek-r1 style: whole stream is reasoning when no </think> comes
        r_s, n_s = drip("deepseek_r1", "only thoughts, no close")
        assert "only thoughts" in r_s
        assert n_s == ""

    def test_split_inside_multibyte_marker(self):
        # kimi's ◁think▷ markers are multi-codepoint; drip by 1 char
        r_s, n_s = drip("kimi", CASES["kimi"], chunk=1)
        assert r_s.strip() == "hmm"
        assert n_s.strip() == "done"

    def test_harmony_channels_char_at_a_time(self):
        """gpt-oss Harmony: analysis -> reasoning, final -> normal, streamed
        one char at a time must equal the complete parse."""
        text = ("<|start|>assistant<|channel|>analysis<|message|>let me think<|end|>"
                "<|start|>assistant<|channel|>final<|message|>The answer is 4.<|end|>")
        r_c, n_c = parse_reasoning_complete("harmony", text)
        assert "let me think" in r_c and "answer is 4" in n_c
        for chunk in (1, 5):
            r_s, n_s = drip("harmony", text, chunk)
            assert r_s.strip() == r_c
            assert n_s.strip() == n_c

    def test_harmony_tool_call_reemitted_for_tool_parser(self):
        """A commentary tool-call segment must survive the reasoning stage in
        raw Harmony framing so the downstream harmony TOOL parser extracts
        it (pipeline order: reasoning -> tool)."""
        from smg_amd.parsers.tool import parse_complete as tool_complete

        text = ("<|start|>assistant<|channel|>analysis<|message|>need weather<|end|>"
                "<|start|>assistant<|channel|>commentary to=functions.get_weather"
                "<|message|>{\"city\": \"Paris\"}<|call|>")
        r_s, n_s = drip("harmony", text, 3)
        assert "need weather" in r_s
        _, calls = tool_complete("harmony", n_s)
        assert calls and calls[0]["name"] == "get_weather"
        assert "Paris" in calls[0]["arguments"]

    def test_harmony_truncated_no_crash(self):
        text = "<|start|>assistant<|channel|>analysis<|message|>thinking ab"
        r_s, n_s = drip("harmony", text)
        p = get_reasoning_parser("harmony")
        # truncation at every prefix must not raise
        for cut in range(0, len(text), 7):
            q = get_reasoning_parser("harmony")
            q.parse_streaming(text[:cut])

    def test_model_pattern_dispatch(self):
        assert get_reasoning_parser("deepseek-r1-distill").name == "deepseek_r1"
        assert get_reasoning_parser("Qwen3-32B").name == "qwen3"
        assert get_reasoning_parser("gpt-oss-20b").name == "harmony"
        assert get_reasoning_parser("unknown-model") is None
