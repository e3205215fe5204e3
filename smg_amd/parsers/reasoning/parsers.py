"""Reasoning parser machine + registry (reference parsers/: base.rs:179,
deepseek_r1.rs:22-26, qwen3.rs:22-26/:85-89, kimi.rs:21-25, glm45.rs:20-24,
minimax.rs:22-26, step3.rs:20-24, nano_v3.rs:25-29, cohere_cmd.rs:23-27,
kimi_k3.rs, inkling.rs:5-11, factory.rs:116-197)."""
from __future__ import annotations

import re
from typing import Dict, Optional, Tuple


class ReasoningParser:
    """Extracts in-band reasoning blocks.  `parse(text)` -> (reasoning, normal);
    `parse_streaming(delta)` -> (reasoning_delta, normal_delta)."""

    def __init__(
        self,
        think_start: str = "<think>",
        think_end: str = "</think>",
        always_in_reasoning: bool = False,
        stream_reasoning: bool = True,
        name: str = "base",
    ):
        self.think_start = think_start
        self.think_end = think_end
        self.always_in_reasoning = always_in_reasoning
        self.stream_reasoning = stream_reasoning
        self.name = name
        self.reset()

    def reset(self) -> None:
        self._in_reasoning = self.always_in_reasoning
        self._buffer = ""
        self._done = False

    def fresh(self) -> "ReasoningParser":
        return ReasoningParser(
            self.think_start, self.think_end, self.always_in_reasoning, self.stream_reasoning, self.name
        )

    # ---- complete --------------------------------------------------------
    def parse(self, text: str) -> Tuple[str, str]:
        reasoning_parts = []
        normal_parts = []
        in_r = self.always_in_reasoning
        rest = text
        while rest:
            if in_r:
                end = rest.find(self.think_end)
                if end < 0:
                    reasoning_parts.append(rest)
                    rest = ""
                else:
                    reasoning_parts.append(rest[:end])
                    rest = rest[end + len(self.think_end):]
                    in_r = False
            else:
                start = rest.find(self.think_start)
                if start < 0:
                    normal_parts.append(rest)
                    rest = ""
                else:
                    normal_parts.append(rest[:start])
                    rest = rest[start + len(self.think_start):]
                    in_r = True
        return "".join(reasoning_parts).strip(), "".join(normal_parts).strip()

    # ---- streaming -------------------------------------------------------
    def _partial_marker_len(self, s: str) -> int:
        """Length of the LONGEST suffix of `s` that is a proper prefix of
        either marker.  Must take the max over both markers: when they share
        a prefix (kimi_k3's `<|open|>…` / `<|close|>…`), returning the
        start-marker's short match let end-marker bytes leak into the
        streamed reasoning and the full end marker was then never found."""
        best = 0
        for marker in (self.think_start, self.think_end):
            for k in range(min(len(marker) - 1, len(s)), 0, -1):
                if s.endswith(marker[:k]):
                    best = max(best, k)
                    break
        return best

    def parse_streaming(self, delta: str) -> Tuple[str, str]:
        self._buffer += delta
        reasoning_out = []
        normal_out = []
        while True:
            if self._done:
                normal_out.append(self._buffer)
                self._buffer = ""
                break
            if self._in_reasoning:
                end = self._buffer.find(self.think_end)
                if end >= 0:
                    reasoning_out.append(self._buffer[:end])
                    self._buffer = self._buffer[end + len(self.think_end):]
                    self._in_reasoning = False
                    continue
                keep = self._partial_marker_len(self._buffer)
                emit = self._buffer[: len(self._buffer) - keep]
                if self.stream_reasoning and emit:
                    reasoning_out.append(emit)
                    self._buffer = self._buffer[len(emit):]
                break
            start = self._buffer.find(self.think_start)
            if start >= 0:
                normal_out.append(self._buffer[:start])
                self._buffer = self._buffer[start + len(self.think_start):]
                self._in_reasoning = True
                continue
            keep = self._partial_marker_len(self._buffer)
            emit = self._buffer[: len(self._buffer) - keep]
            if emit:
                normal_out.append(emit)
                self._buffer = self._buffer[len(emit):]
            break
        return "".join(reasoning_out), "".join(normal_out)

    @property
    def in_reasoning(self) -> bool:
        return self._in_reasoning


class InklingReasoningParser(ReasoningParser):
    """inkling channel tags: <|content_thinking|> reasoning until another
    content tag (inkling.rs:5-11)."""

    def __init__(self):
        super().__init__(
            think_start="<|content_thinking|>", think_end="<|content_text|>", name="inkling"
        )

    def fresh(self) -> "InklingReasoningParser":
        return InklingReasoningParser()

    def parse(self, text: str) -> Tuple[str, str]:
        reasoning, normal = super().parse(text)
        for tag in ("<|end_message|>", "<|message_model|>", "<|content_model_end_sampling|>"):
            normal = normal.replace(tag, "")
            reasoning = reasoning.replace(tag, "")
        return reasoning.strip(), normal.strip()


class HarmonyReasoningParser(ReasoningParser):
    """gpt-oss Harmony channel format (reference grpc/harmony/parser.rs +
    streaming.rs): analysis/commentary channels -> reasoning, final ->
    normal.  Tool-call segments (recipient functions.*) are re-emitted into
    the normal stream in raw Harmony framing so the downstream `harmony`
    tool parser can extract them (pipeline order: reasoning -> tool)."""

    def __init__(self):
        super().__init__(name="harmony")

    def reset(self) -> None:
        from ...protocols import harmony as _h

        self._hs = _h.HarmonyStreamParser()
        self._buffer = ""
        self._done = False
        self._in_reasoning = False

    def fresh(self) -> "HarmonyReasoningParser":
        return HarmonyReasoningParser()

    def _events_to_deltas(self, events) -> Tuple[str, str]:
        from ...protocols import harmony as _h

        reasoning, normal = [], []
        for ev in events:
            t = ev["type"]
            if t == "reasoning":
                reasoning.append(ev["text"])
            elif t == "content":
                normal.append(ev["text"])
            elif t == "tool_call_start":
                normal.append(f"{_h.START}assistant{_h.CHANNEL}commentary to=functions.{ev['name']}{_h.MESSAGE}")
            elif t == "tool_call_args":
                normal.append(ev["arguments"])
            elif t == "tool_call_end":
                normal.append(_h.CALL)
        return "".join(reasoning), "".join(normal)

    def parse(self, text: str) -> Tuple[str, str]:
        r1, n1 = self._events_to_deltas(self._hs.feed(text))
        r2, n2 = self._events_to_deltas(self._hs.finalize())
        self.reset()
        return (r1 + r2).strip(), (n1 + n2).strip()

    def parse_streaming(self, delta: str) -> Tuple[str, str]:
        return self._events_to_deltas(self._hs.feed(delta))


def _mk(name, start="<think>", end="</think>", always=False, stream=True):
    return ReasoningParser(start, end, always, stream, name)


PARSERS: Dict[str, ReasoningParser] = {
    "base": _mk("base"),
    "passthrough": _mk("passthrough", always=False),
    "deepseek_r1": _mk("deepseek_r1", always=True),
    "deepseek_v31": _mk("deepseek_v31", always=False),
    "qwen3": _mk("qwen3"),
    "qwen3_thinking": _mk("qwen3_thinking", always=True),
    "kimi": _mk("kimi", "◁think▷", "◁/think▷"),
    "kimi_k25": _mk("kimi_k25", always=False),
    "kimi_thinking": _mk("kimi_thinking", always=True),
    "kimi_k3": _mk("kimi_k3", "<|open|>think<|sep|>", "<|close|>think<|sep|>"),
    "glm45": _mk("glm45"),
    "step3": _mk("step3", always=True),
    "minimax": _mk("minimax", always=True),
    "nano_v3": _mk("nano_v3"),
    "cohere_cmd": _mk("cohere_cmd", "<|START_THINKING|>", "<|END_THINKING|>"),
    "inkling": InklingReasoningParser(),
    "harmony": HarmonyReasoningParser(),
}

MODEL_PATTERNS = [
    (r"gpt[-_]?oss", "harmony"),
    (r"deepseek-r1", "deepseek_r1"),
    (r"deepseek-v3[.-]1", "deepseek_v31"),
    (r"qwen3?-thinking|qwen-thinking", "qwen3_thinking"),
    (r"qwen", "qwen3"),
    (r"glm-?4", "glm45"),
    (r"kimi-k3", "kimi_k3"),
    (r"kimi", "kimi"),
    (r"minimax", "minimax"),
    (r"step-?3", "step3"),
    (r"command|cohere", "cohere_cmd"),
]


def get_reasoning_parser(name_or_model: Optional[str]) -> Optional[ReasoningParser]:
    if not name_or_model:
        return None
    if name_or_model in PARSERS:
        return PARSERS[name_or_model].fresh()
    low = name_or_model.lower()
    for pat, pname in MODEL_PATTERNS:
        if re.search(pat, low):
            return PARSERS[pname].fresh()
    return None


def parse_reasoning_complete(name: str, text: str) -> Tuple[str, str]:
    parser = get_reasoning_parser(name) or PARSERS["base"].fresh()
    return parser.parse(text)
