"""Streaming tool-call parsing (reference: per-parser parse_streaming_increment
in crates/tool_parser/src/parsers/*.rs, e.g. kimik2.rs:287 argument_diff —
argument deltas are RAW text diffs from the buffer, so their concatenation is
exactly the model's argument JSON).

Machine: buffer + rescan per delta.

  ("text", s)              normal-text delta (partial-marker suffix withheld)
  ("tool_name", i, name)   first sighting of call i
  ("tool_args", i, delta)  raw argument chars of call i; concatenation of the
                           deltas == the final arguments JSON

Raw argument spans are located with a small JSON scanner for the common
formats (qwen/glm tag payloads, bare json, mistral/llama arrays, kimik2).
Parsers without a span extractor stream the name early and the full argument
string at finish() — still a correct concatenation, just less incremental.
"""
from __future__ import annotations

import json
from typing import List, Optional, Tuple

from .factory import TagToolParser, ToolParser, get_parser

Event = Tuple


def _skip_ws(s: str, i: int) -> int:
    while i < len(s) and s[i] in " \t\r\n":
        i += 1
    return i


def _scan_value(s: str, i: int) -> Optional[int]:
    """End index (exclusive) of the JSON value starting at i, or None if it
    extends past the end of s (still streaming)."""
    i = _skip_ws(s, i)
    if i >= len(s):
        return None
    c = s[i]
    if c in "{[":
        depth = 0
        in_str = False
        esc = False
        while i < len(s):
            ch = s[i]
            if in_str:
                if esc:
                    esc = False
                elif ch == "\\":
                    esc = True
                elif ch == '"':
                    in_str = False
            elif ch == '"':
                in_str = True
            elif ch in "{[":
                depth += 1
            elif ch in "}]":
                depth -= 1
                if depth == 0:
                    return i + 1
            i += 1
        return None
    if c == '"':
        i += 1
        esc = False
        while i < len(s):
            if esc:
                esc = False
            elif s[i] == "\\":
                esc = True
            elif s[i] == '"':
                return i + 1
            i += 1
        return None
    # number / literal
    j = i
    while j < len(s) and s[j] not in ",}] \t\r\n":
        j += 1
    return j if j < len(s) else None


def _find_key_value(s: str, keys=("arguments", "parameters")) -> Optional[Tuple[int, Optional[int]]]:
    """(value_start, value_end|None) of the first `keys` entry in object s."""
    for key in keys:
        probe = f'"{key}"'
        k = s.find(probe)
        if k < 0:
            continue
        i = _skip_ws(s, k + len(probe))
        if i < len(s) and s[i] == ":":
            start = _skip_ws(s, i + 1)
            if start >= len(s):
                return len(s), None
            end = _scan_value(s, start)
            return start, end
    return None


def _object_spans(s: str) -> List[Tuple[int, Optional[int]]]:
    """Spans of top-level JSON objects in s (array elements or concatenation)."""
    out = []
    i = 0
    while i < len(s):
        i = _skip_ws(s, i)
        if i >= len(s):
            break
        if s[i] == "{":
            end = _scan_value(s, i)
            out.append((i, end))
            if end is None:
                break
            i = end
        else:
            i += 1
    return out


class StreamingToolParser:
    def __init__(self, parser_or_name):
        self.parser = parser_or_name if isinstance(parser_or_name, ToolParser) else get_parser(parser_or_name)
        self.markers = self._markers_for(self.parser)
        self.buffer = ""
        self.emitted_text = 0
        self.emitted_args: List[int] = []  # per call: raw chars emitted
        self.raw_streamed: List[bool] = []
        self.named: List[bool] = []
        self.in_tool = False

    @staticmethod
    def _markers_for(parser: ToolParser) -> List[str]:
        if isinstance(parser, TagToolParser):
            return [parser.section_start or parser.begin]
        return {
            "mistral": ["[TOOL_CALLS]"],
            "llama": ["<|python_tag|>"],
            "deepseek": ["<｜tool▁calls▁begin｜>", "<｜tool▁call▁begin｜>"],
            "deepseek31": ["<｜tool▁calls▁begin｜>", "<｜tool▁call▁begin｜>"],
            "deepseek32": ["<｜tool▁calls▁begin｜>", "<｜tool▁call▁begin｜>"],
            "deepseek_v4": ["<｜tool▁calls▁begin｜>", "<｜tool▁call▁begin｜>"],
            "kimik2": ["<|tool_calls_section_begin|>", "<|tool_call_begin|>"],
            "kimi_k3": ["<|open|>tools<|sep|>"],
            "step3": ["<｜tool_calls_begin｜>"],
            "qwen_xml": ["<tool_call>", "<function="],
            "qwen_coder": ["<tool_call>", "<function="],
            "minimax_m2": ["<minimax:tool_call>"],
            "cohere": ["<|START_ACTION|>"],
            "sarashina": ["<|tool_calls|>"],
            "inkling": ["<|content_invoke_tool_json|>", "<|content_invoke_tool_text|>"],
            "harmony": ["<|start|>"],
            "json": ["{", "["],
            "pythonic": ["["],
        }.get(parser.name, [])

    # ---- raw argument spans per format ------------------------------------
    def _raw_arg_spans(self) -> Optional[List[Tuple[Optional[str], int, Optional[int]]]]:
        """[(name|None, abs_start, abs_end|None)] of raw argument values for
        each call currently visible in the buffer, or None if unsupported."""
        buf = self.buffer
        p = self.parser
        spans: List[Tuple[Optional[str], int, Optional[int]]] = []
        if isinstance(p, TagToolParser) and p.payload == "json":
            begin, end_tag = p.begin, p.end
            cursor = 0
            while True:
                b = buf.find(begin, cursor)
                if b < 0:
                    break
                payload_start = b + len(begin)
                e = buf.find(end_tag, payload_start)
                payload = buf[payload_start: e if e >= 0 else len(buf)]
                name = self._payload_name(payload)
                kv = _find_key_value(payload)
                if kv is not None:
                    vs, ve = kv
                    spans.append((name, payload_start + vs, payload_start + ve if ve is not None else None))
                else:
                    spans.append((name, payload_start + len(payload), None))
                if e < 0:
                    break
                cursor = e + len(end_tag)
            return spans
        if p.name in ("json", "mistral", "llama"):
            if p.name == "mistral":
                k = buf.find("[TOOL_CALLS]")
                if k < 0:
                    return []
                region_off = buf.find("[", k + len("[TOOL_CALLS]"))
                if region_off < 0:
                    return []
                region = buf[region_off:]
            elif p.name == "llama":
                k = buf.find("<|python_tag|>")
                region_off = (k + len("<|python_tag|>")) if k >= 0 else 0
                region = buf[region_off:]
            else:
                region_off = 0
                region = buf
            out = []
            for os_, oe in _object_spans(region):
                obj = region[os_: oe if oe is not None else len(region)]
                name = self._payload_name(obj)
                kv = _find_key_value(obj)
                if kv is not None:
                    vs, ve = kv
                    out.append((name, region_off + os_ + vs, region_off + os_ + ve if ve is not None else None))
                else:
                    out.append((name, region_off + os_ + len(obj), None))
            return out
        if p.name in ("kimik2",):
            out = []
            cursor = 0
            CALL, ARG, END = "<|tool_call_begin|>", "<|tool_call_argument_begin|>", "<|tool_call_end|>"
            while True:
                b = buf.find(CALL, cursor)
                if b < 0:
                    break
                a = buf.find(ARG, b)
                if a < 0:
                    out.append((None, len(buf), None))
                    break
                fid = buf[b + len(CALL): a].strip()
                name = fid.split(":")[0].replace("functions.", "") if fid else None
                e = buf.find(END, a)
                out.append((name, a + len(ARG), e if e >= 0 else None))
                if e < 0:
                    break
                cursor = e + len(END)
            return out
        return None

    @staticmethod
    def _payload_name(payload: str) -> Optional[str]:
        kv = _find_key_value(payload, keys=("name", "tool_name", "tool"))
        if kv is None:
            return None
        vs, ve = kv
        if ve is None:
            return None
        try:
            v = json.loads(payload[vs:ve])
            return v if isinstance(v, str) else None
        except json.JSONDecodeError:
            return None

    # ---- feed -------------------------------------------------------------
    def _partial_marker_len(self) -> int:
        tail = self.buffer[self.emitted_text:]
        best = 0
        for m in self.markers:
            lim = min(len(m) - 1, len(tail))
            for k in range(lim, 0, -1):
                if tail.endswith(m[:k]):
                    best = max(best, k)
                    break
        return best

    END_MARKERS = {
        "kimik2": ["<|tool_call_end|>", "<|tool_calls_section_end|>"],
    }

    def _open_span_holdback(self) -> int:
        """Chars to withhold from an OPEN raw-argument span (ve=None) so a
        partially-arrived end marker never leaks into the streamed argument
        deltas (e.g. kimik2's `<|tool_call_end|` prefix)."""
        ends = list(self.END_MARKERS.get(self.parser.name, []))
        end_tag = getattr(self.parser, "end", None)
        if end_tag:
            ends.append(end_tag)
        best = 0
        for m in ends:
            lim = min(len(m) - 1, len(self.buffer))
            for k in range(lim, 0, -1):
                if self.buffer.endswith(m[:k]):
                    best = max(best, k)
                    break
        return best

    def _ensure(self, i: int) -> None:
        while len(self.emitted_args) <= i:
            self.emitted_args.append(0)
            self.named.append(False)
            self.raw_streamed.append(False)

    def feed(self, delta: str, tools: Optional[List[dict]] = None) -> List[Event]:
        self.buffer += delta
        events: List[Event] = []
        if not self.in_tool:
            if any(m in self.buffer for m in self.markers):
                self.in_tool = True
                # emit text before the first marker
                first = min(self.buffer.find(m) for m in self.markers if m in self.buffer)
                if first > self.emitted_text:
                    events.append(("text", self.buffer[self.emitted_text: first]))
                    self.emitted_text = first
            else:
                keep = self._partial_marker_len()
                end = len(self.buffer) - keep
                if end > self.emitted_text:
                    events.append(("text", self.buffer[self.emitted_text: end]))
                    self.emitted_text = end
                return events
        spans = self._raw_arg_spans()
        if spans is None:
            return events  # non-incremental format: everything at finish()
        for i, (name, vs, ve) in enumerate(spans):
            self._ensure(i)
            if not self.named[i] and name:
                self.named[i] = True
                events.append(("tool_name", i, name))
            avail = (ve if ve is not None
                     else len(self.buffer) - self._open_span_holdback()) - vs
            prev = self.emitted_args[i]
            if self.named[i] and avail > prev:
                events.append(("tool_args", i, self.buffer[vs + prev: vs + avail]))
                self.emitted_args[i] = avail
                self.raw_streamed[i] = True
        return events

    def finish(self, tools: Optional[List[dict]] = None) -> Tuple[List[Event], str, List[dict]]:
        """End of stream: emits whatever remains; returns (events, normal_text,
        calls).  Raw-streamed calls carry the model's raw argument JSON so the
        streamed concatenation equals calls[i]['arguments']."""
        events: List[Event] = []
        normal, calls = self.parser.parse(self.buffer, tools)
        if not self.in_tool and not calls:
            if len(self.buffer) > self.emitted_text:
                events.append(("text", self.buffer[self.emitted_text:]))
                self.emitted_text = len(self.buffer)
            return events, self.buffer, []
        spans = self._raw_arg_spans()
        for i, call in enumerate(calls):
            self._ensure(i)
            if not self.named[i] and call.get("name"):
                self.named[i] = True
                events.append(("tool_name", i, call["name"]))
            raw = None
            if spans is not None and i < len(spans):
                name, vs, ve = spans[i]
                if ve is not None:
                    raw = self.buffer[vs:ve]
            if raw is not None and (self.raw_streamed[i] or self.emitted_args[i] == 0):
                # keep raw-stream consistency
                prev = self.emitted_args[i]
                if len(raw) > prev:
                    events.append(("tool_args", i, raw[prev:]))
                    self.emitted_args[i] = len(raw)
                call["arguments"] = raw
            elif self.emitted_args[i] == 0:
                args = call.get("arguments") or ""
                if args:
                    events.append(("tool_args", i, args))
                    self.emitted_args[i] = len(args)
        return events, normal, calls
