"""Tool-call parser implementations + factory (reference: crates/tool_parser/src/
factory.rs:311-348 registration list; per-parser formats in parsers/*.rs).

21 registered names.  Most formats share one shape — optional section marker,
per-call begin/end tags, JSON payload — captured by TagToolParser; the
exceptions (pythonic, mistral, llama, deepseek fenced-json, kimik2 id-style,
step3, qwen_xml/minimax XML, cohere, sarashina, kimi_k3, inkling) get their
own classes.  Streaming: StreamingToolParser in stream.py wraps any of these
with a buffer-and-reparse incremental machine.
"""
from __future__ import annotations

import ast
import json
import re
from typing import Any, Dict, List, Optional, Tuple

from .partial_json import parse_partial

ToolCall = Dict[str, Any]  # {"name": str, "arguments": str(json), "id": str}


def _mk_call(name: str, arguments: Any, index: int = 0) -> ToolCall:
    if not isinstance(arguments, str):
        arguments = json.dumps(arguments, ensure_ascii=False)
    return {"name": name, "arguments": arguments, "index": index}


class ToolParser:
    """Complete (non-streaming) parse: returns (normal_text, calls)."""

    name = "base"

    def parse(self, text: str, tools: Optional[List[dict]] = None) -> Tuple[str, List[ToolCall]]:
        raise NotImplementedError

    def has_tool_markers(self, text: str) -> bool:
        return False


class PassthroughParser(ToolParser):
    name = "passthrough"

    def parse(self, text, tools=None):
        return text, []


class JsonParser(ToolParser):
    """Bare JSON object/array of {name, arguments|parameters} (json.rs)."""

    name = "json"

    def parse(self, text, tools=None):
        stripped = text.strip()
        for candidate, rest in self._candidates(stripped):
            calls = self._to_calls(candidate)
            if calls:
                return rest, calls
        return text, []

    def _candidates(self, s: str):
        if s.startswith("{") or s.startswith("["):
            try:
                yield json.loads(s), ""
            except json.JSONDecodeError:
                val, consumed = parse_partial(s)
                if val is not None:
                    yield val, ""
        # embedded object
        m = re.search(r"[\{\[]", s)
        if m:
            try:
                val = json.loads(s[m.start():])
                yield val, s[: m.start()].strip()
            except json.JSONDecodeError:
                pass

    def _to_calls(self, val) -> List[ToolCall]:
        if isinstance(val, dict):
            val = [val]
        if not isinstance(val, list):
            return []
        calls = []
        for i, item in enumerate(val):
            if not isinstance(item, dict) or "name" not in item:
                return []
            args = item.get("arguments", item.get("parameters", {}))
            calls.append(_mk_call(item["name"], args, i))
        return calls

    def has_tool_markers(self, text):
        t = text.strip()
        return t.startswith("{") or t.startswith("[")


class TagToolParser(ToolParser):
    """Generic [section_start] (begin_tag JSON end_tag)* [section_end] parser."""

    def __init__(
        self,
        name: str,
        begin: str,
        end: str,
        section_start: Optional[str] = None,
        section_end: Optional[str] = None,
        payload: str = "json",  # json | python_literal
    ):
        self.name = name
        self.begin = begin
        self.end = end
        self.section_start = section_start
        self.section_end = section_end
        self.payload = payload

    def has_tool_markers(self, text):
        return (self.section_start or self.begin) in text

    def _decode(self, blob: str):
        blob = blob.strip()
        if self.payload == "python_literal":
            try:
                return ast.literal_eval(blob)
            except (ValueError, SyntaxError):
                return None
        try:
            return json.loads(blob)
        except json.JSONDecodeError:
            val, _ = parse_partial(blob)
            return val

    def parse(self, text, tools=None):
        marker = self.section_start or self.begin
        pos = text.find(marker)
        if pos < 0:
            return text, []
        normal = text[:pos]
        rest = text[pos:]
        calls: List[ToolCall] = []
        idx = 0
        cursor = 0
        while True:
            b = rest.find(self.begin, cursor)
            if b < 0:
                break
            b += len(self.begin)
            e = rest.find(self.end, b)
            blob = rest[b:e] if e >= 0 else rest[b:]
            val = self._decode(blob)
            if isinstance(val, dict) and "name" in val:
                calls.append(_mk_call(val["name"], val.get("arguments", val.get("parameters", {})), idx))
                idx += 1
            elif isinstance(val, list):
                for item in val:
                    if isinstance(item, dict) and "name" in item:
                        calls.append(_mk_call(item["name"], item.get("arguments", item.get("parameters", {})), idx))
                        idx += 1
            if e < 0:
                break
            cursor = e + len(self.end)
        if self.section_end and self.section_end in rest:
            tail = rest.split(self.section_end, 1)[1]
            normal += tail
        return normal.strip(), calls


class PythonicParser(ToolParser):
    """`[func(a=1, b="x"), other()]` (pythonic.rs)."""

    name = "pythonic"

    def has_tool_markers(self, text):
        return bool(re.search(r"\[\s*\w+\s*\(", text))

    def parse(self, text, tools=None):
        m = re.search(r"\[\s*[\w\.]+\s*\(", text)
        if not m:
            return text, []
        start = m.start()
        try:
            tree = ast.parse(text[start:].strip(), mode="eval")
        except SyntaxError:
            # find balanced bracket
            depth = 0
            for i in range(start, len(text)):
                if text[i] == "[":
                    depth += 1
                elif text[i] == "]":
                    depth -= 1
                    if depth == 0:
                        try:
                            tree = ast.parse(text[start : i + 1], mode="eval")
                            break
                        except SyntaxError:
                            return text, []
            else:
                return text, []
        if not isinstance(tree.body, ast.List):
            return text, []
        calls = []
        for i, el in enumerate(tree.body.elts):
            if not isinstance(el, ast.Call):
                continue
            fname = el.func.id if isinstance(el.func, ast.Name) else ast.unparse(el.func)
            args = {}
            # positional args map onto the tool schema's parameter order
            # (pythonic.rs: models emit f(1, "x") against a known signature)
            if el.args:
                order = []
                for t in tools or []:
                    fn = t.get("function", t)
                    if fn.get("name") == fname:
                        order = list(((fn.get("parameters") or {}).get("properties") or {}).keys())
                        break
                for pi, pos_arg in enumerate(el.args):
                    key = order[pi] if pi < len(order) else f"arg{pi}"
                    try:
                        args[key] = ast.literal_eval(pos_arg)
                    except (ValueError, SyntaxError):
                        args[key] = ast.unparse(pos_arg)
            for kw in el.keywords:
                try:
                    args[kw.arg] = ast.literal_eval(kw.value)
                except (ValueError, SyntaxError):
                    args[kw.arg] = ast.unparse(kw.value)
            calls.append(_mk_call(fname, args, i))
        return text[:start].strip(), calls


class MistralParser(ToolParser):
    """`[TOOL_CALLS] [{...}, ...]` (mistral.rs:108)."""

    name = "mistral"
    BOT = "[TOOL_CALLS]"

    def has_tool_markers(self, text):
        return "[TOOL_CALLS]" in text

    def parse(self, text, tools=None):
        # the wire format is `[TOOL_CALLS][{...}]`; some templates emit a
        # space before the array — accept both (mistral.rs strips it too)
        pos = text.find(self.BOT)
        if pos < 0:
            return text, []
        arr_start = text.find("[", pos + len(self.BOT))
        if arr_start < 0:
            return text, []
        depth = 0
        in_str = False
        esc = False
        for i in range(arr_start, len(text)):
            c = text[i]
            if in_str:
                if esc:
                    esc = False
                elif c == "\\":
                    esc = True
                elif c == '"':
                    in_str = False
                continue
            if c == '"':
                in_str = True
            elif c == "[":
                depth += 1
            elif c == "]":
                depth -= 1
                if depth == 0:
                    blob = text[arr_start : i + 1]
                    try:
                        val = json.loads(blob)
                    except json.JSONDecodeError:
                        return text, []
                    calls = [
                        _mk_call(d["name"], d.get("arguments", {}), j)
                        for j, d in enumerate(val)
                        if isinstance(d, dict) and "name" in d
                    ]
                    return (text[:pos] + text[i + 1 :]).strip(), calls
        val, _ = parse_partial(text[arr_start:])
        if isinstance(val, list):
            calls = [
                _mk_call(d["name"], d.get("arguments", {}), j)
                for j, d in enumerate(val)
                if isinstance(d, dict) and "name" in d
            ]
            return text[:pos].strip(), calls
        return text, []


class LlamaParser(ToolParser):
    """`<|python_tag|>{json}` or bare JSON; ';' separated multiples (llama.rs)."""

    name = "llama"
    TAG = "<|python_tag|>"

    def has_tool_markers(self, text):
        return self.TAG in text or text.strip().startswith("{")

    def parse(self, text, tools=None):
        pos = text.find(self.TAG)
        if pos >= 0:
            normal, payload = text[:pos], text[pos + len(self.TAG):]
        elif text.strip().startswith("{"):
            normal, payload = "", text.strip()
        else:
            return text, []
        calls = []
        for i, part in enumerate(p for p in payload.split(";") if p.strip()):
            try:
                val = json.loads(part.strip())
            except json.JSONDecodeError:
                val, _ = parse_partial(part.strip())
            if isinstance(val, dict) and "name" in val:
                calls.append(_mk_call(val["name"], val.get("parameters", val.get("arguments", {})), i))
        if not calls:
            return text, []
        return normal.strip(), calls


class DeepSeekParser(ToolParser):
    """DeepSeek V3 unicode-token format (deepseek.rs:16):
    <｜tool▁calls▁begin｜><｜tool▁call▁begin｜>function<｜tool▁sep｜>{name}\\n```json\\n{args}\\n```<｜tool▁call▁end｜>...
    """

    name = "deepseek"
    SECTION = "<｜tool▁calls▁begin｜>"
    CALL_BEGIN = "<｜tool▁call▁begin｜>"
    CALL_END = "<｜tool▁call▁end｜>"
    SEP = "<｜tool▁sep｜>"

    def has_tool_markers(self, text):
        return self.SECTION in text or self.CALL_BEGIN in text

    def parse(self, text, tools=None):
        pos = text.find(self.SECTION)
        base = pos if pos >= 0 else text.find(self.CALL_BEGIN)
        if base < 0:
            return text, []
        normal = text[:base]
        calls = []
        cursor = base
        idx = 0
        while True:
            b = text.find(self.CALL_BEGIN, cursor)
            if b < 0:
                break
            seg_end = text.find(self.CALL_END, b)
            seg = text[b + len(self.CALL_BEGIN): seg_end if seg_end >= 0 else len(text)]
            if self.SEP in seg:
                _, after = seg.split(self.SEP, 1)
                name_part, _, rest = after.partition("\n")
                m = re.search(r"```(?:json)?\n(.*?)(?:\n```|$)", rest, re.S)
                blob = m.group(1) if m else rest
                try:
                    args = json.loads(blob)
                except json.JSONDecodeError:
                    args, _ = parse_partial(blob)
                    args = args if args is not None else {}
                calls.append(_mk_call(name_part.strip(), args, idx))
                idx += 1
            if seg_end < 0:
                break
            cursor = seg_end + len(self.CALL_END)
        return normal.strip(), calls


class KimiK2Parser(ToolParser):
    """`<|tool_calls_section_begin|><|tool_call_begin|>functions.{name}:{idx}
    <|tool_call_argument_begin|>{args}<|tool_call_end|>...` (kimik2.rs)."""

    name = "kimik2"
    SECTION = "<|tool_calls_section_begin|>"
    SECTION_END = "<|tool_calls_section_end|>"
    CALL = "<|tool_call_begin|>"
    ARG = "<|tool_call_argument_begin|>"
    END = "<|tool_call_end|>"

    def has_tool_markers(self, text):
        return self.SECTION in text or self.CALL in text

    def parse(self, text, tools=None):
        base = text.find(self.SECTION)
        if base < 0:
            base = text.find(self.CALL)
        if base < 0:
            return text, []
        normal = text[:base]
        calls = []
        cursor = base
        while True:
            b = text.find(self.CALL, cursor)
            if b < 0:
                break
            a = text.find(self.ARG, b)
            if a < 0:
                break
            fid = text[b + len(self.CALL): a].strip()
            e = text.find(self.END, a)
            blob = text[a + len(self.ARG): e if e >= 0 else len(text)]
            m = re.match(r"functions\.(.+?):(\d+)", fid)
            fname = m.group(1) if m else fid
            fidx = int(m.group(2)) if m else len(calls)
            try:
                args = json.loads(blob.strip())
            except json.JSONDecodeError:
                args, _ = parse_partial(blob.strip())
                args = args if args is not None else {}
            calls.append(_mk_call(fname, args, fidx))
            if e < 0:
                break
            cursor = e + len(self.END)
        if self.SECTION_END in text:
            normal += text.split(self.SECTION_END, 1)[1]
        return normal.strip(), calls


class Step3Parser(ToolParser):
    """step3 format (step3.rs:82-86): steptml invoke blocks inside
    <｜tool_calls_begin｜> sections."""

    name = "step3"
    BOT = "<｜tool_calls_begin｜>"
    EOT = "<｜tool_calls_end｜>"
    CB = "<｜tool_call_begin｜>"
    CE = "<｜tool_call_end｜>"
    SEP = "<｜tool_sep｜>"

    def has_tool_markers(self, text):
        return self.BOT in text

    def parse(self, text, tools=None):
        base = text.find(self.BOT)
        if base < 0:
            return text, []
        normal = text[:base]
        calls = []
        cursor = base
        idx = 0
        while True:
            b = text.find(self.CB, cursor)
            if b < 0:
                break
            e = text.find(self.CE, b)
            seg = text[b + len(self.CB): e if e >= 0 else len(text)]
            if self.SEP in seg:
                kind, _, rest = seg.partition(self.SEP)
                if "function" in kind:
                    inv = re.search(r"<steptml:invoke name=\"([^\"]+)\">(.*?)</steptml:invoke>", rest, re.S)
                    if inv:
                        args = {}
                        for pm in re.finditer(r"<steptml:parameter name=\"([^\"]+)\">(.*?)</steptml:parameter>", inv.group(2), re.S):
                            args[pm.group(1)] = _coerce_scalar(pm.group(2))
                        calls.append(_mk_call(inv.group(1), args, idx))
                        idx += 1
                    else:
                        name_part, _, blob = rest.partition("\n")
                        try:
                            args = json.loads(blob.strip() or "{}")
                        except json.JSONDecodeError:
                            args = {}
                        calls.append(_mk_call(name_part.strip(), args, idx))
                        idx += 1
            if e < 0:
                break
            cursor = e + len(self.CE)
        if self.EOT in text:
            normal += text.split(self.EOT, 1)[1]
        return normal.strip(), calls


class QwenXmlParser(ToolParser):
    """`<tool_call><function=name><parameter=key>value</parameter>...` (qwen_xml.rs)."""

    name = "qwen_xml"

    def has_tool_markers(self, text):
        return "<tool_call>" in text or "<function=" in text

    @staticmethod
    def _param_schema(tools, fname):
        for t in tools or []:
            fn = t.get("function", t)
            if fn.get("name") == fname:
                return (fn.get("parameters") or {}).get("properties") or {}
        return {}

    @classmethod
    def _typed_value(cls, raw: str, schema: Optional[dict]):
        """qwen_xml.rs semantics: the TOOL SCHEMA drives value typing —
        string params keep their text verbatim (one leading/trailing newline
        of tag formatting trimmed), object/array params parse as JSON, and
        numeric/bool coerce; no schema falls back to scalar sniffing."""
        txt = _xml_unescape(raw)
        # the conventional layout puts the value on its own line inside the tag
        if txt.startswith("\n"):
            txt = txt[1:]
        if txt.endswith("\n"):
            txt = txt[:-1]
        ptype = (schema or {}).get("type")
        if ptype == "string":
            return txt
        if ptype in ("object", "array"):
            try:
                return json.loads(txt)
            except json.JSONDecodeError:
                return txt
        if ptype == "integer":
            try:
                return int(txt.strip())
            except ValueError:
                return txt
        if ptype == "number":
            try:
                return float(txt.strip())
            except ValueError:
                return txt
        if ptype == "boolean":
            low = txt.strip().lower()
            if low in ("true", "false"):
                return low == "true"
            return txt
        return _coerce_scalar(txt)

    def parse(self, text, tools=None):
        pos = text.find("<tool_call>")
        if pos < 0:
            pos = text.find("<function=")
        if pos < 0:
            return text, []
        normal = text[:pos]
        calls = []
        # nesting-aware parameter scan: a </parameter> inside a nested value
        # (JSON braces containing tags, or a nested <parameter=...>) must not
        # terminate the outer parameter — split on parameter OPENINGS and
        # take everything up to the LAST closing tag before the next opening
        for i, fm in enumerate(re.finditer(r"<function=([^>]+)>(.*?)(?:</function>|$)", text[pos:], re.S)):
            fname = fm.group(1).strip()
            schema = self._param_schema(tools, fname)
            body = fm.group(2)
            args = {}
            opens = list(re.finditer(r"<parameter=([^>]+)>", body))
            for j, om in enumerate(opens):
                seg_end = opens[j + 1].start() if j + 1 < len(opens) else len(body)
                seg = body[om.end(): seg_end]
                close = seg.rfind("</parameter>")
                raw = seg[:close] if close >= 0 else seg
                key = om.group(1).strip()
                args[key] = self._typed_value(raw, schema.get(key))
            calls.append(_mk_call(fname, args, i))
        # trailing text after the last </tool_call> stays normal content
        tail_m = re.search(r"</tool_call>(?!.*</tool_call>)(.*)$", text[pos:], re.S)
        if tail_m and tail_m.group(1).strip():
            normal = (normal + " " + tail_m.group(1).strip()).strip()
        return normal.strip(), calls


class MinimaxM2Parser(ToolParser):
    """`<minimax:tool_call><invoke name="f"><parameter name="k">v</parameter>...`
    (minimax_m2.rs:104)."""

    name = "minimax_m2"
    START = "<minimax:tool_call>"
    END = "</minimax:tool_call>"

    def has_tool_markers(self, text):
        return self.START in text

    def parse(self, text, tools=None):
        pos = text.find(self.START)
        if pos < 0:
            return text, []
        normal = text[:pos]
        seg_end = text.find(self.END, pos)
        seg = text[pos: seg_end if seg_end >= 0 else len(text)]
        seg = _xml_unescape(seg)
        calls = []
        for i, im in enumerate(re.finditer(r"<invoke name=\"([^\"]+)\">(.*?)(?:</invoke>|$)", seg, re.S)):
            args = {}
            for pm in re.finditer(r"<parameter name=\"([^\"]+)\">(.*?)</parameter>", im.group(2), re.S):
                args[pm.group(1)] = _coerce_scalar(pm.group(2))
            calls.append(_mk_call(im.group(1), args, i))
        if seg_end >= 0:
            normal += text[seg_end + len(self.END):]
        return normal.strip(), calls


class CohereParser(ToolParser):
    """`<|START_ACTION|>[{"tool_name":..,"parameters":..}]<|END_ACTION|>` (cohere.rs)."""

    name = "cohere"

    def has_tool_markers(self, text):
        return "<|START_ACTION|>" in text

    def parse(self, text, tools=None):
        pos = text.find("<|START_ACTION|>")
        if pos < 0:
            normal = text
            for a, b in (("<|START_RESPONSE|>", "<|END_RESPONSE|>"), ("<|START_TEXT|>", "<|END_TEXT|>")):
                normal = normal.replace(a, "").replace(b, "")
            return normal.strip(), []
        end = text.find("<|END_ACTION|>", pos)
        blob = text[pos + len("<|START_ACTION|>"): end if end >= 0 else len(text)]
        try:
            val = json.loads(blob.strip())
        except json.JSONDecodeError:
            val, _ = parse_partial(blob.strip())
        calls = []
        if isinstance(val, dict):
            val = [val]
        if isinstance(val, list):
            for i, d in enumerate(val):
                if isinstance(d, dict) and ("tool_name" in d or "name" in d):
                    calls.append(_mk_call(d.get("tool_name") or d.get("name"), d.get("parameters", d.get("arguments", {})), i))
        normal = text[:pos]
        if end >= 0:
            normal += text[end + len("<|END_ACTION|>"):]
        for a, b in (("<|START_RESPONSE|>", "<|END_RESPONSE|>"), ("<|START_TEXT|>", "<|END_TEXT|>")):
            normal = normal.replace(a, "").replace(b, "")
        return normal.strip(), calls


class SarashinaParser(ToolParser):
    """`<|tool_calls|>[{'name': .., 'arguments': ..}]` python-literal list (sarashina.rs:40)."""

    name = "sarashina"
    MARK = "<|tool_calls|>"

    def has_tool_markers(self, text):
        return self.MARK in text

    def parse(self, text, tools=None):
        pos = text.find(self.MARK)
        if pos < 0:
            return text, []
        blob = text[pos + len(self.MARK):].strip()
        try:
            val = ast.literal_eval(blob)
        except (ValueError, SyntaxError):
            try:
                val = json.loads(blob)
            except json.JSONDecodeError:
                val, _ = parse_partial(blob)
        calls = []
        if isinstance(val, list):
            for i, d in enumerate(val):
                if isinstance(d, dict) and "name" in d:
                    calls.append(_mk_call(d["name"], d.get("arguments", {}), i))
        return text[:pos].strip(), calls


class KimiK3Parser(ToolParser):
    """`<|open|>tools<|sep|> ... <|close|>tools<|sep|>` sections (kimi_k3.rs:63-77)."""

    name = "kimi_k3"
    OPEN = "<|open|>tools<|sep|>"
    CLOSE = "<|close|>tools<|sep|>"

    def has_tool_markers(self, text):
        return self.OPEN in text

    def parse(self, text, tools=None):
        pos = text.find(self.OPEN)
        if pos < 0:
            return text, []
        end = text.find(self.CLOSE, pos)
        seg = text[pos + len(self.OPEN): end if end >= 0 else len(text)]
        calls = []
        idx = 0
        # entries: functions.name:idx{json} or bare {name,arguments} objects
        for m in re.finditer(r"functions\.([\w\.-]+):(\d+)\s*(\{.*?\})(?=\s*(?:functions\.|$))", seg, re.S):
            try:
                args = json.loads(m.group(3))
            except json.JSONDecodeError:
                args, _ = parse_partial(m.group(3))
                args = args if args is not None else {}
            calls.append(_mk_call(m.group(1), args, int(m.group(2))))
            idx += 1
        if not calls:
            val, _ = parse_partial(seg.strip())
            if isinstance(val, dict) and "name" in val:
                calls.append(_mk_call(val["name"], val.get("arguments", {}), 0))
            elif isinstance(val, list):
                for i, d in enumerate(val):
                    if isinstance(d, dict) and "name" in d:
                        calls.append(_mk_call(d["name"], d.get("arguments", {}), i))
        normal = text[:pos]
        if end >= 0:
            normal += text[end + len(self.CLOSE):]
        normal = re.sub(r"<\|(?:open|close)\|>\w+<\|sep\|>", "", normal)
        return normal.strip(), calls


class HarmonyToolParser(ToolParser):
    """gpt-oss Harmony commentary tool calls (reference grpc/harmony/parser.rs:
    messages with recipient `functions.NAME` on the commentary — or, model
    quirk, analysis — channel are tool calls; `final` text is content)."""

    name = "harmony"

    def has_tool_markers(self, text):
        return "to=functions." in text

    def parse(self, text, tools=None):
        from ...protocols.harmony import parse_complete as _hp

        res = _hp(text)
        calls = [
            _mk_call(tc["function"]["name"], tc["function"]["arguments"], i)
            for i, tc in enumerate(res.tool_calls)
        ]
        normal = res.content or ""
        return normal.strip(), calls


class InklingParser(ToolParser):
    """inkling channel format (inkling.rs:14-20): `<|content_invoke_tool_json|>
    {"tool": .., ...}<|end_message|>`."""

    name = "inkling"
    JSON_START = "<|content_invoke_tool_json|>"
    TEXT_START = "<|content_invoke_tool_text|>"
    END = "<|end_message|>"

    def has_tool_markers(self, text):
        return self.JSON_START in text or self.TEXT_START in text

    def parse(self, text, tools=None):
        pos = text.find(self.JSON_START)
        if pos < 0:
            normal = text
            for tag in (self.END, "<|message_model|>", "<|content_text|>", "<|content_thinking|>", "<|content_model_end_sampling|>"):
                normal = normal.replace(tag, "")
            return normal.strip(), []
        blob = text[pos + len(self.JSON_START):]
        if self.END in blob:
            blob = blob.split(self.END, 1)[0]
        try:
            val = json.loads(blob.strip())
        except json.JSONDecodeError:
            val, _ = parse_partial(blob.strip())
        calls = []
        if isinstance(val, dict):
            name = val.get("tool") or val.get("name")
            if name:
                args = {k: v for k, v in val.items() if k not in ("tool", "name")}
                args = val.get("arguments", args)
                calls.append(_mk_call(name, args, 0))
        normal = text[:pos]
        for tag in (self.END, "<|message_model|>", "<|content_text|>", "<|content_thinking|>", "<|content_model_end_sampling|>"):
            normal = normal.replace(tag, "")
        return normal.strip(), calls


def _xml_unescape(s: str) -> str:
    return (
        s.replace("&lt;", "<").replace("&gt;", ">").replace("&amp;", "&").replace("&quot;", '"').replace("&#39;", "'")
    )


def _coerce_scalar(s: str):
    s = s.strip()
    try:
        return json.loads(s)
    except (json.JSONDecodeError, ValueError):
        return s


# ---- registry (factory.rs:311-348) ---------------------------------------
def _build_registry() -> Dict[str, ToolParser]:
    qwen = TagToolParser("qwen", "<tool_call>", "</tool_call>")
    glm45 = TagToolParser("glm45_moe", "<tool_call>", "</tool_call>")
    return {
        "passthrough": PassthroughParser(),
        "json": JsonParser(),
        "qwen": qwen,
        "qwen_xml": QwenXmlParser(),
        "qwen_coder": QwenXmlParser(),
        "pythonic": PythonicParser(),
        "llama": LlamaParser(),
        "mistral": MistralParser(),
        "deepseek": DeepSeekParser(),
        "deepseek31": DeepSeekParser(),
        "deepseek32": DeepSeekParser(),
        "deepseek_v4": DeepSeekParser(),
        "glm45_moe": glm45,
        "glm47_moe": glm45,
        "step3": Step3Parser(),
        "sarashina": SarashinaParser(),
        "kimik2": KimiK2Parser(),
        "kimi_k3": KimiK3Parser(),
        "minimax_m2": MinimaxM2Parser(),
        "cohere": CohereParser(),
        "inkling": InklingParser(),
        "harmony": HarmonyToolParser(),
    }


PARSERS: Dict[str, ToolParser] = _build_registry()

# model-name -> parser auto-mapping (reference factory.rs:73 model_mapping)
MODEL_MAPPING = [
    (r"kimi-k2", "kimik2"),
    (r"kimi-k3", "kimi_k3"),
    (r"deepseek-v3\.?2", "deepseek32"),
    (r"deepseek-v4", "deepseek_v4"),
    (r"deepseek", "deepseek"),
    (r"qwen3.*coder", "qwen_coder"),
    (r"qwen", "qwen"),
    (r"glm-?4\.?5", "glm45_moe"),
    (r"glm-?4\.?7", "glm47_moe"),
    (r"llama", "llama"),
    (r"mistral|mixtral", "mistral"),
    (r"minimax", "minimax_m2"),
    (r"step-?3", "step3"),
    (r"command|cohere", "cohere"),
    (r"sarashina", "sarashina"),
    (r"gpt[-_]?oss", "harmony"),
]


def get_parser(name_or_model: Optional[str]) -> ToolParser:
    if not name_or_model:
        return PARSERS["passthrough"]
    if name_or_model in PARSERS:
        return PARSERS[name_or_model]
    low = name_or_model.lower()
    for pat, pname in MODEL_MAPPING:
        if re.search(pat, low):
            return PARSERS[pname]
    return PARSERS["passthrough"]


def parse_complete(name: str, text: str, tools: Optional[List[dict]] = None) -> Tuple[str, List[ToolCall]]:
    return get_parser(name).parse(text, tools)
