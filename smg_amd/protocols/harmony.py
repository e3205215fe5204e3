"""Harmony (gpt-oss) channel format: builder + streaming parser (reference:
model_gateway/src/routers/grpc/harmony/{builder,parser,streaming}.rs — the
reference renders via the openai-harmony crate; the wire format itself is the
public gpt-oss one and is re-implemented here at the text level).

Rendered form (one message):
    <|start|>{role}[<|channel|>{channel}][ to={recipient}]<|message|>{content}<|end|>
Assistant output channels: `analysis` (reasoning), `commentary` (tool calls,
recipient `functions.NAME`, terminated by <|call|>), `final` (user-visible
text, terminated by <|return|> or <|end|>).  The parser mirrors
parser.rs:98-220 semantics: analysis -> reasoning_content, commentary (or
analysis) with a functions.* recipient -> tool_calls, final -> content.
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

START = "<|start|>"
CHANNEL = "<|channel|>"
MESSAGE = "<|message|>"
END = "<|end|>"
CALL = "<|call|>"
RETURN = "<|return|>"
_SPECIALS = (END, CALL, RETURN)


# ---- builder (builder.rs) ---------------------------------------------------
def _msg(role: str, content: str, channel: Optional[str] = None, recipient: Optional[str] = None) -> str:
    head = role
    if channel:
        head += f"{CHANNEL}{channel}"
    if recipient:
        head += f" to={recipient}"
    return f"{START}{head}{MESSAGE}{content}{END}"


def render_system(reasoning_effort: str = "medium", tools: Optional[List[Dict]] = None) -> str:
    lines = [
        "You are ChatGPT, a large language model trained by OpenAI.",
        f"Reasoning: {reasoning_effort}",
        "# Valid channels: analysis, commentary, final. Channel must be included for every message.",
    ]
    if tools:
        lines.append("Calls to these tools must go to the commentary channel: 'functions'.")
    return _msg("system", "\n".join(lines))


def render_developer(instructions: Optional[str], tools: Optional[List[Dict]] = None) -> Optional[str]:
    parts = []
    if instructions:
        parts.append(f"# Instructions\n\n{instructions}")
    if tools:
        decls = []
        for t in tools:
            fn = t.get("function", t)
            decls.append(
                f"// {fn.get('description', '')}\ntype {fn.get('name')} = "
                f"(_: {json.dumps(fn.get('parameters', {}))}) => any;"
            )
        parts.append("# Tools\n\n## functions\n\nnamespace functions {\n\n" + "\n\n".join(decls) + "\n\n} // namespace functions")
    if not parts:
        return None
    return _msg("developer", "\n\n".join(parts))


def build_harmony_prompt(body: Dict[str, Any]) -> str:
    """Chat-completion body -> Harmony conversation text ending with the
    assistant generation prefix (builder.rs encode path)."""
    out = [render_system(body.get("reasoning_effort", "medium"), body.get("tools"))]
    system_texts = [m.get("content") or "" for m in body.get("messages", []) if m.get("role") in ("system", "developer")]
    dev = render_developer("\n\n".join(t for t in system_texts if t) or None, body.get("tools"))
    if dev:
        out.append(dev)
    for m in body.get("messages", []):
        role = m.get("role")
        if role in ("system", "developer"):
            continue  # folded into the developer message above
        content = m.get("content")
        if isinstance(content, list):
            content = "".join(p.get("text", "") for p in content if isinstance(p, dict))
        if role == "assistant":
            for tc in m.get("tool_calls") or []:
                fn = tc.get("function", {})
                out.append(_msg("assistant", fn.get("arguments") or "{}", channel="commentary",
                                recipient=f"functions.{fn.get('name')}"))
            if content:
                out.append(_msg("assistant", content, channel="final"))
        elif role == "tool":
            name = m.get("name") or "tool"
            out.append(_msg(f"functions.{name} to=assistant", content or "", channel="commentary"))
        else:
            out.append(_msg(role or "user", content or ""))
    out.append(f"{START}assistant")
    return "".join(out)


# ---- parser -----------------------------------------------------------------
@dataclass
class HarmonyMessage:
    channel: Optional[str]
    recipient: Optional[str]
    content: str


@dataclass
class HarmonyResult:
    reasoning_content: Optional[str] = None
    content: Optional[str] = None
    tool_calls: List[Dict[str, Any]] = field(default_factory=list)


_HEAD_RE = re.compile(
    r"(?:(?P<role>[^<\s]+))?"
    r"(?:<\|channel\|>(?P<channel>[^<\s]+))?"
    r"(?:\s+to=(?P<recipient>[^<\s]+))?"
)


def _parse_head(head: str):
    m = _HEAD_RE.fullmatch(head.strip())
    if not m:
        return None, None
    return m.group("channel"), m.group("recipient")


def split_messages(text: str) -> List[HarmonyMessage]:
    """Raw assistant output -> messages.  Accepts output that starts mid-message
    (the prompt already ended with `<|start|>assistant`)."""
    msgs: List[HarmonyMessage] = []
    for chunk in text.split(START):
        if not chunk:
            continue
        head, sep, body = chunk.partition(MESSAGE)
        if not sep:
            head, body = "", chunk
        channel, recipient = _parse_head(head)
        # a terminator ends the message; trailing text (no <|start|> framing)
        # is a new channel-less plain-content message
        cut = None
        for sp in _SPECIALS:
            i = body.find(sp)
            if i >= 0 and (cut is None or i < cut[0]):
                cut = (i, sp)
        if cut is not None:
            tail = body[cut[0] + len(cut[1]):]
            body = body[: cut[0]]
            msgs.append(HarmonyMessage(channel, recipient, body))
            if tail:
                for sp in _SPECIALS:
                    tail = tail.replace(sp, "")
                if tail:
                    msgs.append(HarmonyMessage(None, None, tail))
            continue
        msgs.append(HarmonyMessage(channel, recipient, body))
    return msgs


def parse_complete(text: str) -> HarmonyResult:
    """parser.rs:98-220: analysis -> reasoning; commentary/analysis with a
    functions.* recipient -> tool call; final (or channel-less) -> content."""
    res = HarmonyResult()
    for msg in split_messages(text):
        if msg.recipient and msg.recipient.startswith("functions."):
            res.tool_calls.append(
                {
                    "id": f"call_h{len(res.tool_calls)}",
                    "type": "function",
                    "function": {"name": msg.recipient[len("functions."):], "arguments": msg.content.strip() or "{}"},
                }
            )
        elif msg.channel == "analysis":
            res.reasoning_content = (res.reasoning_content or "") + msg.content
        elif msg.channel == "commentary":
            # commentary without a tool recipient is treated as reasoning
            res.reasoning_content = (res.reasoning_content or "") + msg.content
        else:  # final or channel-less
            res.content = (res.content or "") + msg.content
    return res


class HarmonyStreamParser:
    """Incremental variant (streaming.rs): feed text deltas, get typed events
    [{"type": "reasoning"|"content"|"tool_call_start"|"tool_call_args", ...}].
    Holds back partial special tokens at the buffer tail."""

    def __init__(self):
        self._buf = ""
        self._channel: Optional[str] = None
        self._recipient: Optional[str] = None
        self._in_message = False
        self._tool_idx = -1

    def _emit(self, text: str) -> List[Dict[str, Any]]:
        if not text:
            return []
        if self._recipient and self._recipient.startswith("functions."):
            return [{"type": "tool_call_args", "index": self._tool_idx, "arguments": text}]
        if self._channel in ("analysis", "commentary"):
            return [{"type": "reasoning", "text": text}]
        return [{"type": "content", "text": text}]

    def feed(self, delta: str) -> List[Dict[str, Any]]:
        self._buf += delta
        events: List[Dict[str, Any]] = []
        while self._buf:
            if not self._in_message:
                # consuming a header: wait for <|message|>
                idx = self._buf.find(MESSAGE)
                if idx < 0:
                    # drop a complete leading <|start|> but keep partial header
                    if len(self._buf) > 256 and START not in self._buf and MESSAGE[0] not in self._buf:
                        self._buf = ""
                    return events
                head = self._buf[:idx].replace(START, "").replace("assistant", "", 1)
                self._channel, self._recipient = _parse_head(head)
                if self._recipient and self._recipient.startswith("functions."):
                    self._tool_idx += 1
                    events.append(
                        {
                            "type": "tool_call_start",
                            "index": self._tool_idx,
                            "name": self._recipient[len("functions."):],
                        }
                    )
                self._buf = self._buf[idx + len(MESSAGE):]
                self._in_message = True
                continue
            # inside a message: emit up to the next special token
            cut = None
            for sp in _SPECIALS:
                i = self._buf.find(sp)
                if i >= 0 and (cut is None or i < cut[0]):
                    cut = (i, sp)
            if cut is not None:
                events.extend(self._emit(self._buf[: cut[0]]))
                if self._recipient and self._recipient.startswith("functions."):
                    events.append({"type": "tool_call_end", "index": self._tool_idx})
                self._buf = self._buf[cut[0] + len(cut[1]):]
                self._in_message = False
                self._channel = self._recipient = None
                continue
            # hold back anything that could be the start of a special token
            safe = len(self._buf)
            for k in range(1, min(12, len(self._buf)) + 1):
                tail = self._buf[-k:]
                if any(sp.startswith(tail) for sp in (*_SPECIALS, START)):
                    safe = len(self._buf) - k
            events.extend(self._emit(self._buf[:safe]))
            self._buf = self._buf[safe:]
            return events
        return events

    def finalize(self) -> List[Dict[str, Any]]:
        events = self._emit(self._buf) if self._in_message else []
        self._buf = ""
        return events
