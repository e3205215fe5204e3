"""Model-specific chat encoders (reference: crates/tokenizer/src/encoders/ —
deepseek_v32.rs (DSML function-call format, itself mirroring DeepSeek's
public encoding_dsv32.py), deepseek_v4.rs, kimi_k25_tools.rs, kimi_k3_xtml.rs).

These are chat "templates" too stateful for jinja: rendering depends on the
position of the last user turn, tool-call ordering, and the thinking mode.
The encoder is the inverse of the corresponding tool/reasoning parsers.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Optional


class EncodingError(ValueError):
    pass


# ---- DeepSeek V3.2 (DSML) ---------------------------------------------------
BOS_TOKEN = "<｜begin▁of▁sentence｜>"
EOS_TOKEN = "<｜end▁of▁sentence｜>"
THINK_START = "<think>"
THINK_END = "</think>"
DSML = "｜DSML｜"
_USER = "<｜User｜>"
_ASSISTANT = "<｜Assistant｜>"


def _to_json(v: Any) -> str:
    return json.dumps(v, ensure_ascii=False, separators=(", ", ": "))


def _tools_block(tools: List[Dict]) -> str:
    # prompt text fixed by the model's public encoding spec
    schemas = "\n".join(_to_json(t) for t in tools)
    return (
        "## Tools\n\n"
        "You have access to a set of tools you can use to answer the user's question.\n"
        f'You can invoke functions by writing a "<{DSML}function_calls>" block like the '
        "following as part of your reply to the user:\n"
        f"<{DSML}function_calls>\n"
        f'<{DSML}invoke name="$FUNCTION_NAME">\n'
        f'<{DSML}parameter name="$PARAMETER_NAME" string="true|false">$PARAMETER_VALUE</{DSML}parameter>\n'
        "...\n"
        f"</{DSML}invoke>\n"
        f'<{DSML}invoke name="$FUNCTION_NAME2">\n'
        "...\n"
        f"</{DSML}invoke>\n"
        f"</{DSML}function_calls>\n\n"
        "String and scalar parameters should be specified as is without any escaping or quotes, "
        'while lists and objects should use JSON format. The "string" attribute should be set to '
        '"true" for string type parameters and "false" for other types (numbers, booleans, arrays, '
        "objects).\n\n"
        "Here are the functions available in JSONSchema format:\n"
        f"{schemas}"
    )


def _response_format_block(schema: str) -> str:
    return f"## Response Format\n\nYou must reply with a JSON object matching this schema:\n{schema}"


def _openai_tool(t: Dict) -> Dict:
    return t.get("function", t) if isinstance(t, dict) else t


def _openai_tool_call(tc: Dict) -> Dict:
    fn = tc.get("function", tc)
    return {"name": fn.get("name", ""), "arguments": fn.get("arguments", {})}


def _dsml_arguments(tc: Dict) -> str:
    args = tc.get("arguments")
    if isinstance(args, str):
        try:
            args = json.loads(args)
        except json.JSONDecodeError as e:
            raise EncodingError(f"tool arguments are not valid JSON: {e}")
    if not isinstance(args, dict):
        return ""
    parts = []
    for k, v in args.items():
        if isinstance(v, str):
            is_str, sv = "true", v
        else:
            is_str, sv = "false", _to_json(v)
        parts.append(f'<{DSML}parameter name="{k}" string="{is_str}">{sv}</{DSML}parameter>')
    return "\n".join(parts)


def _last_user_index(messages: List[Dict]) -> int:
    for i in range(len(messages) - 1, -1, -1):
        if messages[i].get("role") in ("user", "developer"):
            return i
    return -1


def _drop_thinking(messages: List[Dict]) -> List[Dict]:
    """Strip reasoning_content from assistant turns before the last user turn."""
    last_user = _last_user_index(messages)
    out = []
    for i, msg in enumerate(messages):
        role = msg.get("role", "")
        if role in ("user", "system", "tool") or (last_user < 0 or i >= last_user):
            out.append(msg)
        elif role == "assistant":
            m = dict(msg)
            m.pop("reasoning_content", None)
            out.append(m)
    return out


def _render_ds32(index: int, messages: List[Dict], thinking: bool) -> str:
    msg = messages[index]
    last_user = _last_user_index(messages)
    role = msg.get("role", "")
    content = msg.get("content") or ""
    tools = [_openai_tool(t) for t in (msg.get("tools") or [])]
    tool_calls = [_openai_tool_call(tc) for tc in (msg.get("tool_calls") or [])]
    reasoning = msg.get("reasoning_content") or ""
    rf = msg.get("response_format")
    p = []

    if role == "system":
        p.append(content)
        if tools:
            p.append("\n\n" + _tools_block(tools))
        if rf is not None:
            p.append("\n\n" + _response_format_block(_to_json(rf)))
    elif role in ("user", "developer"):
        if role == "developer":
            if not content:
                raise EncodingError(f"developer message requires content: {msg}")
            body = ""
            if tools:
                body += "\n\n" + _tools_block(tools)
            if rf is not None:
                body += "\n\n" + _response_format_block(_to_json(rf))
            body += f"\n\n# The user's message is: {content}"
            p.append(f"{_USER}{body}{_ASSISTANT}")
        else:
            p.append(f"{_USER}{content}{_ASSISTANT}")
        p.append(THINK_START if (index == last_user and thinking) else THINK_END)
    elif role == "tool":
        # anchor on the assistant turn that issued the calls
        prev = index - 1
        while prev >= 0 and messages[prev].get("role") == "tool":
            prev -= 1
        if index != 0 and (prev < 0 or messages[prev].get("role") != "assistant"):
            raise EncodingError(f"tool message at {index} has no assistant anchor")
        n_calls = len(messages[prev].get("tool_calls") or []) if prev >= 0 else 0
        order = index - prev
        if n_calls == 0 or n_calls < order:
            raise EncodingError("no tool calls but found tool output")
        if order == 1:
            p.append("\n\n<function_results>")
        p.append(f"\n<result>{content}</result>")
        if order == n_calls:
            p.append("\n</function_results>")
            p.append("\n\n" + (THINK_START if (last_user < 0 or index >= last_user) and thinking else THINK_END))
    elif role == "assistant":
        calls_block = ""
        if tool_calls:
            rendered = "\n".join(
                f'<{DSML}invoke name="{tc["name"]}">\n{_dsml_arguments(tc)}\n</{DSML}invoke>'
                for tc in tool_calls
            )
            calls_block = f"\n\n<{DSML}function_calls>\n{rendered}\n</{DSML}function_calls>"
        think_part = ""
        if thinking and (last_user < 0 or index > last_user):
            if not reasoning and not tool_calls:
                raise EncodingError(f"thinking-mode assistant turn needs reasoning or tool calls: {msg}")
            think_part = reasoning + THINK_END
        p.append(think_part + content + calls_block + EOS_TOKEN)
    else:
        raise EncodingError(f"unknown role {role!r}")
    return "".join(p)


def encode_deepseek_v32(
    messages: List[Dict[str, Any]],
    thinking_mode: str = "thinking",
    add_default_bos: bool = True,
    drop_thinking: bool = True,
    tools: Optional[List[Dict]] = None,
) -> str:
    """OpenAI-style messages -> DeepSeek V3.2 DSML prompt string
    (deepseek_v32.rs encode_messages).  Request-level `tools` are attached to
    the first system message (or an implicit empty one)."""
    if thinking_mode not in ("thinking", "chat"):
        raise EncodingError(f"invalid thinking_mode {thinking_mode!r}")
    thinking = thinking_mode == "thinking"
    msgs = [dict(m) for m in messages]
    if tools:
        if msgs and msgs[0].get("role") == "system":
            msgs[0].setdefault("tools", tools)
        else:
            msgs.insert(0, {"role": "system", "content": "", "tools": tools})
    if thinking and drop_thinking:
        msgs = _drop_thinking(msgs)
    prompt = BOS_TOKEN if add_default_bos else ""
    for i in range(len(msgs)):
        prompt += _render_ds32(i, msgs, thinking)
    return prompt


# ---- Kimi K2.5 tool sections (kimi_k25_tools.rs) ----------------------------
K25_SECTION_BEGIN = "<|tool_calls_section_begin|>"
K25_SECTION_END = "<|tool_calls_section_end|>"
K25_CALL_BEGIN = "<|tool_call_begin|>"
K25_CALL_ARG = "<|tool_call_argument_begin|>"
K25_CALL_END = "<|tool_call_end|>"


def encode_kimi_k25_tool_calls(tool_calls: List[Dict[str, Any]], start_index: int = 0) -> str:
    """Assistant tool calls -> Kimi K2.5 section format: each call is
    `functions.{name}:{counter}` followed by the JSON arguments (the inverse
    of the kimik2 tool parser)."""
    parts = [K25_SECTION_BEGIN]
    for i, tc in enumerate(tool_calls, start=start_index):
        fn = tc.get("function", tc)
        args = fn.get("arguments", "{}")
        if not isinstance(args, str):
            args = _to_json(args)
        parts.append(f"{K25_CALL_BEGIN}functions.{fn.get('name')}:{i}{K25_CALL_ARG}{args}{K25_CALL_END}")
    parts.append(K25_SECTION_END)
    return "".join(parts)


# ---- Kimi K3 XTML chat template (kimi_k3_xtml.rs) ---------------------------
XT_OPEN = "<|open|>"
XT_CLOSE = "<|close|>"
XT_SEP = "<|sep|>"
XT_EOM = "<|end_of_msg|>"
XT_IMAGE = "<|kimi_image_placeholder|>"
K3_EFFORTS = ("low", "high", "max")


def _xt_escape(v: str) -> str:
    return v.replace("&", "&amp;").replace('"', "&quot;")


def _xt_open(tag: str, attrs: Optional[List[tuple]] = None) -> str:
    s = XT_OPEN + tag
    for k, v in attrs or []:
        s += f' {k}="{_xt_escape(str(v))}"'
    return s + XT_SEP


def _xt_close(tag: str) -> str:
    return XT_CLOSE + tag + XT_SEP


def _xt_content(content: Any) -> str:
    if isinstance(content, str):
        return content
    if isinstance(content, list):
        out = []
        for p in content:
            if not isinstance(p, dict):
                continue
            if p.get("type") in ("image", "image_url"):
                out.append(XT_IMAGE)
            elif isinstance(p.get("text"), str):
                out.append(p["text"])
        return "".join(out)
    return ""


def _xt_type(v: Any) -> str:
    if isinstance(v, bool):
        return "boolean"
    if v is None:
        return "null"
    if isinstance(v, (int, float)):
        return "number"
    if isinstance(v, str):
        return "string"
    if isinstance(v, dict):
        return "object"
    return "array"


def _xt_value(v: Any) -> str:
    return v if isinstance(v, str) else _to_json(v)


def _xt_internal_system(kind: str, body: str) -> str:
    return (
        _xt_open("message", [("role", "system"), ("type", kind)])
        + body.strip()
        + _xt_close("message")
        + XT_EOM
    )


def _xt_assistant_segments(msg: Dict, thinking: bool) -> str:
    out = []
    if thinking:
        # the think channel is structural: open/close even when empty
        reasoning = msg.get("reasoning_content") or msg.get("reasoning") or ""
        out.append(_xt_open("think") + (reasoning if str(reasoning).strip() else "") + _xt_close("think"))
    out.append(_xt_open("response") + _xt_content(msg.get("content")) + _xt_close("response"))
    tool_calls = msg.get("tool_calls") or []
    if tool_calls:
        out.append(_xt_open("tools"))
        for i, tc in enumerate(tool_calls, start=1):
            fn = tc.get("function", tc) if isinstance(tc.get("function"), dict) else tc
            name = fn.get("name")
            if not name:
                raise EncodingError("Kimi K3 tool call is missing a function name")
            out.append(_xt_open("call", [("tool", name), ("index", i)]))
            args = fn.get("arguments")
            if isinstance(args, str):
                try:
                    args = json.loads(args)
                except json.JSONDecodeError:
                    out.append(_xt_open("json", [("type", "object")]) + fn["arguments"] + _xt_close("json"))
                    args = None
            if isinstance(args, dict):
                for k, v in args.items():
                    out.append(
                        _xt_open("argument", [("key", k), ("type", _xt_type(v))]) + _xt_value(v) + _xt_close("argument")
                    )
            out.append(_xt_close("call"))
        out.append(_xt_close("tools"))
    return "".join(out)


def encode_kimi_k3_xtml(
    messages: List[Dict[str, Any]],
    tools: Optional[List[Dict]] = None,
    thinking: bool = True,
    thinking_effort: Optional[str] = None,
    tool_choice: Optional[str] = None,
    add_generation_prompt: bool = True,
) -> str:
    """OpenAI-style messages -> Kimi K3 XTML prompt (kimi_k3_xtml.rs
    render_xtml): <|open|>tag attrs<|sep|>…<|close|>tag<|sep|> framing,
    tool-declare + thinking-effort internal system messages, think/response/
    tools assistant channels."""
    if thinking_effort is not None and thinking_effort not in K3_EFFORTS:
        raise EncodingError(f"unsupported thinking_effort {thinking_effort!r}; supported: {list(K3_EFFORTS)}")
    out = []
    if tools:
        body = (
            "# Tools\nHere are the available tools, described in JSONSchema.\n\n"
            f"```json\n{json.dumps(tools, ensure_ascii=False, separators=(',', ':'), sort_keys=True)}\n```"
        )
        out.append(_xt_internal_system("tool-declare", body))
    if thinking and thinking_effort:
        out.append(
            _xt_internal_system(
                "thinking-effort",
                "`thinking_effort` guides on how much to think in your thinking channel "
                "(not including the response channel), supported values include `low`, "
                f"`medium`, `high`, and `max`.\nNow the system is invoked with `thinking_effort={thinking_effort}`.",
            )
        )
    current_tool_calls: List[Dict] = []
    tool_index = 0
    for msg in messages:
        role = msg.get("role", "")
        if role in ("user", "system"):
            attrs = [("role", role)]
            if msg.get("name"):
                attrs.append(("name", msg["name"]))
            out.append(_xt_open("message", attrs) + _xt_content(msg.get("content")) + _xt_close("message") + XT_EOM)
        elif role == "tool":
            tool_index += 1
            name = msg.get("tool") or msg.get("name")
            if not name and tool_index <= len(current_tool_calls):
                tc = current_tool_calls[tool_index - 1]
                fn = tc.get("function", tc) if isinstance(tc.get("function"), dict) else tc
                name = fn.get("name")
            if not name:
                raise EncodingError("Kimi K3 tool messages need a resolvable tool name")
            out.append(
                _xt_open("message", [("role", "tool"), ("tool", name), ("index", tool_index)])
                + _xt_content(msg.get("content"))
                + _xt_close("message")
                + XT_EOM
            )
        elif role == "assistant":
            current_tool_calls = msg.get("tool_calls") or []
            tool_index = 0
            attrs = [("role", "assistant")]
            if msg.get("name"):
                attrs.append(("name", msg["name"]))
            out.append(
                _xt_open("message", attrs) + _xt_assistant_segments(msg, thinking) + _xt_close("message") + XT_EOM
            )
        # unknown roles render nothing (kimi_k3_xtml.rs loop)
    if tool_choice == "required":
        out.append(
            _xt_internal_system(
                "tool-choice",
                "The system is invoked with `tool_choice=required`.\nYou MUST call tools in the next message.",
            )
        )
    elif tool_choice == "none":
        out.append(
            _xt_internal_system(
                "tool-choice",
                "The system is invoked with `tool_choice=none`.\nYou MUST NOT call any tools in the next message.",
            )
        )
    if add_generation_prompt:
        out.append(_xt_open("message", [("role", "assistant")]))
        out.append(_xt_open("think" if thinking else "response"))
    return "".join(out)


ENCODERS = {
    "deepseek_v32": encode_deepseek_v32,
    "deepseek_v4": encode_deepseek_v32,  # v4 shares the DSML surface (deepseek_v4.rs)
    "kimi_k3_xtml": encode_kimi_k3_xtml,
}


def get_encoder(name_or_model: Optional[str]):
    if not name_or_model:
        return None
    low = name_or_model.lower()
    if low in ENCODERS:
        return ENCODERS[low]
    if "deepseek-v3.2" in low or "deepseek_v32" in low or "deepseek-v4" in low:
        return encode_deepseek_v32
    if "kimi-k3" in low or "kimi_k3" in low:
        return encode_kimi_k3_xtml
    return None
