"""Incremental/partial JSON parsing (reference: crates/tool_parser/src/partial_json.rs).

`parse_partial(s)` parses as much of a JSON document as exists, completing
unterminated strings/arrays/objects, and returns (value, consumed_chars).
Used by streaming tool parsers to surface argument fragments before the
closing brace arrives.
"""
from __future__ import annotations

import json
from typing import Any, Optional, Tuple


def parse_partial(s: str, allow_partial_strings: bool = True) -> Tuple[Optional[Any], int]:
    s = s.strip()
    if not s:
        return None, 0
    try:
        return json.loads(s), len(s)
    except json.JSONDecodeError:
        pass
    completed = _complete(s, allow_partial_strings)
    if completed is None:
        return None, 0
    try:
        return json.loads(completed), len(s)
    except json.JSONDecodeError:
        return None, 0


def _complete(s: str, allow_partial_strings: bool) -> Optional[str]:
    stack = []
    in_str = False
    escape = False
    last_sig = 0
    for i, c in enumerate(s):
        if in_str:
            if escape:
                escape = False
            elif c == "\\":
                escape = True
            elif c == '"':
                in_str = False
                last_sig = i
            continue
        if c == '"':
            in_str = True
        elif c in "{[":
            stack.append(c)
        elif c in "}]":
            if not stack:
                return None
            stack.pop()
            last_sig = i
        elif not c.isspace():
            last_sig = i
    out = s
    if in_str:
        if not allow_partial_strings:
            out = out[: out.rfind('"')]
            if not out:
                return None
        else:
            if escape:
                out = out[:-1]
            out += '"'
    # drop trailing comma/colon fragments; a stripped ':' leaves a dangling
    # key ("...,"arguments"") which must be dropped too
    trimmed = out.rstrip()
    while trimmed and trimmed[-1] in ",:":
        had_colon = trimmed[-1] == ":"
        trimmed = trimmed[:-1].rstrip()
        if had_colon and trimmed.endswith('"'):
            close = trimmed.rfind('"', 0, len(trimmed) - 1)
            if close >= 0:
                trimmed = trimmed[:close].rstrip()
                if trimmed and trimmed[-1] == ",":
                    trimmed = trimmed[:-1].rstrip()
    out = trimmed
    for b in reversed(stack):
        out += "}" if b == "{" else "]"
    return out


def is_complete_json(s: str) -> bool:
    try:
        json.loads(s)
        return True
    except json.JSONDecodeError:
        return False
