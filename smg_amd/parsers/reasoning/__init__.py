"""Reasoning-content parsers (reference: crates/reasoning_parser — trait
ReasoningParser traits.rs:45; 16 registered variants, factory.rs:116-197).

One parameterized machine covers the family: (think_start, think_end,
always_in_reasoning, stream_reasoning).  Streaming keeps partial-marker
bytes buffered so split tags across chunks parse correctly.
"""
from .parsers import (
    PARSERS,
    ReasoningParser,
    get_reasoning_parser,
    parse_reasoning_complete,
)

__all__ = ["PARSERS", "ReasoningParser", "get_reasoning_parser", "parse_reasoning_complete"]
