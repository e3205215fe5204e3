"""Tool-call parsers (reference: crates/tool_parser — 21 registered names,
factory.rs:311, trait ToolParser traits.rs:11).

Each parser extracts structured tool calls from model output text, in both
complete and streaming modes.  The streaming machines are incremental FSMs
over the delta text (see stream.py); registration mirrors the reference's
factory names.
"""
from .factory import PARSERS, get_parser, parse_complete

__all__ = ["PARSERS", "get_parser", "parse_complete"]
