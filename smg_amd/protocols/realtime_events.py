"""Realtime API event types (reference: crates/protocols/src/
realtime_events.rs — ClientEvent :36 / ServerEvent :186 tagged unions).

The WS relay uses this registry to validate, classify and construct realtime
events: every event is a JSON object whose `type` field selects the variant;
REQUIRED_FIELDS carries each variant's mandatory payload keys (from the
reference's non-Option struct fields).  parse_event() raises on unknown types
or missing fields; make_event() builds well-formed server events."""
from __future__ import annotations

import uuid
from typing import Any, Dict, List, Optional, Tuple

# variant -> required payload fields (beyond `type`)
CLIENT_EVENTS: Dict[str, Tuple[str, ...]] = {
    "session.update": ("session",),
    "conversation.item.create": ("item",),
    "conversation.item.delete": ("item_id",),
    "conversation.item.retrieve": ("item_id",),
    "conversation.item.truncate": ("audio_end_ms", "content_index", "item_id"),
    "input_audio_buffer.append": ("audio",),
    "input_audio_buffer.clear": (),
    "input_audio_buffer.commit": (),
    "output_audio_buffer.clear": (),
    "response.cancel": (),
    "response.create": (),
}

SERVER_EVENTS: Dict[str, Tuple[str, ...]] = {
    "session.created": ("session",),
    "session.updated": ("session",),
    "conversation.created": ("conversation",),
    "conversation.item.created": ("item",),
    "conversation.item.added": ("item",),
    "conversation.item.done": ("item",),
    "conversation.item.deleted": ("item_id",),
    "conversation.item.retrieved": ("item",),
    "conversation.item.truncated": ("item_id", "audio_end_ms", "content_index"),
    "conversation.item.input_audio_transcription.completed": ("item_id", "transcript"),
    "conversation.item.input_audio_transcription.delta": ("item_id",),
    "conversation.item.input_audio_transcription.failed": ("item_id", "error"),
    "conversation.item.input_audio_transcription.segment": ("item_id",),
    "input_audio_buffer.cleared": (),
    "input_audio_buffer.committed": ("item_id",),
    "input_audio_buffer.speech_started": ("audio_start_ms", "item_id"),
    "input_audio_buffer.speech_stopped": ("audio_end_ms", "item_id"),
    "input_audio_buffer.timeout_triggered": ("audio_start_ms", "audio_end_ms", "item_id"),
    "input_audio_buffer.dtmf_event_received": (),
    "output_audio_buffer.started": ("response_id",),
    "output_audio_buffer.stopped": ("response_id",),
    "output_audio_buffer.cleared": ("response_id",),
    "response.created": ("response",),
    "response.done": ("response",),
    "response.output_item.added": ("output_index", "response_id", "item"),
    "response.output_item.done": ("output_index", "response_id", "item"),
    "response.content_part.added": ("content_index", "item_id", "output_index", "part", "response_id"),
    "response.content_part.done": ("content_index", "item_id", "output_index", "part", "response_id"),
    "response.output_text.delta": ("content_index", "delta", "item_id", "output_index", "response_id"),
    "response.output_text.done": ("content_index", "item_id", "output_index", "response_id", "text"),
    "response.output_audio.delta": ("content_index", "delta", "item_id", "output_index", "response_id"),
    "response.output_audio.done": ("content_index", "item_id", "output_index", "response_id"),
    "response.output_audio_transcript.delta": ("content_index", "delta", "item_id", "output_index", "response_id"),
    "response.output_audio_transcript.done": ("content_index", "item_id", "output_index", "response_id", "transcript"),
    "response.function_call_arguments.delta": ("call_id", "delta", "item_id", "output_index", "response_id"),
    "response.function_call_arguments.done": ("arguments", "call_id", "item_id", "output_index", "response_id"),
    "response.mcp_call_arguments.delta": ("delta", "item_id", "output_index", "response_id"),
    "response.mcp_call_arguments.done": ("arguments", "item_id", "output_index", "response_id"),
    "response.mcp_call.in_progress": ("item_id", "output_index"),
    "response.mcp_call.completed": ("item_id", "output_index"),
    "response.mcp_call.failed": ("item_id", "output_index"),
    "mcp_list_tools.in_progress": ("item_id",),
    "mcp_list_tools.completed": ("item_id",),
    "mcp_list_tools.failed": ("item_id",),
    "rate_limits.updated": ("rate_limits",),
    "error": ("error",),
}


class RealtimeEventError(ValueError):
    pass


def event_type(event: Dict[str, Any]) -> str:
    t = event.get("type")
    if not isinstance(t, str):
        raise RealtimeEventError("realtime event has no `type`")
    return t


def is_client_event(etype: str) -> bool:
    return etype in CLIENT_EVENTS


def is_server_event(etype: str) -> bool:
    return etype in SERVER_EVENTS


def parse_event(event: Dict[str, Any], direction: str = "client") -> str:
    """Validate a realtime event envelope; returns its type.  `direction` is
    'client' (events the gateway receives from the user connection) or
    'server' (events relayed back)."""
    t = event_type(event)
    table = CLIENT_EVENTS if direction == "client" else SERVER_EVENTS
    fields = table.get(t)
    if fields is None:
        raise RealtimeEventError(f"unknown {direction} realtime event type {t!r}")
    missing = [f for f in fields if f not in event]
    if missing:
        raise RealtimeEventError(f"{t} missing required fields {missing}")
    return t


def make_event(etype: str, **payload) -> Dict[str, Any]:
    """Build a well-formed SERVER event (validates against the registry)."""
    ev = {"type": etype, "event_id": payload.pop("event_id", None) or f"event_{uuid.uuid4().hex[:20]}"}
    ev.update(payload)
    parse_event(ev, "server")
    return ev


def audio_bearing(etype: str) -> bool:
    """Events whose payload carries a base64 audio blob — the reference warns
    against Debug-logging these (realtime_events.rs:80); the relay's logging
    path uses this to log type-only."""
    return etype in (
        "input_audio_buffer.append",
        "response.output_audio.delta",
        "response.output_audio.done",
    )
