"""Streaming stop-sequence decoder + incremental detokenization
(reference: crates/tokenizer/src/stop.rs `StopSequenceDecoder` (841 LoC),
sequence.rs / stream.rs incremental detok).

StopSequenceDecoder consumes per-token text deltas and emits safe text:
text that could still be the start of a stop sequence is withheld until
disambiguated; on a stop match the stream ends (optionally emitting the
matched text when `include_stop=True`).  Stop token-ids match immediately.

DecodeStream wraps a tokenizer for incremental detok: token ids in, text
deltas out, using the two-window re-decode approach so multi-token unicode
glyphs emit once complete.
"""
from __future__ import annotations

import enum
from typing import Iterable, List, Optional, Sequence, Tuple


class StopOutcome(enum.Enum):
    NONE = "none"
    STOPPED = "stopped"  # matched a stop string / token
    STOPPED_WITH_TEXT = "stopped_with_text"


class StopSequenceDecoder:
    def __init__(
        self,
        stop_sequences: Optional[Sequence[str]] = None,
        stop_token_ids: Optional[Iterable[int]] = None,
        include_stop: bool = False,
    ):
        self.stops = [s for s in (stop_sequences or []) if s]
        self.stop_token_ids = set(stop_token_ids or [])
        self.include_stop = include_stop
        self._held = ""
        self.stopped = False
        self.matched: Optional[str] = None

    def _longest_partial(self, text: str) -> int:
        """Longest suffix of text that is a proper prefix of any stop string."""
        best = 0
        for stop in self.stops:
            m = min(len(stop) - 1, len(text))
            for k in range(m, 0, -1):
                if text.endswith(stop[:k]):
                    best = max(best, k)
                    break
        return best

    def process_token(self, token_id: int, text: str) -> Tuple[str, StopOutcome]:
        """Feed one decoded token's text; returns (emit_text, outcome)."""
        if self.stopped:
            return "", StopOutcome.STOPPED
        if token_id in self.stop_token_ids:
            self.stopped = True
            out = self._held
            self._held = ""
            return out, StopOutcome.STOPPED
        return self.process_text(text)

    def process_text(self, text: str) -> Tuple[str, StopOutcome]:
        if self.stopped:
            return "", StopOutcome.STOPPED
        buf = self._held + text
        # full match?
        earliest = None
        for stop in self.stops:
            i = buf.find(stop)
            if i >= 0 and (earliest is None or i < earliest[0]):
                earliest = (i, stop)
        if earliest is not None:
            i, stop = earliest
            self.stopped = True
            self.matched = stop
            emit = buf[:i] + (stop if self.include_stop else "")
            self._held = ""
            return emit, StopOutcome.STOPPED_WITH_TEXT if emit else StopOutcome.STOPPED
        # withhold a possible partial match
        keep = self._longest_partial(buf)
        emit = buf[: len(buf) - keep]
        self._held = buf[len(buf) - keep:]
        return emit, StopOutcome.NONE

    def flush(self) -> str:
        """End of stream: release withheld text (no stop ever completed)."""
        out, self._held = self._held, ""
        return out


class DecodeStream:
    """Incremental detokenizer: ids in, text deltas out (reference stream.rs).
    Two-offset re-decode: a delta is emitted only once its UTF-8 is complete,
    with the previous window re-decoded for correct byte-merge context."""

    def __init__(self, tokenizer, skip_special: bool = True):
        self.tok = tokenizer
        self.ids: List[int] = []
        self._prefix_offset = 0  # start of the decode window
        self._read_offset = 0    # ids already surfaced as text

    def push(self, token_id: int) -> str:
        self.ids.append(token_id)
        window = self.tok.decode(self.ids[self._prefix_offset:])
        if window.endswith("�"):
            return ""  # incomplete utf-8 glyph: wait for more tokens
        prev = self.tok.decode(self.ids[self._prefix_offset: self._read_offset])
        if len(window) <= len(prev):
            return ""
        delta = window[len(prev):]
        self._prefix_offset = self._read_offset
        self._read_offset = len(self.ids)
        return delta

    def reset(self) -> None:
        self.ids.clear()
        self._prefix_offset = 0
        self._read_offset = 0
