"""API-format bridge: one registry for every request/response translation
between the gateway's wire formats (reference:
model_gateway/src/routers/common/openai_bridge/ — FormatRegistry +
transformer.rs, which the MCP tool loop and provider interop use to move a
conversation between Chat Completions, the Responses API, Anthropic
Messages and Gemini Interactions shapes).

Chat Completions is the HUB format: every format registers a request
transform INTO chat and a response transform OUT of a chat completion.
Cross-format conversions (e.g. Responses -> Anthropic Messages) compose
through the hub, exactly like the reference's transformer pivots on its
canonical ChatRequest.

The concrete transforms live next to their route handlers
(server/responses_routes.py, routers/anthropic.py,
server/interactions_routes.py) — this module is the registry and the
composition rule, so providers and the MCP tool loop have ONE entry point.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple

CHAT = "chat"
RESPONSES = "responses"
MESSAGES = "messages"
INTERACTIONS = "interactions"


class FormatRegistry:
    """(format -> chat) request transforms and (chat -> format) response
    transforms, with hub composition for format-to-format conversion."""

    def __init__(self):
        #  fmt -> fn(body) -> chat request body
        self._to_chat: Dict[str, Callable[[Dict[str, Any]], Dict[str, Any]]] = {}
        #  fmt -> fn(orig_request, chat_response) -> fmt response body
        self._from_chat: Dict[str, Callable[[Dict[str, Any], Dict[str, Any]], Dict[str, Any]]] = {}
        #  fmt -> fn(chat_request) -> fmt request body (outbound direction)
        self._req_from_chat: Dict[str, Callable[[Dict[str, Any]], Dict[str, Any]]] = {}

    def register(self, fmt: str, to_chat=None, from_chat=None, request_from_chat=None) -> None:
        if to_chat is not None:
            self._to_chat[fmt] = to_chat
        if from_chat is not None:
            self._from_chat[fmt] = from_chat
        if request_from_chat is not None:
            self._req_from_chat[fmt] = request_from_chat

    @property
    def formats(self) -> List[str]:
        return sorted({CHAT, *self._to_chat, *self._from_chat})

    def to_chat_request(self, fmt: str, body: Dict[str, Any]) -> Dict[str, Any]:
        if fmt == CHAT:
            return body
        fn = self._to_chat.get(fmt)
        if fn is None:
            raise KeyError(f"no request transform {fmt} -> chat")
        return fn(body)

    def from_chat_response(self, fmt: str, orig_request: Dict[str, Any],
                           chat_response: Dict[str, Any]) -> Dict[str, Any]:
        if fmt == CHAT:
            return chat_response
        fn = self._from_chat.get(fmt)
        if fn is None:
            raise KeyError(f"no response transform chat -> {fmt}")
        return fn(orig_request, chat_response)

    def convert_request(self, src: str, dst: str, body: Dict[str, Any]) -> Dict[str, Any]:
        """Request in `src` format -> request in `dst` format, pivoting
        through chat.  dst != chat requires a chat->dst request transform,
        which only chat-supersets need; for the gateway's use (dispatch to a
        chat-speaking engine or provider) dst is chat."""
        chat_body = self.to_chat_request(src, body)
        if dst == CHAT:
            return chat_body
        fn = self._req_from_chat.get(dst)
        if fn is None:
            raise KeyError(f"no request transform chat -> {dst} (pivot is chat)")
        return fn(chat_body)

    def convert_response(self, src_request_fmt: str, orig_request: Dict[str, Any],
                         chat_response: Dict[str, Any]) -> Dict[str, Any]:
        return self.from_chat_response(src_request_fmt, orig_request, chat_response)


_registry: Optional[FormatRegistry] = None


def get_registry() -> FormatRegistry:
    """The process-wide registry, built lazily from the route-local
    transforms (import cycles: routes import the bridge for the REGISTRY,
    the bridge imports route modules only inside this builder)."""
    global _registry
    if _registry is None:
        reg = FormatRegistry()
        from ..server.responses_routes import chat_to_response, responses_to_chat

        reg.register(RESPONSES, to_chat=responses_to_chat,
                     from_chat=lambda req, chat: chat_to_response(req, chat))
        from .anthropic import chat_to_message, messages_to_chat
        from .providers import AnthropicProvider

        reg.register(MESSAGES, to_chat=messages_to_chat,
                     from_chat=lambda req, chat: chat_to_message(req, chat),
                     request_from_chat=AnthropicProvider().translate_request)
        from ..server.interactions_routes import chat_to_interaction, interactions_to_chat

        reg.register(INTERACTIONS,
                     to_chat=lambda body: interactions_to_chat(body, []),
                     from_chat=lambda req, chat: chat_to_interaction(req, chat))
        _registry = reg
    return _registry


def cross_convert(src: str, dst: str, body: Dict[str, Any],
                  chat_response: Dict[str, Any]) -> Tuple[Dict[str, Any], Dict[str, Any]]:
    """Full hub pivot: (request in src) + (chat response) -> the request the
    dst endpoint would have received and the response in dst's shape —
    the operation the reference's MCP tool loop performs when it replays a
    Responses conversation against a Messages/Chat provider."""
    reg = get_registry()
    chat_req = reg.to_chat_request(src, body)
    dst_resp = reg.from_chat_response(dst, body if dst == src else chat_req, chat_response)
    return chat_req, dst_resp
