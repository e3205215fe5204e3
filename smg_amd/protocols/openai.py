"""OpenAI-compatible API types (reference: crates/protocols/src/{chat,completions,
embeddings,rerank,generate}.rs).

The HTTP router is a streaming pass-through proxy, so requests are kept as
parsed JSON dicts with typed *views* for the fields the gateway itself reads:
model, stream flag, routing text, token estimates, sampling knobs.  The gRPC
pipeline builds full typed requests; validation errors raise ProtocolError
with the OpenAI error envelope shape.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Optional


class ProtocolError(ValueError):
    def __init__(self, message: str, code: int = 400, err_type: str = "invalid_request_error"):
        super().__init__(message)
        self.code = code
        self.err_type = err_type

    def to_response(self) -> Dict[str, Any]:
        return {"error": {"message": str(self), "type": self.err_type, "code": self.code}}


def error_body(message: str, code: int = 400, err_type: str = "invalid_request_error") -> bytes:
    return json.dumps({"error": {"message": message, "type": err_type, "code": code}}).encode()


class RequestView:
    """Typed view over a parsed request body."""

    endpoint = "unknown"

    def __init__(self, body: Dict[str, Any]):
        if not isinstance(body, dict):
            raise ProtocolError("request body must be a JSON object")
        self.body = body

    @property
    def model(self) -> Optional[str]:
        m = self.body.get("model")
        return m if isinstance(m, str) and m else None

    @property
    def stream(self) -> bool:
        return bool(self.body.get("stream", False))

    @property
    def user(self) -> Optional[str]:
        u = self.body.get("user")
        return u if isinstance(u, str) else None

    def routing_text(self) -> str:
        """Text used for cache-aware / hash routing (reference
        http/router.rs:325 extract_text_for_routing)."""
        return ""

    def est_prompt_tokens(self) -> int:
        # chars/4 heuristic when no tokenizer ran (reference uses tokenizer when available)
        return max(1, len(self.routing_text()) // 4)

    def max_output_tokens(self) -> int:
        for key in ("max_completion_tokens", "max_tokens", "max_output_tokens"):
            v = self.body.get(key)
            if isinstance(v, int) and v > 0:
                return v
        return 0


class ChatCompletionRequest(RequestView):
    endpoint = "chat"

    def __init__(self, body: Dict[str, Any]):
        super().__init__(body)
        msgs = body.get("messages")
        if not isinstance(msgs, list) or not msgs:
            raise ProtocolError("'messages' must be a non-empty array")
        for m in msgs:
            if not isinstance(m, dict) or "role" not in m:
                raise ProtocolError("each message must be an object with a 'role'")

    @property
    def messages(self) -> List[Dict[str, Any]]:
        return self.body["messages"]

    def routing_text(self) -> str:
        parts: List[str] = []
        for m in self.messages:
            content = m.get("content")
            if isinstance(content, str):
                parts.append(content)
            elif isinstance(content, list):  # multimodal parts
                for p in content:
                    if isinstance(p, dict) and p.get("type") == "text":
                        parts.append(p.get("text", ""))
        return "\n".join(parts)


class CompletionRequest(RequestView):
    endpoint = "completion"

    def __init__(self, body: Dict[str, Any]):
        super().__init__(body)
        if "prompt" not in body:
            raise ProtocolError("'prompt' is required")

    def routing_text(self) -> str:
        p = self.body.get("prompt")
        if isinstance(p, str):
            return p
        if isinstance(p, list):
            if all(isinstance(x, str) for x in p):
                return "\n".join(p)
            # token-id prompt
            return ""
        return ""

    def prompt_token_ids(self) -> Optional[List[int]]:
        p = self.body.get("prompt")
        if isinstance(p, list) and p and all(isinstance(x, int) for x in p):
            return p
        return None


class GenerateRequest(RequestView):
    """SGLang-native /generate (reference protocols generate types)."""

    endpoint = "generate"

    def routing_text(self) -> str:
        t = self.body.get("text")
        if isinstance(t, str):
            return t
        if isinstance(t, list) and all(isinstance(x, str) for x in t):
            return "\n".join(t)
        return ""

    def prompt_token_ids(self) -> Optional[List[int]]:
        ids = self.body.get("input_ids")
        if isinstance(ids, list) and ids and all(isinstance(x, int) for x in ids):
            return ids
        return None


class EmbeddingRequest(RequestView):
    endpoint = "embedding"

    def __init__(self, body: Dict[str, Any]):
        super().__init__(body)
        if "input" not in body:
            raise ProtocolError("'input' is required")

    def routing_text(self) -> str:
        i = self.body.get("input")
        if isinstance(i, str):
            return i
        if isinstance(i, list) and all(isinstance(x, str) for x in i):
            return "\n".join(i)
        return ""


class RerankRequest(RequestView):
    endpoint = "rerank"

    def __init__(self, body: Dict[str, Any]):
        super().__init__(body)
        if "query" not in body:
            raise ProtocolError("'query' is required")

    def routing_text(self) -> str:
        return str(self.body.get("query", ""))


class ClassifyRequest(RequestView):
    endpoint = "classify"

    def routing_text(self) -> str:
        i = self.body.get("input") or self.body.get("text") or ""
        return i if isinstance(i, str) else ""


ENDPOINT_VIEWS = {
    "/v1/chat/completions": ChatCompletionRequest,
    "/v1/completions": CompletionRequest,
    "/generate": GenerateRequest,
    "/v1/embeddings": EmbeddingRequest,
    "/v1/rerank": RerankRequest,
    "/rerank": RerankRequest,
    "/v1/classify": ClassifyRequest,
}


def parse_request(path: str, body: Dict[str, Any]) -> RequestView:
    view_cls = ENDPOINT_VIEWS.get(path, RequestView)
    return view_cls(body)
