from .openai import (
    ChatCompletionRequest,
    CompletionRequest,
    EmbeddingRequest,
    GenerateRequest,
    ProtocolError,
    RequestView,
    error_body,
    parse_request,
)

__all__ = [
    "ChatCompletionRequest",
    "CompletionRequest",
    "EmbeddingRequest",
    "GenerateRequest",
    "ProtocolError",
    "RequestView",
    "error_body",
    "parse_request",
]
