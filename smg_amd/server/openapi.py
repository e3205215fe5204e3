"""OpenAPI schema generation (reference: clients/openapi-gen — schemars over
the openai-protocol types, Makefile targets generate-openapi/-java-types).

Here the schema is produced by introspecting the live aiohttp route table
(every registered route appears) plus hand-maintained component schemas for
the core OpenAI-surface payloads; served at GET /openapi.json and writable
via scripts/gen_openapi.py for SDK generation.
"""
from __future__ import annotations

import re
from typing import Any, Dict

OPENAPI_VERSION = "3.0.3"

# Component schemas for the primary payloads (the reference generates these
# from protocol types; the full surface is huge — these cover the endpoints
# SDK generators target).
_SCHEMAS: Dict[str, Dict[str, Any]] = {
    "ChatMessage": {
        "type": "object",
        "properties": {
            "role": {"type": "string", "enum": ["system", "developer", "user", "assistant", "tool", "function"]},
            "content": {},
            "name": {"type": "string"},
            "tool_calls": {"type": "array", "items": {"$ref": "#/components/schemas/ToolCall"}},
        },
        "required": ["role"],
    },
    "ToolCall": {
        "type": "object",
        "properties": {
            "id": {"type": "string"},
            "type": {"type": "string", "enum": ["function"]},
            "function": {
                "type": "object",
                "properties": {"name": {"type": "string"}, "arguments": {"type": "string"}},
            },
        },
    },
    "ChatCompletionRequest": {
        "type": "object",
        "properties": {
            "model": {"type": "string"},
            "messages": {"type": "array", "items": {"$ref": "#/components/schemas/ChatMessage"}},
            "max_tokens": {"type": "integer"},
            "max_completion_tokens": {"type": "integer"},
            "temperature": {"type": "number"},
            "top_p": {"type": "number"},
            "stream": {"type": "boolean"},
            "stop": {},
            "tools": {"type": "array"},
            "tool_choice": {},
            "n": {"type": "integer"},
        },
        "required": ["model", "messages"],
    },
    "ChatCompletionResponse": {
        "type": "object",
        "properties": {
            "id": {"type": "string"},
            "object": {"type": "string"},
            "created": {"type": "integer"},
            "model": {"type": "string"},
            "choices": {"type": "array"},
            "usage": {"$ref": "#/components/schemas/Usage"},
        },
    },
    "CompletionRequest": {
        "type": "object",
        "properties": {
            "model": {"type": "string"},
            "prompt": {},
            "max_tokens": {"type": "integer"},
            "temperature": {"type": "number"},
            "stream": {"type": "boolean"},
        },
        "required": ["model", "prompt"],
    },
    "EmbeddingRequest": {
        "type": "object",
        "properties": {"model": {"type": "string"}, "input": {}},
        "required": ["model", "input"],
    },
    "Usage": {
        "type": "object",
        "properties": {
            "prompt_tokens": {"type": "integer"},
            "completion_tokens": {"type": "integer"},
            "total_tokens": {"type": "integer"},
        },
    },
    "ResponsesRequest": {
        "type": "object",
        "properties": {
            "model": {"type": "string"},
            "input": {},
            "instructions": {"type": "string"},
            "max_output_tokens": {"type": "integer"},
            "stream": {"type": "boolean"},
            "store": {"type": "boolean"},
            "conversation": {"type": "string"},
        },
    },
    "AnthropicMessagesRequest": {
        "type": "object",
        "properties": {
            "model": {"type": "string"},
            "messages": {"type": "array"},
            "max_tokens": {"type": "integer"},
            "system": {},
            "stream": {"type": "boolean"},
        },
        "required": ["model", "messages", "max_tokens"],
    },
    "WorkerSpec": {
        "type": "object",
        "properties": {
            "url": {"type": "string"},
            "model_id": {"type": "string"},
            "worker_type": {"type": "string", "enum": ["regular", "prefill", "decode", "encode"]},
            "labels": {"type": "object"},
            "api_key": {"type": "string"},
            "bootstrap_port": {"type": "integer"},
            "model_aliases": {"type": "array", "items": {"type": "string"}},
        },
        "required": ["url"],
    },
    "ErrorResponse": {
        "type": "object",
        "properties": {
            "error": {
                "type": "object",
                "properties": {
                    "message": {"type": "string"},
                    "type": {"type": "string"},
                    "code": {"type": "integer"},
                },
            }
        },
    },
}

_REQUEST_BODIES = {
    "/v1/chat/completions": "ChatCompletionRequest",
    "/v1/completions": "CompletionRequest",
    "/v1/embeddings": "EmbeddingRequest",
    "/v1/responses": "ResponsesRequest",
    "/v1/messages": "AnthropicMessagesRequest",
    "/workers": "WorkerSpec",
}

_RESPONSES = {
    "/v1/chat/completions": "ChatCompletionResponse",
}


def _path_params(path: str):
    return re.findall(r"\{(\w+)(?::[^}]*)?\}", path)


def build_openapi(app=None, title: str = "smg-amd gateway", version: str = "0.1.0") -> Dict[str, Any]:
    paths: Dict[str, Dict[str, Any]] = {}
    if app is not None:
        for resource in app.router.resources():
            canonical = resource.canonical
            for route in resource:
                method = route.method.lower()
                if method in ("head", "options", "*"):
                    continue
                norm = re.sub(r"\{(\w+):[^}]*\}", r"{\1}", canonical)
                op: Dict[str, Any] = {
                    "operationId": f"{method}_{re.sub(r'[^a-zA-Z0-9]+', '_', norm).strip('_')}",
                    "responses": {
                        "200": {"description": "success"},
                        "default": {
                            "description": "error",
                            "content": {"application/json": {"schema": {"$ref": "#/components/schemas/ErrorResponse"}}},
                        },
                    },
                }
                params = _path_params(canonical)
                if params:
                    op["parameters"] = [
                        {"name": p, "in": "path", "required": True, "schema": {"type": "string"}} for p in params
                    ]
                body = _REQUEST_BODIES.get(norm)
                if body and method == "post":
                    op["requestBody"] = {
                        "required": True,
                        "content": {"application/json": {"schema": {"$ref": f"#/components/schemas/{body}"}}},
                    }
                resp = _RESPONSES.get(norm)
                if resp:
                    op["responses"]["200"] = {
                        "description": "success",
                        "content": {"application/json": {"schema": {"$ref": f"#/components/schemas/{resp}"}}},
                    }
                paths.setdefault(norm, {})[method] = op
    return {
        "openapi": OPENAPI_VERSION,
        "info": {"title": title, "version": version},
        "paths": dict(sorted(paths.items())),
        "components": {
            "schemas": _SCHEMAS,
            "securitySchemes": {
                "bearerAuth": {"type": "http", "scheme": "bearer"},
                "apiKeyAuth": {"type": "apiKey", "in": "header", "name": "x-api-key"},
            },
        },
    }
