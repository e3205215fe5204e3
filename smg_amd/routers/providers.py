"""Cloud provider passthrough + translation (reference:
model_gateway/src/routers/openai/provider/ — provider_trait.rs with openai /
anthropic / gemini / xai / sglang vendors; anthropic passthrough translation;
gemini/driver.rs).

A worker labeled `provider=<name>` is an upstream cloud endpoint; the
ProviderAdapter translates the gateway's OpenAI-chat surface to the vendor
API and back.  openai/xai/sglang are wire-compatible; anthropic and gemini
get request/response translation.
"""
from __future__ import annotations

import json
import uuid
from typing import Any, Dict, Optional

import aiohttp

from ..workers.worker import Worker


class ProviderAdapter:
    name = "openai"
    chat_path = "/v1/chat/completions"

    def auth_headers(self, api_key: Optional[str]) -> Dict[str, str]:
        return {"authorization": f"Bearer {api_key}"} if api_key else {}

    def translate_request(self, chat_body: Dict[str, Any]) -> Dict[str, Any]:
        return chat_body

    def translate_response(self, vendor_resp: Dict[str, Any], chat_body: Dict[str, Any]) -> Dict[str, Any]:
        return vendor_resp


class OpenAIProvider(ProviderAdapter):
    name = "openai"


class XAIProvider(ProviderAdapter):
    name = "xai"


class SglangProvider(ProviderAdapter):
    name = "sglang"


class AnthropicProvider(ProviderAdapter):
    name = "anthropic"
    chat_path = "/v1/messages"

    def auth_headers(self, api_key):
        return {"x-api-key": api_key or "", "anthropic-version": "2023-06-01"}

    def translate_request(self, chat_body):
        # OpenAI chat -> Anthropic Messages (inverse of messages_to_chat)
        messages = []
        system = None
        for m in chat_body.get("messages", []):
            if m.get("role") == "system":
                system = m.get("content")
            else:
                messages.append({"role": m.get("role", "user"), "content": m.get("content") or ""})
        out = {
            "model": chat_body.get("model"),
            "messages": messages,
            "max_tokens": chat_body.get("max_tokens") or 256,
        }
        if system:
            out["system"] = system
        for k in ("temperature", "top_p"):
            if chat_body.get(k) is not None:
                out[k] = chat_body[k]
        if chat_body.get("stop"):
            out["stop_sequences"] = chat_body["stop"] if isinstance(chat_body["stop"], list) else [chat_body["stop"]]
        return out

    def translate_response(self, vendor_resp, chat_body):
        # Anthropic Message -> OpenAI chat completion
        text = "".join(b.get("text", "") for b in vendor_resp.get("content", []) if b.get("type") == "text")
        stop_map = {"end_turn": "stop", "max_tokens": "length", "tool_use": "tool_calls"}
        usage = vendor_resp.get("usage", {})
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
            "object": "chat.completion",
            "model": vendor_resp.get("model") or chat_body.get("model"),
            "choices": [
                {
                    "index": 0,
                    "message": {"role": "assistant", "content": text},
                    "finish_reason": stop_map.get(vendor_resp.get("stop_reason"), "stop"),
                }
            ],
            "usage": {
                "prompt_tokens": usage.get("input_tokens", 0),
                "completion_tokens": usage.get("output_tokens", 0),
                "total_tokens": usage.get("input_tokens", 0) + usage.get("output_tokens", 0),
            },
        }


class GeminiProvider(ProviderAdapter):
    name = "gemini"

    def chat_path_for(self, model: str) -> str:
        return f"/v1beta/models/{model}:generateContent"

    def auth_headers(self, api_key):
        return {"x-goog-api-key": api_key or ""}

    def translate_request(self, chat_body):
        contents = []
        system_instruction = None
        for m in chat_body.get("messages", []):
            role = m.get("role")
            text = m.get("content") or ""
            if role == "system":
                system_instruction = {"parts": [{"text": text}]}
                continue
            contents.append({"role": "user" if role == "user" else "model", "parts": [{"text": text}]})
        out: Dict[str, Any] = {"contents": contents}
        if system_instruction:
            out["systemInstruction"] = system_instruction
        gen_cfg = {}
        if chat_body.get("max_tokens"):
            gen_cfg["maxOutputTokens"] = chat_body["max_tokens"]
        for src, dst in (("temperature", "temperature"), ("top_p", "topP")):
            if chat_body.get(src) is not None:
                gen_cfg[dst] = chat_body[src]
        if gen_cfg:
            out["generationConfig"] = gen_cfg
        return out

    def translate_response(self, vendor_resp, chat_body):
        cand = (vendor_resp.get("candidates") or [{}])[0]
        parts = cand.get("content", {}).get("parts", [])
        text = "".join(p.get("text", "") for p in parts)
        finish = {"STOP": "stop", "MAX_TOKENS": "length"}.get(cand.get("finishReason"), "stop")
        meta = vendor_resp.get("usageMetadata", {})
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
            "object": "chat.completion",
            "model": chat_body.get("model"),
            "choices": [
                {"index": 0, "message": {"role": "assistant", "content": text}, "finish_reason": finish}
            ],
            "usage": {
                "prompt_tokens": meta.get("promptTokenCount", 0),
                "completion_tokens": meta.get("candidatesTokenCount", 0),
                "total_tokens": meta.get("totalTokenCount", 0),
            },
        }


PROVIDERS: Dict[str, ProviderAdapter] = {
    "openai": OpenAIProvider(),
    "xai": XAIProvider(),
    "sglang": SglangProvider(),
    "anthropic": AnthropicProvider(),
    "gemini": GeminiProvider(),
}


def provider_for_worker(worker: Worker) -> Optional[ProviderAdapter]:
    name = worker.labels.get("provider")
    return PROVIDERS.get(name) if name else None


async def dispatch_to_provider(
    session: aiohttp.ClientSession, worker: Worker, chat_body: Dict[str, Any]
) -> Dict[str, Any]:
    """Unary chat completion through a cloud provider worker."""
    adapter = provider_for_worker(worker) or OpenAIProvider()
    vendor_body = adapter.translate_request(chat_body)
    if isinstance(adapter, GeminiProvider):
        path = adapter.chat_path_for(chat_body.get("model") or "gemini")
    else:
        path = adapter.chat_path
    headers = {"content-type": "application/json", **adapter.auth_headers(worker.api_key)}
    async with session.post(worker.url + path, json=vendor_body, headers=headers) as resp:
        data = await resp.json()
        if resp.status != 200:
            raise RuntimeError(f"provider {adapter.name} error {resp.status}: {json.dumps(data)[:200]}")
    return adapter.translate_response(data, chat_body)
