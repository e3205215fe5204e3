"""FormatRegistry hub conversions (reference
routers/common/openai_bridge/transformer.rs: Chat is the pivot format for
Responses / Anthropic Messages / Gemini Interactions interop)."""
import pytest

from smg_amd.routers.openai_bridge import (
    CHAT,
    INTERACTIONS,
    MESSAGES,
    RESPONSES,
    cross_convert,
    get_registry,
)

CHAT_RESP = {
    "id": "chatcmpl-1",
    "model": "m",
    "choices": [{"index": 0, "message": {"role": "assistant", "content": "hi there"},
                 "finish_reason": "stop"}],
    "usage": {"prompt_tokens": 3, "completion_tokens": 2, "total_tokens": 5},
}


class TestFormatRegistry:
    def test_formats_registered(self):
        reg = get_registry()
        assert set(reg.formats) >= {CHAT, RESPONSES, MESSAGES, INTERACTIONS}

    def test_responses_to_chat_request(self):
        reg = get_registry()
        body = {"model": "m", "input": "hello", "instructions": "be brief"}
        chat = reg.convert_request(RESPONSES, CHAT, body)
        assert chat["model"] == "m"
        roles = [m["role"] for m in chat["messages"]]
        assert roles[0] == "system" and "user" in roles

    def test_messages_to_chat_request(self):
        reg = get_registry()
        body = {"model": "m", "max_tokens": 16, "system": "s",
                "messages": [{"role": "user", "content": "hello"}]}
        chat = reg.convert_request(MESSAGES, CHAT, body)
        assert chat["messages"][0]["role"] == "system"
        assert chat["max_tokens"] == 16

    def test_chat_response_to_messages_shape(self):
        reg = get_registry()
        req = {"model": "m", "max_tokens": 8, "messages": [{"role": "user", "content": "x"}]}
        out = reg.from_chat_response(MESSAGES, req, CHAT_RESP)
        assert out["type"] == "message"
        assert out["content"][0]["text"] == "hi there"
        assert out["stop_reason"] in ("end_turn", "max_tokens")

    def test_chat_response_to_responses_shape(self):
        reg = get_registry()
        req = {"model": "m", "input": "x"}
        out = reg.from_chat_response(RESPONSES, req, CHAT_RESP)
        assert out["object"] == "response"
        texts = [c.get("text") for item in out.get("output", [])
                 for c in item.get("content", []) if isinstance(c, dict)]
        assert "hi there" in texts

    def test_cross_convert_responses_to_messages(self):
        """The MCP-loop replay: a Responses request whose chat completion is
        re-shaped as an Anthropic message."""
        body = {"model": "m", "input": "hello"}
        chat_req, msg_resp = cross_convert(RESPONSES, MESSAGES, body, CHAT_RESP)
        assert chat_req["messages"]
        assert msg_resp["type"] == "message"

    def test_request_pivot_to_messages_wire(self):
        reg = get_registry()
        body = {"model": "m", "input": "hello", "instructions": "s"}
        msg_req = reg.convert_request(RESPONSES, MESSAGES, body)
        # anthropic wire: system is a top-level field, not a message
        assert "system" in msg_req
        assert all(m["role"] != "system" for m in msg_req["messages"])

    def test_interactions_to_chat(self):
        reg = get_registry()
        body = {"model": "m", "input": "question"}
        chat = reg.convert_request(INTERACTIONS, CHAT, body)
        assert chat["messages"][-1]["role"] == "user"

    def test_unknown_format_raises(self):
        reg = get_registry()
        with pytest.raises(KeyError):
            reg.convert_request("grpc", CHAT, {})
        with pytest.raises(KeyError):
            reg.from_chat_response("nope", {}, CHAT_RESP)
