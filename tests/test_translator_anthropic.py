"""OpenAI ⇄ Anthropic translation tests (parity targets:
anthropic_helper_test.go, openai_helper_test.go golden behaviors)."""

import json

import pytest

from aigw.filterapi.config import APISchemaName
from aigw.translator import get_translator
from aigw.translator.anthropic_schema import (
    AnthropicToOpenAIStream,
    anthropic_to_openai_response,
    fold_anthropic_stream,
    openai_to_anthropic_request,
)
from aigw.translator.chat_anthropic import (
    anthropic_to_openai_request,
    openai_to_anthropic_response,
)
from aigw.translator.sse import SSEDecoder

OPENAI_REQ = {
    "model": "claude-sonnet",
    "messages": [
        {"role": "system", "content": "be brief"},
        {"role": "user", "content": "hello"},
        {
            "role": "assistant",
            "content": None,
            "tool_calls": [
                {
                    "id": "call_1",
                    "type": "function",
                    "function": {"name": "get_weather", "arguments": '{"city":"SF"}'},
                }
            ],
        },
        {"role": "tool", "tool_call_id": "call_1", "content": "sunny"},
        {
            "role": "user",
            "content": [
                {"type": "text", "text": "and now?"},
                {
                    "type": "image_url",
                    "image_url": {"url": "data:image/jpeg;base64,AAAA"},
                },
            ],
        },
    ],
    "max_tokens": 100,
    "temperature": 0.5,
    "stop": ["END"],
    "tools": [
        {
            "type": "function",
            "function": {
                "name": "get_weather",
                "description": "w",
                "parameters": {"type": "object", "properties": {}},
            },
        }
    ],
    "tool_choice": "auto",
}


def test_request_mapping():
    a = openai_to_anthropic_request(json.loads(json.dumps(OPENAI_REQ)))
    assert a["system"] == "be brief"
    assert a["max_tokens"] == 100
    assert a["temperature"] == 0.5
    assert a["stop_sequences"] == ["END"]
    assert a["tool_choice"] == {"type": "auto"}
    msgs = a["messages"]
    assert msgs[0] == {"role": "user", "content": [{"type": "text", "text": "hello"}]}
    assert msgs[1]["role"] == "assistant"
    assert msgs[1]["content"][0]["type"] == "tool_use"
    assert msgs[1]["content"][0]["input"] == {"city": "SF"}
    assert msgs[2]["content"][0]["type"] == "tool_result"
    assert msgs[2]["content"][0]["tool_use_id"] == "call_1"
    img = msgs[3]["content"][1]
    assert img["source"]["media_type"] == "image/jpeg"
    assert a["tools"][0]["input_schema"] == {"type": "object", "properties": {}}


def test_request_roundtrip_through_reverse():
    """OpenAI→Anthropic→OpenAI keeps the semantic payload."""
    a = openai_to_anthropic_request(json.loads(json.dumps(OPENAI_REQ)))
    o = anthropic_to_openai_request(a)
    assert o["messages"][0] == {"role": "system", "content": "be brief"}
    assert o["max_completion_tokens"] == 100
    assert o["tools"][0]["function"]["name"] == "get_weather"
    roles = [m["role"] for m in o["messages"]]
    assert roles == ["system", "user", "assistant", "tool", "user"]


ANTHROPIC_RESP = {
    "id": "msg_01",
    "type": "message",
    "role": "assistant",
    "model": "claude-sonnet-4",
    "content": [
        {"type": "thinking", "thinking": "hmm"},
        {"type": "text", "text": "It is "},
        {"type": "text", "text": "sunny."},
        {"type": "tool_use", "id": "tu_1", "name": "f", "input": {"x": 1}},
    ],
    "stop_reason": "tool_use",
    "usage": {
        "input_tokens": 10,
        "output_tokens": 20,
        "cache_read_input_tokens": 5,
        "cache_creation_input_tokens": 2,
    },
}


def test_response_mapping():
    o, usage = anthropic_to_openai_response(ANTHROPIC_RESP)
    choice = o["choices"][0]
    assert choice["finish_reason"] == "tool_calls"
    assert choice["message"]["content"] == "It is sunny."
    assert choice["message"]["reasoning_content"] == "hmm"
    assert choice["message"]["tool_calls"][0]["function"]["arguments"] == '{"x": 1}'
    # Anthropic input_tokens excludes cache reads; OpenAI includes them.
    assert usage.input_tokens == 17
    assert usage.output_tokens == 20
    assert o["usage"]["prompt_tokens"] == 17
    assert o["usage"]["prompt_tokens_details"]["cached_tokens"] == 5


def test_response_roundtrip_reverse():
    o, _ = anthropic_to_openai_response(ANTHROPIC_RESP)
    a, usage = openai_to_anthropic_response(o)
    assert a["stop_reason"] == "tool_use"
    assert [b["type"] for b in a["content"]] == ["text", "tool_use"]
    assert a["usage"]["input_tokens"] == 12  # 17 total - 5 cached
    assert a["usage"]["cache_read_input_tokens"] == 5


ANTHROPIC_EVENTS = [
    ("message_start", {"message": {"id": "msg_01", "model": "claude", "usage": {"input_tokens": 7}}}),
    ("content_block_start", {"index": 0, "content_block": {"type": "text", "text": ""}}),
    ("content_block_delta", {"index": 0, "delta": {"type": "text_delta", "text": "Hel"}}),
    ("content_block_delta", {"index": 0, "delta": {"type": "text_delta", "text": "lo"}}),
    ("content_block_stop", {"index": 0}),
    (
        "content_block_start",
        {"index": 1, "content_block": {"type": "tool_use", "id": "tu1", "name": "fn"}},
    ),
    ("content_block_delta", {"index": 1, "delta": {"type": "input_json_delta", "partial_json": '{"a"'}}),
    ("content_block_delta", {"index": 1, "delta": {"type": "input_json_delta", "partial_json": ":1}"}}),
    ("content_block_stop", {"index": 1}),
    ("message_delta", {"delta": {"stop_reason": "tool_use"}, "usage": {"output_tokens": 9}}),
    ("message_stop", {}),
]


def test_stream_machine_anthropic_to_openai():
    m = AnthropicToOpenAIStream()
    chunks = []
    for et, data in ANTHROPIC_EVENTS:
        chunks.extend(m.feed_event(et, data))
    text = "".join(
        c["choices"][0]["delta"].get("content") or "" for c in chunks
    )
    assert text == "Hello"
    args = "".join(
        tc["function"].get("arguments", "")
        for c in chunks
        for tc in c["choices"][0]["delta"].get("tool_calls") or []
    )
    assert args == '{"a":1}'
    final = chunks[-1]
    assert final["choices"][0]["finish_reason"] == "tool_calls"
    assert final["usage"]["prompt_tokens"] == 7
    assert final["usage"]["completion_tokens"] == 9
    assert final["usage"]["total_tokens"] == 16


def test_fold_anthropic_stream():
    resp = fold_anthropic_stream(ANTHROPIC_EVENTS)
    assert resp["id"] == "msg_01"
    assert resp["stop_reason"] == "tool_use"
    blocks = resp["content"]
    assert blocks[0]["text"] == "Hello"
    assert blocks[1]["input"] == {"a": 1}
    assert resp["usage"]["output_tokens"] == 9


def test_fold_hostile_index_guard():
    with pytest.raises(Exception):
        fold_anthropic_stream([("content_block_start", {"index": 2000, "content_block": {}})])


def _anthropic_sse_bytes() -> bytes:
    out = bytearray()
    for et, data in ANTHROPIC_EVENTS:
        d = dict(data)
        d["type"] = et
        out.extend(f"event: {et}\n".encode())
        out.extend(b"data: " + json.dumps(d).encode() + b"\n\n")
    return bytes(out)


@pytest.mark.parametrize("n", [1, 9, 1024])
def test_full_translator_streaming(n):
    """End-to-end: OpenAI client → Anthropic backend streaming, chunked at
    arbitrary boundaries, produces valid OpenAI SSE with usage."""
    t = get_translator("/v1/chat/completions", APISchemaName.ANTHROPIC)
    req = t.request(
        {"model": "claude", "messages": [{"role": "user", "content": "hi"}], "stream": True},
        stream=True,
    )
    assert req.path == "/v1/messages"
    assert json.loads(req.body)["stream"] is True

    blob = _anthropic_sse_bytes()
    out = bytearray()
    usage = None
    for i in range(0, len(blob), n):
        r = t.response_chunk(blob[i : i + n])
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    assert usage is not None and usage.input_tokens == 7 and usage.output_tokens == 9

    # The output must itself parse as OpenAI SSE ending in [DONE].
    d = SSEDecoder()
    evs = d.feed(bytes(out))
    assert evs[-1].data == "[DONE]"
    text = ""
    for ev in evs[:-1]:
        c = json.loads(ev.data)
        assert c["object"] == "chat.completion.chunk"
        text += c["choices"][0]["delta"].get("content") or ""
    assert text == "Hello"


def test_anthropic_native_to_openai_backend_stream():
    """Anthropic client → OpenAI backend: OpenAI chunks re-synthesized as
    Anthropic SSE events (openai_helper.go:516-698)."""
    t = get_translator("/anthropic/v1/messages", APISchemaName.OPENAI)
    req = t.request(
        {"model": "gpt-4o", "max_tokens": 10, "messages": [{"role": "user", "content": "hi"}]},
        stream=True,
    )
    assert req.path == "/v1/chat/completions"
    body = json.loads(req.body)
    assert body["stream"] is True and body["stream_options"]["include_usage"] is True

    chunks = [
        {"id": "c1", "model": "gpt-4o", "choices": [{"index": 0, "delta": {"role": "assistant", "content": "He"}}]},
        {"id": "c1", "model": "gpt-4o", "choices": [{"index": 0, "delta": {"content": "y"}}]},
        {"id": "c1", "model": "gpt-4o", "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}]},
        {"id": "c1", "model": "gpt-4o", "choices": [], "usage": {"prompt_tokens": 3, "completion_tokens": 2, "total_tokens": 5}},
    ]
    blob = b"".join(
        b"data: " + json.dumps(c).encode() + b"\n\n" for c in chunks
    ) + b"data: [DONE]\n\n"
    out = bytearray()
    usage = None
    for i in range(0, len(blob), 13):
        r = t.response_chunk(blob[i : i + 13])
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    assert usage.output_tokens == 2

    d = SSEDecoder()
    evs = d.feed(bytes(out))
    types = [ev.event for ev in evs]
    assert types[0] == "message_start"
    assert "content_block_start" in types
    assert types[-2:] == ["message_delta", "message_stop"]
    text = "".join(
        json.loads(ev.data)["delta"].get("text", "")
        for ev in evs
        if ev.event == "content_block_delta"
    )
    assert text == "Hey"
    md = json.loads(evs[-2].data)
    assert md["delta"]["stop_reason"] == "end_turn"
    assert md["usage"]["output_tokens"] == 2
