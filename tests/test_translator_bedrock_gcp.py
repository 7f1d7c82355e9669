"""Bedrock Converse and GCP Gemini translator tests (parity targets:
openai_awsbedrock_test.go, openai_gcpvertexai_test.go behaviors)."""

import json

import pytest

from aigw.filterapi.config import APISchemaName
from aigw.translator import get_translator
from aigw.translator.chat_bedrock import openai_to_converse_request
from aigw.translator.chat_gcp import openai_to_gemini_request
from aigw.translator.eventstream import encode_event
from aigw.translator.sse import SSEDecoder

REQ = {
    "model": "anthropic.claude-3",
    "messages": [
        {"role": "system", "content": "sys"},
        {"role": "user", "content": "hello"},
        {
            "role": "assistant",
            "tool_calls": [
                {"id": "t1", "type": "function", "function": {"name": "f", "arguments": "{}"}}
            ],
        },
        {"role": "tool", "tool_call_id": "t1", "content": "result"},
    ],
    "max_tokens": 50,
    "temperature": 0.1,
    "tools": [
        {"type": "function", "function": {"name": "f", "parameters": {"type": "object"}}}
    ],
}


def test_converse_request_mapping():
    c = openai_to_converse_request(json.loads(json.dumps(REQ)))
    assert c["system"] == [{"text": "sys"}]
    assert c["inferenceConfig"] == {"maxTokens": 50, "temperature": 0.1}
    assert c["messages"][0] == {"role": "user", "content": [{"text": "hello"}]}
    assert c["messages"][1]["content"][0]["toolUse"]["toolUseId"] == "t1"
    tr = c["messages"][2]["content"][0]["toolResult"]
    assert tr["toolUseId"] == "t1"
    assert c["toolConfig"]["tools"][0]["toolSpec"]["name"] == "f"


def test_bedrock_paths_and_unary():
    t = get_translator("/v1/chat/completions", APISchemaName.AWS_BEDROCK)
    req = t.request(json.loads(json.dumps(REQ)), model_override="anthropic.claude-3-5")
    assert req.path == "/model/anthropic.claude-3-5/converse"
    resp = {
        "output": {
            "message": {
                "role": "assistant",
                "content": [{"text": "hi"}, {"toolUse": {"toolUseId": "x", "name": "f", "input": {"a": 1}}}],
            }
        },
        "stopReason": "tool_use",
        "usage": {"inputTokens": 5, "outputTokens": 6, "totalTokens": 11},
    }
    r = t.response_body(200, json.dumps(resp).encode())
    o = json.loads(r.body)
    assert o["choices"][0]["finish_reason"] == "tool_calls"
    assert o["choices"][0]["message"]["content"] == "hi"
    assert r.usage.input_tokens == 5 and r.usage.total_tokens == 11


@pytest.mark.parametrize("n", [1, 17, 4096])
def test_bedrock_eventstream_to_openai_sse(n):
    t = get_translator("/v1/chat/completions", APISchemaName.AWS_BEDROCK)
    t.request(json.loads(json.dumps(REQ)), stream=True)
    assert t.response_headers(200, {})["content-type"] == "text/event-stream"
    frames = b"".join(
        [
            encode_event("messageStart", json.dumps({"role": "assistant"}).encode()),
            encode_event(
                "contentBlockDelta",
                json.dumps({"contentBlockIndex": 0, "delta": {"text": "Hel"}}).encode(),
            ),
            encode_event(
                "contentBlockDelta",
                json.dumps({"contentBlockIndex": 0, "delta": {"text": "lo"}}).encode(),
            ),
            encode_event("messageStop", json.dumps({"stopReason": "end_turn"}).encode()),
            encode_event(
                "metadata",
                json.dumps({"usage": {"inputTokens": 3, "outputTokens": 2, "totalTokens": 5}}).encode(),
            ),
        ]
    )
    out = bytearray()
    usage = None
    for i in range(0, len(frames), n):
        r = t.response_chunk(frames[i : i + n])
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    assert usage.total_tokens == 5
    evs = SSEDecoder().feed(bytes(out))
    assert evs[-1].data == "[DONE]"
    text = ""
    finish = None
    for ev in evs[:-1]:
        c = json.loads(ev.data)
        text += c["choices"][0]["delta"].get("content") or ""
        finish = c["choices"][0]["finish_reason"] or finish
    assert text == "Hello" and finish == "stop"


def test_gemini_request_mapping():
    g = openai_to_gemini_request(json.loads(json.dumps(REQ)))
    assert g["systemInstruction"] == {"parts": [{"text": "sys"}]}
    assert g["contents"][0] == {"role": "user", "parts": [{"text": "hello"}]}
    assert g["contents"][1]["role"] == "model"
    assert g["contents"][1]["parts"][0]["functionCall"]["name"] == "f"
    fr = g["contents"][2]["parts"][0]["functionResponse"]
    assert fr["name"] == "f" and fr["response"] == {"result": "result"}
    assert g["generationConfig"]["maxOutputTokens"] == 50
    assert g["tools"][0]["functionDeclarations"][0]["name"] == "f"


def test_gemini_paths_and_unary():
    t = get_translator(
        "/v1/chat/completions",
        APISchemaName.GCP_VERTEX_AI,
        gcp_project="proj",
        gcp_region="us-central1",
    )
    req = t.request({"model": "gemini-2.0-flash", "messages": [{"role": "user", "content": "q"}]})
    assert (
        req.path
        == "/v1/projects/proj/locations/us-central1/publishers/google/models/gemini-2.0-flash:generateContent"
    )
    resp = {
        "candidates": [
            {"content": {"role": "model", "parts": [{"text": "ans"}]}, "finishReason": "STOP"}
        ],
        "usageMetadata": {"promptTokenCount": 4, "candidatesTokenCount": 3, "totalTokenCount": 7},
        "modelVersion": "gemini-2.0-flash-001",
    }
    r = t.response_body(200, json.dumps(resp).encode())
    o = json.loads(r.body)
    assert o["choices"][0]["message"]["content"] == "ans"
    assert o["model"] == "gemini-2.0-flash-001"
    assert r.usage.input_tokens == 4 and r.usage.total_tokens == 7


def test_gemini_streaming():
    t = get_translator(
        "/v1/chat/completions", APISchemaName.GCP_VERTEX_AI, gcp_project="p", gcp_region="r"
    )
    req = t.request(
        {"model": "gemini-2.0-flash", "messages": [{"role": "user", "content": "q"}]}, stream=True
    )
    assert req.path.endswith(":streamGenerateContent?alt=sse")
    sse = (
        b'data: {"candidates":[{"content":{"parts":[{"text":"An"}]}}]}\n\n'
        b'data: {"candidates":[{"content":{"parts":[{"text":"swer"}]},"finishReason":"STOP"}],'
        b'"usageMetadata":{"promptTokenCount":4,"candidatesTokenCount":2,"totalTokenCount":6}}\n\n'
    )
    out = bytearray()
    usage = None
    for i in range(0, len(sse), 23):
        r = t.response_chunk(sse[i : i + 23])
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    assert usage.total_tokens == 6
    evs = SSEDecoder().feed(bytes(out))
    assert evs[-1].data == "[DONE]"
    text = "".join(
        json.loads(ev.data)["choices"][0]["delta"].get("content") or "" for ev in evs[:-1]
    )
    assert text == "Answer"


def test_azure_path_rewrite():
    t = get_translator("/v1/chat/completions", APISchemaName.AZURE_OPENAI, api_version="2024-10-01")
    req = t.request({"model": "gpt-4o", "messages": []})
    assert req.path == "/openai/deployments/gpt-4o/chat/completions?api-version=2024-10-01"


def test_openai_passthrough_force_usage_and_scan():
    t = get_translator("/v1/chat/completions", APISchemaName.OPENAI)
    req = t.request(
        {"model": "gpt-4o", "messages": [], "stream": True},
        model_override="gpt-4o-mini",
        stream=True,
        force_include_usage=True,
    )
    body = json.loads(req.body)
    assert body["model"] == "gpt-4o-mini"
    assert body["stream_options"]["include_usage"] is True
    chunk = (
        b'data: {"object":"chat.completion.chunk","model":"gpt-4o-mini","choices":[],'
        b'"usage":{"prompt_tokens":10,"completion_tokens":4,"total_tokens":14,'
        b'"prompt_tokens_details":{"cached_tokens":8}}}\n\ndata: [DONE]\n\n'
    )
    r = t.response_chunk(chunk)
    assert r.body == chunk  # passthrough preserves bytes
    assert r.usage.total_tokens == 14 and r.usage.cached_input_tokens == 8


def test_bedrock_embeddings():
    t = get_translator("/v1/embeddings", APISchemaName.AWS_BEDROCK)
    req = t.request({"model": "amazon.titan-embed-text-v2:0", "input": "hello"})
    assert req.path == "/model/amazon.titan-embed-text-v2:0/invoke"
    assert json.loads(req.body) == {"inputText": "hello"}
    r = t.response_body(
        200, json.dumps({"embedding": [0.1, 0.2], "inputTextTokenCount": 2}).encode()
    )
    o = json.loads(r.body)
    assert o["data"][0]["embedding"] == [0.1, 0.2]
    assert r.usage.input_tokens == 2
