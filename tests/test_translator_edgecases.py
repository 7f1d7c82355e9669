"""Translator edge-case conformance (modelled on the reference's
table-driven translator test corpora: tool_choice variants, multimodal
content, stop-reason maps, usage merging, null-body guards)."""

import json

import pytest

from aigw.filterapi.config import APISchemaName
from aigw.translator import TranslationError, Usage, get_translator
from aigw.translator.anthropic_schema import (
    ANTHROPIC_TO_OPENAI_STOP,
    OPENAI_TO_ANTHROPIC_STOP,
    openai_to_anthropic_request,
)
from aigw.translator.chat_bedrock import BEDROCK_TO_OPENAI_STOP, openai_to_converse_request
from aigw.translator.chat_gcp import GEMINI_TO_OPENAI_FINISH, openai_to_gemini_request


def test_stop_reason_maps_are_total():
    # every provider stop reason maps into the OpenAI finish_reason enum
    valid = {"stop", "length", "tool_calls", "content_filter"}
    assert set(ANTHROPIC_TO_OPENAI_STOP.values()) <= valid
    assert set(BEDROCK_TO_OPENAI_STOP.values()) <= valid
    assert set(GEMINI_TO_OPENAI_FINISH.values()) <= valid
    # and the reverse map round-trips the common cases
    for oa, an in OPENAI_TO_ANTHROPIC_STOP.items():
        assert ANTHROPIC_TO_OPENAI_STOP.get(an) is not None


@pytest.mark.parametrize("choice,expected", [
    ("auto", {"type": "auto"}),
    ("required", {"type": "any"}),
    ({"type": "function", "function": {"name": "f"}}, {"type": "tool", "name": "f"}),
])
def test_anthropic_tool_choice_variants(choice, expected):
    req = {"model": "m", "messages": [], "tools": [
        {"type": "function", "function": {"name": "f", "parameters": {}}}], "tool_choice": choice}
    a = openai_to_anthropic_request(req)
    assert a["tool_choice"] == expected


def test_anthropic_tool_choice_none_drops_tools():
    req = {"model": "m", "messages": [], "tools": [
        {"type": "function", "function": {"name": "f"}}], "tool_choice": "none"}
    a = openai_to_anthropic_request(req)
    assert "tools" not in a


def test_anthropic_reasoning_effort_to_thinking():
    req = {"model": "m", "messages": [], "reasoning_effort": "high"}
    a = openai_to_anthropic_request(req)
    assert a["thinking"] == {"type": "enabled", "budget_tokens": 24576}


def test_anthropic_rejects_audio_content():
    req = {"model": "m", "messages": [
        {"role": "user", "content": [{"type": "input_audio", "input_audio": {}}]}]}
    with pytest.raises(TranslationError):
        openai_to_anthropic_request(req)


def test_bedrock_rejects_remote_image_url():
    req = {"model": "m", "messages": [
        {"role": "user", "content": [
            {"type": "image_url", "image_url": {"url": "https://x/y.png"}}]}]}
    with pytest.raises(TranslationError):
        openai_to_converse_request(req)


def test_bedrock_adjacent_tool_results_merge():
    req = {
        "model": "m",
        "messages": [
            {"role": "assistant", "tool_calls": [
                {"id": "a", "type": "function", "function": {"name": "f", "arguments": "{}"}},
                {"id": "b", "type": "function", "function": {"name": "g", "arguments": "{}"}},
            ]},
            {"role": "tool", "tool_call_id": "a", "content": "1"},
            {"role": "tool", "tool_call_id": "b", "content": "2"},
        ],
    }
    c = openai_to_converse_request(req)
    # both toolResults must land in ONE user message (Converse requirement)
    user_msgs = [m for m in c["messages"] if m["role"] == "user"]
    assert len(user_msgs) == 1
    assert len(user_msgs[0]["content"]) == 2


def test_gemini_response_format_json():
    req = {"model": "m", "messages": [], "response_format": {"type": "json_object"}}
    g = openai_to_gemini_request(req)
    assert g["generationConfig"]["responseMimeType"] == "application/json"


def test_gemini_tool_choice_forced_function():
    req = {"model": "m", "messages": [],
           "tools": [{"type": "function", "function": {"name": "f"}}],
           "tool_choice": {"type": "function", "function": {"name": "f"}}}
    g = openai_to_gemini_request(req)
    fcc = g["toolConfig"]["functionCallingConfig"]
    assert fcc["mode"] == "ANY" and fcc["allowedFunctionNames"] == ["f"]


def test_gemini_multi_candidate_choices():
    t = get_translator("/v1/chat/completions", APISchemaName.GCP_VERTEX_AI,
                       gcp_project="p", gcp_region="r")
    t.request({"model": "g", "messages": [], "n": 2})
    resp = {
        "candidates": [
            {"content": {"parts": [{"text": "a"}]}, "finishReason": "STOP"},
            {"content": {"parts": [{"text": "b"}]}, "finishReason": "MAX_TOKENS"},
        ],
        "usageMetadata": {"promptTokenCount": 1, "candidatesTokenCount": 2,
                          "totalTokenCount": 3},
    }
    r = t.response_body(200, json.dumps(resp).encode())
    o = json.loads(r.body)
    assert [c["finish_reason"] for c in o["choices"]] == ["stop", "length"]
    assert o["choices"][1]["index"] == 1


def test_usage_merge_max_semantics():
    u = Usage()
    u.merge_max(Usage(input_tokens=10, output_tokens=1, total_tokens=11))
    u.merge_max(Usage(input_tokens=10, output_tokens=5, total_tokens=15))
    u.merge_max(Usage(output_tokens=3))  # late partial chunk must not regress
    assert (u.input_tokens, u.output_tokens, u.total_tokens) == (10, 5, 15)


def test_null_content_message():
    # assistant message with null content + tool_calls (null-body guard)
    req = {"model": "m", "messages": [
        {"role": "assistant", "content": None,
         "tool_calls": [{"id": "x", "type": "function",
                         "function": {"name": "f", "arguments": "not-json"}}]}]}
    a = openai_to_anthropic_request(req)
    assert a["messages"][0]["content"][0]["type"] == "tool_use"
    assert a["messages"][0]["content"][0]["input"] == {}  # bad JSON args -> {}
    c = openai_to_converse_request(json.loads(json.dumps(req)))
    assert c["messages"][0]["content"][0]["toolUse"]["input"] == {}


def test_error_translation_wraps_non_openai_errors():
    t = get_translator("/v1/chat/completions", APISchemaName.OPENAI)
    out = json.loads(t.response_error(500, b"plain text failure", {}))
    assert out["error"]["message"] == "plain text failure"
    assert out["error"]["code"] == "500"
    # already-OpenAI-shaped errors pass through
    native = b'{"error": {"message": "x", "type": "invalid_request_error"}}'
    assert t.response_error(400, native, {}) == native


def test_developer_role_treated_as_system():
    req = {"model": "m", "messages": [{"role": "developer", "content": "dev rules"}]}
    a = openai_to_anthropic_request(req)
    assert a["system"] == "dev rules"
    c = openai_to_converse_request(req)
    assert c["system"] == [{"text": "dev rules"}]
    g = openai_to_gemini_request(req)
    assert g["systemInstruction"]["parts"] == [{"text": "dev rules"}]


def test_max_completion_tokens_priority():
    req = {"model": "m", "messages": [], "max_tokens": 10, "max_completion_tokens": 99}
    assert openai_to_anthropic_request(req)["max_tokens"] == 99
    assert openai_to_converse_request(req)["inferenceConfig"]["maxTokens"] == 99
    assert openai_to_gemini_request(req)["generationConfig"]["maxOutputTokens"] == 99


def test_extension_fields_thinking_and_cache_control():
    """vendor-specific-fields.md + prompt-caching.md: the unified
    `thinking` field and `cache_control` breakpoints translate per
    backend."""
    import json as _json

    from aigw.filterapi.config import APISchemaName
    from aigw.translator import get_translator

    base = {
        "model": "m",
        "thinking": {"type": "enabled", "budget_tokens": 2048},
        "messages": [{"role": "user", "content": [
            {"type": "text", "text": "big context",
             "cache_control": {"type": "ephemeral"}},
            {"type": "text", "text": "question"},
        ]}],
    }
    # -> Anthropic: thinking verbatim, cache_control on the block
    t = get_translator("/v1/chat/completions", APISchemaName.ANTHROPIC)
    body = _json.loads(t.request(_json.loads(_json.dumps(base))).body)
    assert body["thinking"] == {"type": "enabled", "budget_tokens": 2048}
    blocks = body["messages"][0]["content"]
    assert blocks[0]["cache_control"] == {"type": "ephemeral"}
    assert "cache_control" not in blocks[1]

    # -> Bedrock Converse: thinking in additionalModelRequestFields,
    # cache_control becomes a cachePoint block after the cached text
    t = get_translator("/v1/chat/completions", APISchemaName.AWS_BEDROCK)
    body = _json.loads(t.request(_json.loads(_json.dumps(base))).body)
    assert body["additionalModelRequestFields"]["thinking"]["budget_tokens"] == 2048
    content = body["messages"][0]["content"]
    assert content[0] == {"text": "big context"}
    assert content[1] == {"cachePoint": {"type": "default"}}
    assert content[2] == {"text": "question"}

    # -> Gemini: thinking -> generationConfig.thinkingConfig;
    # safetySettings passthrough
    t = get_translator("/v1/chat/completions", APISchemaName.GCP_VERTEX_AI,
                       gcp_project="p", gcp_region="r")
    gbase = _json.loads(_json.dumps(base))
    gbase["safetySettings"] = [{"category": "HARM_CATEGORY_HARASSMENT",
                                "threshold": "BLOCK_NONE"}]
    body = _json.loads(t.request(gbase).body)
    assert body["generationConfig"]["thinkingConfig"]["thinkingBudget"] == 2048
    assert body["safetySettings"][0]["threshold"] == "BLOCK_NONE"


def test_gemini_google_search_tool():
    import json as _json

    from aigw.filterapi.config import APISchemaName
    from aigw.translator import get_translator

    t = get_translator("/v1/chat/completions", APISchemaName.GCP_VERTEX_AI,
                       gcp_project="p", gcp_region="r")
    body = _json.loads(t.request({
        "model": "gemini", "messages": [{"role": "user", "content": "q"}],
        "tools": [{"type": "google_search",
                   "google_search": {"excludeDomains": ["x.com"]}}],
    }).body)
    assert body["tools"] == [{"googleSearch": {"excludeDomains": ["x.com"]}}]


def test_streaming_response_headers_rewrites():
    """Streamed responses present client-appropriate content types:
    Bedrock's binary event-stream becomes SSE; SSE backends stay SSE;
    content-length never survives re-chunking (A.8)."""
    from aigw.filterapi.config import APISchemaName
    from aigw.translator import get_translator

    req = {"model": "m", "stream": True,
           "messages": [{"role": "user", "content": "q"}]}

    t = get_translator("/v1/chat/completions", APISchemaName.AWS_BEDROCK)
    t.request(dict(req), stream=True)
    h = t.response_headers(200, {
        "content-type": "application/vnd.amazon.eventstream",
        "content-length": "12345",
    })
    assert h["content-type"] == "text/event-stream"

    t = get_translator("/v1/chat/completions", APISchemaName.GCP_VERTEX_AI,
                       gcp_project="p", gcp_region="r")
    t.request(dict(req), stream=True)
    h = t.response_headers(200, {"content-type": "text/event-stream"})
    assert h.get("content-type", "text/event-stream") == "text/event-stream"
