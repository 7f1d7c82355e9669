"""Typed schema layer (aigw.apischema): every vendored cassette payload
must parse through the pydantic union models (the reference certifies
its 12k-LoC Go types against the same traffic), exotic unions must
round-trip losslessly, and WRONG shapes must fail loudly — the failure
mode the round-1 review flagged for dict-structural handling."""

import json

import pytest
import yaml

from aigw.apischema import SchemaError, validate_request, validate_response
from aigw.apischema.anthropic import MessagesRequest, MessagesResponse
from tests.test_conformance_cassettes import load_cassettes


def test_all_cassette_requests_parse_typed():
    n = 0
    for c in load_cassettes():
        if not c["req_ctype"].startswith("application/json"):
            continue
        try:
            body = json.loads(c["req_body"])
        except ValueError:
            continue
        if not isinstance(body, dict):
            continue
        path = c["path"] if c["path"].startswith("/v1/") else "/v1/chat/completions"
        try:
            validate_request(path, body)
            n += 1
        except SchemaError as e:
            # bad-request cassettes record intentionally malformed traffic
            assert "bad-request" in c["name"] or "no-messages" in c["name"], (
                c["name"], str(e))
    assert n >= 35, n


def test_all_cassette_200_responses_parse_typed():
    n = 0
    for c in load_cassettes():
        if c["status"] != 200 or not c["resp_ctype"].startswith("application/json"):
            continue
        try:
            body = json.loads(c["resp_body"])
        except ValueError:
            continue
        path = c["path"] if c["path"].startswith("/v1/") else "/v1/chat/completions"
        validate_response(path, body)  # raises on shape drift
        n += 1
    assert n >= 25, n


def test_exotic_unions_roundtrip():
    # parallel tool calls with logprobs and detailed usage
    req = validate_request("/v1/chat/completions", {
        "model": "m",
        "messages": [
            {"role": "system", "content": "s"},
            {"role": "user", "content": [
                {"type": "text", "text": "look"},
                {"type": "image_url", "image_url": {"url": "data:image/png;base64,xx",
                                                    "detail": "high"}},
                {"type": "input_audio", "input_audio": {"data": "QQ==", "format": "wav"}},
            ]},
            {"role": "assistant", "content": None, "tool_calls": [
                {"id": "c1", "type": "function",
                 "function": {"name": "f", "arguments": "{\"x\":1}"}},
                {"id": "c2", "type": "function",
                 "function": {"name": "g", "arguments": "{}"}},
            ]},
            {"role": "tool", "tool_call_id": "c1", "content": "42"},
        ],
        "tool_choice": {"type": "function", "function": {"name": "f"}},
        "response_format": {"type": "json_schema",
                            "json_schema": {"name": "s", "schema": {"type": "object"}}},
        "stop": ["a", "b"],
        "logit_bias": {"50256": -100},
    })
    assert req.messages[1].content[1].image_url.detail == "high"
    assert req.messages[2].tool_calls[1].function.name == "g"
    assert req.tool_choice.function.name == "f"
    assert req.response_format.json_schema.schema_ == {"type": "object"}

    # unknown fields survive (tolerant unmarshal, Go parity)
    req = validate_request("/v1/chat/completions", {
        "model": "m", "messages": [{"role": "user", "content": "x"}],
        "brand_new_field": {"nested": True},
    })
    assert req.model_dump(exclude_none=True)["brand_new_field"] == {"nested": True}

    # completions prompt unions
    for prompt in ("x", ["a", "b"], [1, 2, 3], [[1, 2], [3]]):
        validate_request("/v1/completions", {"model": "m", "prompt": prompt})

    # embeddings input unions + base64 response embedding
    for inp in ("x", ["a"], [1, 2], [[1], [2]]):
        validate_request("/v1/embeddings", {"model": "m", "input": inp})
    validate_response("/v1/embeddings", {
        "object": "list", "model": "m",
        "data": [{"object": "embedding", "index": 0, "embedding": "AAAA"}],
    })
    validate_response("/v1/embeddings", {
        "data": [{"embedding": [0.1, 0.2]}]})

    # chunk with logprobs
    validate_response("/v1/chat/completions", {
        "object": "chat.completion.chunk",
        "choices": [{"index": 0, "delta": {"content": "hi"},
                     "logprobs": {"content": [
                         {"token": "hi", "logprob": -0.5,
                          "top_logprobs": [{"token": "hi", "logprob": -0.5}]}]}}],
    })


def test_wrong_shapes_rejected():
    bad_requests = [
        ("/v1/chat/completions", {"model": "m", "messages": "not-a-list"}),
        ("/v1/chat/completions", {"model": "m", "messages": [{"role": "nope",
                                                             "content": "x"}]}),
        ("/v1/chat/completions", {"model": "m", "messages": [
            {"role": "user", "content": [{"type": "teleport", "text": "x"}]}]}),
        ("/v1/chat/completions", {"model": "m", "messages": [
            {"role": "assistant", "tool_calls": "oops"}]}),
        ("/v1/chat/completions", {"model": 7, "messages": []}),
        ("/v1/chat/completions", {"model": "m", "messages": [],
                                  "tool_choice": 5}),
        ("/v1/chat/completions", {"model": "m", "messages": [],
                                  "response_format": {"type": "yaml"}}),
        ("/v1/completions", {"model": "m", "prompt": {"bad": 1}}),
        ("/v1/embeddings", {"model": "m", "input": {"bad": 1}}),
        ("/v1/embeddings", {"model": "m"}),  # input required
    ]
    for path, body in bad_requests:
        with pytest.raises(SchemaError):
            validate_request(path, body)


def test_anthropic_unions():
    req = MessagesRequest.model_validate({
        "model": "claude", "max_tokens": 100,
        "system": [{"type": "text", "text": "be terse",
                    "cache_control": {"type": "ephemeral"}}],
        "messages": [
            {"role": "user", "content": [
                {"type": "text", "text": "look"},
                {"type": "image", "source": {"type": "base64",
                                             "media_type": "image/png",
                                             "data": "QQ=="}},
            ]},
            {"role": "assistant", "content": [
                {"type": "thinking", "thinking": "hmm", "signature": "s"},
                {"type": "tool_use", "id": "t1", "name": "f", "input": {"x": 1}},
            ]},
            {"role": "user", "content": [
                {"type": "tool_result", "tool_use_id": "t1",
                 "content": [{"type": "text", "text": "42"}]},
            ]},
        ],
        "tools": [{"name": "f", "input_schema": {"type": "object"}}],
        "tool_choice": {"type": "tool", "name": "f"},
        "thinking": {"type": "enabled", "budget_tokens": 512},
    })
    assert req.messages[1].content[1].input == {"x": 1}
    assert req.system[0].cache_control == {"type": "ephemeral"}

    resp = MessagesResponse.model_validate({
        "id": "m1", "type": "message", "role": "assistant", "model": "claude",
        "content": [{"type": "text", "text": "hi"}],
        "stop_reason": "end_turn",
        "usage": {"input_tokens": 5, "output_tokens": 2,
                  "cache_read_input_tokens": 3},
    })
    assert resp.usage.cache_read_input_tokens == 3

    with pytest.raises(Exception):
        MessagesRequest.model_validate({
            "model": "claude",
            "messages": [{"role": "user", "content": [{"type": "warp"}]}]})
