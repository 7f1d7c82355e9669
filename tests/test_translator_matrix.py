"""Matrix completeness + the composed
Anthropic→Bedrock-Converse translator + tokenize variants."""

import json

import pytest

from aigw.filterapi.config import APISchemaName
from aigw.translator import get_translator, supported_matrix
from aigw.translator.eventstream import encode_event
from aigw.translator.sse import SSEDecoder


def test_endpoint_schema_matrix_covers_survey_a9():
    m = supported_matrix()
    expect = {
        "/v1/chat/completions": {"OpenAI", "AWSBedrock", "AWSAnthropic", "AzureOpenAI",
                                 "GCPVertexAI", "GCPAnthropic", "Anthropic"},
        "/v1/completions": {"OpenAI"},
        "/v1/embeddings": {"OpenAI", "AzureOpenAI", "GCPVertexAI", "AWSBedrock"},
        "/v1/responses": {"OpenAI", "AzureOpenAI"},
        "/v1/responses/input_tokens": {"OpenAI", "AzureOpenAI"},
        "/v1/images/generations": {"OpenAI"},
        "/v1/audio/speech": {"OpenAI"},
        "/v1/audio/transcriptions": {"OpenAI"},
        "/v1/audio/translations": {"OpenAI"},
        "/tokenize": {"OpenAI", "GCPVertexAI", "GCPAnthropic", "Anthropic"},
        "/anthropic/v1/messages": {"Anthropic", "GCPAnthropic", "AWSAnthropic",
                                   "OpenAI", "AWSBedrock"},
        "/anthropic/v1/messages/count_tokens": {"Anthropic", "GCPAnthropic",
                                                "AWSAnthropic"},
        "/v2/rerank": {"Cohere"},
    }
    for ep, want in expect.items():
        got = set(m.get(ep, []))
        assert want <= got, f"{ep}: missing {want - got}"


ANTHROPIC_REQ = {
    "model": "claude-on-bedrock",
    "max_tokens": 64,
    "system": "be brief",
    "messages": [{"role": "user", "content": "hello"}],
}


def test_anthropic_to_bedrock_unary():
    t = get_translator("/anthropic/v1/messages", APISchemaName.AWS_BEDROCK)
    req = t.request(json.loads(json.dumps(ANTHROPIC_REQ)))
    assert req.path == "/model/claude-on-bedrock/converse"
    creq = json.loads(req.body)
    assert creq["system"] == [{"text": "be brief"}]
    assert creq["inferenceConfig"]["maxTokens"] == 64

    resp = {
        "output": {"message": {"role": "assistant", "content": [{"text": "hi there"}]}},
        "stopReason": "end_turn",
        "usage": {"inputTokens": 5, "outputTokens": 2, "totalTokens": 7},
    }
    r = t.response_body(200, json.dumps(resp).encode())
    a = json.loads(r.body)
    assert a["type"] == "message"
    assert a["content"] == [{"type": "text", "text": "hi there"}]
    assert a["stop_reason"] == "end_turn"
    assert a["usage"]["output_tokens"] == 2
    assert r.usage.input_tokens == 5


def test_anthropic_to_bedrock_streaming():
    t = get_translator("/anthropic/v1/messages", APISchemaName.AWS_BEDROCK)
    req = t.request(json.loads(json.dumps(ANTHROPIC_REQ)), stream=True)
    assert req.path.endswith("/converse-stream")
    frames = b"".join(
        [
            encode_event("messageStart", b'{"role":"assistant"}'),
            encode_event("contentBlockDelta",
                         json.dumps({"contentBlockIndex": 0, "delta": {"text": "yo"}}).encode()),
            encode_event("messageStop", json.dumps({"stopReason": "max_tokens"}).encode()),
            encode_event("metadata",
                         json.dumps({"usage": {"inputTokens": 5, "outputTokens": 1,
                                               "totalTokens": 6}}).encode()),
        ]
    )
    out = bytearray()
    usage = None
    for i in range(0, len(frames), 37):
        r = t.response_chunk(frames[i : i + 37])
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    evs = SSEDecoder().feed(bytes(out))
    types = [e.event for e in evs]
    assert types[0] == "message_start" and types[-1] == "message_stop"
    text = "".join(
        json.loads(e.data)["delta"].get("text", "")
        for e in evs if e.event == "content_block_delta"
    )
    assert text == "yo"
    md = json.loads(evs[-2].data)
    assert md["delta"]["stop_reason"] == "max_tokens"
    assert usage is not None and usage.output_tokens == 1


def test_tokenize_gcp_vertex():
    t = get_translator("/tokenize", APISchemaName.GCP_VERTEX_AI,
                       gcp_project="p", gcp_region="r")
    req = t.request({"model": "gemini-2.0", "prompt": "count me"})
    assert req.path.endswith("models/gemini-2.0:countTokens")
    r = t.response_body(200, b'{"totalTokens": 7}')
    assert json.loads(r.body)["count"] == 7
    assert r.usage.input_tokens == 7


def test_tokenize_anthropic():
    t = get_translator("/tokenize", APISchemaName.ANTHROPIC)
    req = t.request({"model": "claude", "prompt": "count me"})
    assert req.path == "/v1/messages/count_tokens"
    body = json.loads(req.body)
    assert body["messages"][0]["content"] == "count me"
    r = t.response_body(200, b'{"input_tokens": 3}')
    assert json.loads(r.body)["count"] == 3


def test_count_tokens_aws():
    t = get_translator("/anthropic/v1/messages/count_tokens", APISchemaName.AWS_ANTHROPIC)
    req = t.request({"model": "claude-b", "messages": []})
    assert req.path == "/model/claude-b/count-tokens"
    assert json.loads(req.body)["anthropic_version"] == "bedrock-2023-05-31"


@pytest.mark.parametrize("endpoint,schema", [
    ("/v1/chat/completions", APISchemaName.COHERE),
    ("/v2/rerank", APISchemaName.OPENAI),
])
def test_unsupported_combinations_rejected(endpoint, schema):
    from aigw.translator import TranslationError

    with pytest.raises(TranslationError):
        get_translator(endpoint, schema)
