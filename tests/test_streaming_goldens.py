"""Golden streaming transcripts: fixed provider streams -> committed
byte-exact client SSE output, for every streaming conversion family
(the reference's translator test corpus pins streamed output the same
way). Regenerate with:
    python tests/test_streaming_goldens.py regenerate
"""

import json
import os
import pathlib
import struct
import sys
import zlib
from unittest import mock

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from aigw.filterapi.config import APISchemaName
from aigw.translator import get_translator

GOLDEN_PATH = pathlib.Path(__file__).parent / "goldens" / "streaming_transcripts.json"
FIXED_TIME = 1700000000


def _sse(events):
    out = bytearray()
    for ev in events:
        if isinstance(ev, tuple):
            et, d = ev
            out += f"event: {et}\n".encode()
        else:
            d = ev
        data = d if isinstance(d, str) else json.dumps(d, sort_keys=True)
        out += b"data: " + data.encode() + b"\n\n"
    return bytes(out)


def _eventstream(messages):
    """AWS binary event-stream frames (:event-type header + JSON payload)."""
    out = bytearray()
    for etype, payload in messages:
        headers = bytearray()
        for k, v in ((":event-type", etype), (":message-type", "event")):
            kb, vb = k.encode(), v.encode()
            headers += struct.pack(">B", len(kb)) + kb + b"\x07"
            headers += struct.pack(">H", len(vb)) + vb
        body = json.dumps(payload, sort_keys=True).encode()
        total = 12 + len(headers) + len(body) + 4
        prelude = struct.pack(">II", total, len(headers))
        pre_crc = struct.pack(">I", zlib.crc32(prelude) & 0xFFFFFFFF)
        frame = prelude + pre_crc + bytes(headers) + body
        out += frame + struct.pack(">I", zlib.crc32(frame) & 0xFFFFFFFF)
    return bytes(out)


ANTHROPIC_STREAM = _sse([
    ("message_start", {"type": "message_start", "message": {
        "id": "msg_01", "model": "claude-3", "role": "assistant", "content": [],
        "usage": {"input_tokens": 7, "output_tokens": 1}}}),
    ("content_block_start", {"type": "content_block_start", "index": 0,
                             "content_block": {"type": "text", "text": ""}}),
    ("content_block_delta", {"type": "content_block_delta", "index": 0,
                             "delta": {"type": "text_delta", "text": "Hel"}}),
    ("content_block_delta", {"type": "content_block_delta", "index": 0,
                             "delta": {"type": "text_delta", "text": "lo"}}),
    ("content_block_stop", {"type": "content_block_stop", "index": 0}),
    ("message_delta", {"type": "message_delta",
                       "delta": {"stop_reason": "end_turn"},
                       "usage": {"output_tokens": 9}}),
    ("message_stop", {"type": "message_stop"}),
])

OPENAI_STREAM = _sse([
    {"id": "c1", "object": "chat.completion.chunk", "model": "gpt-4o",
     "choices": [{"index": 0, "delta": {"role": "assistant", "content": "He"}}]},
    {"id": "c1", "object": "chat.completion.chunk", "model": "gpt-4o",
     "choices": [{"index": 0, "delta": {"content": "y"}}]},
    {"id": "c1", "object": "chat.completion.chunk", "model": "gpt-4o",
     "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}]},
    {"id": "c1", "object": "chat.completion.chunk", "model": "gpt-4o",
     "choices": [], "usage": {"prompt_tokens": 3, "completion_tokens": 2,
                              "total_tokens": 5}},
    "[DONE]",
])

GEMINI_STREAM = _sse([
    {"candidates": [{"content": {"role": "model",
                                 "parts": [{"text": "Bon"}]}, "index": 0}]},
    {"candidates": [{"content": {"role": "model", "parts": [{"text": "jour"}]},
                     "finishReason": "STOP", "index": 0}],
     "usageMetadata": {"promptTokenCount": 4, "candidatesTokenCount": 2,
                       "totalTokenCount": 6}},
])

BEDROCK_STREAM = _eventstream([
    ("messageStart", {"role": "assistant"}),
    ("contentBlockDelta", {"contentBlockIndex": 0,
                           "delta": {"text": "Hi"}}),
    ("contentBlockDelta", {"contentBlockIndex": 0,
                           "delta": {"text": " there"}}),
    ("contentBlockStop", {"contentBlockIndex": 0}),
    ("messageStop", {"stopReason": "end_turn"}),
    ("metadata", {"usage": {"inputTokens": 5, "outputTokens": 2,
                            "totalTokens": 7}}),
])

# (case name, client endpoint, backend schema, request body, provider stream)
CASES = [
    ("openai_from_anthropic", "/v1/chat/completions", "Anthropic",
     {"model": "m", "messages": [{"role": "user", "content": "hi"}], "stream": True},
     ANTHROPIC_STREAM),
    ("openai_from_gemini", "/v1/chat/completions", "GCPVertexAI",
     {"model": "m", "messages": [{"role": "user", "content": "hi"}], "stream": True},
     GEMINI_STREAM),
    ("openai_from_bedrock", "/v1/chat/completions", "AWSBedrock",
     {"model": "m", "messages": [{"role": "user", "content": "hi"}], "stream": True},
     BEDROCK_STREAM),
    ("anthropic_from_openai", "/anthropic/v1/messages", "OpenAI",
     {"model": "m", "max_tokens": 8, "messages": [{"role": "user", "content": "hi"}]},
     OPENAI_STREAM),
    ("anthropic_from_bedrock", "/anthropic/v1/messages", "AWSBedrock",
     {"model": "m", "max_tokens": 8, "messages": [{"role": "user", "content": "hi"}]},
     BEDROCK_STREAM),
]


def _run_case(endpoint, schema, req, stream, chunk_size):
    with mock.patch("time.time", lambda: FIXED_TIME):
        t = get_translator(endpoint, APISchemaName(schema),
                           gcp_project="proj", gcp_region="region")
        t.request(json.loads(json.dumps(req)), stream=True)
        out = bytearray()
        usage = None
        for i in range(0, len(stream), chunk_size):
            r = t.response_chunk(stream[i : i + chunk_size])
            out.extend(r.body)
            if r.usage:
                usage = r.usage
        r = t.response_flush()
        out.extend(r.body)
        if r.usage:
            usage = r.usage
    return bytes(out), usage


def _transcripts(chunk_size=10**9):
    out = {}
    for name, endpoint, schema, req, stream in CASES:
        body, usage = _run_case(endpoint, schema, req, stream, chunk_size)
        out[name] = {
            "transcript": body.decode("utf-8"),
            "usage": {
                "input_tokens": usage.input_tokens,
                "output_tokens": usage.output_tokens,
            } if usage else None,
        }
    return out


def test_streaming_transcripts_match_goldens():
    got = _transcripts()
    want = json.loads(GOLDEN_PATH.read_text())
    assert set(got) == set(want)
    for name in want:
        assert got[name] == want[name], f"transcript drift in {name}"


def test_transcripts_invariant_to_chunking():
    whole = _transcripts()
    for n in (1, 7):
        rechunked = _transcripts(chunk_size=n)
        for name in whole:
            assert rechunked[name] == whole[name], f"{name} differs at chunk={n}"


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "regenerate":
        GOLDEN_PATH.write_text(json.dumps(_transcripts(), indent=1, sort_keys=True))
        print(f"wrote {GOLDEN_PATH}")
