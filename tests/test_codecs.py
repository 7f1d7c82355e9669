"""SSE + AWS event-stream codec tests, including arbitrary-chunk-boundary
stress (the stateful re-chunking correctness property)."""

import json

import pytest

from aigw.translator.eventstream import (
    EventStreamDecoder,
    EventStreamError,
    encode_event,
    encode_message,
)
from aigw.translator.sse import PySSEDecoder, SSEDecoder, SSEEvent

# exercise BOTH the native (C++) and pure-Python SSE decoders
DECODERS = [SSEDecoder, PySSEDecoder] if SSEDecoder is not PySSEDecoder else [PySSEDecoder]


STREAM = (
    b": comment line\n"
    b"event: message_start\n"
    b'data: {"type":"message_start"}\n'
    b"\n"
    b'data: {"a":1}\n'
    b'data: {"b":2}\n'
    b"\n"
    b"data: [DONE]\n"
    b"\n"
)


def _expected_events():
    return [
        SSEEvent(data='{"type":"message_start"}', event="message_start"),
        SSEEvent(data='{"a":1}\n{"b":2}'),
        SSEEvent(data="[DONE]"),
    ]


@pytest.mark.parametrize("cls", DECODERS)
def test_sse_whole(cls):
    d = cls()
    assert d.feed(STREAM) + d.flush() == _expected_events()


@pytest.mark.parametrize("cls", DECODERS)
@pytest.mark.parametrize("n", [1, 2, 3, 5, 7, 11, 64])
def test_sse_any_chunking(cls, n):
    d = cls()
    events = []
    for i in range(0, len(STREAM), n):
        events.extend(d.feed(STREAM[i : i + n]))
    events.extend(d.flush())
    assert events == _expected_events()


@pytest.mark.parametrize("cls", DECODERS)
def test_sse_crlf(cls):
    d = cls()
    evs = d.feed(b"data: x\r\n\r\n")
    assert evs == [SSEEvent(data="x")]


@pytest.mark.parametrize("cls", DECODERS)
def test_sse_encode_roundtrip(cls):
    ev = SSEEvent(data='{"x":1}\nline2', event="delta")
    d = cls()
    assert d.feed(ev.encode()) == [ev]


@pytest.mark.parametrize("cls", DECODERS)
def test_sse_flush_unterminated(cls):
    d = cls()
    assert d.feed(b"data: partial") == []
    assert d.flush() == [SSEEvent(data="partial")]


def test_eventstream_roundtrip():
    msgs = [
        encode_event("messageStart", b'{"role":"assistant"}'),
        encode_event("contentBlockDelta", json.dumps({"delta": {"text": "hi"}}).encode()),
        encode_message({":message-type": "event", ":event-type": "metadata", "n": 42}, b"{}"),
    ]
    blob = b"".join(msgs)
    d = EventStreamDecoder()
    out = d.feed(blob)
    assert [m.event_type for m in out] == ["messageStart", "contentBlockDelta", "metadata"]
    assert out[0].payload == b'{"role":"assistant"}'
    assert out[2].headers["n"] == 42


@pytest.mark.parametrize("n", [1, 3, 7, 16, 33])
def test_eventstream_any_chunking(n):
    blob = b"".join(
        encode_event(f"ev{i}", json.dumps({"i": i}).encode()) for i in range(10)
    )
    d = EventStreamDecoder()
    out = []
    for i in range(0, len(blob), n):
        out.extend(d.feed(blob[i : i + n]))
    assert [m.event_type for m in out] == [f"ev{i}" for i in range(10)]


def test_eventstream_crc_mismatch():
    msg = bytearray(encode_event("x", b"{}"))
    msg[-1] ^= 0xFF
    with pytest.raises(EventStreamError):
        EventStreamDecoder().feed(bytes(msg))
