"""Property-based tests (hypothesis) for the stateful codecs and security
primitives — the subsystems where arbitrary-boundary and adversarial-input
correctness matter most (SURVEY §7 hard parts 1-2)."""

import json

from hypothesis import given, settings
from hypothesis import strategies as st

from aigw.mcp.session import SessionCrypto
from aigw.translator.eventstream import EventStreamDecoder, encode_event
from aigw.translator.sse import PySSEDecoder, SSEDecoder, SSEEvent

# -- strategies ---------------------------------------------------------------

sse_text = st.text(
    alphabet=st.characters(blacklist_characters="\r\n", blacklist_categories=("Cs",)),
    max_size=80,
)
event_types = st.sampled_from(["", "message_start", "content_block_delta", "done"])


def chunked(data: bytes, cuts: list[int]) -> list[bytes]:
    out = []
    prev = 0
    for c in sorted(set(min(c, len(data)) for c in cuts)):
        if c > prev:
            out.append(data[prev:c])
            prev = c
    out.append(data[prev:])
    return [c for c in out if c]


# -- SSE ----------------------------------------------------------------------


@settings(max_examples=150, deadline=None)
@given(
    events=st.lists(
        st.tuples(event_types, st.lists(sse_text, min_size=1, max_size=3)),
        min_size=1,
        max_size=6,
    ),
    cuts=st.lists(st.integers(min_value=0, max_value=2000), max_size=12),
)
def test_sse_decoders_agree_and_roundtrip(events, cuts):
    blob = bytearray()
    expected = []
    for etype, data_lines in events:
        data = "\n".join(data_lines)
        expected.append(SSEEvent(data=data, event=etype))
        blob.extend(SSEEvent(data=data, event=etype).encode())
    for cls in {SSEDecoder, PySSEDecoder}:
        dec = cls()
        got = []
        for chunk in chunked(bytes(blob), cuts):
            got.extend(dec.feed(chunk))
        got.extend(dec.flush())
        assert got == expected, cls.__name__


# -- AWS event-stream ---------------------------------------------------------


@settings(max_examples=100, deadline=None)
@given(
    payloads=st.lists(st.binary(max_size=200), min_size=1, max_size=6),
    cuts=st.lists(st.integers(min_value=0, max_value=3000), max_size=10),
)
def test_eventstream_roundtrip_any_chunking(payloads, cuts):
    blob = b"".join(encode_event(f"ev{i}", p) for i, p in enumerate(payloads))
    dec = EventStreamDecoder()
    got = []
    for chunk in chunked(blob, cuts):
        got.extend(dec.feed(chunk))
    assert [m.payload for m in got] == payloads
    assert [m.event_type for m in got] == [f"ev{i}" for i in range(len(payloads))]


# -- session crypto -----------------------------------------------------------


@settings(max_examples=100, deadline=None)
@given(
    payload=st.dictionaries(
        st.text(max_size=20), st.text(max_size=50), max_size=5
    ),
    seed=st.text(min_size=1, max_size=20),
    flip=st.integers(min_value=0, max_value=10_000),
)
def test_session_crypto_roundtrip_and_tamper(payload, seed, flip):
    c = SessionCrypto(seed)
    token = c.seal({"s": payload})
    assert c.open(token) == {"s": payload}
    # flip one character -> must never open successfully with wrong content
    i = flip % len(token)
    ch = "A" if token[i] != "A" else "B"
    tampered = token[:i] + ch + token[i + 1 :]
    if tampered != token:
        try:
            out = c.open(tampered)
        except Exception:
            return
        assert out == {"s": payload}  # base64 alias may decode identically


# -- C++ body scanner vs json.loads -------------------------------------------

json_scalars = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-(10**9), max_value=10**9),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=40),
)
json_values = st.recursive(
    json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=10), children, max_size=4),
    ),
    max_leaves=20,
)


@settings(max_examples=200, deadline=None)
@given(
    model=st.text(max_size=30),
    stream=st.booleans(),
    extra=st.dictionaries(st.text(max_size=8), json_values, max_size=4),
    messages=st.lists(
        st.fixed_dictionaries({"role": st.sampled_from(["user", "assistant"]),
                               "content": st.text(max_size=60)}),
        max_size=4,
    ),
)
def test_native_scanner_matches_json(model, stream, extra, messages):
    import aigw_native

    body = dict(extra)
    body["model"] = model
    body["stream"] = stream
    body["messages"] = messages
    raw = json.dumps(body).encode()
    ok, got_model, got_stream, text = aigw_native.scan_chat_body(raw)
    assert ok
    assert got_model == model
    assert got_stream == stream
    expected_text = "".join(m["content"] + "\n" for m in messages)
    assert text.decode() == expected_text


@settings(max_examples=150, deadline=None)
@given(junk=st.binary(max_size=200))
def test_native_scanner_never_crashes_on_garbage(junk):
    import aigw_native

    ok, model, stream, text = aigw_native.scan_chat_body(junk)
    # must not crash; on ok=True for valid JSON the fields are consistent
    assert isinstance(ok, bool)


# -- cost expressions ----------------------------------------------------------


@settings(max_examples=100, deadline=None)
@given(
    a=st.integers(min_value=0, max_value=10**6),
    b=st.integers(min_value=0, max_value=10**6),
    mul=st.integers(min_value=0, max_value=100),
)
def test_cost_arithmetic_matches_python(a, b, mul):
    from aigw.llmcost import CostProgram, CostVars

    v = CostVars(input_tokens=a, output_tokens=b)
    assert CostProgram("input_tokens + output_tokens * {}".format(mul)).evaluate(v) == a + b * mul
    assert CostProgram("max(input_tokens, output_tokens)").evaluate(v) == max(a, b)


@given(
    data=st.binary(min_size=0, max_size=400),
    schema=st.sampled_from(["Anthropic", "AWSBedrock", "GCPVertexAI", "OpenAI"]),
)
@settings(max_examples=200, deadline=None)
def test_response_chunk_never_raises_unhandled(data, schema):
    """Streaming translators fed arbitrary provider bytes may reject them
    (ValueError/TranslationError — the gateway cuts the stream) but must
    never raise anything else or corrupt their internal state."""
    from aigw.filterapi.config import APISchemaName
    from aigw.translator import TranslationError, get_translator

    endpoint = ("/anthropic/v1/messages" if schema == "OpenAI"
                else "/v1/chat/completions")  # covers both SSE machines
    t = get_translator(endpoint, APISchemaName(schema),
                       gcp_project="p", gcp_region="r")
    t.request({"model": "m", "max_tokens": 8,
               "messages": [{"role": "user", "content": "q"}],
               "stream": True}, stream=True)
    for chunk in (data[:137], data[137:]):
        try:
            t.response_chunk(chunk)
        except (ValueError, TranslationError):
            pass  # handled by the gateway as a stream cut
    try:
        t.response_flush()
    except (ValueError, TranslationError):
        pass


@settings(max_examples=200, deadline=None)
@given(
    model=st.text(max_size=30),
    stream=st.booleans(),
    extra=st.dictionaries(st.text(max_size=8), json_values, max_size=4),
    messages=st.lists(
        st.fixed_dictionaries({"role": st.sampled_from(["user", "assistant"]),
                               "content": st.text(max_size=60)}),
        max_size=4,
    ),
    override=st.text(alphabet=st.characters(min_codepoint=33, max_codepoint=126,
                                            blacklist_characters='"\\'),
                     min_size=1, max_size=20),
)
def test_scanner_spans_splice_correctly(model, stream, extra, messages, override):
    """The byte spans the native fast path uses must be EXACT: splicing a
    model override at [model_vs, model_ve) yields JSON whose model is the
    override and nothing else changed; slicing [msgs_vs, msgs_ve) yields
    the messages array; hashing around the messages span keys on every
    other byte (the cache scope fingerprint contract)."""
    import aigw_native

    body = dict(extra)
    body["model"] = model
    body["stream"] = stream
    body["messages"] = messages
    raw = json.dumps(body).encode()
    ok, got_model, _, _, mvs, mve, svs, sve = aigw_native.scan_chat_body_spans(raw)
    assert ok and got_model == model
    # model span: value incl. quotes
    assert raw[mvs:mve] == json.dumps(model).encode()
    spliced = raw[:mvs] + json.dumps(override).encode() + raw[mve:]
    doc = json.loads(spliced)
    assert doc["model"] == override
    doc["model"] = model
    assert doc == body
    # messages span: exactly the array value
    assert sve > svs
    assert json.loads(raw[svs:sve]) == messages
    # bytes outside the messages span are identical for two bodies that
    # differ only in message content
    body2 = dict(body)
    body2["messages"] = [{"role": "user", "content": "DIFFERENT"}]
    raw2 = json.dumps(body2).encode()
    ok2, _, _, _, _, _, svs2, sve2 = aigw_native.scan_chat_body_spans(raw2)
    assert ok2
    assert raw[:svs] + raw[sve:] == raw2[:svs2] + raw2[sve2:]
