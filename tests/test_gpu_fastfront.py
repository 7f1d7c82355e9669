"""GPU: the native fast front's in-process HIP admission path
(csrc/admission.hip) must produce token counts bit-identical to the CPU
BPE oracle, end-to-end through the C++ server under concurrent load."""

import asyncio
import json

import pytest

pytestmark = pytest.mark.gpu


def test_fast_front_direct_admission_counts_match_oracle():
    import yaml

    from aigw.extproc.fast_front import FastFront
    from aigw.extproc.server import GatewayServer
    from aigw.extproc.upstream_client import LeanClient
    from aigw.filterapi import RuntimeConfig, load_config
    from aigw.ops.bpe_ref import BPERef, make_merges
    from aigw.testing.fastmock import start_fast_mock

    # canned response WITHOUT usage so the gateway's own GPU count feeds
    # the accounting (finish_usage fallback, same as the Python path)
    body = json.dumps({"id": "x", "object": "chat.completion",
                       "choices": [{"index": 0,
                                    "message": {"role": "assistant",
                                                "content": "ok"},
                                    "finish_reason": "stop"}]}).encode()
    canned = (b"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
              b"content-length: %d\r\n\r\n" % len(body)) + body

    ref = BPERef(make_merges(8192, 1355))

    texts = [
        b"You are a terse assistant. The quick brown fox jumps over the lazy dog. " * 20,
        b"short one",
        b"Numbers 12345 and punctuation!!! mixed,with.commas " * 40,
        bytes(range(32, 127)) * 10,
    ]

    async def run():
        up_srv, up_port = await start_fast_mock(response=canned)
        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
llmRequestCosts:
  - metadataKey: llm_total_token
    type: TotalToken
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        front = FastFront(server, server.runtime, gpu_direct=True,
                          n_merges=8192)
        port = await front.start("127.0.0.1", 0)
        client = LeanClient()

        expected_total = 0
        reqs = []
        for i, t in enumerate(texts):
            payload = {"model": f"m{i}",
                       "messages": [{"role": "user",
                                     "content": t.decode("latin1")}]}
            # chat text extraction appends one newline per collected value
            chat_text = t + b"\n"
            expected_total += len(ref.encode_batch([chat_text])[0])
            reqs.append(json.dumps(payload).encode())

        async def one(body):
            r = await client.post(host="127.0.0.1", port=port, tls=False,
                                  path="/v1/chat/completions",
                                  headers={"content-type": "application/json"},
                                  body=body, timeout_s=60.0)
            data = await r.read()
            r.release()
            assert r.status == 200, data[:200]

        # concurrent + repeated so several batches coalesce in flight
        rounds = 6
        for _ in range(rounds):
            await asyncio.gather(*(one(b) for b in reqs))

        st = front.stats()
        assert st["gpu_tokens"] == expected_total * rounds, (
            st["gpu_tokens"], expected_total * rounds)
        # the GPU counts fed the rate-limit/accounting totals (usage absent)
        assert st["input_tokens"] == expected_total * rounds
        await client.close()
        await front.stop()
        up_srv.close()

    asyncio.run(run())


@pytest.mark.parametrize("index_dtype", ["bf16", "fp8"])
def test_fast_front_native_semantic_cache(index_dtype):
    """Native cache path: MFMA embed + fused top-k on the admission
    stream, value store + scope fingerprint in the C++ server. A repeat
    request must hit without touching the upstream; model, sampling
    params, and credential changes must miss (same isolation contract as
    tests/test_cache_scope.py for the Python cache)."""
    import aigw_fast
    import yaml

    from aigw.extproc.fast_front import FastFront
    from aigw.extproc.server import GatewayServer
    from aigw.extproc.upstream_client import LeanClient
    from aigw.filterapi import RuntimeConfig, load_config

    body = json.dumps({"id": "r1", "object": "chat.completion", "model": "m",
                       "choices": [{"index": 0,
                                    "message": {"role": "assistant",
                                                "content": "cached answer"},
                                    "finish_reason": "stop"}],
                       "usage": {"prompt_tokens": 4, "completion_tokens": 2,
                                 "total_tokens": 6}}).encode()
    canned = (b"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
              b"content-length: %d\r\n\r\n" % len(body)) + body

    async def run():
        mock = aigw_fast.FastMock()
        up_port = mock.start("127.0.0.1", canned.decode("latin1"))
        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        front = FastFront(server, server.runtime, gpu_direct=True,
                          gpu_cache=True, n_merges=8192,
                          cache_threshold=0.95,
                          cache_index_dtype=index_dtype)
        port = await front.start("127.0.0.1", 0)
        client = LeanClient()

        async def chat(content, model="m", auth="Bearer user-a", **params):
            payload = {"model": model,
                       "messages": [{"role": "user", "content": content}]}
            payload.update(params)
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json",
                         "authorization": auth},
                body=json.dumps(payload).encode(), timeout_s=60.0)
            data = await r.read()
            hit = r.headers.get("x-aigw-cache") == "hit"
            r.release()
            assert r.status == 200, data[:200]
            return hit, data

        text = "what is the airspeed velocity of an unladen swallow " * 20
        hit, _ = await chat(text)
        assert not hit  # cold
        hit, data = await chat(text)
        assert hit, front.stats()
        assert json.loads(data)["choices"][0]["message"]["content"] == "cached answer"
        served_before = mock.requests()
        hit, _ = await chat(text)
        assert hit and mock.requests() == served_before  # upstream untouched

        # isolation: model / sampling params / credential all miss
        hit, _ = await chat(text, model="m2")
        assert not hit
        hit, _ = await chat(text, temperature=0.9)
        assert not hit
        hit, _ = await chat(text, auth="Bearer user-b")
        assert not hit
        # and a hit again for the original scope
        hit, _ = await chat(text)
        assert hit

        st = front.stats()
        assert st["cache_hits"] >= 3 and st["cache_misses"] >= 4, st
        await client.close()
        await front.stop()
        mock.stop()

    asyncio.run(run())


def test_fast_front_stream_deferred_counts_match_oracle():
    """Streamed requests ENQUEUE their admission count before upstream
    dispatch and collect it at stream end (deferred path); the deferred
    count must still be bit-identical to the CPU oracle and feed the
    usage accounting when the stream carries no usage chunk."""
    import yaml

    import aigw_fast
    from aigw.extproc.fast_front import FastFront
    from aigw.extproc.server import GatewayServer
    from aigw.extproc.upstream_client import LeanClient
    from aigw.ops.bpe_ref import BPERef, make_merges

    # SSE stream WITHOUT a usage chunk -> gateway's deferred GPU count
    # is the only token source
    frames = (b"data: {\"id\":\"c\",\"object\":\"chat.completion.chunk\","
              b"\"choices\":[{\"index\":0,\"delta\":{\"content\":\"hi\"},"
              b"\"finish_reason\":null}]}\n\n"
              b"data: [DONE]\n\n")
    head = (b"HTTP/1.1 200 OK\r\ncontent-type: text/event-stream\r\n"
            b"transfer-encoding: chunked\r\n\r\n")
    chunked = (b"%x\r\n" % len(frames)) + frames + b"\r\n0\r\n\r\n"
    canned = head + chunked

    ref = BPERef(make_merges(8192, 1355))
    text = b"stream accounting check with some longer body text " * 30

    async def run():
        mock = aigw_fast.FastMock()
        up_port = mock.start("127.0.0.1", canned.decode("latin1"))
        cfg = load_config_yaml = yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
llmRequestCosts:
  - metadataKey: llm_total_token
    type: TotalToken
""")
        from aigw.filterapi import RuntimeConfig, load_config

        server = GatewayServer(RuntimeConfig(load_config(cfg)))
        front = FastFront(server, server.runtime, gpu_direct=True,
                          n_merges=8192)
        port = await front.start("127.0.0.1", 0)
        client = LeanClient()
        payload = json.dumps({
            "model": "m", "stream": True,
            "messages": [{"role": "user",
                          "content": text.decode("latin1")}]}).encode()
        expected = len(ref.encode_batch([text + b"\n"])[0])

        async def one():
            r = await client.post(host="127.0.0.1", port=port, tls=False,
                                  path="/v1/chat/completions",
                                  headers={"content-type": "application/json"},
                                  body=payload, timeout_s=60.0)
            data = await r.read()
            r.release()
            assert r.status == 200 and b"[DONE]" in data

        n = 24
        await asyncio.gather(*(one() for _ in range(n)))
        st = front.stats()
        assert st["gpu_tokens"] == expected * n, (st["gpu_tokens"], expected, n)
        assert st["input_tokens"] == expected * n
        await client.close()
        await front.stop()
        mock.stop()

    asyncio.run(run())
