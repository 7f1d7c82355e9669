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
