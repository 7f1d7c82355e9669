"""GPU-path gateway e2e (runs on MI355X): real GPUServices in the serving
loop — GPU token accounting, semantic cache hit/miss, GPU tokenize API."""

import asyncio

import aiohttp
import pytest

pytestmark = pytest.mark.gpu


def _cfg(up_port):
    return {
        "version": "v1",
        "uuid": "gpu-e2e",
        "llmRequestCosts": [{"metadataKey": "llm_total_token", "type": "TotalToken"}],
        "routes": [
            {
                "name": "r",
                "backends": [
                    {"name": "openai", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": up_port}}
                ],
            }
        ],
        "rateLimits": [
            {"name": "budget", "metadataKey": "llm_total_token",
             "limit": 10**9, "windowS": 3600}
        ],
    }


def test_gpu_gateway_cache_and_tokenize():
    from aigw.extproc.server import GatewayServer, run_server
    from aigw.filterapi import RuntimeConfig, load_config
    from aigw.gpu import GPUServices
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        gpu = GPUServices(device="cuda", n_merges=8192, enable_cache=True,
                          cache_threshold=0.95, window_ms=0.5)
        server = GatewayServer(RuntimeConfig(load_config(_cfg(up_port))),
                               gpu_services=gpu)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        base = f"http://127.0.0.1:{port}"
        payload = {
            "model": "m",
            "messages": [{"role": "user", "content": "what is the capital of france " * 20}],
        }
        async with aiohttp.ClientSession() as client:
            # miss -> upstream -> insert
            async with client.post(f"{base}/v1/chat/completions", json=payload) as r:
                assert r.status == 200
                assert "x-aigw-cache" not in r.headers
                first_body = await r.read()
            # identical request -> cache hit, no upstream call
            n_up = len(mock.requests)
            async with client.post(f"{base}/v1/chat/completions", json=payload) as r:
                assert r.status == 200
                assert r.headers.get("x-aigw-cache") == "hit"
                assert await r.read() == first_body
            assert len(mock.requests) == n_up

            # different prompt -> miss again
            other = {"model": "m",
                     "messages": [{"role": "user", "content": "completely different topic: rust gpus"}]}
            async with client.post(f"{base}/v1/chat/completions", json=other) as r:
                assert r.status == 200
                assert "x-aigw-cache" not in r.headers

            # gateway-local GPU tokenizer endpoint
            async with client.post(f"{base}/v1/gateway/tokenize",
                                   json={"text": "hello brave new world"}) as r:
                body = await r.json()
                assert body["count"] == len(body["tokens"]) > 0

            # GPU token accounting flowed into metrics
            async with client.get(f"{base}/metrics") as r:
                text = await r.text()
                assert "gen_ai_client_token_usage" in text
        await gw.cleanup()
        await up_runner.cleanup()
        gpu.close()

    asyncio.run(main())


def test_gpu_batched_token_counting_concurrent():
    from aigw.gpu import GPUServices

    async def main():
        gpu = GPUServices(device="cuda", n_merges=8192, window_ms=1.0, max_batch=64)
        bodies = [
            {"messages": [{"role": "user", "content": f"request number {i} " * (i + 1)}]}
            for i in range(40)
        ]
        counts = await asyncio.gather(*(gpu.count_request_tokens(b) for b in bodies))
        assert all(c > 0 for c in counts)
        # longer texts count more tokens
        assert counts[-1] > counts[0]
        # cross-check a few against the CPU oracle
        from aigw.gpu.services import extract_chat_text

        ref = gpu.tokenizer.reference()
        for i in (0, 7, 39):
            want = sum(len(x) for x in ref.encode_batch([extract_chat_text(bodies[i])]))
            assert counts[i] == want
        gpu.close()

    asyncio.run(main())


def test_gpu_admission_service_path():
    """Production topology on hardware: gateway worker -> unix-socket RPC ->
    shard GPU service -> kernels (the aigw run --workers path)."""
    from aigw.extproc.server import GatewayServer, run_server
    from aigw.filterapi import RuntimeConfig, load_config
    from aigw.gpu import GPUServices
    from aigw.gpu.service import GPUServiceHost, RemoteGPUClient
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        import tempfile

        sock = tempfile.mktemp(suffix=".sock")
        mock, up_runner, up_port = await start_mock_upstream()
        gpu = GPUServices(device="cuda", n_merges=8192, enable_cache=True,
                          cache_threshold=0.95, window_ms=0.5)
        host = GPUServiceHost(gpu, sock)
        await host.start()
        client = RemoteGPUClient(sock, enable_cache=True, window_ms=0.5)
        server = GatewayServer(RuntimeConfig(load_config(_cfg(up_port))),
                               gpu_services=client)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        payload = {"model": "m",
                   "messages": [{"role": "user", "content": "service path test " * 30}]}
        async with aiohttp.ClientSession() as c:
            async with c.post(f"http://127.0.0.1:{port}/v1/chat/completions",
                              json=payload) as r:
                assert r.status == 200
                body = await r.read()
            # identical request -> cache hit SERVED FROM THE SHARD SERVICE
            async with c.post(f"http://127.0.0.1:{port}/v1/chat/completions",
                              json=payload) as r:
                assert r.status == 200
                assert r.headers.get("x-aigw-cache") == "hit"
                assert await r.read() == body
            # direct tokenize through the RPC
            ids = await client.tokenize("hello world")
            ref = gpu.tokenizer.reference().encode_batch([b"hello world"])[0]
            assert ids == ref
        client.close()
        await host.stop()
        await gw.cleanup()
        await up_runner.cleanup()
        gpu.close()

    asyncio.run(main())


def test_gpu_streaming_cache_replay():
    """Streamed requests hit the semantic cache too: the translated SSE
    transcript is stored on miss and replayed on hit."""
    from aigw.extproc.server import GatewayServer, run_server
    from aigw.filterapi import RuntimeConfig, load_config
    from aigw.gpu import GPUServices
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        gpu = GPUServices(device="cuda", n_merges=8192, enable_cache=True,
                          cache_threshold=0.95, window_ms=0.5)
        server = GatewayServer(RuntimeConfig(load_config(_cfg(up_port))),
                               gpu_services=gpu)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        payload = {"model": "m", "stream": True,
                   "messages": [{"role": "user", "content": "stream cache test " * 25}]}
        async with aiohttp.ClientSession() as c:
            async with c.post(f"http://127.0.0.1:{port}/v1/chat/completions",
                              json=payload,
                              headers={"x-mock-response-tokens": "6"}) as r:
                assert r.status == 200
                first = await r.read()
                assert first.endswith(b"data: [DONE]\n\n")
            n_up = len(mock.requests)
            async with c.post(f"http://127.0.0.1:{port}/v1/chat/completions",
                              json=payload) as r:
                assert r.status == 200
                assert r.headers.get("x-aigw-cache") == "hit"
                assert await r.read() == first
            assert len(mock.requests) == n_up
            # the UNARY form of the same prompt must NOT hit the stream entry
            unary = dict(payload)
            unary.pop("stream")
            async with c.post(f"http://127.0.0.1:{port}/v1/chat/completions",
                              json=unary) as r:
                assert r.status == 200
                assert "x-aigw-cache" not in r.headers
        await gw.cleanup()
        await up_runner.cleanup()
        gpu.close()

    asyncio.run(main())


@pytest.mark.gpu
def test_oversized_unary_response_not_cached():
    """Responses over the 2 MiB cap are served but never inserted into
    the semantic cache (bounded host/HBM value storage)."""
    import asyncio

    import yaml as _yaml
    from aiohttp import web

    from aigw.extproc.server import GatewayServer, run_server
    from aigw.extproc.upstream_client import LeanClient
    from aigw.filterapi.config import load_config
    from aigw.filterapi.runtime import RuntimeConfig
    from aigw.gpu import GPUServices

    async def run():
        import json

        big = "x" * (3 << 20)

        async def chat(request):
            return web.json_response(
                {"id": "b", "object": "chat.completion", "model": "m",
                 "choices": [{"index": 0, "message": {"role": "assistant",
                                                      "content": big},
                              "finish_reason": "stop"}],
                 "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                           "total_tokens": 2}})

        app = web.Application()
        app.router.add_post("/v1/chat/completions", chat)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        up_port = site._server.sockets[0].getsockname()[1]
        cfg = load_config(_yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
        gpu = GPUServices(device="cuda:0", enable_cache=True)
        server = GatewayServer(RuntimeConfig(cfg), gpu_services=gpu)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        client = LeanClient()
        body = json.dumps({"model": "m",
                           "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"},
                              body=body)
        data = await r.read()
        assert r.status == 200 and len(data) > (3 << 20)
        r.release()
        assert gpu.cache.size == 0  # oversized body was not inserted
        await client.close()
        await gw.cleanup()
        await server.close()
        await runner.cleanup()
        gpu.close()

    asyncio.run(run())
