"""Lean raw-asyncio HTTP front: the full hot-path behavior matrix must
match the aiohttp front (unary, streaming via chunked encoding, errors,
admin GETs, rate limiting) and cold paths must ride the loopback fallback
(MCP through the proxy)."""

import asyncio
import json

import aiohttp

from aigw.extproc.lean_front import serve_lean
from aigw.extproc.server import GatewayServer
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream
from tests.test_mcp import FakeMCPServer, _start


def _cfg(up_port, mcp_port=None):
    cfg = {
        "version": "v1",
        "llmRequestCosts": [{"metadataKey": "llm_total_token", "type": "TotalToken"}],
        "routes": [
            {
                "name": "r",
                "backends": [
                    {"name": "b", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": up_port},
                     "auth": {"apiKey": "sk-lean"}}
                ],
            }
        ],
        "models": [{"name": "lean-model"}],
        "rateLimits": [
            {"name": "rl", "metadataKey": "llm_total_token", "limit": 10**9,
             "windowS": 3600}
        ],
    }
    if mcp_port:
        cfg["mcp"] = {
            "routes": [
                {"name": "m", "path": "/mcp",
                 "backends": [{"name": "alpha",
                               "upstream": {"host": "127.0.0.1", "port": mcp_port}}]}
            ]
        }
    return load_config(cfg)


def test_lean_front_full_matrix():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        mcp_srv = FakeMCPServer("alpha", ["tool_a"])
        mcp_runner, mcp_port = await _start(mcp_srv.handle)
        server = GatewayServer(RuntimeConfig(_cfg(up_port, mcp_port)))
        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0)
        base = f"http://127.0.0.1:{port}"
        async with aiohttp.ClientSession() as c:
            # unary chat w/ auth injection + usage extraction
            async with c.post(f"{base}/v1/chat/completions",
                              json={"model": "m", "messages": [
                                  {"role": "user", "content": "hi"}]},
                              headers={"x-expected-auth": "Bearer sk-lean"}) as r:
                assert r.status == 200
                body = await r.json()
                assert body["choices"][0]["message"]["content"]

            # streaming via chunked transfer-encoding
            async with c.post(f"{base}/v1/chat/completions",
                              json={"model": "m", "stream": True,
                                    "messages": [{"role": "user", "content": "s"}]},
                              headers={"x-mock-response-tokens": "5"}) as r:
                assert r.status == 200
                assert r.headers.get("transfer-encoding") == "chunked"
                raw = await r.read()
                assert raw.endswith(b"data: [DONE]\n\n")
                lines = [l for l in raw.split(b"\n\n") if l.startswith(b"data: ")]
                usage = json.loads(lines[-2][6:])["usage"]
                assert usage["completion_tokens"] == 5

            # error translation preserved
            async with c.post(f"{base}/v1/chat/completions",
                              json={"model": "m", "messages": []},
                              headers={"x-mock-status": "400"}) as r:
                assert r.status == 400
                assert "error" in await r.json()
            # missing model -> 400 local reply
            async with c.post(f"{base}/v1/chat/completions",
                              json={"messages": []}) as r:
                assert r.status == 400

            # admin GETs
            async with c.get(f"{base}/health") as r:
                assert (await r.json())["status"] == "ok"
            async with c.get(f"{base}/v1/models") as r:
                assert (await r.json())["data"][0]["id"] == "lean-model"
            async with c.get(f"{base}/metrics") as r:
                assert "gen_ai_client_token_usage" in await r.text()

            # cold path rides the loopback fallback: MCP initialize
            async with c.post(f"{base}/mcp",
                              json={"jsonrpc": "2.0", "id": 1, "method": "initialize",
                                    "params": {}}) as r:
                assert r.status == 200
                assert (await r.json())["result"]["serverInfo"]["name"] == "aigw-mcp-gateway"

            # keep-alive: many sequential requests on one session
            for i in range(20):
                async with c.post(f"{base}/v1/chat/completions",
                                  json={"model": "m", "messages": [
                                      {"role": "user", "content": f"n{i}"}]}) as r:
                    assert r.status == 200
        await cleanup()
        await up_runner.cleanup()
        await mcp_runner.cleanup()

    asyncio.run(main())


def test_lean_front_concurrent_storm():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        server = GatewayServer(RuntimeConfig(_cfg(up_port)))
        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0, with_fallback=False)
        base = f"http://127.0.0.1:{port}"

        async def one(c, i):
            stream = i % 3 == 0
            payload = {"model": "m", "messages": [{"role": "user", "content": f"q{i}"}]}
            if stream:
                payload["stream"] = True
            async with c.post(f"{base}/v1/chat/completions", json=payload) as r:
                assert r.status == 200
                raw = await r.read()
                if stream:
                    assert raw.endswith(b"data: [DONE]\n\n")

        async with aiohttp.ClientSession() as c:
            await asyncio.gather(*(one(c, i) for i in range(120)))
        await cleanup()
        await up_runner.cleanup()

    asyncio.run(main())
