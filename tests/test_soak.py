"""Concurrency soak: mixed streaming/unary/fallback/limited traffic in
flight simultaneously — guards the shared-state paths (limiter, metrics,
per-request translators) against cross-request interference."""

import asyncio
import random

import aiohttp

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream


def test_mixed_concurrent_traffic():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        up = {"host": "127.0.0.1", "port": up_port}
        cfg = load_config(
            {
                "version": "v1",
                "llmRequestCosts": [
                    {"metadataKey": "llm_total_token", "type": "TotalToken"}
                ],
                "routes": [
                    {"name": "openai", "headers": [{"name": "x-ai-eg-model", "value": "m-openai"}],
                     "backends": [{"name": "b1", "schema": "OpenAI", "upstream": up}]},
                    {"name": "anthropic", "headers": [{"name": "x-ai-eg-model", "value": "m-claude"}],
                     "backends": [{"name": "b2", "schema": "Anthropic", "upstream": up}]},
                    {"name": "flaky", "headers": [{"name": "x-ai-eg-model", "value": "m-flaky"}],
                     "retries": 3,
                     "backends": [
                         {"name": "bad", "schema": "OpenAI", "upstream": up, "priority": 0,
                          "headerMutation": {"set": {"x-mock-fail-times": "999",
                                                     "x-mock-fail-key": "soak-bad"}}},
                         {"name": "good", "schema": "OpenAI", "upstream": up, "priority": 1},
                     ]},
                ],
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        base = f"http://127.0.0.1:{port}"
        rng = random.Random(7)
        results = {"ok": 0, "fail": 0}

        async def one(i):
            kind = rng.choice(["openai", "openai-stream", "anthropic", "flaky"])
            try:
                async with aiohttp.ClientSession() as c:
                    if kind == "openai-stream":
                        payload = {"model": "m-openai", "stream": True,
                                   "messages": [{"role": "user", "content": f"q{i}"}]}
                        async with c.post(f"{base}/v1/chat/completions", json=payload) as r:
                            assert r.status == 200
                            raw = await r.read()
                            assert raw.endswith(b"data: [DONE]\n\n")
                    elif kind == "anthropic":
                        payload = {"model": "m-claude", "max_tokens": 8,
                                   "messages": [{"role": "user", "content": f"q{i}"}]}
                        async with c.post(f"{base}/anthropic/v1/messages", json=payload) as r:
                            assert r.status == 200
                            assert (await r.json())["type"] == "message"
                    else:
                        model = "m-flaky" if kind == "flaky" else "m-openai"
                        payload = {"model": model,
                                   "messages": [{"role": "user", "content": f"q{i}"}]}
                        async with c.post(f"{base}/v1/chat/completions", json=payload) as r:
                            assert r.status == 200
                            assert (await r.json())["choices"]
                results["ok"] += 1
            except Exception:
                results["fail"] += 1
                raise

        await asyncio.gather(*(one(i) for i in range(150)))
        assert results == {"ok": 150, "fail": 0}

        # metrics must be internally consistent after the storm
        async with aiohttp.ClientSession() as c:
            async with c.get(f"{base}/metrics") as r:
                text = await r.text()
        total = 0.0
        for line in text.splitlines():
            if line.startswith("aigw_requests_total{"):
                total += float(line.rsplit(" ", 1)[1])
        assert total >= 150  # >= because fallback retries add attempts
        await gw.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())
