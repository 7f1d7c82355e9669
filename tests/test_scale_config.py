"""Large-config scale (the reference's tests/e2e/large_config_test.go:
100 AIGatewayRoutes must keep working): 120 routes end-to-end + routing
cost stays flat."""

import asyncio
import time

import aiohttp

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream


def test_120_route_config():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        routes = [
            {
                "name": f"route-{i}",
                "headers": [{"name": "x-ai-eg-model", "value": f"model-{i}"}],
                "backends": [
                    {"name": f"b-{i}", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": up_port}}
                ],
                "requestCosts": [
                    {"metadataKey": f"cost_{i}", "type": "CEL",
                     "cel": "input_tokens + output_tokens"}
                ],
            }
            for i in range(120)
        ]
        t0 = time.perf_counter()
        cfg = load_config({"version": "v1", "routes": routes})
        rt = RuntimeConfig(cfg)
        compile_s = time.perf_counter() - t0
        assert compile_s < 5.0, f"config compile too slow: {compile_s:.1f}s"

        server = GatewayServer(rt)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        async with aiohttp.ClientSession() as c:
            for i in (0, 60, 119):
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json={"model": f"model-{i}", "messages": [{"role": "user", "content": "q"}]},
                ) as r:
                    assert r.status == 200, (i, r.status)
            async with c.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "unknown-model", "messages": []},
            ) as r:
                assert r.status == 404
        await gw.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())
