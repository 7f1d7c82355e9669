"""Graceful drain: SIGTERM semantics — /health flips to 503, new
requests are refused with connection:close, in-flight requests finish
(the Envoy drain-sequence analogue; k8s preStop/terminationGracePeriod
lifecycle)."""

import asyncio
import json

import aiohttp
import pytest
import yaml
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi.config import load_config
from aigw.filterapi.runtime import RuntimeConfig

CFG = """\
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {host: 127.0.0.1, port: %d}
"""


async def _slow_upstream():
    async def chat(request):
        await asyncio.sleep(0.4)
        return web.json_response(
            {"id": "x", "object": "chat.completion", "model": "m",
             "choices": [{"index": 0, "message": {"role": "assistant", "content": "hi"},
                          "finish_reason": "stop"}],
             "usage": {"prompt_tokens": 1, "completion_tokens": 1, "total_tokens": 2}}
        )

    app = web.Application()
    app.router.add_post("/v1/chat/completions", chat)
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    port = site._server.sockets[0].getsockname()[1]
    return runner, port


@pytest.mark.timeout(60)
def test_drain_finishes_inflight_and_rejects_new():
    async def run():
        up_runner, up_port = await _slow_upstream()
        cfg = load_config(yaml.safe_load(CFG % up_port))
        server = GatewayServer(RuntimeConfig(cfg))
        runner = await run_server(server, host="127.0.0.1", port=0)
        gw_port = runner.addresses and runner.addresses[0][1]
        body = {"model": "m", "messages": [{"role": "user", "content": "q"}]}
        url = f"http://127.0.0.1:{gw_port}/v1/chat/completions"
        async with aiohttp.ClientSession() as sess:
            # start an in-flight request, then begin draining mid-request
            task = asyncio.create_task(sess.post(url, json=body))
            await asyncio.sleep(0.1)
            drain = asyncio.create_task(server.drain(timeout_s=10))
            await asyncio.sleep(0.05)
            # new work refused while draining
            r = await sess.post(url, json=body)
            assert r.status == 503
            assert (await r.json())["error"]["type"] == "unavailable"
            h = await sess.get(f"http://127.0.0.1:{gw_port}/health")
            assert h.status == 503
            assert (await h.json())["status"] == "draining"
            # the in-flight request completes normally
            resp = await task
            assert resp.status == 200
            data = json.loads(await resp.read())
            assert data["choices"][0]["message"]["content"] == "hi"
            left = await drain
            assert left == 0
        await runner.cleanup()
        await up_runner.cleanup()
        await server.close()

    asyncio.run(run())


@pytest.mark.timeout(60)
def test_drain_immediate_when_idle():
    async def run():
        cfg = load_config(yaml.safe_load(CFG % 9))
        server = GatewayServer(RuntimeConfig(cfg))
        assert await server.drain(timeout_s=1) == 0
        assert server.draining

    asyncio.run(run())


@pytest.mark.timeout(60)
def test_drain_rejects_on_lean_front():
    from aigw.extproc.lean_front import serve_lean
    from aigw.extproc.upstream_client import LeanClient

    async def run():
        up_runner, up_port = await _slow_upstream()
        cfg = load_config(yaml.safe_load(CFG % up_port))
        server = GatewayServer(RuntimeConfig(cfg))
        await server.start()
        _, gw_port, gw_cleanup = await serve_lean(
            server, "127.0.0.1", 0, with_fallback=False
        )
        body = json.dumps(
            {"model": "m", "messages": [{"role": "user", "content": "q"}]}
        ).encode()
        client = LeanClient()

        async def post():
            return await client.post(
                host="127.0.0.1", port=gw_port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )

        task = asyncio.create_task(post())
        await asyncio.sleep(0.1)
        drain = asyncio.create_task(server.drain(timeout_s=10))
        await asyncio.sleep(0.05)
        r = await post()
        assert r.status == 503
        assert json.loads(await r.read())["error"]["type"] == "unavailable"
        r.close()
        resp = await task  # in-flight request still completes
        assert resp.status == 200
        assert json.loads(await resp.read())["choices"][0]["message"]["content"] == "hi"
        resp.release()
        assert await drain == 0
        await gw_cleanup()
        await up_runner.cleanup()
        await client.close()
        await server.close()

    asyncio.run(run())
