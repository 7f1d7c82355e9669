"""Per-try stream idle timeout (ai_gateway_route.go StreamIdleTimeout →
route.retry_policy.per_try_idle_timeout): before the first response byte
an idle upstream resets the try and fallback proceeds; mid-stream the
stream is cut; unary bodies that stall also fail over."""

import asyncio
import json

import pytest
import yaml
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.extproc.upstream_client import LeanClient, UpstreamError
from aigw.filterapi.config import load_config
from aigw.filterapi.runtime import RuntimeConfig

CFG = """\
routes:
  - name: r
    backends:
      - name: stall
        schema: OpenAI
        upstream: {host: 127.0.0.1, port: %d}
        streamIdleTimeoutS: 0.5
      - name: good
        priority: 1
        schema: OpenAI
        upstream: {host: 127.0.0.1, port: %d}
        streamIdleTimeoutS: 0.5
"""

OK_UNARY = {
    "id": "x", "object": "chat.completion", "model": "m",
    "choices": [{"index": 0, "message": {"role": "assistant", "content": "ok"},
                 "finish_reason": "stop"}],
    "usage": {"prompt_tokens": 1, "completion_tokens": 1, "total_tokens": 2},
}


async def _stall_upstream(mode: str):
    """mode: 'headers-only' (send status+headers then stall) or
    'mid-stream' (send one SSE chunk then stall) or 'slow-unary'."""

    async def chat(request):
        resp = web.StreamResponse(status=200)
        if mode == "slow-unary":
            resp.headers["content-type"] = "application/json"
        else:
            resp.headers["content-type"] = "text/event-stream"
        await resp.prepare(request)
        if mode == "mid-stream":
            await resp.write(b'data: {"id":"c","object":"chat.completion.chunk",'
                             b'"model":"m","choices":[{"index":0,"delta":'
                             b'{"content":"partial"}}]}\n\n')
        await asyncio.sleep(4)
        return resp

    app = web.Application()
    app.router.add_post("/v1/chat/completions", chat)
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    return runner, site._server.sockets[0].getsockname()[1]


async def _good_upstream():
    async def chat(request):
        body = await request.json()
        if body.get("stream"):
            resp = web.StreamResponse(status=200)
            resp.headers["content-type"] = "text/event-stream"
            await resp.prepare(request)
            await resp.write(b'data: {"id":"g","object":"chat.completion.chunk",'
                             b'"model":"m","choices":[{"index":0,"delta":'
                             b'{"content":"good"}}]}\n\ndata: [DONE]\n\n')
            await resp.write_eof()
            return resp
        return web.json_response(OK_UNARY)

    app = web.Application()
    app.router.add_post("/v1/chat/completions", chat)
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    return runner, site._server.sockets[0].getsockname()[1]


async def _gateway(stall_port, good_port):
    cfg = load_config(yaml.safe_load(CFG % (stall_port, good_port)))
    assert cfg.routes[0].backends[0].stream_idle_timeout_s == 0.5
    server = GatewayServer(RuntimeConfig(cfg))
    runner = await run_server(server, host="127.0.0.1", port=0)
    return server, runner, runner.addresses[0][1]


@pytest.mark.timeout(60)
def test_first_byte_idle_falls_back():
    async def run():
        s_runner, s_port = await _stall_upstream("headers-only")
        g_runner, g_port = await _good_upstream()
        server, runner, port = await _gateway(s_port, g_port)
        client = LeanClient()
        body = json.dumps({"model": "m", "stream": True,
                           "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        data = await r.read()
        assert r.status == 200
        assert b"good" in data  # served by the fallback backend
        r.release()
        await client.close()
        await runner.cleanup()
        await server.close()
        await s_runner.cleanup()
        await g_runner.cleanup()

    asyncio.run(run())


@pytest.mark.timeout(60)
def test_unary_idle_falls_back():
    async def run():
        s_runner, s_port = await _stall_upstream("slow-unary")
        g_runner, g_port = await _good_upstream()
        server, runner, port = await _gateway(s_port, g_port)
        client = LeanClient()
        body = json.dumps({"model": "m",
                           "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        data = await r.read()
        assert r.status == 200
        assert json.loads(data)["choices"][0]["message"]["content"] == "ok"
        r.release()
        await client.close()
        await runner.cleanup()
        await server.close()
        await s_runner.cleanup()
        await g_runner.cleanup()

    asyncio.run(run())


@pytest.mark.timeout(60)
def test_mid_stream_idle_cuts_stream():
    async def run():
        s_runner, s_port = await _stall_upstream("mid-stream")
        g_runner, g_port = await _good_upstream()
        server, runner, port = await _gateway(s_port, g_port)
        client = LeanClient()
        body = json.dumps({"model": "m", "stream": True,
                           "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        assert r.status == 200
        got = bytearray()
        # bytes already flowed -> no fallback; the stream must END
        # (truncated framing -> UpstreamError or clean close), not hang
        # until the stalled upstream gives up
        try:
            await asyncio.wait_for(_collect(r, got), timeout=10)
        except UpstreamError:
            pass
        assert b"partial" in got  # the pre-stall chunk was delivered
        assert b"good" not in got  # never switched backends mid-stream
        r.close()
        await client.close()
        await runner.cleanup()
        await server.close()
        await s_runner.cleanup()
        await g_runner.cleanup()

    asyncio.run(run())


async def _collect(resp, sink):
    async for c in resp.iter_chunks():
        sink.extend(c)


@pytest.mark.timeout(60)
def test_malformed_provider_stream_cuts_not_500():
    """A provider emitting garbage mid-stream (here: a corrupt Bedrock
    event-stream frame) truncates the client stream instead of crashing
    the worker or emitting a second response."""

    async def run():
        async def chat(request):
            resp = web.StreamResponse(status=200)
            resp.headers["content-type"] = "application/vnd.amazon.eventstream"
            await resp.prepare(request)
            await resp.write(b"\xde\xad\xbe\xef" * 64)  # not a valid frame
            await resp.write_eof()
            return resp

        app = web.Application()
        app.router.add_post("/model/m/converse-stream", chat)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        up_port = site._server.sockets[0].getsockname()[1]

        cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: bedrock
        schema: AWSBedrock
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        client = LeanClient()
        body = json.dumps({"model": "m", "stream": True,
                           "messages": [{"role": "user", "content": "q"}]}).encode()
        r = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"}, body=body)
        assert r.status == 200  # headers were already committed
        try:
            await asyncio.wait_for(r.read(), timeout=10)
        except UpstreamError:
            pass  # truncated framing is the expected signal
        r.close()
        # the worker still serves (no crash): same request round-trips
        # again, truncating the same way
        h = await client.post(host="127.0.0.1", port=port, tls=False,
                              path="/v1/chat/completions",
                              headers={"content-type": "application/json"},
                              body=body)
        assert h.status == 200
        try:
            await asyncio.wait_for(h.read(), timeout=10)
        except UpstreamError:
            pass
        h.close()
        await client.close()
        await gw.cleanup()
        await server.close()
        await runner.cleanup()

    asyncio.run(run())
