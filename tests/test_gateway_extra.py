"""Additional data-plane behaviors: gzip upstream bodies (unary and
streamed — the reference's streaming-decompression invariant A.8),
legacy completions, CLI `run` end-to-end, translate->run config flow."""

import asyncio
import gzip
import json
import os
import signal
import socket
import subprocess
import sys
import time

import aiohttp
import pytest
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


class GzipUpstream:
    async def chat(self, request: web.Request) -> web.StreamResponse:
        body = json.loads(await request.read())
        out = {
            "id": "c",
            "object": "chat.completion",
            "model": body.get("model", "m"),
            "choices": [
                {"index": 0, "message": {"role": "assistant", "content": "zipped"},
                 "finish_reason": "stop"}
            ],
            "usage": {"prompt_tokens": 2, "completion_tokens": 1, "total_tokens": 3},
        }
        if body.get("stream"):
            # gzip-compressed SSE stream, flushed across chunk boundaries
            resp = web.StreamResponse()
            resp.headers["content-encoding"] = "gzip"
            resp.content_type = "text/event-stream"
            await resp.prepare(request)
            import zlib

            co = zlib.compressobj(wbits=31)  # gzip container
            chunks = [
                b'data: {"choices":[{"index":0,"delta":{"content":"zip"},"finish_reason":null}]}\n\n',
                b'data: {"choices":[{"index":0,"delta":{},"finish_reason":"stop"}],'
                b'"usage":{"prompt_tokens":2,"completion_tokens":1,"total_tokens":3}}\n\n',
                b"data: [DONE]\n\n",
            ]
            for c in chunks:
                data = co.compress(c) + co.flush(zlib.Z_SYNC_FLUSH)
                await resp.write(data)
            await resp.write(co.flush())
            await resp.write_eof()
            return resp
        payload = gzip.compress(json.dumps(out).encode())
        return web.Response(
            body=payload,
            headers={"content-encoding": "gzip", "content-type": "application/json"},
        )


def test_gzip_upstream_unary_and_streamed():
    async def main():
        up = GzipUpstream()
        app = web.Application()
        app.router.add_post("/v1/chat/completions", up.chat)
        runner = web.AppRunner(app, access_log=None)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = runner.addresses[0][1]

        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {"name": "r", "backends": [
                        {"name": "b", "schema": "OpenAI",
                         "upstream": {"host": "127.0.0.1", "port": port}}]}
                ],
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        gport = gw.addresses[0][1]
        async with aiohttp.ClientSession() as client:
            async with client.post(
                f"http://127.0.0.1:{gport}/v1/chat/completions",
                json={"model": "m", "messages": []},
            ) as r:
                body = await r.json()
                assert body["choices"][0]["message"]["content"] == "zipped"
            async with client.post(
                f"http://127.0.0.1:{gport}/v1/chat/completions",
                json={"model": "m", "messages": [], "stream": True},
            ) as r:
                raw = await r.read()
                assert b'"content":"zip"' in raw
                assert raw.endswith(b"data: [DONE]\n\n")
        await gw.cleanup()
        await runner.cleanup()

    asyncio.run(main())


def test_completions_endpoint():
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        mock, runner, port = await start_mock_upstream()
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {"name": "r", "backends": [
                        {"name": "b", "schema": "OpenAI",
                         "upstream": {"host": "127.0.0.1", "port": port}}]}
                ],
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        gport = gw.addresses[0][1]
        async with aiohttp.ClientSession() as client:
            async with client.post(
                f"http://127.0.0.1:{gport}/v1/completions",
                json={"model": "m", "prompt": "say hi"},
            ) as r:
                assert r.status == 200
                assert (await r.json())["choices"]
        await gw.cleanup()
        await runner.cleanup()

    asyncio.run(main())


@pytest.mark.timeout(90)
def test_cli_run_end_to_end(tmp_path):
    """`python -m aigw run` as a subprocess: serves, healthchecks, reloads."""
    import threading

    from aigw.testing.mockupstream import start_mock_upstream

    # the mock upstream needs a RUNNING loop for the whole test: host it in
    # a background thread
    loop = asyncio.new_event_loop()
    started = threading.Event()
    state = {}

    def loop_main():
        asyncio.set_event_loop(loop)

        async def boot():
            state["mock"], state["runner"], state["port"] = await start_mock_upstream()
            started.set()

        loop.create_task(boot())
        loop.run_forever()

    t = threading.Thread(target=loop_main, daemon=True)
    t.start()
    assert started.wait(timeout=30)
    up_port = state["port"]

    cfg = {
        "version": "v1",
        "uuid": "cli-1",
        "routes": [
            {"name": "r", "backends": [
                {"name": "b", "schema": "OpenAI",
                 "upstream": {"host": "127.0.0.1", "port": up_port}}]}
        ],
        "models": [{"name": "cli-model"}],
    }
    cfg_path = tmp_path / "gw.yaml"
    import yaml

    cfg_path.write_text(yaml.safe_dump(cfg))
    port = _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "aigw", "run", "--config", str(cfg_path),
         "--host", "127.0.0.1", "--port", str(port)],
        env=env,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
    )
    try:
        deadline = time.time() + 60
        import urllib.request

        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/health", timeout=2
                ) as r:
                    if json.loads(r.read())["status"] == "ok":
                        break
            except Exception:
                time.sleep(0.3)
        else:
            raise AssertionError("gateway did not become healthy")

        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/v1/chat/completions",
            data=json.dumps({"model": "m", "messages": [{"role": "user", "content": "q"}]}).encode(),
            headers={"content-type": "application/json"},
        )
        with urllib.request.urlopen(req, timeout=10) as r:
            body = json.loads(r.read())
            assert body["choices"][0]["message"]["content"]
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/v1/models", timeout=5) as r:
            assert json.loads(r.read())["data"][0]["id"] == "cli-model"
        # healthcheck subcommand against the live server
        rc = subprocess.run(
            [sys.executable, "-m", "aigw", "healthcheck", "--port", str(port)],
            env=env, capture_output=True, timeout=15,
        )
        assert rc.returncode == 0, rc.stdout + rc.stderr
    finally:
        proc.send_signal(signal.SIGINT)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        asyncio.run_coroutine_threadsafe(state["runner"].cleanup(), loop).result(10)
        loop.call_soon_threadsafe(loop.stop)
        t.join(timeout=10)


def test_endpoint_prefixes():
    """--endpointPrefixes parity: alternate mount points per API family."""
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        mock, runner, port = await start_mock_upstream()
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {"name": "r", "backends": [
                        {"name": "b", "schema": "OpenAI",
                         "upstream": {"host": "127.0.0.1", "port": port}}]}
                ],
            }
        )
        server = GatewayServer(
            RuntimeConfig(cfg),
            endpoint_prefixes={"openai": "/openai", "anthropic": "/claude"},
        )
        gw = await run_server(server, host="127.0.0.1", port=0)
        gport = gw.addresses[0][1]
        async with aiohttp.ClientSession() as c:
            async with c.post(
                f"http://127.0.0.1:{gport}/openai/v1/chat/completions",
                json={"model": "m", "messages": [{"role": "user", "content": "q"}]},
            ) as r:
                assert r.status == 200
            # canonical path still works
            async with c.post(
                f"http://127.0.0.1:{gport}/v1/chat/completions",
                json={"model": "m", "messages": []},
            ) as r:
                assert r.status == 200
        await gw.cleanup()
        await runner.cleanup()

    asyncio.run(main())


def test_root_prefix_mounts_everything():
    """mainlib --rootPrefix: every endpoint is served under the global
    prefix (path.Join(flags.rootPrefix, ...)), on both fronts."""
    import asyncio
    import json as _json

    import yaml as _yaml

    from aigw.extproc.lean_front import serve_lean
    from aigw.extproc.server import GatewayServer, run_server
    from aigw.extproc.upstream_client import LeanClient
    from aigw.filterapi.config import load_config
    from aigw.filterapi.runtime import RuntimeConfig
    from aigw.testing.fastmock import start_fast_mock

    async def run():
        up_srv, up_port = await start_fast_mock("127.0.0.1", 0)
        cfg = load_config(_yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
models: [{{name: m}}]
"""))
        server = GatewayServer(RuntimeConfig(cfg), root_prefix="/ai/v2")
        await server.start()
        runner = await run_server(server, host="127.0.0.1", port=0)
        aio_port = runner.addresses[0][1]
        _, lean_port, lean_cleanup = await serve_lean(
            server, "127.0.0.1", 0, with_fallback=False
        )
        body = _json.dumps(
            {"model": "m", "messages": [{"role": "user", "content": "q"}]}
        ).encode()
        client = LeanClient()
        for port in (aio_port, lean_port):
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/ai/v2/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )
            data = await r.read()
            assert r.status == 200, (port, r.status, data[:200])
            r.release()
            # unprefixed path 404s
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )
            await r.read()
            assert r.status == 404, port
            r.close()
        import aiohttp

        async with aiohttp.ClientSession() as s:
            async with s.get(f"http://127.0.0.1:{aio_port}/ai/v2/v1/models") as r:
                assert r.status == 200
                assert (await r.json())["data"][0]["id"] == "m"
        await client.close()
        await lean_cleanup()
        await runner.cleanup()
        await server.close()
        up_srv.close()

    asyncio.run(run())
