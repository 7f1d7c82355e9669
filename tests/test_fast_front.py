"""Native C++ fast front (csrc/fastpath.cpp): the passthrough hot loop in
native threads must reproduce the Python pipeline's observable behavior —
auth injection, model override splice, forced include_usage, retries on
dead backends, rate limiting, desync hardening — and relay every cold
path to the Python fallback app unchanged."""

import asyncio
import json
import socket
import threading

import aiohttp
import pytest
import yaml

from aigw.extproc.fast_front import FastFront, FastFrontUnsupported, build_fast_server
from aigw.extproc.server import GatewayServer
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream

pytestmark = pytest.mark.timeout(120)


def _cfg(up_port, *, dead_port=1, rate_limit=None):
    cfg = {
        "version": "v1",
        "llmRequestCosts": [{"metadataKey": "llm_total_token", "type": "TotalToken"}],
        "routes": [
            {
                "name": "fast",
                "headers": [{"name": "x-ai-eg-model", "value": "fast-model"}],
                "backends": [
                    {"name": "mock", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": up_port},
                     "modelNameOverride": "real-model",
                     "auth": {"apiKey": "sk-fast"}},
                ],
            },
            {
                "name": "retry",
                "headers": [{"name": "x-ai-eg-model", "value": "retry-model"}],
                "retries": 2,
                "backends": [
                    {"name": "dead", "schema": "OpenAI", "priority": 0,
                     "upstream": {"host": "127.0.0.1", "port": dead_port}},
                    {"name": "live", "schema": "OpenAI", "priority": 1,
                     "upstream": {"host": "127.0.0.1", "port": up_port},
                     "auth": {"apiKey": "sk-live"}},
                ],
            },
            {
                "name": "cold",
                "headers": [{"name": "x-ai-eg-model", "value": "cold-model"}],
                "backends": [
                    # header mutation makes this route ineligible for the
                    # native path -> per-request fallback to Python
                    {"name": "mut", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": up_port},
                     "headerMutation": {"set": {"x-extra": "1"}},
                     "auth": {"apiKey": "sk-cold"}},
                ],
            },
        ],
        "models": [{"name": "fast-model"}],
    }
    if rate_limit:
        cfg["rateLimits"] = [rate_limit]
    return load_config(cfg)


async def _start(cfg):
    server = GatewayServer(RuntimeConfig(cfg))
    front = FastFront(server, server.runtime)
    port = await front.start("127.0.0.1", 0)
    return front, port


def test_fast_front_full_matrix():
    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        front, port = await _start(_cfg(up_port))
        base = f"http://127.0.0.1:{port}"
        async with aiohttp.ClientSession() as c:
            # unary passthrough: model override spliced, auth injected,
            # spoofable/override headers stripped
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "fast-model",
                      "messages": [{"role": "user", "content": "hello world"}]},
                headers={"x-ai-eg-model": "spoof", "x-aigw-aws-access-key-id": "AK",
                         "x-client-note": "keep"},
            ) as r:
                body = await r.json()
                assert r.status == 200, body
                assert body["model"] == "real-model"
                assert body["usage"]["total_tokens"] > 0
            rec = mock.requests[-1]
            assert rec["headers"]["Authorization"] == "Bearer sk-fast"
            assert "x-ai-eg-model" not in {k.lower() for k in rec["headers"]}
            assert "x-aigw-aws-access-key-id" not in {k.lower() for k in rec["headers"]}
            assert {k.lower(): v for k, v in rec["headers"].items()}["x-client-note"] == "keep"
            assert json.loads(rec["body"])["model"] == "real-model"

            # streaming: SSE relayed, include_usage forced (route has costs)
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "fast-model", "stream": True,
                      "messages": [{"role": "user", "content": "hi"}]},
            ) as r:
                assert r.status == 200
                text = (await r.read()).decode()
                assert text.rstrip().endswith("data: [DONE]")
                assert '"usage"' in text  # forced include_usage
            assert json.loads(mock.requests[-1]["body"])["stream_options"][
                "include_usage"] is True

            # ineligible route (header mutation) -> Python fallback serves it
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "cold-model",
                      "messages": [{"role": "user", "content": "x"}]},
            ) as r:
                assert r.status == 200
            rec = mock.requests[-1]
            assert {k.lower(): v for k, v in rec["headers"].items()}["x-extra"] == "1"

            # cold paths ride the fallback app
            async with c.get(f"{base}/v1/models") as r:
                data = await r.json()
                assert r.status == 200
                assert data["data"][0]["id"] == "fast-model"
            async with c.get(f"{base}/health") as r:
                assert (await r.json())["front"] == "fast"

            # local replies
            async with c.post(f"{base}/v1/chat/completions",
                              data=b"{broken",
                              headers={"content-type": "application/json"}) as r:
                assert r.status == 400
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "nope", "messages": []},
            ) as r:
                assert r.status == 404

        st = front.stats()
        assert st["responses_2xx"] >= 3
        assert st["fallback"] >= 2
        assert st["total_tokens"] > 0
        metrics = front.render_metrics_extra()
        assert "aigw_fast_requests_total" in metrics
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_retry_and_rate_limit():
    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        cfg = _cfg(up_port, rate_limit={
            "name": "budget", "metadataKey": "llm_total_token",
            "limit": 10, "windowS": 3600,
        })
        front, port = await _start(cfg)
        base = f"http://127.0.0.1:{port}"
        async with aiohttp.ClientSession() as c:
            # first backend is a dead port: native retry moves to tier 1
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "retry-model",
                      "messages": [{"role": "user", "content": "fallback"}]},
            ) as r:
                assert r.status == 200
            assert mock.requests[-1]["headers"]["Authorization"] == "Bearer sk-live"
            assert front.stats()["retries"] >= 1

            # the first response charged ~17 tokens against limit 10:
            # next request must be denied locally with retry-after
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "fast-model",
                      "messages": [{"role": "user", "content": "x"}]},
            ) as r:
                assert r.status == 429
                assert int(r.headers["retry-after"]) > 0
        assert front.stats()["local_429"] == 1
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_desync_hardening():
    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        front, port = await _start(_cfg(up_port))

        async def send(payload: bytes) -> bytes:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            writer.write(payload)
            try:
                await writer.drain()
                return await asyncio.wait_for(reader.read(65536), timeout=6)
            finally:
                writer.close()

        data = await send(
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"transfer-encoding: chunked\r\n\r\n2\r\n{}\r\n0\r\n\r\n"
            b"GET /health HTTP/1.1\r\n\r\n"
        )
        assert data.startswith(b"HTTP/1.1 501"), data[:60]
        assert data.count(b"HTTP/1.1") == 1
        data = await send(
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"content-length: 2\r\ncontent-length: 9\r\n\r\n{}1234567"
        )
        assert data.startswith(b"HTTP/1.1 400"), data[:60]
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_concurrent_storm():
    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        front, port = await _start(_cfg(up_port))
        base = f"http://127.0.0.1:{port}"

        async def one(c, i):
            stream = i % 3 == 0
            async with c.post(
                f"{base}/v1/chat/completions",
                json={"model": "fast-model", "stream": stream,
                      "messages": [{"role": "user", "content": f"req {i}"}]},
            ) as r:
                await r.read()
                assert r.status == 200

        async with aiohttp.ClientSession(
            connector=aiohttp.TCPConnector(limit=64)) as c:
            await asyncio.gather(*(one(c, i) for i in range(200)))
        st = front.stats()
        assert st["responses_2xx"] == 200
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_unsupported_configs():
    cfg = load_config(yaml.safe_load("""
routes:
  - name: r
    headers: [{name: x-custom, value: v}]
    backends:
      - {name: b, schema: OpenAI, upstream: {host: h, port: 80}}
"""))
    with pytest.raises(FastFrontUnsupported):
        build_fast_server(RuntimeConfig(cfg))

    cfg = load_config({
        "routes": [{"name": "r", "backends": [
            {"name": "b", "schema": "OpenAI",
             "upstream": {"host": "h", "port": 80}}]}],
        "rateLimits": [{"name": "k", "metadataKey": "llm_total_token",
                        "limit": 10, "windowS": 60,
                        "keyHeaders": ["x-user"]}],
    })
    with pytest.raises(FastFrontUnsupported):
        build_fast_server(RuntimeConfig(cfg))


def test_cli_run_fast_front():
    """`aigw run --front fast` serves through the native server, with
    cold paths on its fallback."""
    import os
    import subprocess
    import sys
    import tempfile
    import time as _time
    import urllib.request

    import torch  # noqa: F401
    import aigw_fast

    body = json.dumps({"id": "x", "object": "chat.completion", "model": "m",
                       "choices": [], "usage": {"total_tokens": 3}}).encode()
    resp = (b"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(body)) + body
    mock = aigw_fast.FastMock()
    up_port = mock.start("127.0.0.1", resp.decode("latin1"))
    cfg = {
        "routes": [{"name": "r", "backends": [
            {"name": "b", "schema": "OpenAI",
             "upstream": {"host": "127.0.0.1", "port": up_port},
             "auth": {"apiKey": "sk-cli"}}]}],
        "models": [{"name": "cli-model"}],
    }
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        yaml.safe_dump(cfg, f)
        cfg_path = f.name
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "aigw", "run", "--config", cfg_path,
         "--front", "fast", "--host", "127.0.0.1", "--port", "0"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        port = None
        for _ in range(300):
            line = proc.stdout.readline()
            if "aigw shard" in line and "listening on" in line:
                port = int(line.rsplit(":", 1)[1].strip())
                break
        assert port, "gateway did not start"
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/v1/chat/completions",
            data=json.dumps({"model": "cli-model",
                             "messages": [{"role": "user", "content": "x"}]}).encode(),
            headers={"content-type": "application/json"})
        for _ in range(50):
            try:
                with urllib.request.urlopen(req, timeout=5) as r:
                    assert r.status == 200
                break
            except (ConnectionError, OSError):
                _time.sleep(0.1)
        else:
            raise AssertionError("hot path never answered")
        assert mock.requests() >= 1
        with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/v1/models", timeout=5) as r:
            data = json.loads(r.read())
            assert data["data"][0]["id"] == "cli-model"
        with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/health", timeout=5) as r:
            assert json.loads(r.read())["front"] == "fast"
    finally:
        proc.terminate()
        proc.wait(timeout=15)
        mock.stop()
        os.unlink(cfg_path)


def test_fast_front_azure_deployments_rewrite():
    """AzureOpenAI backends serve natively: deployments-path rewrite with
    the effective model, api-version from the schema version, api-key
    header (openai_azureopenai.go parity)."""

    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        cfg = load_config({
            "routes": [{
                "name": "az",
                "headers": [{"name": "x-ai-eg-model", "value": "gpt-4o"}],
                "backends": [{
                    "name": "azure", "schema": {"name": "AzureOpenAI",
                                                "version": "2024-10-21"},
                    "upstream": {"host": "127.0.0.1", "port": up_port,
                                 "pathPrefix": ""},
                    "modelNameOverride": "my-deployment",
                    "auth": {"azureApiKey": "az-key"},
                }],
            }],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        front = FastFront(server, server.runtime)
        port = await front.start("127.0.0.1", 0)
        async with aiohttp.ClientSession() as c:
            async with c.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "gpt-4o",
                      "messages": [{"role": "user", "content": "hi"}]},
            ) as r:
                assert r.status == 200, await r.text()
        rec = mock.requests[-1]
        assert rec["path"] == ("/openai/deployments/my-deployment/"
                               "chat/completions?api-version=2024-10-21"), rec["path"]
        hdrs = {k.lower(): v for k, v in rec["headers"].items()}
        assert hdrs["api-key"] == "az-key"
        assert "authorization" not in hdrs
        assert json.loads(rec["body"])["model"] == "my-deployment"
        assert front.stats()["fallback"] == 0
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_hot_reload_and_drain():
    """swap_routes under live traffic: new requests see the new table,
    nothing 5xxs during the swap; drain() stops accepting but lets
    in-flight requests finish."""

    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        mock2, up_runner2, up_port2 = await start_mock_upstream()
        mock.record = mock2.record = True
        front, port = await _start(_cfg(up_port))
        base = f"http://127.0.0.1:{port}"

        stop_flag = {"stop": False}
        errors = []

        async def pound(c):
            while not stop_flag["stop"]:
                try:
                    async with c.post(
                        f"{base}/v1/chat/completions",
                        json={"model": "fast-model",
                              "messages": [{"role": "user", "content": "x"}]},
                    ) as r:
                        if r.status >= 500:
                            errors.append(r.status)
                        await r.read()
                except Exception as e:
                    errors.append(repr(e))

        async with aiohttp.ClientSession() as c:
            pounders = [asyncio.ensure_future(pound(c)) for _ in range(8)]
            await asyncio.sleep(0.2)
            # hot-swap: fast-model now routes to mock2 with a new key
            new_cfg = load_config({
                "version": "v1",
                "llmRequestCosts": [{"metadataKey": "llm_total_token",
                                     "type": "TotalToken"}],
                "routes": [{
                    "name": "fast",
                    "headers": [{"name": "x-ai-eg-model",
                                 "value": "fast-model"}],
                    "backends": [{"name": "mock2", "schema": "OpenAI",
                                  "upstream": {"host": "127.0.0.1",
                                               "port": up_port2},
                                  "auth": {"apiKey": "sk-new"}}],
                }],
            })
            from aigw.filterapi import RuntimeConfig as _RC

            front.reload(_RC(new_cfg))
            await asyncio.sleep(0.3)
            stop_flag["stop"] = True
            await asyncio.gather(*pounders)
        assert not errors, errors[:5]
        assert mock2.requests, "reloaded backend never hit"
        assert mock2.requests[-1]["headers"]["Authorization"] == "Bearer sk-new"

        # drain: no new connections, zero left after idle traffic stops
        left = front.fast.drain(2.0)
        assert left == 0, left
        await front.stop()
        await up_runner.cleanup()
        await up_runner2.cleanup()

    asyncio.run(run())


def test_fast_front_metrics_exposed_on_fallback():
    """/metrics (served by the fallback app) carries the native
    counters alongside the Python registry."""

    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        front, port = await _start(_cfg(up_port))
        async with aiohttp.ClientSession() as c:
            async with c.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "fast-model",
                      "messages": [{"role": "user", "content": "x"}]},
            ) as r:
                assert r.status == 200
            async with c.get(f"http://127.0.0.1:{port}/metrics") as r:
                text = await r.text()
                assert "aigw_fast_requests_total" in text
                assert "aigw_fast_responses_2xx_total 1" in text
        await front.stop()
        await up_runner.cleanup()

    asyncio.run(run())


def test_fast_front_follows_pool_membership(tmp_path):
    """Dynamic pool membership reaches the NATIVE front: PoolManager
    swaps through the reload chain, so the C++ route table follows
    members-file changes (scale-out lands new replicas in rotation)."""

    async def run():
        mock, up_runner, up_port = await start_mock_upstream()
        mock2, up_runner2, up_port2 = await start_mock_upstream()
        mock.record = mock2.record = True
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{up_port}\n")
        cfg = load_config({
            "version": "v1",
            "routes": [{
                "name": "pool-route",
                "headers": [{"name": "x-ai-eg-model", "value": "fast-model"}],
                "pool": {"membersFile": str(f), "intervalS": 60,
                         "schema": "OpenAI"},
            }],
        })
        from aigw.extproc.pool import PoolManager
        from aigw.extproc.server import GatewayServer
        from aigw.filterapi import RuntimeConfig as _RC

        server = GatewayServer(_RC(cfg))
        await server.start()
        mgr = PoolManager(server)
        await mgr.resolve_once()  # inject member 1 BEFORE the front builds

        from aigw.extproc.fast_front import FastFront

        front = FastFront(server, server.runtime)
        port = await front.start("127.0.0.1", 0)
        mgr.swap_cb = lambda rc: (server.swap_runtime(rc), front.reload(rc))[0]

        async def post(c):
            async with c.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "fast-model",
                      "messages": [{"role": "user", "content": "x"}]},
            ) as r:
                assert r.status == 200
                await r.read()

        async with aiohttp.ClientSession() as c:
            for _ in range(4):
                await post(c)
            assert len(mock.requests) == 4 and not mock2.requests
            # scale out: second member appears in the native route table
            f.write_text(f"127.0.0.1:{up_port}\n127.0.0.1:{up_port2}\n")
            assert await mgr.resolve_once() is True
            for _ in range(40):
                await post(c)
            assert mock2.requests, "new pool member never hit via fast front"
        await front.stop()
        await server.close()
        await up_runner.cleanup()
        await up_runner2.cleanup()

    asyncio.run(run())


def test_fast_front_truncated_upstream_is_5xx_not_hang():
    """An upstream that dies mid-body (content-length promised, half
    delivered, connection closed) must produce a clean 503 — never a
    hang, never a truncated 200 relayed to the client — and the
    connection pool must recover for subsequent requests."""

    async def run():
        body = json.dumps({
            "id": "x", "object": "chat.completion", "choices": [],
            "usage": {"prompt_tokens": 5, "completion_tokens": 2,
                      "total_tokens": 7}}).encode()
        head = (f"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
                f"content-length: {len(body)}\r\n\r\n").encode()
        srv = socket.socket()
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", 0))
        srv.listen(16)
        port_up = srv.getsockname()[1]

        def serve():
            while True:
                try:
                    c, _ = srv.accept()
                except OSError:
                    return
                try:
                    c.recv(65536)
                    c.sendall(head + body[: len(body) // 2])
                    c.close()
                except OSError:
                    pass

        t = threading.Thread(target=serve, daemon=True)
        t.start()
        front, port = await _start(_cfg(port_up))
        payload = {"model": "fast-model",
                   "messages": [{"role": "user", "content": "x"}]}
        async with aiohttp.ClientSession() as c:
            for _ in range(5):
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json=payload,
                    timeout=aiohttp.ClientTimeout(total=5),
                ) as r:
                    assert r.status == 503
                    await r.read()
        st = front.fast.stats()
        assert st["responses_5xx"] == 5 and st["responses_2xx"] == 0
        await front.stop()
        srv.close()

    asyncio.run(run())


def test_fast_front_client_abort_mid_stream():
    """A client that disconnects mid-SSE-relay must not wedge the relay
    thread or poison the server: subsequent requests on fresh
    connections keep succeeding."""

    async def run():
        from aigw.testing.fastmock import canned_chat_sse

        import aigw_fast

        mock = aigw_fast.FastMock()
        port_up = mock.start("127.0.0.1",
                             canned_chat_sse(prompt_tokens=64,
                                             n_chunks=64).decode("latin1"))
        front, port = await _start(_cfg(port_up))
        payload = json.dumps({
            "model": "fast-model", "stream": True,
            "messages": [{"role": "user", "content": "x"}]}).encode()
        req = (b"POST /v1/chat/completions HTTP/1.1\r\n"
               b"host: x\r\ncontent-type: application/json\r\n"
               b"content-length: " + str(len(payload)).encode() +
               b"\r\n\r\n" + payload)
        loop = asyncio.get_running_loop()

        def abort_once():
            s = socket.create_connection(("127.0.0.1", port), timeout=5)
            s.sendall(req)
            s.recv(256)   # first bytes of the relay
            s.close()     # abort mid-stream

        for _ in range(8):
            await loop.run_in_executor(None, abort_once)
        # server still healthy for well-behaved clients
        async with aiohttp.ClientSession() as c:
            for _ in range(4):
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json={"model": "fast-model", "stream": True,
                          "messages": [{"role": "user", "content": "x"}]},
                    timeout=aiohttp.ClientTimeout(total=5),
                ) as r:
                    assert r.status == 200
                    body = await r.read()
                    assert b"[DONE]" in body
        # aborted handlers notice the dead socket on their next write;
        # give them a moment to drain
        for _ in range(50):
            if front.fast.stats()["active_connections"] == 0:
                break
            await asyncio.sleep(0.05)
        assert front.fast.stats()["active_connections"] == 0
        await front.stop()
        mock.stop()

    asyncio.run(run())


def test_gpu_direct_stats_shape_contract():
    """bench.py zips gpu_direct_stats() with 9 field names; the shape is
    a cross-component contract (csrc/fastpath.h), so lock it."""
    import aigw_fast

    srv = aigw_fast.FastServer()
    stats = srv.gpu_direct_stats()
    assert list(stats) == [0] * 9
