"""Robustness fuzz of the gateway HTTP surface (CPU-only): malformed
JSON bodies and broken HTTP framing must produce clean 4xx responses or
connection closes — never unhandled exceptions, hangs, or 5xx (the
reference inherits this hardening from Envoy's codec; the lean front
implements framing itself, so it gets fuzzed here)."""

import asyncio
import json

import pytest
import yaml

from aigw.extproc.lean_front import serve_lean
from aigw.extproc.server import GatewayServer
from aigw.extproc.upstream_client import LeanClient
from aigw.filterapi.config import load_config
from aigw.filterapi.runtime import RuntimeConfig
from aigw.testing.fastmock import start_fast_mock


def _rng(seed=0x5EED):
    state = seed or 1

    def nxt(n):
        nonlocal state
        state ^= (state << 13) & 0xFFFFFFFFFFFFFFFF
        state ^= state >> 7
        state ^= (state << 17) & 0xFFFFFFFFFFFFFFFF
        return state % n

    return nxt


async def _gateway(idle_timeout_s: float = 2.0):
    up_srv, up_port = await start_fast_mock("127.0.0.1", 0)
    cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
    server = GatewayServer(RuntimeConfig(cfg))
    await server.start()
    _, gw_port, gw_cleanup = await serve_lean(
        server, "127.0.0.1", 0, with_fallback=False, idle_timeout_s=idle_timeout_s
    )

    async def cleanup():
        await gw_cleanup()
        await server.close()
        up_srv.close()

    return server, gw_port, cleanup


@pytest.mark.timeout(120)
def test_malformed_json_bodies_get_4xx():
    async def run():
        _, port, cleanup = await _gateway()
        client = LeanClient()
        nxt = _rng()
        valid = json.dumps({"model": "m", "messages": [{"role": "user", "content": "hi"}]})
        alphabet = '{}[]",:null true false 0123456789.eE-\\u00'
        cases = [b"", b"{", b"[1,", b'{"model":}', b'{"model":"m"',
                 b"\xff\xfe\x00", b'{"model": "m", "messages": "not-a-list"}',
                 b"[" * 100, json.dumps({"messages": []}).encode()]
        for _ in range(150):
            b = bytearray(valid.encode())
            for _m in range(1 + nxt(4)):
                b[nxt(len(b))] = ord(alphabet[nxt(len(alphabet))])
            cases.append(bytes(b))
        for body in cases:
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )
            data = await r.read()
            # mutations can leave the body valid (-> 200 via mock) or
            # make it invalid (-> 4xx); a 5xx or protocol error = bug
            assert r.status < 500, (r.status, body[:120], data[:200])
            r.release()
        await client.close()
        await cleanup()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_broken_http_framing_never_hangs():
    async def run():
        _, port, cleanup = await _gateway()
        frames = [
            b"GARBAGE\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\ncontent-length: -5\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\ncontent-length: zzz\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\n" + b"x: y\r\n" * 5000 + b"\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\ncontent-length: 10\r\n\r\nshort",
            # ^ stalls mid-body: the idle guard must answer 408 and close
            b"POST /nope HTTP/1.1\r\ncontent-length: 2\r\n\r\n{}",
            b"GET /health HTTP/1.1\r\n\r\n" * 3,  # pipelined
            b"\r\n\r\n\r\n",
        ]
        for payload in frames:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            writer.write(payload)
            try:
                await writer.drain()
                # server must either answer or close within the deadline
                await asyncio.wait_for(reader.read(65536), timeout=6)
            except (asyncio.TimeoutError,) as e:
                raise AssertionError(f"hang on frame {payload[:60]!r}") from e
            except (ConnectionResetError, BrokenPipeError):
                pass
            finally:
                writer.close()
        # the server must still be healthy afterwards
        client = LeanClient()
        r = await client.post(
            host="127.0.0.1", port=port, tls=False, path="/v1/chat/completions",
            headers={"content-type": "application/json"},
            body=json.dumps({"model": "m", "messages": [{"role": "user", "content": "ok"}]}).encode(),
        )
        assert r.status == 200
        await r.read()
        r.release()
        await client.close()
        await cleanup()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_transfer_encoding_and_duplicate_content_length_rejected():
    """Request-desync hardening: this front frames by Content-Length only,
    so any Transfer-Encoding must be refused (501) and conflicting
    Content-Length values must be refused (400) — the body bytes of a
    chunked request must NEVER be parsed as pipelined follow-up requests."""

    async def run():
        _, port, cleanup = await _gateway()

        async def send(payload: bytes) -> bytes:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            writer.write(payload)
            try:
                await writer.drain()
                return await asyncio.wait_for(reader.read(65536), timeout=6)
            finally:
                writer.close()

        # chunked request whose body smuggles a pipelined GET /health —
        # must be rejected wholesale, not answered twice
        smuggle = (
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"transfer-encoding: chunked\r\n\r\n"
            b"2\r\n{}\r\n0\r\n\r\n"
            b"GET /health HTTP/1.1\r\n\r\n"
        )
        data = await send(smuggle)
        assert data.startswith(b"HTTP/1.1 501"), data[:80]
        assert data.count(b"HTTP/1.1") == 1, "smuggled request was answered"

        for te in (b"chunked", b"identity", b"gzip, chunked", b"ChUnKeD"):
            data = await send(
                b"POST /v1/chat/completions HTTP/1.1\r\n"
                b"Transfer-Encoding: " + te + b"\r\ncontent-length: 2\r\n\r\n{}"
            )
            assert data.startswith(b"HTTP/1.1 501"), (te, data[:80])

        # conflicting Content-Length values
        data = await send(
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"content-length: 2\r\ncontent-length: 5\r\n\r\n{}123"
        )
        assert data.startswith(b"HTTP/1.1 400"), data[:80]

        # duplicate but AGREEING Content-Length is tolerated (last-win is
        # safe when the values match)
        body = json.dumps({"model": "m",
                           "messages": [{"role": "user", "content": "ok"}]}).encode()
        data = await send(
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"content-length: %d\r\ncontent-length: %d\r\n"
            b"content-type: application/json\r\n\r\n" % (len(body), len(body)) + body
        )
        assert data.startswith(b"HTTP/1.1 200"), data[:80]
        await cleanup()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_slow_and_partial_requests():
    """Byte-at-a-time delivery and mid-body disconnects must not wedge
    the worker or leak the connection slot."""

    async def run():
        _, port, cleanup = await _gateway()
        body = json.dumps({"model": "m", "messages": [{"role": "user", "content": "x"}]}).encode()
        head = (b"POST /v1/chat/completions HTTP/1.1\r\n"
                b"content-type: application/json\r\n"
                b"content-length: %d\r\n\r\n" % len(body))
        # trickle a full request
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        for i in range(0, len(head + body), 7):
            writer.write((head + body)[i : i + 7])
            await writer.drain()
            await asyncio.sleep(0)
        resp = await asyncio.wait_for(reader.readline(), timeout=5)
        assert b"200" in resp
        writer.close()
        # disconnect mid-body x20
        for _ in range(20):
            _, w = await asyncio.open_connection("127.0.0.1", port)
            w.write(head + body[: len(body) // 2])
            await w.drain()
            w.close()
        # server still serves
        client = LeanClient()
        r = await client.post(
            host="127.0.0.1", port=port, tls=False, path="/v1/chat/completions",
            headers={"content-type": "application/json"}, body=body,
        )
        assert r.status == 200
        await r.read()
        r.release()
        await client.close()
        await cleanup()

    asyncio.run(run())


async def _fast_gateway():
    """Same surface as _gateway() but served by the native C++ front."""
    from aigw.extproc.fast_front import FastFront

    up_srv, up_port = await start_fast_mock("127.0.0.1", 0)
    cfg = load_config(yaml.safe_load(f"""
routes:
  - name: r
    backends:
      - name: b
        schema: OpenAI
        upstream: {{host: 127.0.0.1, port: {up_port}}}
"""))
    server = GatewayServer(RuntimeConfig(cfg))
    front = FastFront(server, server.runtime)
    port = await front.start("127.0.0.1", 0)

    async def cleanup():
        await front.stop()
        up_srv.close()

    return server, port, cleanup


@pytest.mark.timeout(120)
def test_fast_front_malformed_json_bodies_get_4xx():
    """The native server gets the same mutation corpus as the lean front:
    clean 4xx or 200, never 5xx/hangs."""

    async def run():
        _, port, cleanup = await _fast_gateway()
        client = LeanClient()
        nxt = _rng(0xFA57)
        valid = json.dumps({"model": "m", "messages": [{"role": "user", "content": "hi"}]})
        alphabet = '{}[]",:null true false 0123456789.eE-\\u00'
        cases = [b"", b"{", b"[1,", b'{"model":}', b'{"model":"m"',
                 b"\xff\xfe\x00", b'{"model": "m", "messages": "not-a-list"}',
                 b"[" * 100, json.dumps({"messages": []}).encode()]
        for _ in range(150):
            b = bytearray(valid.encode())
            for _m in range(1 + nxt(4)):
                b[nxt(len(b))] = ord(alphabet[nxt(len(alphabet))])
            cases.append(bytes(b))
        for body in cases:
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )
            data = await r.read()
            assert r.status < 500, (r.status, body[:120], data[:200])
            r.release()
        await client.close()
        await cleanup()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_fast_front_broken_framing_never_hangs():
    async def run():
        _, port, cleanup = await _fast_gateway()
        frames = [
            b"GARBAGE\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\ncontent-length: -5\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\ncontent-length: zzz\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\n" + b"x: y\r\n" * 5000 + b"\r\n",
            b"POST /nope HTTP/1.1\r\ncontent-length: 2\r\n\r\n{}",
            b"GET /health HTTP/1.1\r\n\r\n" * 3,  # pipelined
            b"\r\n\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"transfer-encoding: chunked\r\n\r\n2\r\n{}\r\n0\r\n\r\n",
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"content-length: 2\r\ncontent-length: 9\r\n\r\n{}1234567",
        ]
        for payload in frames:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            writer.write(payload)
            try:
                await writer.drain()
                await asyncio.wait_for(reader.read(65536), timeout=6)
            except (asyncio.TimeoutError,) as e:
                raise AssertionError(f"hang on frame {payload[:60]!r}") from e
            except (ConnectionResetError, BrokenPipeError):
                pass
            finally:
                writer.close()
        # healthy afterwards
        client = LeanClient()
        r = await client.post(
            host="127.0.0.1", port=port, tls=False, path="/v1/chat/completions",
            headers={"content-type": "application/json"},
            body=json.dumps({"model": "m",
                             "messages": [{"role": "user", "content": "ok"}]}).encode(),
        )
        assert r.status == 200
        await r.read()
        r.release()
        await client.close()
        await cleanup()

    asyncio.run(run())
