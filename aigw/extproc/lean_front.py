"""Raw-asyncio HTTP/1.1 front for the gateway.

The processing core is transport-agnostic (RequestView); this front
replaces aiohttp's server machinery (~100-150 µs of parse/dispatch per
request) with a purpose-built protocol for the gateway's actual surface:

- POST with Content-Length bodies (every JSON endpoint);
- GET admin endpoints (/health, /metrics, /v1/models, /debug/tasks);
- streamed responses via chunked transfer-encoding;
- keep-alive, sequential per-connection request handling.

Paths it does not serve natively (multipart audio, MCP) are proxied to a
loopback aiohttp instance running the full app in the same process, so a
lean-front deployment still exposes the complete surface on one port.
"""

from __future__ import annotations

import asyncio
import json
import logging
from typing import Optional

from aiohttp import web

from aigw.extproc.server import JSON_ENDPOINTS, GatewayServer, RequestView
from aigw.extproc.upstream_client import LeanClient

logger = logging.getLogger("aigw.lean_front")

_GET_ROUTES = {"/v1/models", "/anthropic/v1/models", "/health", "/metrics", "/debug/tasks"}


class _Shim:
    """Duck-typed stand-in for aiohttp's request in the simple GET/admin
    handlers (they use .headers.get, .host, .remote, .json())."""

    __slots__ = ("headers", "host", "remote", "_body")

    def __init__(self, headers: dict, host: str, remote: str, body: bytes):
        self.headers = headers
        self.host = host
        self.remote = remote
        self._body = body

    async def json(self):
        return json.loads(self._body or b"{}")


class _LeanStreamWriter:
    __slots__ = ("transport", "_started")

    def __init__(self, transport):
        self.transport = transport
        self._started = False

    def start(self, status: int, headers: dict[str, str]) -> None:
        head = [f"HTTP/1.1 {status} OK".encode()]
        for k, v in headers.items():
            head.append(f"{k}: {v}".encode())
        head.append(b"transfer-encoding: chunked")
        head.append(b"")
        head.append(b"")
        self.transport.write(b"\r\n".join(head))
        self._started = True

    async def write(self, data: bytes) -> None:
        if self.transport.is_closing():
            raise ConnectionResetError("client closed")
        self.transport.write(b"%x\r\n" % len(data) + data + b"\r\n")

    async def finish(self):
        if not self.transport.is_closing():
            self.transport.write(b"0\r\n\r\n")

    def abort(self) -> None:
        """Close without the terminal chunk: the client sees truncated
        chunked framing (mid-stream idle cut)."""
        self.transport.close()

    def result(self):
        return _STREAMED


_STREAMED = object()  # sentinel: response already written to the transport


class LeanFront(asyncio.Protocol):
    def __init__(self, server: GatewayServer, fallback_port: Optional[int],
                 fallback_client: Optional[LeanClient],
                 idle_timeout_s: float = 60.0):
        self.server = server
        self.fallback_port = fallback_port
        self.fallback = fallback_client
        self.idle_timeout_s = idle_timeout_s
        self._buf = bytearray()
        self._transport = None
        self._chain: Optional[asyncio.Future] = None
        self._peer = ""
        # slowloris guard: armed only while an INCOMPLETE request sits in
        # the buffer; fires -> if no new bytes arrived since, close
        self._idle_handle = None
        self._rx_total = 0          # monotonic; snapshot taken at arm time
        self._rx_at_arm = -1

    # ---- protocol ------------------------------------------------------------

    def connection_made(self, transport):
        self._transport = transport
        peer = transport.get_extra_info("peername")
        self._peer = peer[0] if isinstance(peer, tuple) else "127.0.0.1"
        sock = transport.get_extra_info("socket")
        if sock is not None:
            import socket as _s

            sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)

    def connection_lost(self, exc):
        if self._idle_handle is not None:
            self._idle_handle.cancel()
            self._idle_handle = None
        if self._chain is not None and not self._chain.done():
            self._chain.cancel()

    def data_received(self, data: bytes):
        self._rx_total += len(data)
        self._buf.extend(data)
        while True:
            req = self._try_parse()
            if req is None:
                if self._buf and self._idle_handle is None:
                    self._arm_idle_guard()
                return
            prev = self._chain
            self._chain = asyncio.ensure_future(self._handle_ordered(prev, req))

    def _arm_idle_guard(self):
        self._rx_at_arm = self._rx_total
        self._idle_handle = asyncio.get_running_loop().call_later(
            self.idle_timeout_s, self._idle_check
        )

    def _idle_check(self):
        self._idle_handle = None
        if not self._buf:
            return
        if self._rx_total == self._rx_at_arm:
            # partial request made zero progress for a full window
            if self._transport is not None and not self._transport.is_closing():
                self._transport.write(
                    b"HTTP/1.1 408 Request Timeout\r\ncontent-length: 0\r\n"
                    b"connection: close\r\n\r\n"
                )
                self._transport.close()
            return
        self._arm_idle_guard()

    def _reject(self, status_line: bytes) -> None:
        self._transport.write(
            b"HTTP/1.1 " + status_line + b"\r\ncontent-length: 0\r\n"
            b"connection: close\r\n\r\n"
        )
        self._transport.close()

    def _try_parse(self):
        end = self._buf.find(b"\r\n\r\n")
        if end < 0:
            if len(self._buf) > 65536:
                self._transport.close()
            return None
        head = bytes(self._buf[:end])
        lines = head.split(b"\r\n")
        try:
            method, path, _ = lines[0].split(b" ", 2)
        except ValueError:
            self._transport.close()
            return None
        headers: dict[str, str] = {}
        clen_values: list[str] = []
        te_present = False
        for line in lines[1:]:
            k, _, v = line.partition(b":")
            key = k.decode("latin1").strip().lower()
            val = v.decode("latin1").strip()
            if key == "content-length":
                clen_values.append(val)
            elif key == "transfer-encoding":
                te_present = True
            headers[key] = val
        if te_present:
            # This front frames requests by Content-Length only. Accepting a
            # chunked body would reinterpret its bytes as pipelined requests
            # (request smuggling through any intermediary that forwards TE),
            # so chunked uploads are refused outright rather than mis-framed.
            self._reject(b"501 Not Implemented")
            return None
        if len(set(clen_values)) > 1:
            # conflicting Content-Length values: classic desync vector
            self._reject(b"400 Bad Request")
            return None
        try:
            clen = int(clen_values[0]) if clen_values else 0
        except ValueError:
            clen = -1
        if clen < 0 or clen > 64 * 1024 * 1024:
            self._reject(b"400 Bad Request")
            return None
        total = end + 4 + clen
        if len(self._buf) < total:
            return None
        body = bytes(self._buf[end + 4 : total])
        del self._buf[:total]
        return (method.decode(), path.decode(), headers, body)

    # ---- dispatch ------------------------------------------------------------

    async def _handle_ordered(self, prev, req):
        if prev is not None:
            try:
                await prev
            except Exception:
                pass
        try:
            await self._handle(req)
        except asyncio.CancelledError:
            raise
        except (ConnectionResetError, BrokenPipeError):
            pass
        except Exception:
            logger.exception("lean front request failed")
            self._write_simple(500, {"content-type": "application/json"},
                               b'{"type":"error","error":{"type":"internal_error","code":"500","message":"internal error"}}')

    def _write_simple(self, status: int, headers: dict, body: bytes,
                      keep_alive: bool = True):
        if self._transport.is_closing():
            return
        head = [f"HTTP/1.1 {status} OK".encode()]
        for k, v in headers.items():
            if k.lower() in ("content-length", "transfer-encoding"):
                continue
            head.append(f"{k}: {v}".encode())
        head.append(b"content-length: %d" % len(body))
        if not keep_alive:
            head.append(b"connection: close")
        head.append(b"")
        head.append(b"")
        self._transport.write(b"\r\n".join(head) + body)
        if not keep_alive:
            self._transport.close()

    async def _handle(self, req):
        method, path, headers, body = req
        server = self.server
        if server.draining and path not in ("/health", "/metrics"):
            self._write_simple(
                503, {"content-type": "application/json"},
                b'{"type":"error","error":{"type":"unavailable","code":"503","message":"shutting down"}}',
                keep_alive=False,
            )
            return
        server.inflight += 1
        try:
            await self._handle_inner(req)
        finally:
            server.inflight -= 1
            if server.inflight == 0 and server.draining:
                server._idle_event.set()

    async def _handle_inner(self, req):
        method, path, headers, body = req
        server = self.server
        root = server.root_prefix
        if root:
            if path.startswith(root + "/"):
                path = path[len(root):]
            elif path not in ("/health", "/metrics", "/debug/tasks"):
                # admin endpoints stay unprefixed; everything else must
                # carry the root prefix (mainlib --rootPrefix semantics)
                self._write_simple(
                    404, {"content-type": "application/json"},
                    b'{"type":"error","error":{"type":"not_found",'
                    b'"code":"404","message":"not found"}}',
                )
                return
        if method == "POST" and path in JSON_ENDPOINTS:
            view = RequestView(
                method=method,
                path=path,
                host=headers.get("host", ""),
                remote=self._peer,
                headers=headers,
                body=body,
                stream_factory=self._stream_factory,
            )
            result = await server._process(view, path)
            if result is _STREAMED:
                return
            self._emit(result)
            return
        if method == "GET" and path in _GET_ROUTES:
            shim = _Shim(headers, headers.get("host", ""), self._peer, body)
            if path == "/health":
                self._emit(await server._handle_health(shim))
            elif path == "/metrics":
                self._emit(await server._handle_metrics(shim))
            elif path == "/v1/models":
                self._emit(await server._handle_models(shim))
            elif path == "/anthropic/v1/models":
                self._emit(await server._handle_anthropic_models(shim))
            else:
                self._emit(await server._handle_debug_tasks(shim))
            return
        if method == "POST" and path == "/v1/gateway/tokenize" and server.gpu is not None:
            shim = _Shim(headers, headers.get("host", ""), self._peer, body)
            self._emit(await server._handle_gpu_tokenize(shim))
            return
        await self._fallback(method, path, headers, body)

    async def _stream_factory(self, status: int, headers: dict[str, str]):
        w = _LeanStreamWriter(self._transport)
        w.start(status, headers)
        return w

    def _emit(self, resp: web.Response) -> None:
        headers = {"content-type": resp.content_type or "application/json"}
        for k, v in resp.headers.items():
            if k.lower() not in ("content-length", "content-type"):
                headers[k] = v
        self._write_simple(resp.status, headers, resp.body or b"")

    async def _fallback(self, method, path, headers, body):
        """Proxy cold paths (multipart audio, MCP, unknown) to the loopback
        aiohttp instance serving the full app."""
        if self.fallback_port is None:
            self._write_simple(404, {"content-type": "application/json"},
                               b'{"type":"error","error":{"type":"not_found","code":"404","message":"not found"}}')
            return
        if method != "POST":
            self._write_simple(405, {"content-type": "application/json"},
                               b'{"error":{"message":"method not allowed"}}')
            return
        r = await self.fallback.post(
            host="127.0.0.1", port=self.fallback_port, tls=False, path=path,
            headers=headers, body=body, timeout_s=300.0,
        )
        data = await r.read()
        r.release()
        out_headers = {k: v for k, v in r.headers.items()
                       if k not in ("content-length", "transfer-encoding",
                                    "content-encoding", "connection")}
        self._write_simple(r.status, out_headers, data)


async def serve_lean(server: GatewayServer, host: str, port: int,
                     *, with_fallback: bool = True, reuse_port: bool = False,
                     idle_timeout_s: float = 60.0):
    """Start the lean front (and, when with_fallback, a loopback aiohttp
    app for the cold paths). Returns (asyncio.Server, actual_port, cleanup)."""
    fallback_port = None
    fallback_runner = None
    fallback_client = None
    if with_fallback:
        app = server.make_app()
        fallback_runner = web.AppRunner(app, access_log=None)
        await fallback_runner.setup()
        site = web.TCPSite(fallback_runner, "127.0.0.1", 0)
        await site.start()
        fallback_port = fallback_runner.addresses[0][1]
        fallback_client = LeanClient()
    else:
        await server.start()

    loop = asyncio.get_running_loop()
    srv = await loop.create_server(
        lambda: LeanFront(server, fallback_port, fallback_client,
                          idle_timeout_s=idle_timeout_s),
        host, port, reuse_port=reuse_port, backlog=4096,
    )
    actual = srv.sockets[0].getsockname()[1]

    async def cleanup():
        srv.close()
        await srv.wait_closed()
        if fallback_client is not None:
            await fallback_client.close()
        if fallback_runner is not None:
            await fallback_runner.cleanup()
        await server.close()

    return srv, actual, cleanup
