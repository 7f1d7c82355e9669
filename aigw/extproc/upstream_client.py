"""Lean pooled HTTP/1.1 upstream client.

Purpose-built replacement for the general-purpose aiohttp client on the
gateway's upstream hop (the reference gets this layer from Envoy's C++
cluster/codec machinery). Profiling showed aiohttp's client call path —
URL re-parsing, CIMultiDict construction, middleware/trace hooks, timer
wheels — costing ~35% of a worker's loop time per request, all of it
avoidable for this fixed workload: POST with a known body to a fixed
origin, keep-alive, HTTP/1.1.

Supports exactly what the data plane needs:
- per-origin keep-alive connection pool (unbounded; sized by concurrency);
- request bodies with Content-Length;
- response framing: Content-Length, chunked, or close-delimited;
- streamed reads (iter_chunks) for SSE / event-stream responses;
- transparent gzip/deflate decompression (incremental, so streamed
  compressed bodies re-chunk correctly — invariant A.8);
- per-request total timeout; TLS origins.

Everything else (redirects, cookies, proxies, HTTP/2, 100-continue) is
deliberately out of scope.
"""

from __future__ import annotations

import asyncio
import ssl as ssl_mod
import zlib
from typing import AsyncIterator

_MAX_HEADER_BYTES = 64 * 1024


class UpstreamError(Exception):
    pass


class LeanResponse:
    def __init__(self, conn: "_Connection", status: int, headers: dict[str, str]):
        self._conn = conn
        self.status = status
        self.headers = headers
        self._decomp = None
        enc = headers.get("content-encoding", "").lower()
        if enc in ("gzip", "deflate"):
            self._decomp = zlib.decompressobj(wbits=47 if enc == "gzip" else 15)
        elif enc == "br":
            # streamed brotli decode via the system libbrotlidec
            # (extproc/util.go:57 parity)
            from aigw.utils.brotli_dec import BrotliDecompressor

            self._decomp = BrotliDecompressor()
        clen = headers.get("content-length")
        self._remaining = int(clen) if clen is not None else None
        self._chunked = headers.get("transfer-encoding", "").lower() == "chunked"
        self._eof = False
        if status in (204, 304) or 100 <= status < 200:
            # statuses defined to carry no body: EOF immediately, whatever
            # the framing headers say — otherwise a keep-alive 204 with no
            # Content-Length would block as close-delimited until timeout
            # and poison the connection
            self._eof = True
            self._chunked = False
            self._remaining = 0

    def _maybe_decompress(self, data: bytes) -> bytes:
        if self._decomp is None or not data:
            return data
        return self._decomp.decompress(data)

    async def _read_raw_chunk(self) -> bytes:
        """One framing-level chunk; b'' at end of body."""
        r = self._conn.reader
        if self._eof:
            return b""
        if self._chunked:
            size_line = await r.readline()
            if not size_line:
                raise UpstreamError("connection closed mid-chunk")
            try:
                size = int(size_line.split(b";")[0].strip(), 16)
            except ValueError as e:
                raise UpstreamError(f"bad chunk size {size_line!r}") from e
            if size == 0:
                # trailers until blank line
                while True:
                    line = await r.readline()
                    if line in (b"\r\n", b"\n", b""):
                        break
                self._eof = True
                return b""
            data = await r.readexactly(size)
            await r.readexactly(2)  # CRLF
            return data
        if self._remaining is not None:
            if self._remaining == 0:
                self._eof = True
                return b""
            data = await r.read(min(self._remaining, 262144))
            if not data:
                raise UpstreamError("connection closed before content-length")
            self._remaining -= len(data)
            if self._remaining == 0:
                self._eof = True
            return data
        # close-delimited
        data = await r.read(262144)
        if not data:
            self._eof = True
            self._conn.reusable = False
        return data

    async def iter_chunks(self, idle_timeout: float = 0) -> AsyncIterator[bytes]:
        """Yield decompressed body chunks. idle_timeout > 0 bounds the gap
        between consecutive raw chunks (per-try idle timeout — the
        reference's route.retry_policy.per_try_idle_timeout); on expiry
        the connection is dropped and UpstreamError raised."""
        while True:
            if idle_timeout > 0:
                try:
                    raw = await asyncio.wait_for(
                        self._read_raw_chunk(), timeout=idle_timeout
                    )
                except asyncio.TimeoutError:
                    self._conn.reusable = False
                    raise UpstreamError(
                        f"no upstream bytes within {idle_timeout}s (idle timeout)"
                    )
            else:
                raw = await self._read_raw_chunk()
            if not raw and self._eof:
                tail = self._decomp.flush() if self._decomp is not None else b""
                if tail:
                    yield tail
                return
            out = self._maybe_decompress(raw)
            if out:
                yield out

    async def read(self, idle_timeout: float = 0) -> bytes:
        parts = []
        async for c in self.iter_chunks(idle_timeout):
            parts.append(c)
        return b"".join(parts)

    def release(self) -> None:
        """Return the connection to the pool (body must be fully read)."""
        self._conn.pool._release(self._conn)

    def close(self) -> None:
        self._conn.reusable = False
        self._conn.pool._release(self._conn)


class _Connection:
    def __init__(self, pool: "_OriginPool", reader, writer):
        self.pool = pool
        self.reader = reader
        self.writer = writer
        self.reusable = True

    def alive(self) -> bool:
        return self.reusable and not self.reader.at_eof() and not self.writer.is_closing()


class _OriginPool:
    def __init__(self, host: str, port: int, tls: bool, server_name: str = "",
                 ca_file: str = "", ca_pem: str = ""):
        self.host = host
        self.port = port
        self.tls = tls
        self.server_name = server_name or host
        self.ca_file = ca_file
        self.ca_pem = ca_pem
        self._ssl_ctx = None
        self._idle: list[_Connection] = []

    async def acquire(self) -> _Connection:
        while self._idle:
            conn = self._idle.pop()
            if conn.alive():
                return conn
            conn.writer.close()
        ssl_ctx = None
        if self.tls:
            if self._ssl_ctx is None:
                # custom trust anchor for private upstreams (BackendTLSPolicy
                # caCertificateRefs analogue); system store otherwise
                self._ssl_ctx = ssl_mod.create_default_context(
                    cafile=self.ca_file or None,
                    cadata=self.ca_pem or None,
                )
            ssl_ctx = self._ssl_ctx
        reader, writer = await asyncio.open_connection(
            self.host, self.port, ssl=ssl_ctx,
            server_hostname=self.server_name if self.tls else None,
        )
        sock = writer.get_extra_info("socket")
        if sock is not None:
            import socket as _s

            sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)
        return _Connection(self, reader, writer)

    def _release(self, conn: _Connection) -> None:
        if conn.alive():
            self._idle.append(conn)
        else:
            conn.writer.close()

    def close(self) -> None:
        for c in self._idle:
            c.writer.close()
        self._idle.clear()


class LeanClient:
    """Pooled client over all upstream origins of one worker."""

    def __init__(self):
        self._pools: dict[tuple[str, int, bool], _OriginPool] = {}

    def _pool(self, host: str, port: int, tls: bool, server_name: str = "",
              ca_file: str = "", ca_pem: str = "") -> _OriginPool:
        key = (host, port, tls, ca_file, bool(ca_pem))
        p = self._pools.get(key)
        if p is None:
            p = _OriginPool(host, port, tls, server_name, ca_file, ca_pem)
            self._pools[key] = p
        return p

    async def post(
        self,
        *,
        host: str,
        port: int,
        tls: bool,
        path: str,
        headers: dict[str, str],
        body: bytes,
        timeout_s: float = 60.0,
        server_name: str = "",
        ca_file: str = "",
        ca_pem: str = "",
    ) -> LeanResponse:
        return await asyncio.wait_for(
            self._post(host, port, tls, path, headers, body, server_name,
                       ca_file, ca_pem),
            timeout=timeout_s,
        )

    async def _post(self, host, port, tls, path, headers, body, server_name,
                    ca_file="", ca_pem="") -> LeanResponse:
        pool = self._pool(host, port, tls, server_name, ca_file, ca_pem)
        conn = await pool.acquire()
        try:
            authority = headers.get("host") or (
                host if port in (80, 443) else f"{host}:{port}"
            )
            buf = bytearray()
            buf += b"POST " + path.encode() + b" HTTP/1.1\r\n"
            buf += b"host: " + authority.encode() + b"\r\n"
            buf += b"content-length: " + str(len(body)).encode() + b"\r\n"
            for k, v in headers.items():
                if k in ("host", "content-length", "connection", "transfer-encoding"):
                    continue
                buf += k.encode() + b": " + v.encode() + b"\r\n"
            buf += b"\r\n"
            conn.writer.write(bytes(buf) + body)
            await conn.writer.drain()

            status, resp_headers = await self._read_head(conn)
            if resp_headers.get("connection", "").lower() == "close":
                conn.reusable = False
            return LeanResponse(conn, status, resp_headers)
        except Exception:
            conn.reusable = False
            pool._release(conn)
            raise

    @staticmethod
    async def _read_head(conn: _Connection) -> tuple[int, dict[str, str]]:
        r = conn.reader
        status_line = await r.readline()
        if not status_line:
            raise UpstreamError("connection closed before response")
        parts = status_line.split(None, 2)
        if len(parts) < 2 or not parts[0].startswith(b"HTTP/1."):
            raise UpstreamError(f"bad status line {status_line!r}")
        status = int(parts[1])
        headers: dict[str, str] = {}
        total = len(status_line)
        while True:
            line = await r.readline()
            total += len(line)
            if total > _MAX_HEADER_BYTES:
                raise UpstreamError("response headers too large")
            if line in (b"\r\n", b"\n"):
                break
            if not line:
                raise UpstreamError("connection closed in headers")
            k, _, v = line.partition(b":")
            key = k.decode("latin1").strip().lower()
            if key in headers:
                headers[key] += ", " + v.decode("latin1").strip()
            else:
                headers[key] = v.decode("latin1").strip()
        if status == 100:  # skip interim responses
            return await LeanClient._read_head(conn)
        return status, headers

    async def close(self) -> None:
        for p in self._pools.values():
            p.close()
        self._pools.clear()
