"""Lean HTTP/1.1 upstream client: framing unit tests against a scripted
raw server (content-length, chunked + trailers, close-delimited, gzip over
chunked, keep-alive reuse, interim 100-continue, oversized headers)."""

import asyncio
import zlib

import pytest

from aigw.extproc.upstream_client import LeanClient, UpstreamError


class ScriptedServer:
    """Serves a fixed list of raw response blobs, one per request."""

    def __init__(self, responses):
        self.responses = list(responses)
        self.connections = 0
        self.requests = []
        self._server = None

    async def start(self):
        async def handle(reader, writer):
            self.connections += 1
            try:
                while self.responses:
                    # read request head + content-length body
                    head = b""
                    while b"\r\n\r\n" not in head:
                        chunk = await reader.read(4096)
                        if not chunk:
                            return
                        head += chunk
                    head_part, _, rest = head.partition(b"\r\n\r\n")
                    clen = 0
                    for line in head_part.split(b"\r\n"):
                        if line.lower().startswith(b"content-length:"):
                            clen = int(line.split(b":")[1])
                    while len(rest) < clen:
                        rest += await reader.read(4096)
                    self.requests.append(head_part + b"\r\n\r\n" + rest)
                    resp, close = self.responses.pop(0)
                    writer.write(resp)
                    await writer.drain()
                    if close:
                        writer.close()
                        return
            except (ConnectionResetError, asyncio.IncompleteReadError):
                pass

        self._server = await asyncio.start_server(handle, "127.0.0.1", 0)
        return self._server.sockets[0].getsockname()[1]

    async def stop(self):
        self._server.close()
        await self._server.wait_closed()


def run(coro):
    asyncio.run(coro)


def test_content_length_and_keepalive():
    async def main():
        body = b'{"ok":true}'
        resp = (b"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
                b"content-length: %d\r\n\r\n" % len(body)) + body
        srv = ScriptedServer([(resp, False), (resp, False)])
        port = await srv.start()
        c = LeanClient()
        for _ in range(2):
            r = await c.post(host="127.0.0.1", port=port, tls=False, path="/x",
                             headers={}, body=b"{}")
            assert r.status == 200
            assert await r.read() == body
            r.release()
        assert srv.connections == 1  # keep-alive reused the connection
        await c.close()
        await srv.stop()

    run(main())


def test_chunked_with_trailers_and_gzip():
    async def main():
        payload = b"data: hello\n\ndata: [DONE]\n\n"
        co = zlib.compressobj(wbits=31)
        gz = co.compress(payload) + co.flush()
        # split compressed bytes across chunks, add a trailer header
        mid = len(gz) // 2
        chunks = b"".join(
            b"%x\r\n%s\r\n" % (len(part), part) for part in (gz[:mid], gz[mid:])
        )
        resp = (b"HTTP/1.1 200 OK\r\ncontent-encoding: gzip\r\n"
                b"transfer-encoding: chunked\r\n\r\n"
                + chunks + b"0\r\nx-trailer: v\r\n\r\n")
        srv = ScriptedServer([(resp, False)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/s",
                         headers={}, body=b"")
        out = b""
        async for ch in r.iter_chunks():
            out += ch
        assert out == payload
        r.release()
        await c.close()
        await srv.stop()

    run(main())


def test_close_delimited_body():
    async def main():
        resp = b"HTTP/1.1 200 OK\r\nconnection: close\r\n\r\nstream-until-close"
        srv = ScriptedServer([(resp, True)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/c",
                         headers={}, body=b"")
        assert await r.read() == b"stream-until-close"
        r.release()
        # next request must open a NEW connection
        srv.responses.append(
            (b"HTTP/1.1 204 OK\r\ncontent-length: 0\r\n\r\n", False)
        )
        r2 = await c.post(host="127.0.0.1", port=port, tls=False, path="/c2",
                          headers={}, body=b"")
        assert r2.status == 204
        assert srv.connections == 2
        await c.close()
        await srv.stop()

    run(main())


def test_interim_100_continue_skipped():
    async def main():
        resp = (b"HTTP/1.1 100 Continue\r\n\r\n"
                b"HTTP/1.1 201 Created\r\ncontent-length: 2\r\n\r\nok")
        srv = ScriptedServer([(resp, False)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/i",
                         headers={}, body=b"")
        assert r.status == 201
        assert await r.read() == b"ok"
        await c.close()
        await srv.stop()

    run(main())


def test_bad_status_line_raises():
    async def main():
        srv = ScriptedServer([(b"NOT-HTTP garbage\r\n\r\n", True)])
        port = await srv.start()
        c = LeanClient()
        with pytest.raises(UpstreamError):
            await c.post(host="127.0.0.1", port=port, tls=False, path="/b",
                         headers={}, body=b"")
        await c.close()
        await srv.stop()

    run(main())


def test_request_wire_format():
    async def main():
        resp = b"HTTP/1.1 200 OK\r\ncontent-length: 0\r\n\r\n"
        srv = ScriptedServer([(resp, False)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/v1/x?q=1",
                         headers={"authorization": "Bearer k", "host": "override.example"},
                         body=b"BODY")
        await r.read()
        raw = srv.requests[0]
        assert raw.startswith(b"POST /v1/x?q=1 HTTP/1.1\r\n")
        assert b"host: override.example\r\n" in raw
        assert b"authorization: Bearer k\r\n" in raw
        assert b"content-length: 4\r\n" in raw
        assert raw.endswith(b"\r\n\r\nBODY")
        # host header not duplicated
        assert raw.lower().count(b"host:") == 1
        await c.close()
        await srv.stop()

    run(main())


def test_bodyless_204_304_keepalive_not_close_delimited():
    """204/304 carry no body by definition: with neither Content-Length nor
    chunked framing they must yield EOF immediately (not block as
    close-delimited) and keep the connection reusable."""

    async def main():
        ok = b"HTTP/1.1 200 OK\r\ncontent-length: 2\r\n\r\nok"
        for status in (204, 304):
            resp = b"HTTP/1.1 %d X\r\ndate: now\r\n\r\n" % status
            srv = ScriptedServer([(resp, False), (ok, False)])
            port = await srv.start()
            c = LeanClient()
            r = await c.post(host="127.0.0.1", port=port, tls=False, path="/x",
                             headers={}, body=b"{}")
            assert r.status == status
            body = await asyncio.wait_for(r.read(), timeout=2)
            assert body == b""
            r.release()
            # connection stays alive and framed: the next request reuses it
            r2 = await c.post(host="127.0.0.1", port=port, tls=False, path="/x",
                              headers={}, body=b"{}")
            assert r2.status == 200 and await r2.read() == b"ok"
            r2.release()
            assert srv.connections == 1
            await c.close()
            await srv.stop()

    run(main())


def _brotli_compress(data: bytes) -> bytes:
    """One-shot brotli encode via the system libbrotlienc (test-only)."""
    import ctypes
    import ctypes.util

    lib = ctypes.CDLL(ctypes.util.find_library("brotlienc") or "libbrotlienc.so.1")
    lib.BrotliEncoderCompress.restype = ctypes.c_int
    out_cap = ctypes.c_size_t(len(data) + 1024)
    out_buf = (ctypes.c_uint8 * out_cap.value)()
    ok = lib.BrotliEncoderCompress(
        5, 22, 0, ctypes.c_size_t(len(data)),
        (ctypes.c_uint8 * len(data)).from_buffer_copy(data),
        ctypes.byref(out_cap), out_buf,
    )
    assert ok == 1
    return bytes(out_buf[: out_cap.value])


def test_brotli_response_decompression():
    """content-encoding: br bodies decode incrementally like gzip
    (extproc/util.go:57 parity), including chunked SSE split mid-stream."""
    from aigw.utils.brotli_dec import BrotliDecompressor, available

    assert available(), "libbrotlidec must ship with the image"
    payload = b'{"usage": {"total_tokens": 42}, "pad": "' + b"x" * 5000 + b'"}'
    br = _brotli_compress(payload)

    # unit: incremental feed, byte at a time
    d = BrotliDecompressor()
    out = b"".join(d.decompress(br[i : i + 1]) for i in range(len(br)))
    assert out + d.flush() == payload and d.eof

    async def main():
        resp = (b"HTTP/1.1 200 OK\r\ncontent-encoding: br\r\n"
                b"content-length: %d\r\n\r\n" % len(br)) + br
        srv = ScriptedServer([(resp, False)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/x",
                         headers={}, body=b"{}")
        assert await r.read() == payload
        r.release()
        await c.close()
        await srv.stop()

        # chunked SSE stream, compressed, split across chunk boundaries
        sse = b"".join(b"data: {\"n\": %d}\n\n" % i for i in range(50))
        sbr = _brotli_compress(sse)
        mid = len(sbr) // 3
        chunks = [sbr[:mid], sbr[mid : 2 * mid], sbr[2 * mid :]]
        blob = (b"HTTP/1.1 200 OK\r\ncontent-encoding: br\r\n"
                b"content-type: text/event-stream\r\n"
                b"transfer-encoding: chunked\r\n\r\n")
        for ch in chunks:
            blob += b"%x\r\n" % len(ch) + ch + b"\r\n"
        blob += b"0\r\n\r\n"
        srv = ScriptedServer([(blob, False)])
        port = await srv.start()
        c = LeanClient()
        r = await c.post(host="127.0.0.1", port=port, tls=False, path="/x",
                         headers={}, body=b"{}")
        got = bytearray()
        async for piece in r.iter_chunks():
            got.extend(piece)
        assert bytes(got) == sse
        r.release()
        await c.close()
        await srv.stop()

    run(main())
