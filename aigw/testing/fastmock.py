"""Minimal raw-asyncio mock provider for the benchmark harness.

The reference's bench upstream (testupstream) serves canned responses; the
shared aiohttp MockUpstream in mockupstream.py stays for functional tests,
but in the benchmark the upstream shares a core with the gateway under
test, so its cost directly eats measured gateway capacity. This server is
an asyncio.Protocol with a fixed-keepalive HTTP/1.1 loop and a precomputed
response — a few microseconds per request.
"""

from __future__ import annotations

import asyncio
import json
import time


def canned_chat_response(model: str = "bench-llm", prompt_tokens: int = 4096,
                         completion_tokens: int = 16) -> bytes:
    body = json.dumps(
        {
            "id": "chatcmpl-bench",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": model,
            "choices": [
                {
                    "index": 0,
                    "message": {"role": "assistant", "content": "ok"},
                    "finish_reason": "stop",
                }
            ],
            "usage": {
                "prompt_tokens": prompt_tokens,
                "completion_tokens": completion_tokens,
                "total_tokens": prompt_tokens + completion_tokens,
            },
        },
        separators=(",", ":"),
    ).encode()
    head = (
        b"HTTP/1.1 200 OK\r\n"
        b"content-type: application/json\r\n"
        b"content-length: " + str(len(body)).encode() + b"\r\n"
        b"\r\n"
    )
    return head + body


class _FastMockProtocol(asyncio.Protocol):
    __slots__ = ("response", "_buf", "_need", "_transport")

    def __init__(self, response: bytes):
        self.response = response
        self._buf = bytearray()
        self._need = -1  # body bytes still expected; -1 = parsing headers
        self._transport = None

    def connection_made(self, transport):
        self._transport = transport
        sock = transport.get_extra_info("socket")
        if sock is not None:
            import socket as _s

            sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)

    def data_received(self, data: bytes):
        self._buf.extend(data)
        while True:
            if self._need < 0:
                end = self._buf.find(b"\r\n\r\n")
                if end < 0:
                    return
                head = bytes(self._buf[:end]).lower()
                clen = 0
                idx = head.find(b"content-length:")
                if idx >= 0:
                    eol = head.find(b"\r\n", idx)
                    clen = int(head[idx + 15 : eol if eol > 0 else None].strip())
                del self._buf[: end + 4]
                self._need = clen
            if len(self._buf) < self._need:
                return
            del self._buf[: self._need]
            self._need = -1
            self._transport.write(self.response)


async def start_fast_mock(host: str = "127.0.0.1", port: int = 0,
                          response: bytes | None = None):
    """Returns (asyncio.Server, actual_port)."""
    resp = response or canned_chat_response()
    loop = asyncio.get_running_loop()
    server = await loop.create_server(lambda: _FastMockProtocol(resp), host, port)
    actual = server.sockets[0].getsockname()[1]
    return server, actual


def canned_chat_sse(model: str = "bench-llm", prompt_tokens: int = 4096,
                    n_chunks: int = 16) -> bytes:
    """Chunked SSE chat stream (content deltas + usage chunk + [DONE]) as
    one raw HTTP byte blob for the native mock (streamed-serving bench)."""
    events = []
    for i in range(n_chunks):
        delta = {"role": "assistant", "content": ""} if i == 0 else {"content": f"tok{i} "}
        events.append({"id": "chatcmpl-bench", "object": "chat.completion.chunk",
                       "model": model,
                       "choices": [{"index": 0, "delta": delta,
                                    "finish_reason": None}]})
    events.append({"id": "chatcmpl-bench", "object": "chat.completion.chunk",
                   "model": model,
                   "choices": [{"index": 0, "delta": {},
                                "finish_reason": "stop"}]})
    events.append({"id": "chatcmpl-bench", "object": "chat.completion.chunk",
                   "model": model, "choices": [],
                   "usage": {"prompt_tokens": prompt_tokens,
                             "completion_tokens": n_chunks,
                             "total_tokens": prompt_tokens + n_chunks}})
    body = b""
    for ev in events:
        data = b"data: " + json.dumps(ev, separators=(",", ":")).encode() + b"\n\n"
        body += b"%x\r\n" % len(data) + data + b"\r\n"
    done = b"data: [DONE]\n\n"
    body += b"%x\r\n" % len(done) + done + b"\r\n"
    body += b"0\r\n\r\n"
    head = (b"HTTP/1.1 200 OK\r\ncontent-type: text/event-stream\r\n"
            b"transfer-encoding: chunked\r\n\r\n")
    return head + body
