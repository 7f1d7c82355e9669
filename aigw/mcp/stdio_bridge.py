"""stdio → streamable-HTTP MCP bridge.

Parity with cmd/aigw/stdio2http.go: wraps a stdio MCP server (JSON-RPC
messages on stdin/stdout, one per line) as an HTTP endpoint the MCP
gateway — or any streamable-HTTP client — can talk to. Responses are
matched to requests by JSON-RPC id; notifications are fire-and-forget 202s.
"""

from __future__ import annotations

import asyncio
import json
import logging
from typing import Optional

from aiohttp import web

logger = logging.getLogger("aigw.mcp.stdio")


class StdioMCPBridge:
    def __init__(self, command: list[str]):
        self.command = command
        self._proc: Optional[asyncio.subprocess.Process] = None
        self._waiters: dict[object, asyncio.Future] = {}
        self._reader_task: Optional[asyncio.Task] = None

    async def start(self) -> None:
        self._proc = await asyncio.create_subprocess_exec(
            *self.command,
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.DEVNULL,
        )
        self._reader_task = asyncio.create_task(self._read_loop(), name="mcp-stdio-reader")

    async def stop(self) -> None:
        if self._reader_task:
            self._reader_task.cancel()
            try:
                await self._reader_task
            except asyncio.CancelledError:
                pass
        if self._proc and self._proc.returncode is None:
            self._proc.terminate()
            try:
                await asyncio.wait_for(self._proc.wait(), timeout=5)
            except asyncio.TimeoutError:
                self._proc.kill()

    async def _read_loop(self) -> None:
        assert self._proc and self._proc.stdout
        while True:
            line = await self._proc.stdout.readline()
            if not line:
                for fut in self._waiters.values():
                    if not fut.done():
                        fut.set_exception(RuntimeError("stdio MCP server exited"))
                self._waiters.clear()
                return
            try:
                msg = json.loads(line)
            except ValueError:
                logger.warning("non-JSON line from stdio server: %r", line[:120])
                continue
            fut = self._waiters.pop(msg.get("id"), None)
            if fut is not None and not fut.done():
                fut.set_result(msg)

    async def call(self, payload: dict, timeout_s: float = 60.0) -> Optional[dict]:
        assert self._proc and self._proc.stdin
        id_ = payload.get("id")
        fut = None
        if id_ is not None:
            fut = asyncio.get_running_loop().create_future()
            self._waiters[id_] = fut
        self._proc.stdin.write(json.dumps(payload, separators=(",", ":")).encode() + b"\n")
        await self._proc.stdin.drain()
        if fut is None:
            return None
        try:
            return await asyncio.wait_for(fut, timeout=timeout_s)
        finally:
            self._waiters.pop(id_, None)

    async def handle_http(self, request: web.Request) -> web.Response:
        try:
            payload = json.loads(await request.read())
        except ValueError:
            return web.json_response(
                {"jsonrpc": "2.0", "id": None,
                 "error": {"code": -32700, "message": "parse error"}},
                status=400,
            )
        try:
            msg = await self.call(payload)
        except Exception as e:
            return web.json_response(
                {"jsonrpc": "2.0", "id": payload.get("id"),
                 "error": {"code": -32603, "message": str(e)}}
            )
        if msg is None:
            return web.Response(status=202)
        return web.json_response(msg)


async def serve_stdio_bridge(command: list[str], host: str, port: int):
    bridge = StdioMCPBridge(command)
    await bridge.start()
    app = web.Application()
    app.router.add_post("/mcp", bridge.handle_http)

    async def _cleanup(_app):
        await bridge.stop()

    app.on_cleanup.append(_cleanup)
    runner = web.AppRunner(app, access_log=None)
    await runner.setup()
    site = web.TCPSite(runner, host, port, reuse_address=True)
    await site.start()
    return bridge, runner
