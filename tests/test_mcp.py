"""MCP gateway tests: session crypto, tool aggregation/prefixing/selector,
call routing, end-to-end through the gateway app (parity:
internal/mcpproxy handlers_test behaviors)."""

import asyncio
import json

import aiohttp
import pytest
from aiohttp import web

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config
from aigw.mcp.session import SessionCrypto, SessionError


def test_session_roundtrip_and_auth():
    c = SessionCrypto("seed-1")
    token = c.seal({"s": {"b1": "sess-abc"}})
    assert c.open(token) == {"s": {"b1": "sess-abc"}}
    # wrong seed fails closed
    with pytest.raises(SessionError):
        SessionCrypto("other").open(token)
    # tamper detection
    bad = token[:-4] + ("AAAA" if not token.endswith("AAAA") else "BBBB")
    with pytest.raises(SessionError):
        c.open(bad)
    # rotation: new primary seed + old fallback still opens
    rotated = SessionCrypto("seed-2", fallback_seeds=["seed-1"])
    assert rotated.open(token)["s"]["b1"] == "sess-abc"


class FakeMCPServer:
    """Minimal streamable-HTTP MCP server."""

    def __init__(self, name: str, tools: list[str], sse: bool = False):
        self.name = name
        self.tools = tools
        self.sse = sse
        self.calls: list[dict] = []

    async def handle(self, request: web.Request) -> web.StreamResponse:
        payload = json.loads(await request.read())
        self.calls.append(
            {"payload": payload, "session": request.headers.get("mcp-session-id", "")}
        )
        method = payload.get("method")
        id_ = payload.get("id")
        if method == "initialize":
            result = {
                "protocolVersion": "2025-06-18",
                "capabilities": {"tools": {"listChanged": True}},
                "serverInfo": {"name": self.name},
            }
            resp = web.json_response({"jsonrpc": "2.0", "id": id_, "result": result})
            resp.headers["mcp-session-id"] = f"{self.name}-session"
            return resp
        if method and method.startswith("notifications/"):
            return web.Response(status=202)
        if method == "tools/list":
            result = {
                "tools": [
                    {"name": t, "description": f"{t} on {self.name}",
                     "inputSchema": {"type": "object"}}
                    for t in self.tools
                ]
            }
        elif method == "tools/call":
            result = {
                "content": [
                    {"type": "text",
                     "text": f"{self.name}:{payload['params']['name']} ok"}
                ]
            }
        else:
            return web.json_response(
                {"jsonrpc": "2.0", "id": id_, "error": {"code": -32601, "message": "nope"}}
            )
        msg = {"jsonrpc": "2.0", "id": id_, "result": result}
        if self.sse:
            resp = web.StreamResponse()
            resp.content_type = "text/event-stream"
            await resp.prepare(request)
            await resp.write(b"data: " + json.dumps(msg).encode() + b"\n\n")
            await resp.write_eof()
            return resp
        return web.json_response(msg)


async def _start(app_handler):
    app = web.Application()
    app.router.add_post("/mcp", app_handler)
    runner = web.AppRunner(app, access_log=None)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    return runner, runner.addresses[0][1]


def test_mcp_authorization_gate():
    async def main():
        s1 = FakeMCPServer("a", ["t"])
        r1, p1 = await _start(s1.handle)
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {"name": "d", "backends": [
                        {"name": "d", "upstream": {"host": "127.0.0.1", "port": 9}}]}
                ],
                "mcp": {
                    "routes": [
                        {"name": "m", "path": "/mcp",
                         "bearerToken": "tok-123",
                         "resourceMetadataUrl": "https://gw/.well-known/oauth-protected-resource",
                         "backends": [
                             {"name": "a", "upstream": {"host": "127.0.0.1", "port": p1}}]}
                    ]
                },
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        base = f"http://127.0.0.1:{port}/mcp"
        async with aiohttp.ClientSession() as client:
            async with client.post(base, json={"jsonrpc": "2.0", "id": 1, "method": "ping"}) as r:
                assert r.status == 401
                assert "resource_metadata=" in r.headers["www-authenticate"]
            async with client.post(
                base, json={"jsonrpc": "2.0", "id": 1, "method": "ping"},
                headers={"authorization": "Bearer tok-123"},
            ) as r:
                assert r.status == 200
        await gw.cleanup()
        await r1.cleanup()

    asyncio.run(main())


def test_mcp_gateway_end_to_end():
    async def main():
        s1 = FakeMCPServer("alpha", ["search", "fetch", "secret_tool"])
        s2 = FakeMCPServer("beta", ["compute"], sse=True)
        r1, p1 = await _start(s1.handle)
        r2, p2 = await _start(s2.handle)
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {"name": "dummy", "backends": [
                        {"name": "d", "upstream": {"host": "127.0.0.1", "port": 9}}]}
                ],
                "mcp": {
                    "sessionSeed": "test-seed",
                    "routes": [
                        {
                            "name": "m",
                            "path": "/mcp",
                            "backends": [
                                {"name": "alpha",
                                 "upstream": {"host": "127.0.0.1", "port": p1},
                                 "toolExclude": ["secret_.*"]},
                                {"name": "beta",
                                 "upstream": {"host": "127.0.0.1", "port": p2}},
                            ],
                        }
                    ],
                },
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        base = f"http://127.0.0.1:{port}/mcp"
        async with aiohttp.ClientSession() as client:
            # initialize: merged capabilities + encrypted session header
            async with client.post(
                base,
                json={"jsonrpc": "2.0", "id": 1, "method": "initialize",
                      "params": {"protocolVersion": "2025-06-18", "capabilities": {}}},
            ) as r:
                assert r.status == 200
                body = await r.json()
                assert body["result"]["capabilities"]["tools"]["listChanged"] is True
                token = r.headers["mcp-session-id"]
                assert token and "alpha-session" not in token  # encrypted, not raw

            hdrs = {"mcp-session-id": token}
            # tools/list: prefixed, selector-filtered aggregation
            async with client.post(
                base, json={"jsonrpc": "2.0", "id": 2, "method": "tools/list"},
                headers=hdrs,
            ) as r:
                tools = {t["name"] for t in (await r.json())["result"]["tools"]}
                assert tools == {"alpha__search", "alpha__fetch", "beta__compute"}
                assert "alpha__secret_tool" not in tools

            # tools/call routes to the right backend with ITS session id
            async with client.post(
                base,
                json={"jsonrpc": "2.0", "id": 3, "method": "tools/call",
                      "params": {"name": "beta__compute", "arguments": {}}},
                headers=hdrs,
            ) as r:
                body = await r.json()
                assert body["result"]["content"][0]["text"] == "beta:compute ok"
            assert s2.calls[-1]["payload"]["params"]["name"] == "compute"
            assert s2.calls[-1]["session"] == "beta-session"

            # excluded tool cannot be called even with the prefix
            async with client.post(
                base,
                json={"jsonrpc": "2.0", "id": 4, "method": "tools/call",
                      "params": {"name": "alpha__secret_tool", "arguments": {}}},
                headers=hdrs,
            ) as r:
                assert "error" in await r.json()

            # ping is answered locally
            async with client.post(
                base, json={"jsonrpc": "2.0", "id": 5, "method": "ping"}
            ) as r:
                assert (await r.json())["result"] == {}

            # tampered session -> 404 invalid session
            async with client.post(
                base, json={"jsonrpc": "2.0", "id": 6, "method": "tools/list"},
                headers={"mcp-session-id": "not-a-session"},
            ) as r:
                assert r.status == 404
        await gw.cleanup()
        await r1.cleanup()
        await r2.cleanup()

    asyncio.run(main())


def test_session_seed_rotation():
    """mainlib --mcpFallbackSessionEncryptionSeed: sessions minted under
    the OLD seed keep decrypting after the seed rotates."""
    from aigw.mcp.session import SessionCrypto

    old = SessionCrypto("seed-v1")
    token = old.seal({"backends": {"a": "s1"}})
    rotated = SessionCrypto("seed-v2", fallback_seeds=["seed-v1"])
    assert rotated.open(token) == {"backends": {"a": "s1"}}
    # and new sessions use the new seed
    t2 = rotated.seal({"x": 1})
    assert SessionCrypto("seed-v2").open(t2) == {"x": 1}
    import pytest as _pytest

    with _pytest.raises(Exception):
        SessionCrypto("seed-v3").open(token)
