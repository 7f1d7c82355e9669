"""stdio → HTTP MCP bridge test: a tiny stdio MCP server implemented as a
python -c child, bridged to HTTP, driven by the MCP gateway client path."""

import asyncio
import sys
import textwrap

import aiohttp

from aigw.mcp.stdio_bridge import serve_stdio_bridge

STDIO_SERVER = textwrap.dedent(
    """
    import json, sys
    for line in sys.stdin:
        msg = json.loads(line)
        mid = msg.get("id")
        if mid is None:
            continue  # notification
        if msg["method"] == "initialize":
            result = {"protocolVersion": "2025-06-18",
                      "capabilities": {"tools": {}},
                      "serverInfo": {"name": "stdio-fake"}}
        elif msg["method"] == "tools/list":
            result = {"tools": [{"name": "echo", "inputSchema": {"type": "object"}}]}
        elif msg["method"] == "tools/call":
            result = {"content": [{"type": "text",
                                   "text": msg["params"]["arguments"]["text"]}]}
        else:
            print(json.dumps({"jsonrpc": "2.0", "id": mid,
                              "error": {"code": -32601, "message": "nope"}}), flush=True)
            continue
        print(json.dumps({"jsonrpc": "2.0", "id": mid, "result": result}), flush=True)
    """
)


def test_stdio_bridge_end_to_end():
    async def main():
        bridge, runner = await serve_stdio_bridge(
            [sys.executable, "-u", "-c", STDIO_SERVER], "127.0.0.1", 0
        )
        port = runner.addresses[0][1]
        base = f"http://127.0.0.1:{port}/mcp"
        async with aiohttp.ClientSession() as c:
            async with c.post(base, json={"jsonrpc": "2.0", "id": 1,
                                          "method": "initialize", "params": {}}) as r:
                body = await r.json()
                assert body["result"]["serverInfo"]["name"] == "stdio-fake"
            async with c.post(base, json={"jsonrpc": "2.0", "id": 2,
                                          "method": "tools/list"}) as r:
                assert (await r.json())["result"]["tools"][0]["name"] == "echo"
            async with c.post(base, json={"jsonrpc": "2.0", "id": 3,
                                          "method": "tools/call",
                                          "params": {"name": "echo",
                                                     "arguments": {"text": "hi"}}}) as r:
                assert (await r.json())["result"]["content"][0]["text"] == "hi"
            # notification -> 202, no body
            async with c.post(base, json={"jsonrpc": "2.0",
                                          "method": "notifications/initialized"}) as r:
                assert r.status == 202
        await runner.cleanup()

    asyncio.run(main())
