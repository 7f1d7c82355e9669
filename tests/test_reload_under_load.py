"""Config hot reload under live traffic: swapping RuntimeConfig while
requests are in flight drops nothing (server.go:81-88 semantics — in-
flight requests keep the runtime they resolved; new requests see the new
one)."""

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

CFG = """\
uuid: cfg-%d
routes:
  - name: r%d
    backends:
      - name: b
        schema: OpenAI
        upstream: {host: 127.0.0.1, port: %d}
"""


@pytest.mark.timeout(120)
def test_swap_mid_traffic_drops_nothing():
    async def run():
        up_srv, up_port = await start_fast_mock("127.0.0.1", 0)
        server = GatewayServer(
            RuntimeConfig(load_config(yaml.safe_load(CFG % (0, 0, up_port))))
        )
        await server.start()
        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0, with_fallback=False)
        client = LeanClient()
        body = json.dumps(
            {"model": "m", "messages": [{"role": "user", "content": "x" * 2000}]}
        ).encode()
        results = {"ok": 0, "bad": []}

        async def one():
            r = await client.post(
                host="127.0.0.1", port=port, tls=False,
                path="/v1/chat/completions",
                headers={"content-type": "application/json"}, body=body,
            )
            await r.read()
            r.release()
            if r.status == 200:
                results["ok"] += 1
            else:
                results["bad"].append(r.status)

        async def load():
            for _ in range(40):
                await asyncio.gather(*[one() for _ in range(16)])

        async def swapper():
            for i in range(1, 30):
                await asyncio.sleep(0.01)
                server.swap_runtime(
                    RuntimeConfig(load_config(yaml.safe_load(CFG % (i, i, up_port))))
                )

        await asyncio.gather(load(), swapper())
        assert results["bad"] == []
        assert results["ok"] == 640
        # the final runtime is the last swapped one
        assert server.runtime.config.routes[0].name.startswith("r2")
        await client.close()
        await cleanup()
        up_srv.close()

    asyncio.run(run())
