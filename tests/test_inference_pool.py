"""Dynamic InferencePool member resolution (aigw/extproc/pool.py).

Parity target: the reference resolves pool members behind a k8s selector
and rewrites the cluster endpoints per member
(extensionserver/inferencepool.go:39-54, post_cluster_modify.go:32).
Here: a members file (and DNS) is re-resolved; membership changes swap
the runtime, traffic follows the current member set, telemetry rows
re-register, and resolution failures keep the last-known members.
"""

import asyncio
import json

import aiohttp
from aiohttp import web

from aigw.extproc.pool import PoolManager, resolve_members
from aigw.extproc.server import GatewayServer
from aigw.filterapi import RuntimeConfig, load_config
from aigw.filterapi.config import ConfigError, InferencePool

import pytest


class FakeReplica:
    def __init__(self, name):
        self.name = name
        self.served = 0
        self.kv_usage = 0.1

    async def chat(self, request):
        self.served += 1
        return web.json_response({
            "id": "r", "object": "chat.completion", "model": self.name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant", "content": self.name},
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 2, "completion_tokens": 1,
                      "total_tokens": 3},
        })

    async def metrics(self, request):
        return web.Response(
            text=f'vllm:gpu_cache_usage_perc{{m="x"}} {self.kv_usage}\n',
            content_type="text/plain")

    async def start(self):
        app = web.Application()
        app.router.add_post("/v1/chat/completions", self.chat)
        app.router.add_get("/metrics", self.metrics)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        return runner, runner.addresses[0][1]


def _pool_cfg(members_file, telemetry=False):
    pool = {"membersFile": str(members_file), "intervalS": 0.05,
            "schema": "OpenAI"}
    if telemetry:
        pool["telemetry"] = {"path": "/metrics", "intervalS": 0.05,
                             "kvTotal": 1000}
    return load_config({
        "routes": [{"name": "pool-route", "pool": pool}],
    })


def test_pool_config_validation():
    with pytest.raises(ConfigError):
        load_config({"routes": [{"name": "r", "pool": {}}]})
    with pytest.raises(ConfigError):  # DNS mode requires a port
        load_config({"routes": [{"name": "r",
                                 "pool": {"service": "svc.local"}}]})
    cfg = load_config({"routes": [{
        "name": "r", "pool": {"service": "svc.local", "port": 9000}}]})
    assert cfg.routes[0].pool.service == "svc.local"


def test_members_file_resolution(tmp_path):
    f = tmp_path / "members"
    f.write_text("# comment\n127.0.0.1:9001\n127.0.0.1:9002\n\n")
    pool = InferencePool(members_file=str(f))
    members = asyncio.run(resolve_members(pool))
    assert members == [("127.0.0.1", 9001), ("127.0.0.1", 9002)]


def test_dns_resolution():
    pool = InferencePool(service="localhost", port=9000)
    members = asyncio.run(resolve_members(pool))
    assert ("127.0.0.1", 9000) in members


def test_membership_changes_route_traffic(tmp_path):
    async def run():
        rep_a, rep_b = FakeReplica("a"), FakeReplica("b")
        runner_a, port_a = await rep_a.start()
        runner_b, port_b = await rep_b.start()
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{port_a}\n")

        server = GatewayServer(RuntimeConfig(_pool_cfg(f)))
        await server.start()
        mgr = PoolManager(server)
        assert await mgr.resolve_once() is True

        route = server.runtime.config.routes[0]
        assert [b.upstream.port for b in route.backends] == [port_a]

        from aigw.extproc.lean_front import serve_lean

        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                            with_fallback=False)

        async def burst(n=6):
            async with aiohttp.ClientSession() as c:
                for _ in range(n):
                    async with c.post(
                        f"http://127.0.0.1:{port}/v1/chat/completions",
                        json={"model": "m",
                              "messages": [{"role": "user", "content": "x"}]},
                    ) as r:
                        assert r.status == 200

        await burst()
        assert rep_a.served == 6 and rep_b.served == 0

        # scale out: add replica B -> traffic spreads to the new member
        f.write_text(f"127.0.0.1:{port_a}\n127.0.0.1:{port_b}\n")
        assert await mgr.resolve_once() is True
        ports = sorted(b.upstream.port
                       for b in server.runtime.config.routes[0].backends)
        assert ports == sorted([port_a, port_b])
        rep_a.served = rep_b.served = 0
        await burst(40)
        assert rep_a.served > 0 and rep_b.served > 0
        assert rep_a.served + rep_b.served == 40

        # scale in: drop A -> only B serves
        f.write_text(f"127.0.0.1:{port_b}\n")
        assert await mgr.resolve_once() is True
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_b.served == 6 and rep_a.served == 0

        # no change -> no swap
        assert await mgr.resolve_once() is False

        # resolution failure keeps the last-known member set
        f.unlink()
        assert await mgr.resolve_once() is False
        assert "pool-route" in mgr.resolve_errors
        rep_b.served = 0
        await burst()
        assert rep_b.served == 6

        await cleanup()
        await server.close()
        await runner_a.cleanup()
        await runner_b.cleanup()

    asyncio.run(run())


def test_pool_members_register_telemetry(tmp_path):
    async def run():
        rep = FakeReplica("a")
        runner, port_a = await rep.start()
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{port_a}\n")
        server = GatewayServer(RuntimeConfig(_pool_cfg(f, telemetry=True)))
        await server.start()
        assert server.telemetry is None  # no static telemetry backends
        mgr = PoolManager(server)
        await mgr.resolve_once()
        assert server.telemetry is not None
        name = f"pool:pool-route:127.0.0.1:{port_a}"
        assert name in server.telemetry.rows
        row = server.telemetry.fresh_row(name, max_age_s=5.0)
        assert row is not None and row[1] == 1000  # kv_total from template
        await server.close()
        await runner.cleanup()

    asyncio.run(run())


def test_members_file_edge_cases(tmp_path):
    f = tmp_path / "m"
    # dedup, default port from pool.port, comments, whitespace
    f.write_text("""
# pool members
10.0.0.1:8000
10.0.0.1:8000
10.0.0.2
  10.0.0.3:9001
""")
    pool = InferencePool(members_file=str(f), port=8000)
    members = asyncio.run(resolve_members(pool))
    assert members == [("10.0.0.1", 8000), ("10.0.0.2", 8000),
                       ("10.0.0.3", 9001)]

    # malformed port -> resolution error (caller keeps last-known set)
    f.write_text("10.0.0.1:not-a-port\n")
    with pytest.raises(ValueError):
        asyncio.run(resolve_members(pool))

    # missing file -> RuntimeError
    f.unlink()
    with pytest.raises(RuntimeError):
        asyncio.run(resolve_members(pool))


class GreedyPickGPU:
    """CPU stand-in for the KV scorer kernel (same greedy objective;
    the kernel is oracle-tested on hardware in test_gpu_kernels)."""

    cache_enabled = False

    async def pick_endpoint(self, stats_rows, predicted):
        def score(row):
            used, total, waiting, running = row
            return ((used + predicted) / max(total, 1.0)
                    + 0.1 * waiting + 0.05 * running)

        return min(range(len(stats_rows)), key=lambda i: score(stats_rows[i]))

    async def count_text_tokens(self, text):
        return 8


def test_picker_over_dynamic_pool_members(tmp_path):
    """The full integration: members resolved from a file, telemetry
    scraped from each member, KV-occupancy picker routing to the least
    loaded — and a freshly scaled-out member joining the SCORED set."""

    async def run():
        rep_a, rep_b = FakeReplica("a"), FakeReplica("b")
        runner_a, port_a = await rep_a.start()
        runner_b, port_b = await rep_b.start()
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{port_a}\n")
        cfg = load_config({
            "routes": [{
                "name": "pool-route", "endpointPicker": True,
                "pool": {"membersFile": str(f), "intervalS": 60,
                         "schema": "OpenAI",
                         "telemetry": {"path": "/metrics",
                                       "intervalS": 0.05,
                                       "kvTotal": 1000}},
            }],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        server.gpu = GreedyPickGPU()
        await server.start()
        mgr = PoolManager(server)
        await mgr.resolve_once()

        from aigw.extproc.lean_front import serve_lean

        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                            with_fallback=False)

        async def burst(n=8):
            async with aiohttp.ClientSession() as c:
                for _ in range(n):
                    async with c.post(
                        f"http://127.0.0.1:{port}/v1/chat/completions",
                        json={"model": "m",
                              "messages": [{"role": "user", "content": "x"}]},
                    ) as r:
                        assert r.status == 200

        await burst(4)
        assert rep_a.served == 4  # only member

        # scale out; B reports LOW occupancy, A HIGH -> picker moves to B
        rep_a.kv_usage, rep_b.kv_usage = 0.9, 0.05
        f.write_text(f"127.0.0.1:{port_a}\n127.0.0.1:{port_b}\n")
        await mgr.resolve_once()  # re-registers telemetry for both
        await asyncio.sleep(0.15)  # let the poller scrape both members
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_b.served == 8 and rep_a.served == 0, (
            rep_a.served, rep_b.served)

        # occupancy flips -> assignment flips on the next scrape
        rep_a.kv_usage, rep_b.kv_usage = 0.05, 0.9
        await asyncio.sleep(0.15)
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_a.served == 8 and rep_b.served == 0

        await cleanup()
        await server.close()
        await runner_a.cleanup()
        await runner_b.cleanup()

    asyncio.run(run())


def test_fast_front_relays_picker_pool_route_to_fallback(tmp_path):
    """--front fast with an endpointPicker pool route: the route is
    ineligible for the native hot path (the picker runs in Python), so
    the C++ server must relay it per-request to the fallback app — and
    the picker must still see scraped occupancy."""

    async def run():
        rep = FakeReplica("a")
        runner, port_a = await rep.start()
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{port_a}\n")
        cfg = load_config({
            "routes": [{
                "name": "pool-route", "endpointPicker": True,
                "pool": {"membersFile": str(f), "intervalS": 60,
                         "schema": "OpenAI",
                         "telemetry": {"path": "/metrics",
                                       "intervalS": 0.05,
                                       "kvTotal": 1000}},
            }],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        server.gpu = GreedyPickGPU()
        await server.start()
        mgr = PoolManager(server)
        await mgr.resolve_once()

        from aigw.extproc.fast_front import FastFront

        front = FastFront(server, server.runtime)
        port = await front.start("127.0.0.1", 0)
        async with aiohttp.ClientSession() as c:
            for _ in range(4):
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json={"model": "m",
                          "messages": [{"role": "user", "content": "x"}]},
                ) as r:
                    assert r.status == 200
                    body = await r.json()
                    assert body["model"] == "a"
        assert rep.served == 4
        assert front.fast.stats()["fallback"] >= 4
        await front.stop()
        await server.close()
        await runner.cleanup()

    asyncio.run(run())


def test_empty_members_file_is_scale_to_zero(tmp_path):
    """An empty members file empties the dynamic set: with no static
    backends the route 503s; repopulating the file restores service."""

    async def run():
        rep = FakeReplica("a")
        runner, port_a = await rep.start()
        f = tmp_path / "members"
        f.write_text(f"127.0.0.1:{port_a}\n")
        server = GatewayServer(RuntimeConfig(_pool_cfg(f)))
        await server.start()
        mgr = PoolManager(server)
        await mgr.resolve_once()

        from aigw.extproc.lean_front import serve_lean

        _, port, cleanup = await serve_lean(server, "127.0.0.1", 0,
                                            with_fallback=False)

        async def status():
            async with aiohttp.ClientSession() as c:
                async with c.post(
                    f"http://127.0.0.1:{port}/v1/chat/completions",
                    json={"model": "m",
                          "messages": [{"role": "user", "content": "x"}]},
                ) as r:
                    return r.status

        assert await status() == 200
        f.write_text("")  # scale to zero
        assert await mgr.resolve_once() is True
        assert await status() >= 500
        f.write_text(f"127.0.0.1:{port_a}\n")
        assert await mgr.resolve_once() is True
        assert await status() == 200

        await cleanup()
        await server.close()
        await runner.cleanup()

    asyncio.run(run())
