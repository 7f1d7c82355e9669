"""Endpoint-picker telemetry: the KV-occupancy scorer must run on REAL
replica state scraped from vLLM-style /metrics endpoints (parity:
extensionserver/inferencepool.go:39-54 — the reference feeds its EPP the
same inputs via the inference-extension protocol). E2E: two fake
replicas report different KV occupancy; assignment must follow the
scraped numbers, and flip when they flip."""

import asyncio
import json

import aiohttp
from aiohttp import web

from aigw.extproc.server import GatewayServer
from aigw.extproc.telemetry import ReplicaTelemetry, parse_prometheus_gauge
from aigw.filterapi import RuntimeConfig, load_config


def test_prometheus_gauge_parsing():
    text = """# HELP vllm:gpu_cache_usage_perc GPU KV-cache usage
# TYPE vllm:gpu_cache_usage_perc gauge
vllm:gpu_cache_usage_perc{model_name="m1"} 0.25
vllm:gpu_cache_usage_perc{model_name="m2"} 0.15
vllm:num_requests_running{model_name="m1"} 3.0
vllm:num_requests_waiting{model_name="m1"} 7
other_metric 99
"""
    assert parse_prometheus_gauge(text, "vllm:gpu_cache_usage_perc") == 0.40
    assert parse_prometheus_gauge(text, "vllm:num_requests_running") == 3.0
    assert parse_prometheus_gauge(text, "vllm:num_requests_waiting") == 7.0
    assert parse_prometheus_gauge(text, "vllm:num_requests") is None  # prefix
    assert parse_prometheus_gauge(text, "missing") is None


class FakeReplica:
    """Chat endpoint + vLLM-style /metrics with settable occupancy."""

    def __init__(self, name: str):
        self.name = name
        self.kv_usage = 0.0
        self.running = 0
        self.waiting = 0
        self.served = 0

    async def chat(self, request: web.Request) -> web.Response:
        self.served += 1
        return web.json_response({
            "id": "r", "object": "chat.completion", "model": self.name,
            "choices": [{"index": 0, "message": {"role": "assistant",
                                                 "content": self.name},
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 2, "completion_tokens": 1,
                      "total_tokens": 3},
        })

    async def metrics(self, request: web.Request) -> web.Response:
        body = (
            f'vllm:gpu_cache_usage_perc{{model_name="m"}} {self.kv_usage}\n'
            f'vllm:num_requests_running{{model_name="m"}} {self.running}\n'
            f'vllm:num_requests_waiting{{model_name="m"}} {self.waiting}\n'
        )
        return web.Response(text=body, content_type="text/plain")

    async def start(self):
        app = web.Application()
        app.router.add_post("/v1/chat/completions", self.chat)
        app.router.add_get("/metrics", self.metrics)
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        return runner, runner.addresses[0][1]


class GreedyPickGPU:
    """CPU stand-in for the KV scorer kernel: identical greedy objective
    (lowest projected occupancy + load penalties) — the kernel itself is
    oracle-tested on hardware in tests/test_gpu_kernels.py."""

    cache_enabled = False

    async def pick_endpoint(self, stats_rows, predicted):
        def score(row):
            used, total, waiting, running = row
            return (used + predicted) / max(total, 1.0) + 0.1 * waiting + 0.05 * running

        return min(range(len(stats_rows)), key=lambda i: score(stats_rows[i]))

    async def count_text_tokens(self, text):
        return 8


def test_endpoint_picker_follows_scraped_occupancy():
    async def run():
        rep_a, rep_b = FakeReplica("replica-a"), FakeReplica("replica-b")
        runner_a, port_a = await rep_a.start()
        runner_b, port_b = await rep_b.start()
        cfg = load_config({
            "routes": [{
                "name": "pool",
                "endpointPicker": True,
                "backends": [
                    {"name": "replica-a", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": port_a},
                     "telemetry": {"path": "/metrics", "intervalS": 0.05,
                                   "kvTotal": 1000}},
                    {"name": "replica-b", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": port_b},
                     "telemetry": {"path": "/metrics", "intervalS": 0.05,
                                   "kvTotal": 1000}},
                ],
            }],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        server.gpu = GreedyPickGPU()
        await server.start()
        assert server.telemetry is not None
        assert set(server.telemetry.rows) == {"replica-a", "replica-b"}

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

        # replica-a nearly full, replica-b empty -> traffic goes to b
        rep_a.kv_usage, rep_b.kv_usage = 0.9, 0.1
        await asyncio.sleep(0.15)  # let the poller scrape
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_b.served == 8 and rep_a.served == 0, (rep_a.served, rep_b.served)

        # occupancy flips -> assignment flips after the next scrape
        rep_a.kv_usage, rep_b.kv_usage = 0.05, 0.95
        await asyncio.sleep(0.15)
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_a.served == 8 and rep_b.served == 0, (rep_a.served, rep_b.served)

        # queue depth matters too: same KV, b has a deep waiting queue
        rep_a.kv_usage = rep_b.kv_usage = 0.5
        rep_a.waiting, rep_b.waiting = 0, 50
        await asyncio.sleep(0.15)
        rep_a.served = rep_b.served = 0
        await burst()
        assert rep_a.served == 8, (rep_a.served, rep_b.served)

        await cleanup()
        await runner_a.cleanup()
        await runner_b.cleanup()

    asyncio.run(run())


def test_telemetry_scrape_failure_degrades_gracefully():
    """Dead metrics endpoint: rows go stale, the picker falls back to
    local estimates, errors are recorded, traffic keeps flowing."""

    async def run():
        rep = FakeReplica("only")
        runner, port = await rep.start()
        cfg = load_config({
            "routes": [{
                "name": "pool", "endpointPicker": True,
                "backends": [
                    {"name": "only", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": port},
                     "telemetry": {"path": "/metrics", "intervalS": 0.05}},
                    {"name": "dead", "schema": "OpenAI",
                     "upstream": {"host": "127.0.0.1", "port": 9},
                     "telemetry": {"path": "/metrics", "intervalS": 0.05}},
                ],
            }],
        })
        server = GatewayServer(RuntimeConfig(cfg))
        server.gpu = GreedyPickGPU()
        await server.start()
        t = server.telemetry
        assert "only" in t.rows
        assert "dead" in t.scrape_errors
        assert t.fresh_row("dead", max_age_s=1.0) is None
        assert t.fresh_row("only", max_age_s=1.0) is not None
        # stale check: a row older than max_age is not served
        import time as _time

        ts, row = t.rows["only"]
        t.rows["only"] = (ts - 100.0, row)
        assert t.fresh_row("only", max_age_s=1.0) is None
        await server.close()
        await runner.cleanup()

    asyncio.run(run())


def test_prometheus_parser_never_crashes():
    """The scraper feeds arbitrary replica output into the parser; any
    bytes must parse to a float-or-None, never raise (a misbehaving
    replica must not take down the poller)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=300), st.text(min_size=1, max_size=40))
    def check(text, name):
        out = parse_prometheus_gauge(text, name)
        assert out is None or isinstance(out, float)

    check()


def test_prometheus_parser_hostile_lines():
    hostile = (
        "m NaN\nm +Inf\nm -Inf\n"          # non-finite -> ignored
        "m{label=\"}\"} 1.5\n"              # brace inside label value
        "m\n"                                # no value at all
        "m 1 2 3\n"                          # extra fields (last wins per split)
        "\x00\x01\x02 m 7\n"                 # binary garbage
    )
    out = parse_prometheus_gauge(hostile, "m")
    # the finite samples sum; NaN/Inf are dropped so the scorer's min()
    # comparison can never be poisoned
    assert out is not None and out == out and abs(out) != float("inf")
    assert parse_prometheus_gauge("m NaN\n", "m") is None
