"""Endpoint-picker routing (InferencePool analogue): the KV-occupancy
scorer's choice drives backend selection; live stats are tracked per
backend. Uses a fake GPU service on CPU (the scorer kernel itself is
covered by tests/test_gpu_kernels.py::test_kv_scorer_matches_greedy_ref)."""

import asyncio
import json

import aiohttp

from aigw.extproc.server import GatewayServer, run_server
from aigw.filterapi import RuntimeConfig, load_config
from aigw.testing.mockupstream import start_mock_upstream


class FakeGPU:
    cache_enabled = False

    def __init__(self, pick):
        self.pick = pick
        self.seen_stats = []

    async def count_request_tokens(self, body):
        return 100

    async def count_text_tokens(self, text):
        return 100

    async def pick_endpoint(self, stats_rows, predicted):
        self.seen_stats.append((stats_rows, predicted))
        return self.pick

    async def tokenize(self, text):
        return [1]


def test_picker_selects_scored_backend():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        up = {"host": "127.0.0.1", "port": up_port}
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {
                        "name": "pool",
                        "endpointPicker": True,
                        "backends": [
                            {"name": "replica-0", "schema": "OpenAI", "upstream": up,
                             "headerMutation": {"set": {"x-replica": "0"}}},
                            {"name": "replica-1", "schema": "OpenAI", "upstream": up,
                             "headerMutation": {"set": {"x-replica": "1"}}},
                            {"name": "replica-2", "schema": "OpenAI", "upstream": up,
                             "headerMutation": {"set": {"x-replica": "2"}}},
                        ],
                    }
                ],
            }
        )
        gpu = FakeGPU(pick=2)
        server = GatewayServer(RuntimeConfig(cfg), gpu_services=gpu)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        async with aiohttp.ClientSession() as client:
            async with client.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "m", "messages": [{"role": "user", "content": "q"}]},
            ) as r:
                assert r.status == 200
        # the scorer's pick (replica-2) got the request
        assert mock.requests[-1]["headers"]["x-replica"] == "2"
        # scorer saw 3 replica rows and the GPU-counted predicted tokens
        stats_rows, predicted = gpu.seen_stats[0]
        assert len(stats_rows) == 3 and predicted == 100.0
        # stats were decremented after completion
        assert server._ep_stats["replica-2"][0] == 0
        await gw.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())


def test_picker_fallback_on_picked_failure():
    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        up = {"host": "127.0.0.1", "port": up_port}
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {
                        "name": "pool",
                        "endpointPicker": True,
                        "retries": 2,
                        "backends": [
                            {"name": "r0", "schema": "OpenAI", "upstream": up,
                             "headerMutation": {"set": {"x-replica": "0"}}},
                            {"name": "r1", "schema": "OpenAI", "upstream": up,
                             "headerMutation": {"set": {
                                 "x-replica": "1",
                                 "x-mock-fail-times": "99",
                                 "x-mock-fail-key": "r1"}}},
                        ],
                    }
                ],
            }
        )
        gpu = FakeGPU(pick=1)  # picker chooses the failing replica
        server = GatewayServer(RuntimeConfig(cfg), gpu_services=gpu)
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]
        async with aiohttp.ClientSession() as client:
            async with client.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "m", "messages": [{"role": "user", "content": "q"}]},
            ) as r:
                assert r.status == 200  # fell back to r0
        replicas = [
            json.loads(json.dumps(req["headers"].get("x-replica")))
            for req in mock.requests
        ]
        assert replicas == ["1", "0"]
        await gw.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())


def test_circuit_breaker_spills_to_fallback():
    """Backend.maxConcurrency opens the circuit and fallback takes over
    (Envoy cluster circuit-breaker parity)."""
    from aigw.testing.mockupstream import start_mock_upstream

    async def main():
        mock, up_runner, up_port = await start_mock_upstream()
        mock.record = True
        up = {"host": "127.0.0.1", "port": up_port}
        cfg = load_config(
            {
                "version": "v1",
                "routes": [
                    {
                        "name": "cb",
                        "retries": 2,
                        "backends": [
                            {"name": "tiny", "schema": "OpenAI", "upstream": up,
                             "maxConcurrency": 1,
                             "headerMutation": {"set": {"x-replica": "tiny",
                                                        "x-mock-delay-ms": "300"}}},
                            {"name": "spill", "schema": "OpenAI", "upstream": up,
                             "priority": 1,
                             "headerMutation": {"set": {"x-replica": "spill"}}},
                        ],
                    }
                ],
            }
        )
        server = GatewayServer(RuntimeConfig(cfg))
        gw = await run_server(server, host="127.0.0.1", port=0)
        port = gw.addresses[0][1]

        async def one(c):
            async with c.post(
                f"http://127.0.0.1:{port}/v1/chat/completions",
                json={"model": "m", "messages": [{"role": "user", "content": "q"}]},
            ) as r:
                assert r.status == 200

        async with aiohttp.ClientSession() as c:
            await asyncio.gather(*(one(c) for _ in range(4)))
        replicas = [r["headers"].get("x-replica") for r in mock.requests]
        assert replicas.count("tiny") >= 1
        assert replicas.count("spill") >= 1  # overflow spilled to fallback
        await gw.cleanup()
        await up_runner.cleanup()

    asyncio.run(main())
