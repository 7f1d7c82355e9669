"""Cross-shard rate-limit counter sync over torch.distributed.

Runs world_size=2 with gloo on CPU here (the same code path runs RCCL/nccl
over xGMI on the 8-GPU node — backend string is the only difference)."""

import multiprocessing as mp
import os

import pytest

from aigw.filterapi.config import RateLimitRule
from aigw.ratelimit import RateLimiter


def _worker(rank: int, world: int, port: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        from aigw.parallel import StateSync

        dist.init_process_group("gloo", rank=rank, world_size=world)
        rules = [
            RateLimitRule(name="budget", metadata_key="llm_total_token", limit=100,
                          window_s=3600),
            RateLimitRule(name="per-user", metadata_key="llm_total_token", limit=50,
                          window_s=3600, key_headers=["x-user-id"]),
        ]
        limiter = RateLimiter(rules)
        sync = StateSync(limiter)

        if rank == 0:
            # rank 0 spends 80 of the global budget and 30 of user u1's
            limiter.charge({"x-user-id": "u1"}, {"llm_total_token": 80})
        sync.tick_sync()  # collective: both ranks participate

        if rank == 1:
            # rank 1 must now see rank 0's spend: 80 + 30 > remaining
            d = limiter.check({"x-user-id": "u1"})
            assert not d.allowed or limiter._buckets, "bucket state missing"
            # global budget bucket ("" descriptor): 80 spent of 100 -> allowed
            assert limiter.check({}).allowed
            # charge 30 more here -> 110 total after next sync
            limiter.charge({"x-user-id": "u2"}, {"llm_total_token": 30})
        sync.tick_sync()
        sync.tick_sync()

        # after both ticks every rank sees 110 global spend -> denied
        d = limiter.check({})
        assert not d.allowed, f"rank {rank}: expected denial, got {d}"
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def test_counter_sync_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29765
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            pytest.fail("worker hung")
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


class FakeCache:
    """CPU stand-in matching SemanticCache's sync interface (the GPU cache
    itself is covered by tests/test_gpu_kernels.py)."""

    dim = 8

    def __init__(self):
        import torch

        self.device = torch.device("cpu")
        self.rows = []
        self.values = []
        self.pending = []
        self.record_pending = True

    def insert(self, vec, value):
        self.rows.append(vec)
        self.values.append(value)
        if self.record_pending:
            self.pending.append((vec, value))

    def insert_remote(self, vec, value):
        self.rows.append(vec)
        self.values.append(value)

    def drain_pending(self, max_n=32):
        import torch

        batch, self.pending = self.pending[:max_n], self.pending[max_n:]
        if not batch:
            return torch.zeros(0, self.dim, dtype=torch.bfloat16), []
        return torch.stack([v for v, _ in batch]).to(torch.bfloat16), [r for _, r in batch]


def _cache_worker(rank: int, world: int, port: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch
        import torch.distributed as dist

        from aigw.parallel import StateSync

        dist.init_process_group("gloo", rank=rank, world_size=world)
        limiter = RateLimiter(
            [RateLimitRule(name="b", metadata_key="llm_total_token", limit=10, window_s=60)]
        )
        cache = FakeCache()
        sync = StateSync(limiter, cache=cache)
        if rank == 0:
            cache.insert(torch.arange(8, dtype=torch.float32), b"response-zero")
            cache.insert(torch.ones(8), b"response-one")
        sync.tick_sync()
        sync.tick_sync()  # second tick: nothing pending -> no-op collectives
        if rank == 1:
            assert len(cache.rows) == 2, len(cache.rows)
            assert cache.values == [b"response-zero", b"response-one"]
            assert torch.allclose(
                cache.rows[0].float(), torch.arange(8, dtype=torch.float32), atol=0.1
            )
        if rank == 0:
            assert len(cache.rows) == 2  # own rows only, not duplicated
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def test_cache_allgather_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_cache_worker, args=(r, 2, 29767, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            pytest.fail("worker hung")
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_slot_remote_claimed_on_first_local_sight():
    rules = [RateLimitRule(name="r", metadata_key="llm_total_token", limit=100, window_s=60,
                           key_headers=["x-user-id"])]
    limiter = RateLimiter(rules)
    from aigw.ratelimit.limiter import bucket_slot

    slot = bucket_slot("r", "alice")
    limiter.apply_remote_slot(slot, 150)  # remote shard spent alice's budget
    d = limiter.check({"x-user-id": "alice"})
    assert not d.allowed


def test_counter_sync_world8():
    """The full node shape: 8 shards, same collective sequence the
    round-end SCALE bench runs (gloo here, RCCL on hardware)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29781
    procs = [ctx.Process(target=_world8_worker, args=(r, 8, port, q)) for r in range(8)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(8)]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            pytest.fail("worker hung")
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _world8_worker(rank: int, world: int, port: int, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        from aigw.parallel import StateSync

        dist.init_process_group("gloo", rank=rank, world_size=world)
        limiter = RateLimiter([
            RateLimitRule(name="node", metadata_key="llm_total_token",
                          limit=100, window_s=3600),
        ])
        sync = StateSync(limiter)
        # every shard spends 14: below the limit locally, 112 > 100 globally
        limiter.charge({}, {"llm_total_token": 14})
        sync.tick_sync()
        sync.tick_sync()
        d = limiter.check({})
        assert not d.allowed, f"rank {rank}: global budget not enforced: {d}"
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def _fastfront_worker(rank: int, world: int, port: int, q):
    """One native fast-front shard per rank, real requests charging its
    C++ rate counters, StateSync over the FastLimiterBridge."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import json as _json

        import torch.distributed as dist

        import aigw_fast
        from aigw.parallel import FastLimiterBridge, StateSync

        dist.init_process_group("gloo", rank=rank, world_size=world)
        body = _json.dumps({
            "id": "x", "object": "chat.completion", "model": "m",
            "choices": [],
            "usage": {"prompt_tokens": 5, "completion_tokens": 2,
                      "total_tokens": 7},
        }).encode()
        resp = (b"HTTP/1.1 200 OK\r\ncontent-type: application/json\r\n"
                b"content-length: %d\r\n\r\n" % len(body)) + body
        mock = aigw_fast.FastMock()
        up_port = mock.start("127.0.0.1", resp.decode("latin1"))
        srv = aigw_fast.FastServer()
        srv.add_route("r", "", 1, True, True,
                      [{"name": "b", "host": "127.0.0.1", "port": up_port}])
        srv.add_rate_rule("node", 50, 3600.0, "llm_total_token")
        gw_port = srv.start("127.0.0.1", 0)
        payload = _json.dumps({"model": "m", "messages": [
            {"role": "user", "content": "hi"}]}).encode()
        # each shard serves 1 request -> 7 tokens; 56 > 50 node-wide
        r = aigw_fast.run_load("127.0.0.1", gw_port,
                               "/v1/chat/completions", payload, 1, 1)
        assert r["completed"] == 1, r
        sync = StateSync(FastLimiterBridge(srv, ["node"]))
        sync.tick_sync()
        sync.tick_sync()
        # every shard must now deny locally (global spend 56 >= 50)
        r = aigw_fast.run_load("127.0.0.1", gw_port,
                               "/v1/chat/completions", payload, 1, 1)
        assert r["completed"] == 0 and r["errors"] == 1, r
        st = srv.stats()
        assert st["local_429"] == 1, st
        dist.barrier()
        dist.destroy_process_group()
        srv.stop()
        mock.stop()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def test_fastfront_counter_sync_world8():
    """8 native shards: real HTTP requests charge the C++ counters, one
    fused gloo all-reduce folds the node-wide spend, every shard denies."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29783
    procs = [ctx.Process(target=_fastfront_worker, args=(r, 8, port, q))
             for r in range(8)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(8)]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            pytest.fail("worker hung")
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"
