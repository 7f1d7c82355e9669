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
