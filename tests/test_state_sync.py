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


def test_slot_remote_claimed_on_first_local_sight():
    rules = [RateLimitRule(name="r", metadata_key="llm_total_token", limit=100, window_s=60,
                           key_headers=["x-user-id"])]
    limiter = RateLimiter(rules)
    from aigw.ratelimit.limiter import bucket_slot

    slot = bucket_slot("r", "alice")
    limiter.apply_remote_slot(slot, 150)  # remote shard spent alice's budget
    d = limiter.check({"x-user-id": "alice"})
    assert not d.allowed
