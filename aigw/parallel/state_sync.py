"""Cross-shard state synchronization over RCCL/xGMI.

Replaces the reference's Redis-backed global rate-limit state (the only
cross-replica shared state in Envoy AI Gateway — SURVEY.md §5.8) with
collectives among the 8 per-GPU gateway shards:

- token-budget counters: one fused int64 all-reduce(sum) per tick carrying
  every bucket's spend delta. The payload is tiny, so the collective is
  latency-bound — exactly why it is a single fused buffer on a periodic
  tick, OFF the request critical path, rather than a per-request op. The
  window of over-admission this allows mirrors the reference's own
  eventual consistency (rate-limit cost applies at stream completion).
- semantic-cache index rows: an all-gather of newly inserted embedding
  rows per tick so every shard can serve every shard's cache hits.

Bucket identity across shards uses a stable hash (md5) of
``rule|descriptor`` into a fixed slot table so every rank all-reduces the
same tensor shape regardless of which descriptors it has seen locally.

Backend: "nccl" (RCCL over xGMI) when shards run on GPUs; "gloo" for
CPU-only tests (tests/test_state_sync.py runs world_size=2 here).
"""

from __future__ import annotations

import asyncio
import logging
from typing import Optional

import torch
import torch.distributed as dist

from aigw.ratelimit.limiter import SLOTS, bucket_slot

logger = logging.getLogger("aigw.parallel")


class StateSync:
    def __init__(
        self,
        limiter,
        cache=None,
        *,
        interval_s: float = 0.25,
        device: Optional[torch.device] = None,
        group=None,
    ):
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed must be initialized for StateSync")
        self.limiter = limiter
        self.cache = cache
        self.interval_s = interval_s
        self.group = group
        backend = dist.get_backend(group)
        if device is None:
            device = torch.device("cuda") if backend == "nccl" else torch.device("cpu")
        self.device = device
        self._buf = torch.zeros(SLOTS, dtype=torch.int64, device=device)
        self._task: Optional[asyncio.Task] = None
        self.ticks = 0

    # ---- one synchronization round ------------------------------------------

    def tick_sync(self) -> None:
        """One fused all-reduce round (blocking; called from the async loop
        via a thread or directly in tests)."""
        self._buf.zero_()
        deltas = self.limiter.collect_deltas()
        own: dict[int, int] = {}
        slot_keys: dict[int, list] = {}
        for key, d in deltas.items():
            s = bucket_slot(key[0], key[1])
            own[s] = own.get(s, 0) + d
            slot_keys.setdefault(s, []).append(key)
        if own:
            idx = torch.tensor(list(own.keys()), dtype=torch.long)
            val = torch.tensor(list(own.values()), dtype=torch.int64)
            self._buf[idx.to(self.device)] = val.to(self.device)
        dist.all_reduce(self._buf, op=dist.ReduceOp.SUM, group=self.group)
        if self.cache is not None:
            self._sync_cache()
        nonzero = torch.nonzero(self._buf, as_tuple=False).flatten()
        if nonzero.numel():
            host = self._buf[nonzero].cpu().tolist()
            slots = nonzero.cpu().tolist()
            for s, total in zip(slots, host):
                mine = own.get(s, 0)
                remote = total - mine
                if remote <= 0:
                    continue
                keys = slot_keys.get(s)
                if keys:
                    # attribute the remote spend to the first local bucket in
                    # this slot (collisions are rare at 4096 slots)
                    self.limiter.apply_remote(keys[0], total, mine)
                else:
                    self.limiter.apply_remote_slot(s, remote)
        self.ticks += 1

    # ---- semantic-cache index replication ------------------------------------

    MAX_CACHE_ROWS_PER_TICK = 32

    def _sync_cache(self) -> None:
        """All-gather newly inserted cache rows (embedding vector + cached
        response body) so every shard serves every shard's hits
        (RCCL all-gather of semantic-cache index updates).
        Collective sequence is fixed per tick, so shards stay aligned:
        sizes all-gather, then padded vec + payload all-gathers."""
        cache = self.cache
        rank = dist.get_rank(self.group)
        world = dist.get_world_size(self.group)
        vecs, values = cache.drain_pending(self.MAX_CACHE_ROWS_PER_TICK)
        payload = bytearray()
        for v in values:
            payload.extend(len(v).to_bytes(4, "little"))
            payload.extend(v)
        sizes = torch.tensor([vecs.shape[0], len(payload)], dtype=torch.int64,
                             device=self.device)
        all_sizes = [torch.zeros_like(sizes) for _ in range(world)]
        dist.all_gather(all_sizes, sizes, group=self.group)
        max_k = max(int(s[0]) for s in all_sizes)
        max_p = max(int(s[1]) for s in all_sizes)
        if max_k == 0:
            return
        dim = vecs.shape[1] if vecs.numel() else cache.dim
        vpad = torch.zeros(max_k, dim, dtype=torch.bfloat16, device=self.device)
        if vecs.shape[0]:
            vpad[: vecs.shape[0]] = vecs
        ppad = torch.zeros(max(max_p, 1), dtype=torch.uint8, device=self.device)
        if payload:
            ppad[: len(payload)] = torch.frombuffer(bytes(payload), dtype=torch.uint8).to(
                self.device
            )
        all_vecs = [torch.zeros_like(vpad) for _ in range(world)]
        all_payload = [torch.zeros_like(ppad) for _ in range(world)]
        dist.all_gather(all_vecs, vpad, group=self.group)
        dist.all_gather(all_payload, ppad, group=self.group)
        for r in range(world):
            k = int(all_sizes[r][0])
            if r == rank or k == 0:
                continue
            blob = bytes(all_payload[r][: int(all_sizes[r][1])].cpu().numpy().tobytes())
            off = 0
            rows = all_vecs[r]
            for i in range(k):
                ln = int.from_bytes(blob[off : off + 4], "little")
                off += 4
                value = blob[off : off + ln]
                off += ln
                cache.insert_remote(rows[i], value)

    # ---- background loop -----------------------------------------------------

    async def start(self) -> None:
        loop = asyncio.get_running_loop()

        async def run():
            while True:
                await asyncio.sleep(self.interval_s)
                try:
                    await loop.run_in_executor(None, self.tick_sync)
                except asyncio.CancelledError:
                    raise
                except Exception:
                    logger.exception("state sync tick failed")

        self._task = asyncio.create_task(run(), name="aigw-state-sync")

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None


class FastLimiterBridge:
    """Adapts the native fast front's per-rule rate counters
    (aigw_fast.FastServer rl_collect_deltas / rl_apply_remote) to the
    limiter interface StateSync drives, so one code path covers both
    fronts: the C++ server keeps its fixed-window atomics, this bridge
    maps them onto the cross-shard slot table (rule name, "" descriptor)
    and folds remote spend back in after each fused all-reduce."""

    def __init__(self, fast_server, rule_names: list):
        self.fast = fast_server
        self.rule_names = list(rule_names)
        self._slot_to_idx = {
            bucket_slot(name, ""): i for i, name in enumerate(self.rule_names)
        }

    def collect_deltas(self) -> dict:
        deltas = self.fast.rl_collect_deltas()
        return {
            (self.rule_names[i], ""): d
            for i, d in enumerate(deltas)
            if d
        }

    def _apply(self, idx: int, remote: int) -> None:
        vec = [0] * len(self.rule_names)
        vec[idx] = remote
        self.fast.rl_apply_remote(vec)

    def apply_remote(self, key, total: int, mine: int) -> None:
        remote = total - mine
        if remote > 0 and key[0] in self.rule_names:
            self._apply(self.rule_names.index(key[0]), remote)

    def apply_remote_slot(self, slot: int, remote: int) -> None:
        idx = self._slot_to_idx.get(slot)
        if idx is not None and remote > 0:
            self._apply(idx, remote)
