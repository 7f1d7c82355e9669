"""Usage-based (token) rate limiting.

The reference delegates enforcement to Envoy's ratelimit service backed by
Redis, fed with request costs via dynamic metadata
(extproc/processor_impl.go:917-1003, internal/ratelimit/). This gateway
enforces in-process instead: fixed-window token buckets charged with the
SAME cost semantics (cost applied at stream completion, i.e. the request
that crosses the boundary succeeds and subsequent ones are denied —
eventually-consistent, exactly like the reference's Redis flow).

Cross-shard consistency on an 8×GPU node is maintained by
aigw.parallel.StateSync: each shard accumulates a local delta per bucket
and a periodic RCCL all-reduce folds deltas into a globally consistent
spent-count (SURVEY.md §5.8 — tiny payloads, one fused buffer per tick,
off the request critical path).
"""

from __future__ import annotations

import hashlib
import time
from dataclasses import dataclass

from aigw.filterapi.config import RateLimitRule

# fixed slot table for cross-shard bucket identity (see aigw.parallel)
SLOTS = 4096


def bucket_slot(rule: str, descriptor: str) -> int:
    h = hashlib.md5(f"{rule}|{descriptor}".encode()).digest()
    return int.from_bytes(h[:4], "little") % SLOTS


@dataclass
class RateLimitDecision:
    allowed: bool
    rule: str = ""
    remaining: int = 0
    retry_after_s: float = 0.0


class _Bucket:
    __slots__ = ("window_start", "local_spent", "remote_spent", "pending_delta")

    def __init__(self):
        self.window_start = 0.0
        self.local_spent = 0  # tokens charged by THIS shard this window
        self.remote_spent = 0  # tokens from other shards (via all-reduce)
        self.pending_delta = 0  # local spend not yet folded into the sync

    @property
    def total(self) -> int:
        return self.local_spent + self.remote_spent


class RateLimiter:
    """Fixed-window token buckets keyed by (rule, descriptor)."""

    def __init__(self, rules: list[RateLimitRule], clock=time.monotonic):
        self.rules = {r.name: r for r in rules}
        self._buckets: dict[tuple[str, str], _Bucket] = {}
        self._slot_remote: dict[int, int] = {}  # remote spend with no local bucket yet
        self._clock = clock

    def _bucket(self, rule: RateLimitRule, descriptor: str) -> _Bucket:
        key = (rule.name, descriptor)
        b = self._buckets.get(key)
        now = self._clock()
        if b is None:
            b = _Bucket()
            b.window_start = now
            # claim any remote spend that arrived before we saw this bucket
            b.remote_spent = self._slot_remote.pop(bucket_slot(*key), 0)
            self._buckets[key] = b
        elif now - b.window_start >= rule.window_s:
            b.window_start = now
            b.local_spent = 0
            b.remote_spent = 0
        return b

    @staticmethod
    def _descriptor(rule: RateLimitRule, headers: dict[str, str]) -> str:
        if not rule.key_headers:
            return ""
        return "|".join(headers.get(h.lower(), "") for h in rule.key_headers)

    def check(self, headers: dict[str, str]) -> RateLimitDecision:
        """Pre-admission check: deny when a bucket is already exhausted."""
        for rule in self.rules.values():
            b = self._bucket(rule, self._descriptor(rule, headers))
            if b.total >= rule.limit:
                retry = rule.window_s - (self._clock() - b.window_start)
                return RateLimitDecision(
                    allowed=False, rule=rule.name, remaining=0, retry_after_s=max(retry, 0.0)
                )
        return RateLimitDecision(allowed=True)

    def charge(self, headers: dict[str, str], costs: dict[str, int]) -> None:
        """Apply per-request costs at stream completion."""
        for rule in self.rules.values():
            cost = costs.get(rule.metadata_key)
            if not cost:
                continue
            b = self._bucket(rule, self._descriptor(rule, headers))
            b.local_spent += cost
            b.pending_delta += cost

    # --- cross-shard sync hooks (driven by aigw.parallel.StateSync) ---------

    def collect_deltas(self) -> dict[tuple[str, str], int]:
        """Drain pending local deltas for the next all-reduce tick."""
        out = {}
        for key, b in self._buckets.items():
            if b.pending_delta:
                out[key] = b.pending_delta
                b.pending_delta = 0
        return out

    def apply_remote(self, key: tuple[str, str], global_delta: int, own_delta: int) -> None:
        """Fold an all-reduced global delta into the bucket: the global sum
        includes our own contribution, which is already in local_spent."""
        rule = self.rules.get(key[0])
        if rule is None:
            return
        b = self._bucket(rule, key[1])
        b.remote_spent += global_delta - own_delta

    def apply_remote_slot(self, slot: int, amount: int) -> None:
        """Remote spend for a bucket this shard has not seen locally; held
        per-slot and claimed when the bucket first appears."""
        for key, b in self._buckets.items():
            if bucket_slot(*key) == slot:
                b.remote_spent += amount
                return
        self._slot_remote[slot] = self._slot_remote.get(slot, 0) + amount
