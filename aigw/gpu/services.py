"""Async facade over the GPU ops for the gateway hot path.

Requests never launch kernels individually: a micro-batching collector
coalesces concurrent requests (up to ``max_batch`` or ``window_ms``) into
one packed tokenizer launch — the LDS-staged "batched 4k-token chat
payloads" admission model from BASELINE.json — and, for cache-enabled
requests, one embedding GEMM + one fused index lookup. GPU work runs on a
dedicated single-thread executor so kernel syncs never stall the event
loop.
"""

from __future__ import annotations

import asyncio
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass
from typing import Optional

import torch


def extract_chat_text(body: dict) -> bytes:
    """Concatenated message text of a chat request (OpenAI or Anthropic
    shape) — the tokenizer input for admission accounting."""
    parts: list[str] = []
    system = body.get("system")
    if isinstance(system, str):
        parts.append(system)
    for msg in body.get("messages") or []:
        c = msg.get("content")
        if isinstance(c, str):
            parts.append(c)
        elif isinstance(c, list):
            for p in c:
                if isinstance(p, dict):
                    t = p.get("text") or ""
                    if t:
                        parts.append(t)
    return "\n".join(parts).encode("utf-8", "replace")[:262144] or b" "


@dataclass
class _Pending:
    """One queued work item: a BATCH of texts sharing a future (single
    requests are 1-element batches; the shard GPU service submits whole
    worker batches so the per-text future overhead disappears)."""

    texts: list
    want_vec: bool
    future: asyncio.Future = None  # type: ignore[assignment]


class GPUServices:
    def __init__(
        self,
        device: str = "cuda",
        *,
        n_merges: int = 32768,
        enable_cache: bool = False,
        cache_capacity: int = 65536,
        cache_threshold: float = 0.92,
        cache_index_dtype: str = "bf16",  # "bf16" | "fp8" (2x rows/GB, faster lookups)
        window_ms: float = 0.1,
        max_batch: int = 128,
    ):
        from aigw.ops.semcache import SemanticCache
        from aigw.ops.tokenizer import GPUTokenizer

        self.device = device
        self.tokenizer = GPUTokenizer(n_merges=n_merges, device=device)
        self.cache: Optional[SemanticCache] = (
            SemanticCache(
                self.tokenizer.vocab_size,
                capacity=cache_capacity,
                threshold=cache_threshold,
                device=device,
                index_dtype=cache_index_dtype,
            )
            if enable_cache
            else None
        )
        self.window_ms = window_ms
        self.max_batch = max_batch
        import os as _os

        # Measured NEGATIVE (profiles/r01 ab.jsonl): the sync-free
        # event-polling count path is ~20% slower at 12 workers — executor
        # threads may freely spin on this many-core node, while 0.2 ms
        # event polling adds per-batch latency. Kept behind an opt-in for
        # core-constrained deployments.
        self.async_counts = _os.environ.get("AIGW_GPU_ASYNC_PATH", "") == "1"
        self._pending: list[_Pending] = []
        self._pending_texts = 0
        self._flush_handle = None
        self._executor = ThreadPoolExecutor(max_workers=1, thread_name_prefix="aigw-gpu")
        self._lock = asyncio.Lock()

    @property
    def cache_enabled(self) -> bool:
        return self.cache is not None

    # ---- batching core -------------------------------------------------------

    async def _submit_batch(self, texts: list, want_vec: bool):
        loop = asyncio.get_running_loop()
        item = _Pending(texts=texts, want_vec=want_vec, future=loop.create_future())
        self._pending.append(item)
        self._pending_texts += len(texts)
        if self._pending_texts >= self.max_batch:
            if self._flush_handle:
                self._flush_handle.cancel()
                self._flush_handle = None
            asyncio.ensure_future(self._flush())
        elif self._flush_handle is None:
            self._flush_handle = loop.call_later(
                self.window_ms / 1000.0, lambda: asyncio.ensure_future(self._flush())
            )
        return await item.future

    async def _submit(self, text: bytes, want_vec: bool):
        out = await self._submit_batch([text], want_vec)
        return out[0]  # (count, vec, cached_value)

    async def _flush(self):
        self._flush_handle = None
        batch, self._pending = self._pending, []
        self._pending_texts = 0
        if not batch:
            return
        loop = asyncio.get_running_loop()
        if self.async_counts and not any(it.want_vec for it in batch):
            # count-only fast path: launch without ANY host sync and await
            # the completion event cooperatively — a blocking torch sync on
            # ROCm busy-spins a core per worker, which halved 12-worker
            # serving throughput (profiles/r01)
            try:
                counts_gpu, spans = await loop.run_in_executor(
                    self._executor, self._launch_counts, batch
                )
                ev = counts_gpu["ev"]
                while not ev.query():
                    await asyncio.sleep(0.0002)
                counts_host = counts_gpu["counts"].cpu().tolist()
            except Exception as e:  # pragma: no cover - defensive
                for it in batch:
                    if not it.future.done():
                        it.future.set_exception(e)
                return
            for it, (lo, hi) in zip(batch, spans):
                if not it.future.done():
                    it.future.set_result(
                        [(counts_host[i], None, None) for i in range(lo, hi)]
                    )
            return
        try:
            results = await loop.run_in_executor(self._executor, self._run_batch, batch)
        except Exception as e:  # pragma: no cover - defensive
            for it in batch:
                if not it.future.done():
                    it.future.set_exception(e)
            return
        for it, res in zip(batch, results):
            if not it.future.done():
                it.future.set_result(res)

    def _launch_counts(self, batch: list[_Pending]):
        """GPU worker thread: pack + H2D + kernel launches, NO sync."""
        texts = []
        spans = []
        for it in batch:
            spans.append((len(texts), len(texts) + len(it.texts)))
            texts.extend(it.texts)
        counts_gpu, _ids, _off, ev = self.tokenizer.encode_batch_async(texts)
        return {"counts": counts_gpu, "ev": ev}, spans

    def _run_batch(self, batch: list[_Pending]):
        """Executed on the GPU worker thread: one packed tokenizer launch
        over every text of every queued batch; for vector-wanting items,
        ONE batched embedding GEMM and ONE fused index lookup (a lookup per
        request would mean a single-query kernel launch each). Returns one
        result list per item of (count, vec, cached_value_or_None)."""
        texts = []
        spans = []
        for it in batch:
            spans.append((len(texts), len(texts) + len(it.texts)))
            texts.extend(it.texts)
        counts, _, state = self.tokenizer.encode_batch(texts)
        counts_host = counts.cpu().tolist()
        qvecs = None
        hits = None
        if any(it.want_vec for it in batch) and self.cache is not None:
            qvecs = self.cache.embed(state["out_ids"], state["req_off"])
            found = self.cache.lookup(qvecs)  # one fused kernel, all queries
            hits = [self.cache.get(h[0]) if h is not None else None for h in found]
        results = []
        for it, (lo, hi) in zip(batch, spans):
            if it.want_vec and qvecs is not None:
                results.append(
                    [(counts_host[i], qvecs[i], hits[i]) for i in range(lo, hi)]
                )
            else:
                results.append([(counts_host[i], None, None) for i in range(lo, hi)])
        return results

    # ---- public API ----------------------------------------------------------

    async def count_request_tokens(self, body: dict) -> int:
        count, _, _ = await self._submit(extract_chat_text(body), want_vec=False)
        return count

    async def count_text_tokens(self, text: bytes) -> int:
        count, _, _ = await self._submit(text or b" ", want_vec=False)
        return count

    async def count_texts_batch(self, texts: list[bytes]) -> list[int]:
        """Batch entry used by the shard GPU service: one future per worker
        batch; texts from many workers coalesce into one kernel launch."""
        results = await self._submit_batch([t or b" " for t in texts], want_vec=False)
        return [r[0] for r in results]

    async def cache_lookup_text(self, text: bytes):
        """Like cache_lookup but takes pre-extracted text (from the C++
        scanner) instead of a parsed body. Lookup is part of the batched
        flush — no per-request kernel launches."""
        _, vec, hit = await self._submit(text or b" ", want_vec=True)
        return hit, vec

    async def lookup_texts_batch(self, texts: list[bytes]):
        """Batch entry for the shard GPU service: one future per worker
        batch; returns [(hit_or_None, vec)] aligned with texts."""
        results = await self._submit_batch([t or b" " for t in texts], want_vec=True)
        return [(hit, vec) for _c, vec, hit in results]

    async def tokenize(self, text) -> list[int]:
        if isinstance(text, str):
            text = text.encode("utf-8", "replace")
        loop = asyncio.get_running_loop()

        def run():
            _, ids, _ = self.tokenizer.encode_batch([text or b" "], return_ids=True)
            return ids[0]

        return await loop.run_in_executor(self._executor, run)

    async def cache_lookup(self, body: dict):
        """Returns (cached_response_bytes | None, query_vec)."""
        _, vec, hit = await self._submit(extract_chat_text(body), want_vec=True)
        return hit, vec

    async def cache_insert(self, vec, response: bytes) -> None:
        if vec is None or self.cache is None:
            return
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(self._executor, self.cache.insert, vec, response)

    async def pick_endpoint(self, stats_rows: list[list[float]], predicted: float) -> int:
        """One KV-occupancy scoring call for a single request: returns the
        index of the replica to try first (InferencePool EPP analogue)."""
        loop = asyncio.get_running_loop()

        def run():
            stats = torch.tensor(stats_rows, dtype=torch.float32, device=self.device)
            pred = torch.tensor([predicted], dtype=torch.float32, device=self.device)
            hip = self.tokenizer.hip
            return int(hip.kv_score_assign(stats, pred, 1.0, 0.1, 0.05).cpu()[0])

        return await loop.run_in_executor(self._executor, run)

    async def assign_replicas(self, stats: torch.Tensor, predicted: torch.Tensor):
        loop = asyncio.get_running_loop()
        hip = self.tokenizer.hip

        def run():
            return hip.kv_score_assign(stats, predicted, 1.0, 0.1, 0.05).cpu().tolist()

        return await loop.run_in_executor(self._executor, run)

    def close(self):
        self._executor.shutdown(wait=False)
