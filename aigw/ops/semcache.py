"""Semantic response cache: MFMA embedding + HBM-resident cosine index.

The embedding pipeline (mean-pooled token embeddings → bf16 MFMA projection
→ L2 normalize) runs entirely on the GPU from the tokenizer's on-device
output. The index is a ring buffer of bf16 rows sized for the 288 GB HBM3E
budget; lookup is the fused MFMA similarity + argmax kernel (one pass over
the index, no materialized score matrix).
"""

from __future__ import annotations

from typing import Optional

import torch

from aigw.ops import load_hip_module


class SemanticCache:
    def __init__(
        self,
        vocab_size: int,
        dim: int = 384,
        capacity: int = 65536,
        threshold: float = 0.92,
        device: str = "cuda",
        seed: int = 7,
        index_dtype: str = "bf16",  # "bf16" | "fp8" (OCP e4m3: half the
        # HBM per row -> 2x rows in the 288 GB budget, 2x lookup bandwidth;
        # ~1% cosine error from 3-bit mantissas averaged over 384 dims)
        _hip=None,
    ):
        self.hip = _hip if _hip is not None else load_hip_module()
        if self.hip is None:
            raise RuntimeError("aigw_hip extension unavailable")
        self.device = torch.device(device)
        self.dim = dim
        self.capacity = capacity
        self.threshold = threshold
        g = torch.Generator().manual_seed(seed)
        # random-init bge-small-shaped weights (no network for checkpoints;
        # BASELINE.json: "synthetic OpenAI-schema payloads with random-init
        # embedding weights")
        self.emb = (
            torch.randn(vocab_size, dim, generator=g).to(torch.bfloat16).to(self.device)
        )
        self.proj = (
            (torch.randn(dim, dim, generator=g) / dim**0.5)
            .to(torch.bfloat16)
            .to(self.device)
        )
        self.index_dtype = (
            torch.float8_e4m3fn if index_dtype == "fp8" else torch.bfloat16
        )
        self.index = torch.zeros(capacity, dim, dtype=self.index_dtype, device=self.device)
        self.size = 0
        self.head = 0
        self.values: dict[int, bytes] = {}  # slot -> cached response body
        # inserts not yet all-gathered to the other shards (aigw.parallel)
        self.pending: list[tuple[torch.Tensor, bytes]] = []
        self.record_pending = False

    def embed(self, out_ids: torch.Tensor, req_off: torch.Tensor) -> torch.Tensor:
        """(B, dim) bf16 L2-normalized query vectors from tokenizer output."""
        pooled = self.hip.meanpool(out_ids, req_off, self.emb)  # fp32 (B, dim)
        pooled_bf = pooled.to(torch.bfloat16)
        proj = self.hip.gemm_bf16_nt(pooled_bf, self.proj, None, False)  # fp32
        return self.hip.l2norm_rows(proj)

    def lookup(self, queries: torch.Tensor) -> list[Optional[tuple[int, float]]]:
        """Per query: (slot, score) of the best index row, or None."""
        if self.size == 0:
            return [None] * queries.shape[0]
        view = self.index[: self.size]
        hi, idx = [], []
        for q0 in range(0, queries.shape[0], 256):  # kernel cap: 256 q/call
            qs = queries[q0 : q0 + 256].to(self.index_dtype).contiguous()
            h, i = self.hip.cache_topk(view, qs)
            hi.extend(h.cpu().tolist())
            idx.extend(i.cpu().tolist())
        out = []
        for h, i in zip(hi, idx):
            score = _unorder(h)
            out.append((i, score) if score >= self.threshold else None)
        return out

    def get(self, slot: int) -> Optional[bytes]:
        return self.values.get(slot)

    def insert(self, query_vec: torch.Tensor, response: bytes) -> int:
        slot = self.head
        self.index[slot] = query_vec.to(self.index_dtype)
        self.values[slot] = response
        self.head = (self.head + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)
        if self.record_pending:
            self.pending.append((query_vec.detach(), response))
        return slot

    def insert_remote(self, query_vec: torch.Tensor, response: bytes) -> int:
        """Insert a row replicated from another shard (never re-pended)."""
        slot = self.head
        self.index[slot] = query_vec.to(self.index.dtype).to(self.device)
        self.values[slot] = response
        self.head = (self.head + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)
        return slot

    def drain_pending(self, max_n: int = 32):
        """Take up to max_n locally inserted rows for the next all-gather
        tick: returns (vecs [k, dim] on device, [response bytes])."""
        batch, self.pending = self.pending[:max_n], self.pending[max_n:]
        if not batch:
            return torch.zeros(0, self.dim, dtype=torch.bfloat16, device=self.device), []
        vecs = torch.stack([v for v, _ in batch]).to(torch.bfloat16)
        return vecs, [r for _, r in batch]


def _unorder(u: int) -> float:
    """Reverse of pack_score's orderable-float transform (hi 32 bits)."""
    import struct

    u &= 0xFFFFFFFF
    bits = (u ^ 0x80000000) if u >= 0x80000000 else (~u & 0xFFFFFFFF)
    return struct.unpack("<f", struct.pack("<I", bits))[0]
