"""Synthetic byte-level BPE tokenizer: vocab generation + CPU reference.

There is no network access, so the tokenizer is a deterministic synthetic
Llama-3-scale byte-level BPE (256 byte tokens + N merges). The merge table
and segmentation rule are the single source of truth shared by:

- this CPU reference implementation (the numerics oracle), and
- the CDNA4 kernels in csrc/aigw_kernels.hip (seg_flags_kernel /
  bpe_encode_kernel), which must produce IDENTICAL ids and counts.

Segmentation rule (pre-tokenizer): a new segment starts at position i iff
``i == 0``, ``b[i] == ' '``, or ``class(b[i]) != class(b[i-1]) and
b[i-1] != ' '`` — a branch-free, one-pass approximation of the GPT-style
"space attaches to the following word" regex, chosen for wave-parallel
evaluation. Request boundaries always force a segment start. Segments are
BPE-merged in independent 64-byte chunks (the wave width), identically on
both implementations.

BPE semantics (HF tokenizers-compatible): repeatedly merge ALL occurrences
of the lowest-rank adjacent pair present, resolving overlapping occurrences
leftmost-first, until no pair is in the merge table.
"""

from __future__ import annotations

import numpy as np

VOCAB_BYTES = 256
CHUNK = 64  # wave width: segments are merged in independent 64-byte chunks

_SPACE, _WS, _DIGIT, _LETTER, _OTHER = 0, 1, 2, 3, 4


def byte_class(b: int) -> int:
    if b == 0x20:
        return _SPACE
    if b in (0x09, 0x0A, 0x0D, 0x0B, 0x0C):
        return _WS
    if 0x30 <= b <= 0x39:
        return _DIGIT
    if 0x41 <= b <= 0x5A or 0x61 <= b <= 0x7A or b >= 0x80:
        return _LETTER
    return _OTHER


def segment_starts(data: bytes, req_offsets: list[int]) -> list[int]:
    """Positions where segments start (mirrors seg_flags_kernel +
    seg_force_starts_kernel)."""
    n = len(data)
    flags = bytearray(n)
    for i in range(n):
        if i == 0:
            flags[i] = 1
        else:
            b, p = data[i], data[i - 1]
            flags[i] = 1 if (b == 0x20 or (byte_class(b) != byte_class(p) and p != 0x20)) else 0
    for off in req_offsets:
        if off < n:
            flags[off] = 1
    return [i for i in range(n) if flags[i]]


def make_merges(n_merges: int = 32768, seed: int = 1355) -> np.ndarray:
    """Deterministic synthetic merge table: (n_merges, 2) int32 of
    (left_id, right_id); rank = row index, new_id = 256 + rank.

    Generation is biased toward ASCII text: early merges pair common
    letters/space-letter so typical English-ish payloads actually compress
    (≈3-4 bytes/token), which is what makes the GPU merge loop do realistic
    work in the benchmark.
    """
    rng = np.random.default_rng(seed)
    letters = np.frombuffer(b"etaoinshrdlucmfwypvbgkjqxz", dtype=np.uint8).astype(np.int64)
    merges: list[tuple[int, int]] = []
    seen: set[tuple[int, int]] = set()
    next_id = VOCAB_BYTES

    def add(a: int, b: int) -> int:
        nonlocal next_id
        key = (int(a), int(b))
        if key in seen:
            return -1
        seen.add(key)
        merges.append(key)
        nid = next_id
        next_id += 1
        return nid

    # seed merges: letter pairs and space+letter
    for a in letters[:16]:
        for b in letters[:16]:
            add(int(a), int(b))
    for b in letters:
        add(0x20, int(b))
    # grow compounds: left operand biased toward recently created tokens
    while len(merges) < n_merges:
        hi = next_id
        if rng.random() < 0.7 and hi > VOCAB_BYTES + 64:
            a = int(rng.integers(VOCAB_BYTES, hi))
        else:
            a = int(letters[rng.integers(0, len(letters))])
        if rng.random() < 0.5:
            b = int(letters[rng.integers(0, len(letters))])
        else:
            b = int(rng.integers(VOCAB_BYTES, hi)) if hi > VOCAB_BYTES else 0x20
        add(a, b)
    return np.asarray(merges[:n_merges], dtype=np.int64)


def build_hash_table(merges: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
    """Open-addressing table (keys int64, ranks int32); size = next pow2 with
    load factor <= 0.5. Mirrors the probe sequence in bpe_encode_kernel."""
    n = len(merges)
    size = 1
    while size < n * 2:
        size *= 2
    keys = np.full(size, -1, dtype=np.int64)
    ranks = np.zeros(size, dtype=np.int32)
    mask = size - 1
    pair_keys = (merges[:, 0].astype(np.uint64) << np.uint64(32)) | merges[:, 1].astype(
        np.uint64
    )
    for rank, key in enumerate(pair_keys):
        h = (int(key) * 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
        idx = (h >> 40) & mask
        while keys[idx] != -1:
            idx = (idx + 1) & mask
        keys[idx] = np.int64(np.uint64(key))
        ranks[idx] = rank
    return keys, ranks


class BPERef:
    """CPU reference encoder over a merge table."""

    def __init__(self, merges: np.ndarray):
        self.rank_of: dict[tuple[int, int], int] = {
            (int(a), int(b)): r for r, (a, b) in enumerate(merges)
        }

    def encode_chunk(self, chunk: bytes) -> list[int]:
        toks = list(chunk)
        while True:
            best_rank = None
            for i in range(len(toks) - 1):
                r = self.rank_of.get((toks[i], toks[i + 1]))
                if r is not None and (best_rank is None or r < best_rank):
                    best_rank = r
            if best_rank is None:
                return toks
            # merge ALL occurrences of the best pair, leftmost-first
            new_id = VOCAB_BYTES + best_rank
            out = []
            i = 0
            while i < len(toks):
                if (
                    i + 1 < len(toks)
                    and self.rank_of.get((toks[i], toks[i + 1])) == best_rank
                ):
                    out.append(new_id)
                    i += 2
                else:
                    out.append(toks[i])
                    i += 1
            toks = out

    def encode_segment(self, seg: bytes) -> list[int]:
        ids: list[int] = []
        for c in range(0, len(seg), CHUNK):
            ids.extend(self.encode_chunk(seg[c : c + CHUNK]))
        return ids

    def encode_batch(self, texts: list[bytes]) -> list[list[int]]:
        data = b"".join(texts)
        req_offsets = []
        off = 0
        for t in texts:
            req_offsets.append(off)
            off += len(t)
        starts = segment_starts(data, req_offsets)
        out: list[list[int]] = [[] for _ in texts]
        req = 0
        for si, s in enumerate(starts):
            e = starts[si + 1] if si + 1 < len(starts) else len(data)
            while req + 1 < len(texts) and s >= req_offsets[req + 1]:
                req += 1
            out[req].extend(self.encode_segment(data[s:e]))
        return out
