"""CPU BPE oracle self-tests (the numerics reference the GPU kernels are
compared against in test_gpu_kernels.py)."""

import numpy as np

from aigw.ops.bpe_ref import BPERef, build_hash_table, make_merges, segment_starts


def test_segmentation_rule():
    data = b"hello world  42x!"
    starts = segment_starts(data, [0])
    # h e l l o | ' world' starts at ' ' | second ' ' | ' 42'? -> ' ' starts,
    # then '42' attaches to the space; 'x' letter after digit splits; '!' splits
    segs = [
        data[s : (starts[i + 1] if i + 1 < len(starts) else len(data))]
        for i, s in enumerate(starts)
    ]
    assert segs == [b"hello", b" world", b" ", b" 42", b"x", b"!"]


def test_request_boundary_forces_start():
    a, b = b"abc", b"def"
    starts = segment_starts(a + b, [0, 3])
    assert 3 in starts  # "abcdef" would be one segment without the boundary


def test_merges_deterministic_and_unique():
    m1 = make_merges(2048, seed=1)
    m2 = make_merges(2048, seed=1)
    assert (m1 == m2).all()
    pairs = {(int(a), int(b)) for a, b in m1}
    assert len(pairs) == len(m1)


def test_hash_table_lookup():
    merges = make_merges(4096)
    keys, ranks = build_hash_table(merges)
    mask = len(keys) - 1
    for rank in (0, 17, 4095):
        a, b = merges[rank]
        key = (int(a) << 32) | int(b)
        h = (key * 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
        idx = (h >> 40) & mask
        while keys[idx] != key:
            assert keys[idx] != -1
            idx = (idx + 1) & mask
        assert ranks[idx] == rank


def test_encode_lowest_rank_first():
    # hand-built table: rank0=(h,e), rank1=(he,l), rank2=(l,l)
    merges = np.array([[ord("h"), ord("e")], [256, ord("l")], [ord("l"), ord("l")]],
                      dtype=np.int64)
    ref = BPERef(merges)
    # "hell": (h,e)->256 first (rank 0), then (256,l)->257, leaving 257,'l'
    assert ref.encode_chunk(b"hell") == [257, ord("l")]


def test_overlapping_merges_leftmost():
    merges = np.array([[ord("a"), ord("a")]], dtype=np.int64)
    ref = BPERef(merges)
    assert ref.encode_chunk(b"aaaa") == [256, 256]
    assert ref.encode_chunk(b"aaa") == [256, ord("a")]


def test_batch_roundtrip_counts():
    merges = make_merges(8192)
    ref = BPERef(merges)
    texts = [b"The quick brown fox jumps over the lazy dog. " * 3, b"hi", b" " * 70]
    out = ref.encode_batch(texts)
    assert len(out) == 3
    assert all(len(ids) > 0 for ids in out)
    # compression happens on English-like text
    assert len(out[0]) < len(texts[0])
    # all ids valid
    for ids in out:
        assert all(0 <= t < 256 + len(merges) for t in ids)
