"""Kernel microbenchmarks for the GPU tier (run on MI355X, usually under
rocprofv3 --stats). Prints one JSON line per kernel with achieved rates."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--payload-tokens", type=int, default=4096)
    ap.add_argument("--index-rows", type=int, default=1_000_000)
    args = ap.parse_args()

    from aigw.ops.semcache import SemanticCache
    from aigw.ops.tokenizer import GPUTokenizer

    tok = GPUTokenizer(n_merges=32768, device="cuda")
    words = "the quick brown fox jumps over the lazy dog benchmark".split()
    text = " ".join(words[i % len(words)] for i in range(args.payload_tokens)).encode()
    texts = [text] * args.batch
    arr, offs = tok.pack(texts)
    bytes_t = torch.from_numpy(arr).cuda()
    off_t = torch.from_numpy(offs).cuda()
    total_bytes = len(arr)

    def run_tok():
        tok.hip.bpe_encode(bytes_t, off_t, tok.htab_keys, tok.htab_ranks)

    dt = timeit(run_tok)
    counts = tok.hip.bpe_encode(bytes_t, off_t, tok.htab_keys, tok.htab_ranks)[1]
    total_tokens = int(counts.sum().item())
    print(json.dumps({
        "kernel": "bpe_encode(batch)",
        "batch": args.batch,
        "bytes": total_bytes,
        "tokens": total_tokens,
        "ms": round(dt * 1e3, 3),
        "MB_per_s": round(total_bytes / dt / 1e6, 1),
        "Mtok_per_s": round(total_tokens / dt / 1e6, 2),
    }))

    cache = SemanticCache(tok.vocab_size, capacity=1024, device="cuda")
    _, _, state = tok.encode_batch(texts)

    def run_embed():
        cache.embed(state["out_ids"], state["req_off"])

    dt = timeit(run_embed)
    print(json.dumps({
        "kernel": "embed(meanpool+mfma_gemm+l2norm)",
        "batch": args.batch,
        "ms": round(dt * 1e3, 3),
        "req_per_s": round(args.batch / dt, 1),
    }))

    # standalone MFMA GEMM rate: square compute-bound shape (tiled path)
    m, n, k = 4096, 4096, 4096
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, k, device="cuda").to(torch.bfloat16)

    def run_gemm_sq():
        tok.hip.gemm_bf16_nt(a, bt, None, False)

    dt = timeit(run_gemm_sq)
    print(json.dumps({
        "kernel": "gemm_bf16_nt_tiled",
        "shape": [m, n, k],
        "ms": round(dt * 1e3, 3),
        "TFLOPs": round(2 * m * n * k / dt / 1e12, 2),
    }))

    # skinny embed-projection-like shape
    m, n, k = 4096, 4096, 384
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    bt = torch.randn(n, k, device="cuda").to(torch.bfloat16)

    def run_gemm():
        tok.hip.gemm_bf16_nt(a, bt, None, False)

    dt = timeit(run_gemm)
    print(json.dumps({
        "kernel": "gemm_bf16_nt",
        "shape": [m, n, k],
        "ms": round(dt * 1e3, 3),
        "TFLOPs": round(2 * m * n * k / dt / 1e12, 2),
    }))

    index = torch.randn(args.index_rows, 384, device="cuda").to(torch.bfloat16)
    q = torch.randn(args.batch, 384, device="cuda").to(torch.bfloat16)

    def run_topk():
        tok.hip.cache_topk(index, q)

    dt = timeit(run_topk)
    bytes_read = args.index_rows * 384 * 2
    print(json.dumps({
        "kernel": "cache_topk",
        "rows": args.index_rows,
        "queries": args.batch,
        "ms": round(dt * 1e3, 3),
        "TB_per_s_index_read": round(bytes_read / dt / 1e12, 3),
        "TFLOPs": round(2 * args.index_rows * args.batch * 384 / dt / 1e12, 2),
    }))

    index8 = index.to(torch.float8_e4m3fn)
    q8 = q.to(torch.float8_e4m3fn)

    def run_topk8():
        tok.hip.cache_topk(index8, q8)

    dt = timeit(run_topk8)
    print(json.dumps({
        "kernel": "cache_topk_fp8",
        "rows": args.index_rows,
        "queries": args.batch,
        "ms": round(dt * 1e3, 3),
        "TB_per_s_index_read": round(args.index_rows * 384 / dt / 1e12, 3),
        "TFLOPs": round(2 * args.index_rows * args.batch * 384 / dt / 1e12, 2),
    }))

    stats = torch.rand(8, 4, device="cuda")
    stats[:, 1] = 1000
    pred = torch.rand(args.batch, device="cuda") * 100

    def run_scorer():
        tok.hip.kv_score_assign(stats, pred, 1.0, 0.1, 0.05)

    dt = timeit(run_scorer)
    print(json.dumps({"kernel": "kv_score_assign", "batch": args.batch,
                      "us": round(dt * 1e6, 1)}))


if __name__ == "__main__":
    main()
