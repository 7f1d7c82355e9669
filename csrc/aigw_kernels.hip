// aigw MI355X (gfx950 / CDNA4) kernels.
//
// GPU tier of the gateway (BASELINE.json north star; no counterpart code in
// the reference, which does this work CPU-side or not at all —
// SURVEY.md §2.4):
//
//   1. Byte-level BPE tokenizer (segmenter + wave-per-segment merge loop)
//      — replaces the reference's provider-usage-JSON token accounting
//      (translator/openai_openai.go:185-223) and /tokenize passthrough
//      (translator/tokenize.go:24-81) with gateway-local counting.
//   2. Mean-pool embedding + MFMA bf16 projection GEMM — semantic response
//      cache (the reference's cache is provider-side passthrough).
//   3. Fused MFMA similarity + argmax over the HBM-resident cache index.
//   4. KV-occupancy endpoint scorer — replaces the external EPP service
//      (extensionserver/inferencepool.go:39-54).
//
// CDNA4 specifics used (per the MI355X HIP guide): 64-wide wavefronts
// (64-bit ballot masks), __builtin_amdgcn_mfma_f32_16x16x32_bf16 with the
// C/D mapping col=lane&15 / row=(lane>>4)*4+reg, LDS staging, short8
// vectorized bf16 loads, grid-stride loops sized for 256 CUs.

#include <hip/hip_runtime.h>
#include <algorithm>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>

#define AIGW_CHECK(cond, msg) TORCH_CHECK(cond, msg)

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(4))) short short4v;

static inline hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// 1. BPE tokenizer
// ---------------------------------------------------------------------------

#include "bpe_kernels.cuh"

// ---------------------------------------------------------------------------
// 2. Embedding: mean-pool token embeddings per request (memory-bound)
// ---------------------------------------------------------------------------

// Position-parallel accumulation: grid (n_req, P) — a single block walking
// a 16 KiB request sequentially is latency-bound (~150 ns per token row);
// P blocks stride the id array and atomicAdd partial sums (1 atomic per
// column per block, contention 1/P). A tiny second kernel divides by the
// token count.
__global__ void meanpool_accum_kernel(const int32_t* __restrict__ ids,
                                      const int64_t* __restrict__ req_off, int n_req,
                                      int n_bytes, const bf16* __restrict__ emb,
                                      int dim, int P,
                                      float* __restrict__ out /* n_req x dim */,
                                      int32_t* __restrict__ cnt /* n_req */) {
  int r = blockIdx.x;
  int p = blockIdx.y;
  if (r >= n_req) return;
  int col = threadIdx.x;
  if (col >= dim) return;
  long long s = req_off[r];
  long long e = (r + 1 < n_req) ? req_off[r + 1] : n_bytes;
  float acc = 0.f;
  int c = 0;
  for (long long i = s + p; i < e; i += P) {
    int tok = ids[i];
    if (tok < 0) continue;
    acc += __bfloat162float(emb[(long long)tok * dim + col]);
    ++c;
  }
  if (c) {
    atomicAdd(&out[(long long)r * dim + col], acc);
    if (col == 0) atomicAdd(&cnt[r], c);
  }
}

__global__ void meanpool_div_kernel(float* __restrict__ out,
                                    const int32_t* __restrict__ cnt, int n_req,
                                    int dim) {
  int r = blockIdx.x;
  int col = threadIdx.x;
  if (r >= n_req || col >= dim) return;
  int c = cnt[r];
  if (c) out[(long long)r * dim + col] /= (float)c;
}

// ---------------------------------------------------------------------------
// MFMA bf16 GEMM (NT): C[M,N] = A[M,K] * B[N,K]^T, fp32 out.
// 16x16x32 MFMA; per-wave 16x16 C tile; block = 4 waves covering 16x64.
// Operand layout (verified on HW by tests/test_gpu_kernels.py::test_mfma_probe
// against torch.matmul with random asymmetric inputs):
//   A: lane l holds A[l&15][(l>>4)*8 + j], j=0..7  (8 contiguous bf16)
//   B: lane l holds B^T[l&15][(l>>4)*8 + j]        (8 contiguous bf16 of B^T)
//   C: lane l, reg r -> C[(l>>4)*4 + r][l&15]
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
gemm_bf16_nt_kernel(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                    float* __restrict__ C, int M, int N, int K,
                    const float* __restrict__ bias, int relu) {
  int m0 = blockIdx.x * 16;
  int n0 = blockIdx.y * 64 + (threadIdx.x >> 6) * 16;
  int lane = threadIdx.x & 63;
  int row = lane & 15;     // A row within tile / C col group
  int kgrp = lane >> 4;    // 0..3
  floatx4 acc = {0.f, 0.f, 0.f, 0.f};
  bool a_ok = (m0 + row) < M;
  bool b_ok = (n0 + row) < N;
  for (int k = 0; k < K; k += 32) {
    short8 a = {0, 0, 0, 0, 0, 0, 0, 0}, b = {0, 0, 0, 0, 0, 0, 0, 0};
    int kk = k + kgrp * 8;
    if (a_ok && kk < K)
      a = *reinterpret_cast<const short8*>(&A[(long long)(m0 + row) * K + kk]);
    if (b_ok && kk < K)
      b = *reinterpret_cast<const short8*>(&Bt[(long long)(n0 + row) * K + kk]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  int c_row = m0 + kgrp * 4;
  int c_col = n0 + row;
  if (c_col >= N) return;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    int cr = c_row + r;
    if (cr < M) {
      float v = acc[r];
      if (bias) v += bias[c_col];
      if (relu && v < 0.f) v = 0.f;
      C[(long long)cr * N + c_col] = v;
    }
  }
}

// Large-shape MFMA GEMM: 128x128 tile, BK=32, 4 waves (2x2), LDS staged
// via async global_load_lds width-16 (the HIP guide's m97 structure:
// naive direct-load was 71 TF; this structure reaches ~900 TF at 4096^3
// on the guide's ladder). Both operands NT ([row][k] contiguous) so each
// lane's fragment is one ds_read_b128. Requires M%128==0, N%128==0,
// K%32==0 (host dispatches the simple kernel otherwise). blockIdx is
// XCD-swizzled with the bijective m204 mapping so neighbor tiles share a
// per-XCD L2.
__global__ void __launch_bounds__(256)
gemm_bf16_nt_tiled_kernel(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                          float* __restrict__ C, int M, int N, int K,
                          const float* __restrict__ bias, int relu) {
  __shared__ short As[128 * 32];
  __shared__ short Bs[128 * 32];
  int nwg = (int)(gridDim.x * gridDim.y);
  int orig = (int)(blockIdx.y * gridDim.x + blockIdx.x);
  int wg = orig;
  if (nwg >= 8) {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = orig & 7, seq = orig >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + seq;
  }
  int bx = wg % (int)gridDim.x;  // N tile
  int by = wg / (int)gridDim.x;  // M tile
  long long m0 = (long long)by * 128;
  long long n0 = (long long)bx * 128;

  int t = threadIdx.x;
  int lane = t & 63;
  int wave = t >> 6;
  int wr = wave >> 1, wc = wave & 1;
  int frow = lane & 15;
  int kgrp = lane >> 4;

  floatx4 acc[4][4];
  #pragma unroll
  for (int m = 0; m < 4; ++m)
    #pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = floatx4{0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += 32) {
    // stage A and B tiles: 512 16-byte chunks each; chunk f covers
    // row = f>>2, ks = f&3 of the [128][32] tile. Each wave's 64 lanes
    // write one contiguous 1 KiB LDS span (gload_lds dest is uniform
    // base + lane*16).
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int f = wave * 128 + i * 64 + lane;
      int row = f >> 2, ks = f & 3;
      const bf16* ga = &A[(m0 + row) * K + k0 + ks * 8];
      const bf16* gb = &Bt[(n0 + row) * K + k0 + ks * 8];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)&As[(wave * 128 + i * 64) * 8],
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)&Bs[(wave * 128 + i * 64) * 8],
          16, 0, 0);
    }
    __syncthreads();  // drains vmcnt: staged tiles visible

    short8 a[4], b[4];
    #pragma unroll
    for (int m = 0; m < 4; ++m)
      a[m] = *reinterpret_cast<const short8*>(
          &As[(wr * 64 + m * 16 + frow) * 32 + kgrp * 8]);
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      b[n] = *reinterpret_cast<const short8*>(
          &Bs[(wc * 64 + n * 16 + frow) * 32 + kgrp * 8]);
    #pragma unroll
    for (int m = 0; m < 4; ++m)
      #pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[m], b[n], acc[m][n], 0, 0, 0);
    __syncthreads();  // tile fully consumed before restaging
  }

  #pragma unroll
  for (int m = 0; m < 4; ++m) {
    long long crow_base = m0 + wr * 64 + m * 16 + kgrp * 4;
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      long long ccol = n0 + wc * 64 + n * 16 + frow;
      float bv = bias ? bias[ccol] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[m][n][r] + bv;
        if (relu && v < 0.f) v = 0.f;
        C[(crow_base + r) * N + ccol] = v;
      }
    }
  }
}

// L2-normalize rows, fp32 -> bf16 (one wave per row)
__global__ void l2norm_rows_kernel(const float* __restrict__ in, bf16* __restrict__ out,
                                   int rows, int dim) {
  int r = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  if (r >= rows) return;
  int lane = threadIdx.x & 63;
  float ss = 0.f;
  for (int c = lane; c < dim; c += 64) {
    float v = in[(long long)r * dim + c];
    ss += v * v;
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) ss += __shfl_xor(ss, off);
  float inv = rsqrtf(ss + 1e-12f);
  for (int c = lane; c < dim; c += 64) {
    out[(long long)r * dim + c] = __float2bfloat16(in[(long long)r * dim + c] * inv);
  }
}

// ---------------------------------------------------------------------------
// 3. Fused cache similarity + argmax: scores[i,q] = Index[i,:].Q[q,:]
//    (rows pre-normalized -> cosine). Each wave: MFMA over a 16-row index
//    tile x 16 queries, K-loop over dim; per-query max folded via shfl and
//    packed (orderable-float<<32 | row) into a global 64-bit atomicMax.
// ---------------------------------------------------------------------------

__device__ __forceinline__ unsigned long long shfl_xor_u64(unsigned long long v,
                                                           int off) {
  int lo = __shfl_xor((int)(v & 0xFFFFFFFFull), off);
  int hi = __shfl_xor((int)(v >> 32), off);
  return ((unsigned long long)(unsigned)hi << 32) | (unsigned)lo;
}

__device__ __forceinline__ unsigned long long pack_score(float s, unsigned idx) {
  unsigned u = __float_as_uint(s);
  u = (u & 0x80000000u) ? ~u : (u | 0x80000000u);  // orderable float
  return ((unsigned long long)u << 32) | idx;
}

// fp8 LDS-staged variant — same structure as cache_topk_lds_kernel_t
// below (query block in LDS, A-tile ping-pong, ks-outer interleave) with
// 1-byte elements: the 128-query stage is only ~50 KB, so 3 CTAs/CU fit
// and occupancy recovers on top of the latency fixes. Row stride padded
// to KSTEPS*32+16 bytes (100 dwords = 36 mod 64 -> 16 distinct banks).
template <int KSTEPS, int ROWTILES>
__global__ void __launch_bounds__(256)
cache_topk_fp8_lds_kernel_t(const uint8_t* __restrict__ index, long long n_rows,
                            const uint8_t* __restrict__ q, int n_q /* <= 128 */,
                            int dim, unsigned long long* __restrict__ best) {
  constexpr int DIMP = KSTEPS * 32 + 16;
  extern __shared__ unsigned char smem[];
  uint8_t* qs = smem;
  unsigned long long* blk_best =
      reinterpret_cast<unsigned long long*>(smem + 128 * DIMP);
  if (threadIdx.x < 128) blk_best[threadIdx.x] = 0;
  int chunks_per_row = dim / 8;
  for (int idx = threadIdx.x; idx < n_q * chunks_per_row; idx += 256) {
    int r = idx / chunks_per_row, c = (idx - r * chunks_per_row) * 8;
    *reinterpret_cast<long*>(&qs[r * DIMP + c]) =
        *reinterpret_cast<const long*>(&q[(long long)r * dim + c]);
  }
  __syncthreads();
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kgrp = lane >> 4;
  auto load_tile = [&](long (&frag)[KSTEPS], int t) {
    long long i0 = ((long long)blockIdx.x * 4 * ROWTILES + wave * ROWTILES + t) * 16;
    bool i_ok = i0 < n_rows && (i0 + row) < n_rows;
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      int kk = ks * 32 + kgrp * 8;
      frag[ks] = i_ok
          ? *reinterpret_cast<const long*>(&index[(i0 + row) * dim + kk])
          : 0L;
    }
  };
  constexpr int QT = 8;
  auto compute_tile = [&](long (&frag)[KSTEPS], int t) {
    long long i0 = ((long long)blockIdx.x * 4 * ROWTILES + wave * ROWTILES + t) * 16;
    if (i0 >= n_rows) return;
    floatx4 acc[QT];
    #pragma unroll
    for (int qt = 0; qt < QT; ++qt) acc[qt] = floatx4{0.f, 0.f, 0.f, 0.f};
    int qrow = min(row, n_q - 1);
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      int kk = ks * 32 + kgrp * 8;
      #pragma unroll
      for (int qt = 0; qt < QT; ++qt) {
        long b = *reinterpret_cast<const long*>(&qs[(qt * 16 + qrow) * DIMP + kk]);
        acc[qt] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(frag[ks], b, acc[qt], 0, 0, 0);
      }
    }
    int lim = (int)min((long long)16, n_rows - i0);
    #pragma unroll
    for (int qt = 0; qt < QT; ++qt) {
      int q0 = qt * 16;
      bool q_in = (q0 + row) < n_q;
      float best_s = -1e30f;
      int best_r = 0;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sc = (kgrp * 4 + r < lim) ? acc[qt][r] : -1e30f;
        if (sc > best_s) { best_s = sc; best_r = r; }
      }
      long long irow = i0 + kgrp * 4 + best_r;
      if (!q_in) best_s = -1e30f;
      unsigned long long p = pack_score(best_s, (unsigned)(irow & 0xFFFFFFFF));
      #pragma unroll
      for (int off = 16; off < 64; off <<= 1) {
        unsigned long long o = shfl_xor_u64(p, off);
        if (o > p) p = o;
      }
      if (kgrp == 0 && q_in) atomicMax(&blk_best[q0 + row], p);
    }
  };
  long frag_a[KSTEPS], frag_b[KSTEPS];
  load_tile(frag_a, 0);
  static_assert(ROWTILES % 2 == 0, "pipeline assumes even ROWTILES");
  for (int t = 0; t < ROWTILES; t += 2) {
    load_tile(frag_b, t + 1);
    compute_tile(frag_a, t);
    if (t + 2 < ROWTILES) load_tile(frag_a, t + 2);
    compute_tile(frag_b, t + 1);
  }
  __syncthreads();
  if (threadIdx.x < (unsigned)n_q && blk_best[threadIdx.x])
    atomicMax(&best[threadIdx.x], blk_best[threadIdx.x]);
}

// LDS-staged variant: PMC on the register-tile kernel above shows a
// 27:1 SQ_WAIT:SQ_BUSY ratio — the serial L2 B-load -> MFMA dependency
// chain in its query loop leaves waves stalled on ~300-cycle L2 hits.
// Staging the whole (<=128-query) block in LDS once per CTA turns those
// into ~30-cycle ds_reads the scheduler hides behind MFMAs; the index
// still streams HBM once per 128-query pass (the host chunks 256-query
// calls into two passes). Row stride padded by 8 halves so the 16 query
// rows of a q-tile land on distinct banks (KSTEPS*32+8 halves = 4 mod 64
// dwords -> banks 4*row mod 64, conflict-free for 16 rows).
template <int KSTEPS, int ROWTILES>
__global__ void __launch_bounds__(256)
cache_topk_lds_kernel_t(const bf16* __restrict__ index, long long n_rows,
                        const bf16* __restrict__ q, int n_q /* <= 128 */, int dim,
                        unsigned long long* __restrict__ best) {
  constexpr int DIMP = KSTEPS * 32 + 8;
  extern __shared__ unsigned char smem[];
  bf16* qs = reinterpret_cast<bf16*>(smem);
  unsigned long long* blk_best =
      reinterpret_cast<unsigned long long*>(smem + 128 * DIMP * sizeof(bf16));
  if (threadIdx.x < 128) blk_best[threadIdx.x] = 0;
  int chunks_per_row = dim / 8;
  for (int idx = threadIdx.x; idx < n_q * chunks_per_row; idx += 256) {
    int r = idx / chunks_per_row, c = (idx - r * chunks_per_row) * 8;
    *reinterpret_cast<short8*>(&qs[r * DIMP + c]) =
        *reinterpret_cast<const short8*>(&q[(long long)r * dim + c]);
  }
  __syncthreads();
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  int row = lane & 15;
  int kgrp = lane >> 4;
  // A-tile software pipeline: at 1 CTA/CU (the 100 KB query stage) a
  // wave that loads a row tile and only then computes exposes the full
  // HBM latency every tile. Ping-pong buffers (compile-time indexed —
  // rule #20) let tile t+1's 12 dwordx4 loads fly while tile t's 96
  // MFMAs issue.
  auto load_tile = [&](short8 (&frag)[KSTEPS], int t) {
    long long i0 = ((long long)blockIdx.x * 4 * ROWTILES + wave * ROWTILES + t) * 16;
    bool i_ok = i0 < n_rows && (i0 + row) < n_rows;
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      int kk = ks * 32 + kgrp * 8;
      if (i_ok)
        frag[ks] = *reinterpret_cast<const short8*>(&index[(i0 + row) * dim + kk]);
      else
        frag[ks] = short8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };
  // ks-outer with one accumulator per query tile: at 1 CTA/CU the q-inner
  // form serializes on each MFMA's ~5-cycle result latency (12-deep acc
  // dependency chain per q-tile, nothing else in flight); QT independent
  // chains issued back-to-back hide it completely.
  constexpr int QT = 8;  // 128 queries / 16
  auto compute_tile = [&](short8 (&frag)[KSTEPS], int t) {
    long long i0 = ((long long)blockIdx.x * 4 * ROWTILES + wave * ROWTILES + t) * 16;
    if (i0 >= n_rows) return;
    floatx4 acc[QT];
    #pragma unroll
    for (int qt = 0; qt < QT; ++qt) acc[qt] = floatx4{0.f, 0.f, 0.f, 0.f};
    int qrow = min(row, n_q - 1);  // LDS reads always in-bounds
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      int kk = ks * 32 + kgrp * 8;
      #pragma unroll
      for (int qt = 0; qt < QT; ++qt) {
        short8 b = *reinterpret_cast<const short8*>(&qs[(qt * 16 + qrow) * DIMP + kk]);
        acc[qt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag[ks], b, acc[qt], 0, 0, 0);
      }
    }
    int lim = (int)min((long long)16, n_rows - i0);
    #pragma unroll
    for (int qt = 0; qt < QT; ++qt) {
      int q0 = qt * 16;
      bool q_in = (q0 + row) < n_q;
      float best_s = -1e30f;
      int best_r = 0;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = (kgrp * 4 + r < lim) ? acc[qt][r] : -1e30f;
        if (s > best_s) { best_s = s; best_r = r; }
      }
      long long irow = i0 + kgrp * 4 + best_r;
      if (!q_in) best_s = -1e30f;
      unsigned long long p = pack_score(best_s, (unsigned)(irow & 0xFFFFFFFF));
      #pragma unroll
      for (int off = 16; off < 64; off <<= 1) {
        unsigned long long o = shfl_xor_u64(p, off);
        if (o > p) p = o;
      }
      if (kgrp == 0 && q_in) atomicMax(&blk_best[q0 + row], p);
    }
  };
  short8 frag_a[KSTEPS], frag_b[KSTEPS];
  load_tile(frag_a, 0);
  static_assert(ROWTILES % 2 == 0, "pipeline assumes even ROWTILES");
  for (int t = 0; t < ROWTILES; t += 2) {
    load_tile(frag_b, t + 1);
    compute_tile(frag_a, t);
    if (t + 2 < ROWTILES) load_tile(frag_a, t + 2);
    compute_tile(frag_b, t + 1);
  }
  __syncthreads();
  if (threadIdx.x < (unsigned)n_q && blk_best[threadIdx.x])
    atomicMax(&best[threadIdx.x], blk_best[threadIdx.x]);
}

// ---------------------------------------------------------------------------
// 4. KV-occupancy endpoint scorer: greedy sequential assignment of a request
//    batch to replicas. One wave; lane = replica. score = w_kv*(free KV frac
//    after assignment) - w_q*queue_depth - w_a*active. Mirrors the EPP
//    "prefix-cache + queue depth" scoring the reference delegates to an
//    external endpoint-picker service.
// ---------------------------------------------------------------------------

__global__ void kv_scorer_kernel(const float* __restrict__ stats,  // R x 4
                                 int n_rep, const float* __restrict__ pred_tokens,
                                 int n_req, float w_kv, float w_q, float w_a,
                                 int32_t* __restrict__ assign) {
  int lane = threadIdx.x & 63;
  bool ok = lane < n_rep;
  float kv_used = ok ? stats[lane * 4 + 0] : 0.f;
  float kv_total = ok ? fmaxf(stats[lane * 4 + 1], 1.f) : 1.f;
  float queue = ok ? stats[lane * 4 + 2] : 0.f;
  float active = ok ? stats[lane * 4 + 3] : 0.f;
  for (int i = 0; i < n_req; ++i) {
    float p = pred_tokens[i];
    float score = ok ? (w_kv * (1.f - (kv_used + p) / kv_total) - w_q * queue - w_a * active)
                     : -1e30f;
    if (ok && kv_used + p > kv_total) score -= 1e6f;  // avoid overflowing a replica
    // wave argmax
    float best = score;
    int best_lane = lane;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float os = __shfl_xor(best, off);
      int ol = __shfl_xor(best_lane, off);
      if (os > best || (os == best && ol < best_lane)) { best = os; best_lane = ol; }
    }
    if (lane == 0) assign[i] = best_lane;
    if (lane == best_lane) {
      kv_used += p;
      active += 1.f;
      queue += 1.f;
    }
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static void check_cuda(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// Returns (seg_start, seg_req, n_segs is seg_start.numel())
std::vector<at::Tensor> bpe_segment(at::Tensor bytes, at::Tensor req_off) {
  check_cuda(bytes, "bytes");
  check_cuda(req_off, "req_off");
  int n = (int)bytes.numel();
  TORCH_CHECK(n > 0, "empty batch");
  auto stream = current_stream();
  auto u8 = at::TensorOptions().dtype(at::kByte).device(bytes.device());
  auto i32 = at::TensorOptions().dtype(at::kInt).device(bytes.device());
  at::Tensor flags = at::empty({n}, u8);
  int blocks = (n + 255) / 256;
  hipLaunchKernelGGL(seg_flags_kernel, dim3(blocks), dim3(256), 0, stream,
                     bytes.data_ptr<uint8_t>(), n, flags.data_ptr<uint8_t>());
  int n_req = (int)req_off.numel();
  hipLaunchKernelGGL(seg_force_starts_kernel, dim3((n_req + 255) / 256), dim3(256), 0,
                     stream, req_off.data_ptr<int64_t>(), n_req,
                     flags.data_ptr<uint8_t>());
  at::Tensor blk_counts = at::empty({blocks}, i32);
  hipLaunchKernelGGL(seg_block_count_kernel, dim3(blocks), dim3(256), 0, stream,
                     flags.data_ptr<uint8_t>(), n, blk_counts.data_ptr<int32_t>());
  at::Tensor scan = blk_counts.cumsum(0, at::kInt);
  at::Tensor blk_excl = at::zeros({blocks}, i32);
  if (blocks > 1)
    blk_excl.narrow(0, 1, blocks - 1).copy_(scan.narrow(0, 0, blocks - 1));
  int n_segs = scan[blocks - 1].item<int>();  // one D2H sync per batch
  at::Tensor seg_start = at::empty({n_segs}, i32);
  at::Tensor seg_req = at::empty({n_segs}, i32);
  hipLaunchKernelGGL(seg_write_kernel, dim3(blocks), dim3(256), 0, stream,
                     flags.data_ptr<uint8_t>(), n, blk_excl.data_ptr<int32_t>(),
                     req_off.data_ptr<int64_t>(), n_req,
                     seg_start.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>());
  return {seg_start, seg_req};
}

std::vector<at::Tensor> bpe_encode(at::Tensor bytes, at::Tensor req_off,
                                   at::Tensor htab_keys, at::Tensor htab_rank) {
  check_cuda(bytes, "bytes");
  check_cuda(req_off, "req_off");
  check_cuda(htab_keys, "htab_keys");
  check_cuda(htab_rank, "htab_rank");
  auto segs = bpe_segment(bytes, req_off);
  at::Tensor seg_start = segs[0], seg_req = segs[1];
  int n_segs = (int)seg_start.numel();
  int n = (int)bytes.numel();
  int n_req = (int)req_off.numel();
  auto i32 = at::TensorOptions().dtype(at::kInt).device(bytes.device());
  at::Tensor out_ids = at::full({n}, -1, i32);
  at::Tensor req_counts = at::zeros({n_req}, i32);
  int htab_mask = (int)htab_keys.numel() - 1;
  TORCH_CHECK((htab_keys.numel() & htab_mask) == 0, "htab size must be power of 2");
  auto stream = current_stream();

  // group segments (scheduling-only wave packing; see
  // bpe_encode_grouped_kernel)
  auto u8 = at::TensorOptions().dtype(at::kByte).device(bytes.device());
  at::Tensor gflags = at::empty({n_segs}, i32.dtype(at::kByte));
  int seg_blocks = (n_segs + 255) / 256;
  hipLaunchKernelGGL(group_head_flags_kernel, dim3(seg_blocks), dim3(256), 0, stream,
                     seg_start.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>(),
                     n_segs, req_off.data_ptr<int64_t>(), gflags.data_ptr<uint8_t>());
  at::Tensor gblk = at::empty({seg_blocks}, i32);
  hipLaunchKernelGGL(seg_block_count_kernel, dim3(seg_blocks), dim3(256), 0, stream,
                     gflags.data_ptr<uint8_t>(), n_segs, gblk.data_ptr<int32_t>());
  at::Tensor gscan = gblk.cumsum(0, at::kInt);
  at::Tensor gexcl = at::zeros({seg_blocks}, i32);
  if (seg_blocks > 1)
    gexcl.narrow(0, 1, seg_blocks - 1).copy_(gscan.narrow(0, 0, seg_blocks - 1));
  int n_groups = gscan[seg_blocks - 1].item<int>();
  at::Tensor ghead = at::empty({n_groups}, i32);
  hipLaunchKernelGGL(flag_compact_write_kernel, dim3(seg_blocks), dim3(256), 0, stream,
                     gflags.data_ptr<uint8_t>(), n_segs, gexcl.data_ptr<int32_t>(),
                     ghead.data_ptr<int32_t>());
  (void)u8;

  int blocks = (n_groups + 3) / 4;  // 4 waves per 256-thread block
  hipLaunchKernelGGL(bpe_encode_grouped_kernel, dim3(blocks), dim3(256), 0, stream,
                     bytes.data_ptr<uint8_t>(), seg_start.data_ptr<int32_t>(),
                     seg_req.data_ptr<int32_t>(), n_segs, n,
                     ghead.data_ptr<int32_t>(), n_groups,
                     reinterpret_cast<long long*>(htab_keys.data_ptr<int64_t>()),
                     htab_rank.data_ptr<int32_t>(), htab_mask,
                     out_ids.data_ptr<int32_t>(), req_counts.data_ptr<int32_t>());
  return {out_ids, req_counts, seg_start, seg_req};
}

// Fully async tokenize: ZERO host syncs. Segment/group totals stay in
// device scalars (scan tails) and every kernel self-bounds; intermediate
// arrays use upper bounds (#segments <= n bytes; #groups <= n/32 + n_req).
// The caller awaits a hipEvent instead of a blocking sync — on ROCm a
// blocking host sync busy-spins a core, which was measured to halve the
// serving throughput with 12 gateway workers per GPU (profiles/r01).
std::vector<at::Tensor> bpe_count_async(at::Tensor bytes, at::Tensor req_off,
                                        at::Tensor htab_keys, at::Tensor htab_rank) {
  check_cuda(bytes, "bytes");
  check_cuda(req_off, "req_off");
  int n = (int)bytes.numel();
  int n_req = (int)req_off.numel();
  TORCH_CHECK(n > 0, "empty batch");
  int htab_mask = (int)htab_keys.numel() - 1;
  auto stream = current_stream();
  auto u8 = at::TensorOptions().dtype(at::kByte).device(bytes.device());
  auto i32 = at::TensorOptions().dtype(at::kInt).device(bytes.device());
  int blocks = (n + 255) / 256;

  at::Tensor flags = at::empty({n}, u8);
  hipLaunchKernelGGL(seg_flags_kernel, dim3(blocks), dim3(256), 0, stream,
                     bytes.data_ptr<uint8_t>(), n, flags.data_ptr<uint8_t>());
  hipLaunchKernelGGL(seg_force_starts_kernel, dim3((n_req + 255) / 256), dim3(256), 0,
                     stream, req_off.data_ptr<int64_t>(), n_req,
                     flags.data_ptr<uint8_t>());
  at::Tensor blk_counts = at::empty({blocks}, i32);
  hipLaunchKernelGGL(seg_block_count_kernel, dim3(blocks), dim3(256), 0, stream,
                     flags.data_ptr<uint8_t>(), n, blk_counts.data_ptr<int32_t>());
  at::Tensor scan = blk_counts.cumsum(0, at::kInt);
  at::Tensor blk_excl = at::zeros({blocks}, i32);
  if (blocks > 1)
    blk_excl.narrow(0, 1, blocks - 1).copy_(scan.narrow(0, 0, blocks - 1));
  const int32_t* n_segs_dev = scan.data_ptr<int32_t>() + (blocks - 1);

  at::Tensor seg_start = at::empty({n}, i32);
  at::Tensor seg_req = at::empty({n}, i32);
  hipLaunchKernelGGL(seg_write_kernel, dim3(blocks), dim3(256), 0, stream,
                     flags.data_ptr<uint8_t>(), n, blk_excl.data_ptr<int32_t>(),
                     req_off.data_ptr<int64_t>(), n_req,
                     seg_start.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>());

  at::Tensor gflags = at::zeros({n}, u8);  // beyond n_segs stays 0
  hipLaunchKernelGGL(group_head_flags_dev_kernel, dim3(blocks), dim3(256), 0, stream,
                     seg_start.data_ptr<int32_t>(), seg_req.data_ptr<int32_t>(),
                     n_segs_dev, req_off.data_ptr<int64_t>(),
                     gflags.data_ptr<uint8_t>());
  at::Tensor gblk = at::empty({blocks}, i32);
  hipLaunchKernelGGL(seg_block_count_kernel, dim3(blocks), dim3(256), 0, stream,
                     gflags.data_ptr<uint8_t>(), n, gblk.data_ptr<int32_t>());
  at::Tensor gscan = gblk.cumsum(0, at::kInt);
  at::Tensor gexcl = at::zeros({blocks}, i32);
  if (blocks > 1)
    gexcl.narrow(0, 1, blocks - 1).copy_(gscan.narrow(0, 0, blocks - 1));
  const int32_t* n_groups_dev = gscan.data_ptr<int32_t>() + (blocks - 1);
  at::Tensor ghead = at::empty({n}, i32);
  hipLaunchKernelGGL(flag_compact_write_kernel, dim3(blocks), dim3(256), 0, stream,
                     gflags.data_ptr<uint8_t>(), n, gexcl.data_ptr<int32_t>(),
                     ghead.data_ptr<int32_t>());

  at::Tensor out_ids = at::full({n}, -1, i32);
  at::Tensor req_counts = at::zeros({n_req}, i32);
  long long group_bound = (long long)n / 32 + n_req + 1;
  int blocks2 = (int)((group_bound + 3) / 4);
  hipLaunchKernelGGL(bpe_encode_grouped_dev_kernel, dim3(blocks2), dim3(256), 0,
                     stream, bytes.data_ptr<uint8_t>(), seg_start.data_ptr<int32_t>(),
                     seg_req.data_ptr<int32_t>(), n_segs_dev, n,
                     ghead.data_ptr<int32_t>(), n_groups_dev,
                     reinterpret_cast<long long*>(htab_keys.data_ptr<int64_t>()),
                     htab_rank.data_ptr<int32_t>(), htab_mask,
                     out_ids.data_ptr<int32_t>(), req_counts.data_ptr<int32_t>());
  return {out_ids, req_counts};
}

at::Tensor meanpool(at::Tensor ids, at::Tensor req_off, at::Tensor emb) {
  check_cuda(ids, "ids");
  check_cuda(req_off, "req_off");
  check_cuda(emb, "emb");
  TORCH_CHECK(emb.scalar_type() == at::kBFloat16, "emb must be bf16");
  int n_req = (int)req_off.numel();
  int dim = (int)emb.size(1);
  TORCH_CHECK(dim <= 1024, "dim too large for one block");
  auto out = at::zeros({n_req, dim},
                       at::TensorOptions().dtype(at::kFloat).device(ids.device()));
  auto cnt = at::zeros({n_req},
                       at::TensorOptions().dtype(at::kInt).device(ids.device()));
  constexpr int P = 8;
  hipLaunchKernelGGL(meanpool_accum_kernel, dim3(n_req, P), dim3(dim), 0,
                     current_stream(), ids.data_ptr<int32_t>(),
                     req_off.data_ptr<int64_t>(), n_req, (int)ids.numel(),
                     reinterpret_cast<bf16*>(emb.data_ptr<at::BFloat16>()), dim, P,
                     out.data_ptr<float>(), cnt.data_ptr<int32_t>());
  hipLaunchKernelGGL(meanpool_div_kernel, dim3(n_req), dim3(dim), 0, current_stream(),
                     out.data_ptr<float>(), cnt.data_ptr<int32_t>(), n_req, dim);
  return out;
}

at::Tensor gemm_bf16_nt(at::Tensor a, at::Tensor bt, c10::optional<at::Tensor> bias,
                        bool relu) {
  check_cuda(a, "a");
  check_cuda(bt, "bt");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 && bt.scalar_type() == at::kBFloat16,
              "bf16 required");
  int M = (int)a.size(0), K = (int)a.size(1), N = (int)bt.size(0);
  TORCH_CHECK(bt.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
  auto c = at::empty({M, N}, at::TensorOptions().dtype(at::kFloat).device(a.device()));
  const float* bias_ptr = nullptr;
  if (bias.has_value()) {
    check_cuda(*bias, "bias");
    bias_ptr = bias->data_ptr<float>();
  }
  if (M % 128 == 0 && N % 128 == 0 && K % 32 == 0) {
    dim3 grid(N / 128, M / 128);
    hipLaunchKernelGGL(gemm_bf16_nt_tiled_kernel, grid, dim3(256), 0,
                       current_stream(),
                       reinterpret_cast<bf16*>(a.data_ptr<at::BFloat16>()),
                       reinterpret_cast<bf16*>(bt.data_ptr<at::BFloat16>()),
                       c.data_ptr<float>(), M, N, K, bias_ptr, relu ? 1 : 0);
    return c;
  }
  dim3 grid((M + 15) / 16, (N + 63) / 64);
  hipLaunchKernelGGL(gemm_bf16_nt_kernel, grid, dim3(256), 0, current_stream(),
                     reinterpret_cast<bf16*>(a.data_ptr<at::BFloat16>()),
                     reinterpret_cast<bf16*>(bt.data_ptr<at::BFloat16>()),
                     c.data_ptr<float>(), M, N, K, bias_ptr, relu ? 1 : 0);
  return c;
}

at::Tensor l2norm_rows(at::Tensor x) {
  check_cuda(x, "x");
  TORCH_CHECK(x.scalar_type() == at::kFloat, "fp32 required");
  int rows = (int)x.size(0), dim = (int)x.size(1);
  auto out = at::empty({rows, dim},
                       at::TensorOptions().dtype(at::kBFloat16).device(x.device()));
  int blocks = (rows + 3) / 4;
  hipLaunchKernelGGL(l2norm_rows_kernel, dim3(blocks), dim3(256), 0, current_stream(),
                     x.data_ptr<float>(),
                     reinterpret_cast<bf16*>(out.data_ptr<at::BFloat16>()), rows, dim);
  return out;
}

std::vector<at::Tensor> cache_topk(at::Tensor index, at::Tensor q) {
  check_cuda(index, "index");
  check_cuda(q, "q");
  bool fp8 = index.scalar_type() == at::kFloat8_e4m3fn;
  TORCH_CHECK(index.scalar_type() == q.scalar_type(), "index/q dtype mismatch");
  TORCH_CHECK(fp8 || index.scalar_type() == at::kBFloat16,
              "bf16 or float8_e4m3fn required");
  long long n_rows = index.size(0);
  int n_q = (int)q.size(0), dim = (int)q.size(1);
  TORCH_CHECK(index.size(1) == dim, "dim mismatch");
  TORCH_CHECK(dim % 32 == 0 && dim <= 512, "dim must be a multiple of 32, <= 512");
  auto best = at::zeros({n_q}, at::TensorOptions()
                                   .dtype(at::kLong)
                                   .device(q.device()));
  TORCH_CHECK(n_q <= 256, "cache_topk: at most 256 queries per call");
  constexpr int ROWTILES = 16;
  long long blocks = (n_rows + 64 * ROWTILES - 1) / (64 * ROWTILES);
  auto* best_p = reinterpret_cast<unsigned long long*>(best.data_ptr<int64_t>());
  if (fp8) {
    auto* ip = reinterpret_cast<const uint8_t*>(index.data_ptr());
    auto* q_base = reinterpret_cast<const uint8_t*>(q.data_ptr());
    auto launch_fp8 = [&](auto kernel, int kq, const uint8_t* qp,
                          unsigned long long* bp, size_t lds_bytes) {
      (void)hipFuncSetAttribute(reinterpret_cast<const void*>(kernel),
                                hipFuncAttributeMaxDynamicSharedMemorySize,
                                160 * 1024);
      hipLaunchKernelGGL(kernel, dim3((unsigned)blocks), dim3(256), lds_bytes,
                         current_stream(), ip, n_rows, qp, kq, dim, bp);
    };
    for (int q0 = 0; q0 < n_q; q0 += 128) {
      int kq = std::min(128, n_q - q0);
      const uint8_t* qp = q_base + (long long)q0 * dim;
      unsigned long long* bp = best_p + q0;
      switch (dim >> 5) {
        case 12: {
          constexpr size_t L = 128 * (12 * 32 + 16) + 128 * 8;
          launch_fp8(&cache_topk_fp8_lds_kernel_t<12, ROWTILES>, kq, qp, bp, L);
          break;
        }
        case 8: {
          constexpr size_t L = 128 * (8 * 32 + 16) + 128 * 8;
          launch_fp8(&cache_topk_fp8_lds_kernel_t<8, ROWTILES>, kq, qp, bp, L);
          break;
        }
        case 16: {
          constexpr size_t L = 128 * (16 * 32 + 16) + 128 * 8;
          launch_fp8(&cache_topk_fp8_lds_kernel_t<16, ROWTILES>, kq, qp, bp, L);
          break;
        }
        case 4: {
          constexpr size_t L = 128 * (4 * 32 + 16) + 128 * 8;
          launch_fp8(&cache_topk_fp8_lds_kernel_t<4, ROWTILES>, kq, qp, bp, L);
          break;
        }
        default:
          TORCH_CHECK(false, "cache_topk fp8: unsupported dim ", dim);
      }
    }
    auto hi = best.bitwise_right_shift(32).to(at::kLong);
    auto idx = best.bitwise_and(0xFFFFFFFFLL).to(at::kInt);
    return {hi, idx};
  }
  auto* index_p = reinterpret_cast<bf16*>(index.data_ptr<at::BFloat16>());
  auto* q_p = reinterpret_cast<bf16*>(q.data_ptr<at::BFloat16>());
  // bf16 path: LDS-staged kernel, <=128 queries per pass (the query block
  // lives in LDS; see cache_topk_lds_kernel_t). The index streams once
  // per pass — two passes for 256 queries still beat the register-tile
  // kernel's latency stalls by a wide margin.
  auto launch_lds = [&](auto kernel, int kq, const bf16* qp, unsigned long long* bp,
                        size_t lds_bytes) {
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(kernel),
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              160 * 1024);
    hipLaunchKernelGGL(kernel, dim3((unsigned)blocks), dim3(256), lds_bytes,
                       current_stream(), index_p, n_rows, qp, kq, dim, bp);
  };
  for (int q0 = 0; q0 < n_q; q0 += 128) {
    int kq = std::min(128, n_q - q0);
    const bf16* qp = q_p + (long long)q0 * dim;
    unsigned long long* bp = best_p + q0;
    switch (dim >> 5) {
      case 12: {  // dim = 384 (bge-small) — the hot path
        constexpr size_t L = 128 * (12 * 32 + 8) * sizeof(bf16) + 128 * 8;
        launch_lds(&cache_topk_lds_kernel_t<12, ROWTILES>, kq, qp, bp, L);
        break;
      }
      case 8: {  // dim = 256
        constexpr size_t L = 128 * (8 * 32 + 8) * sizeof(bf16) + 128 * 8;
        launch_lds(&cache_topk_lds_kernel_t<8, ROWTILES>, kq, qp, bp, L);
        break;
      }
      case 16: {  // dim = 512
        constexpr size_t L = 128 * (16 * 32 + 8) * sizeof(bf16) + 128 * 8;
        launch_lds(&cache_topk_lds_kernel_t<16, ROWTILES>, kq, qp, bp, L);
        break;
      }
      case 4: {  // dim = 128
        constexpr size_t L = 128 * (4 * 32 + 8) * sizeof(bf16) + 128 * 8;
        launch_lds(&cache_topk_lds_kernel_t<4, ROWTILES>, kq, qp, bp, L);
        break;
      }
      default:
        TORCH_CHECK(false, "cache_topk: unsupported dim ", dim,
                    " (supported: 128/256/384/512)");
    }
  }
  // unpack: score = orderable^-1(hi32), idx = lo32
  auto hi = best.bitwise_right_shift(32).to(at::kLong);
  auto idx = best.bitwise_and(0xFFFFFFFFLL).to(at::kInt);
  return {hi, idx};
}

at::Tensor kv_score_assign(at::Tensor stats, at::Tensor pred_tokens, double w_kv,
                           double w_q, double w_a) {
  check_cuda(stats, "stats");
  check_cuda(pred_tokens, "pred_tokens");
  int n_rep = (int)stats.size(0);
  TORCH_CHECK(n_rep <= 64, "at most 64 replicas");
  int n_req = (int)pred_tokens.numel();
  auto assign = at::empty({n_req},
                          at::TensorOptions().dtype(at::kInt).device(stats.device()));
  hipLaunchKernelGGL(kv_scorer_kernel, dim3(1), dim3(64), 0, current_stream(),
                     stats.data_ptr<float>(), n_rep, pred_tokens.data_ptr<float>(),
                     n_req, (float)w_kv, (float)w_q, (float)w_a,
                     assign.data_ptr<int32_t>());
  return assign;
}

// MFMA layout probe: C = A(16x32) @ B(32x16) as one intrinsic call; used by
// the HW test to verify the documented fragment layouts against torch.
__global__ void mfma_probe_kernel(const bf16* __restrict__ A, const bf16* __restrict__ Bt,
                                  float* __restrict__ C) {
  int lane = threadIdx.x & 63;
  short8 a = *reinterpret_cast<const short8*>(&A[(lane & 15) * 32 + (lane >> 4) * 8]);
  short8 b = *reinterpret_cast<const short8*>(&Bt[(lane & 15) * 32 + (lane >> 4) * 8]);
  floatx4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r) C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor bt) {
  check_cuda(a, "a");
  check_cuda(bt, "bt");
  auto c = at::empty({16, 16}, at::TensorOptions().dtype(at::kFloat).device(a.device()));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, current_stream(),
                     reinterpret_cast<bf16*>(a.data_ptr<at::BFloat16>()),
                     reinterpret_cast<bf16*>(bt.data_ptr<at::BFloat16>()),
                     c.data_ptr<float>());
  return c;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bpe_segment", &bpe_segment, "segment bytes (GPU)");
  m.def("bpe_encode", &bpe_encode, "BPE encode a packed byte batch (GPU)");
  m.def("bpe_count_async", &bpe_count_async,
        "sync-free BPE encode: returns (out_ids, req_counts) without host syncs");
  m.def("meanpool", &meanpool, "mean-pool token embeddings per request");
  m.def("gemm_bf16_nt", &gemm_bf16_nt, "C = A @ Bt^T (MFMA bf16)",
        py::arg("a"), py::arg("bt"), py::arg("bias") = py::none(),
        py::arg("relu") = false);
  m.def("l2norm_rows", &l2norm_rows, "row-wise L2 normalize fp32->bf16");
  m.def("cache_topk", &cache_topk, "fused cosine-sim argmax over cache index");
  m.def("kv_score_assign", &kv_score_assign, "greedy KV-occupancy assignment");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
}
