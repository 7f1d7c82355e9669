// Embedding / semantic-cache / scorer kernels (gfx950), shared between
// the torch extension (csrc/aigw_kernels.hip) and the torch-free native
// cache path (csrc/admission.hip). Same sharing rule as bpe_kernels.cuh:
// header-defined because the users are separate shared objects.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <climits>
#include <cstdint>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(4))) short short4v;

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

