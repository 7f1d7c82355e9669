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

#include "cache_kernels.cuh"

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
