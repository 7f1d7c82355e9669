// Torch-free GPU admission path for the native fast front.
//
// The round-1 serving design shipped every chat text to a Python GPU
// service over UDS (msgpack) for BPE token counting; measured on
// hardware, one host process decodes ~900 MB/s and sharding hosts
// thrashes GPU contexts (4 processes: 36k -> 10k req/s). This module
// removes the whole detour: the C++ gateway owns ONE HIP stream and
// launches the same gfx950 BPE kernels (csrc/bpe_kernels.cuh) directly —
// pack texts into a pinned staging buffer, one H2D copy, the sync-free
// segmentation/scan/merge pipeline (device-resident totals, every kernel
// self-bounds), one D2H of per-request counts, one event wait. Zero
// Python, zero IPC, zero extra GPU contexts.
//
// The merge table is supplied by the caller (aigw.ops.tokenizer
// make_merges) so counts stay bit-identical to the Python/CPU oracle.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <vector>

#include "bpe_kernels.cuh"

namespace aigw_fast {

namespace {

// Exclusive block-scan over int32 counts with the total appended:
// single-block Hillis-Steele over 256-element chunks with a running
// carry. The input is per-256-byte-block flag counts (<= ~256k entries
// for a 64 MiB batch) — microseconds of work on one CU, and keeping it
// on-device is what keeps the pipeline free of host syncs.
__global__ void excl_scan_kernel(const int32_t* __restrict__ in, int n,
                                 int32_t* __restrict__ excl,
                                 int32_t* __restrict__ total) {
  __shared__ int32_t buf[256];
  __shared__ int32_t carry;
  if (threadIdx.x == 0) carry = 0;
  __syncthreads();
  for (int base = 0; base < n; base += 256) {
    int i = base + (int)threadIdx.x;
    int32_t v = (i < n) ? in[i] : 0;
    buf[threadIdx.x] = v;
    __syncthreads();
    // inclusive scan of buf
    for (int off = 1; off < 256; off <<= 1) {
      int32_t t = (threadIdx.x >= (unsigned)off) ? buf[threadIdx.x - off] : 0;
      __syncthreads();
      buf[threadIdx.x] += t;
      __syncthreads();
    }
    if (i < n) excl[i] = carry + buf[threadIdx.x] - v;  // exclusive
    __syncthreads();
    if (threadIdx.x == 0) carry += buf[255];
    __syncthreads();
  }
  if (threadIdx.x == 0) *total = carry;
}

#define HIP_OK(expr)                                                   \
  do {                                                                 \
    hipError_t _e = (expr);                                            \
    if (_e != hipSuccess) {                                            \
      fprintf(stderr, "aigw admission: %s failed: %s\n", #expr,        \
              hipGetErrorString(_e));                                  \
      return false;                                                    \
    }                                                                  \
  } while (0)

}  // namespace

// One instance per fast server; init once, count() called by the
// admission batcher thread (single-threaded use of the stream).
class GpuAdmissionDirect {
 public:
  bool init(const long long* htab_keys, const int32_t* htab_rank, int htab_n,
            size_t max_bytes, int max_req, int device) {
    int count = 0;
    if (hipGetDeviceCount(&count) != hipSuccess || count == 0) return false;
    device_ = device >= 0 && device < count ? device : 0;
    HIP_OK(hipSetDevice(device_));
    max_bytes_ = max_bytes;
    max_req_ = max_req;
    htab_mask_ = htab_n - 1;
    HIP_OK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    // hipEventBlockingSync: the batcher thread YIELDS while the batch
    // runs instead of busy-spinning — the serving container is CPU-quota
    // bound (16 CPUs for the whole gateway), so a spinning core is ~6%
    // of the entire budget
    HIP_OK(hipEventCreateWithFlags(&event_,
                                   hipEventDisableTiming | hipEventBlockingSync));
    HIP_OK(hipHostMalloc(&h_bytes_, max_bytes, hipHostMallocDefault));
    HIP_OK(hipHostMalloc(&h_off_, sizeof(int64_t) * (max_req + 1),
                         hipHostMallocDefault));
    HIP_OK(hipHostMalloc(&h_counts_, sizeof(int32_t) * max_req,
                         hipHostMallocDefault));
    HIP_OK(hipMalloc(&d_htab_keys_, sizeof(long long) * htab_n));
    HIP_OK(hipMalloc(&d_htab_rank_, sizeof(int32_t) * htab_n));
    HIP_OK(hipMemcpy(d_htab_keys_, htab_keys, sizeof(long long) * htab_n,
                     hipMemcpyHostToDevice));
    HIP_OK(hipMemcpy(d_htab_rank_, htab_rank, sizeof(int32_t) * htab_n,
                     hipMemcpyHostToDevice));
    HIP_OK(hipMalloc(&d_bytes_, max_bytes));
    HIP_OK(hipMalloc(&d_off_, sizeof(int64_t) * (max_req + 1)));
    HIP_OK(hipMalloc(&d_flags_, max_bytes));
    HIP_OK(hipMalloc(&d_gflags_, max_bytes));
    int max_blocks = (int)((max_bytes + 255) / 256);
    HIP_OK(hipMalloc(&d_blk_, sizeof(int32_t) * max_blocks));
    HIP_OK(hipMalloc(&d_excl_, sizeof(int32_t) * max_blocks));
    HIP_OK(hipMalloc(&d_totals_, sizeof(int32_t) * 2));  // [n_segs, n_groups]
    HIP_OK(hipMalloc(&d_seg_start_, sizeof(int32_t) * max_bytes));
    HIP_OK(hipMalloc(&d_seg_req_, sizeof(int32_t) * max_bytes));
    HIP_OK(hipMalloc(&d_ghead_, sizeof(int32_t) * max_bytes));
    HIP_OK(hipMalloc(&d_out_ids_, sizeof(int32_t) * max_bytes));
    HIP_OK(hipMalloc(&d_counts_, sizeof(int32_t) * max_req));
    ready_ = true;
    return true;
  }

  // texts: concatenated; offsets[i] is text i's start (n_req entries,
  // matching the kernel contract in aigw/ops/tokenizer.py pack()).
  // Returns per-request token counts. Single caller thread.
  bool count(const char* bytes, size_t n, const int64_t* offsets, int n_req,
             int32_t* counts_out) {
    if (!ready_ || n == 0 || n_req == 0 || n > max_bytes_ || n_req > max_req_)
      return false;
    // launches follow the CALLER thread's current device; the batcher
    // thread differs from the init thread, so pin it here (no-op when
    // already current) — rank N of an 8-GPU node must stay on device N
    HIP_OK(hipSetDevice(device_));
    memcpy(h_bytes_, bytes, n);
    memcpy(h_off_, offsets, sizeof(int64_t) * (size_t)n_req);
    HIP_OK(hipMemcpyAsync(d_bytes_, h_bytes_, n, hipMemcpyHostToDevice, stream_));
    HIP_OK(hipMemcpyAsync(d_off_, h_off_, sizeof(int64_t) * (size_t)n_req,
                          hipMemcpyHostToDevice, stream_));
    int blocks = (int)((n + 255) / 256);
    hipLaunchKernelGGL(seg_flags_kernel, dim3(blocks), dim3(256), 0, stream_,
                       d_bytes_, (int)n, d_flags_);
    hipLaunchKernelGGL(seg_force_starts_kernel, dim3((n_req + 255) / 256),
                       dim3(256), 0, stream_, d_off_, n_req, d_flags_);
    hipLaunchKernelGGL(seg_block_count_kernel, dim3(blocks), dim3(256), 0,
                       stream_, d_flags_, (int)n, d_blk_);
    hipLaunchKernelGGL(excl_scan_kernel, dim3(1), dim3(256), 0, stream_,
                       d_blk_, blocks, d_excl_, d_totals_);
    hipLaunchKernelGGL(seg_write_kernel, dim3(blocks), dim3(256), 0, stream_,
                       d_flags_, (int)n, d_excl_, d_off_, n_req, d_seg_start_,
                       d_seg_req_);
    // groups (device-bounded; gflags beyond n_segs cleared so counts stay 0)
    HIP_OK(hipMemsetAsync(d_gflags_, 0, n, stream_));
    hipLaunchKernelGGL(group_head_flags_dev_kernel, dim3(blocks), dim3(256), 0,
                       stream_, d_seg_start_, d_seg_req_, d_totals_, d_off_,
                       d_gflags_);
    hipLaunchKernelGGL(seg_block_count_kernel, dim3(blocks), dim3(256), 0,
                       stream_, d_gflags_, (int)n, d_blk_);
    hipLaunchKernelGGL(excl_scan_kernel, dim3(1), dim3(256), 0, stream_,
                       d_blk_, blocks, d_excl_, d_totals_ + 1);
    hipLaunchKernelGGL(flag_compact_write_kernel, dim3(blocks), dim3(256), 0,
                       stream_, d_gflags_, (int)n, d_excl_, d_ghead_);
    HIP_OK(hipMemsetAsync(d_counts_, 0, sizeof(int32_t) * n_req, stream_));
    long long group_bound = (long long)n / 32 + n_req + 1;
    int blocks2 = (int)((group_bound + 3) / 4);
    hipLaunchKernelGGL(bpe_encode_grouped_dev_kernel, dim3(blocks2), dim3(256),
                       0, stream_, d_bytes_, d_seg_start_, d_seg_req_,
                       d_totals_, (int)n, d_ghead_, d_totals_ + 1,
                       d_htab_keys_, d_htab_rank_, htab_mask_, d_out_ids_,
                       d_counts_);
    HIP_OK(hipMemcpyAsync(h_counts_, d_counts_, sizeof(int32_t) * n_req,
                          hipMemcpyDeviceToHost, stream_));
    HIP_OK(hipEventRecord(event_, stream_));
    HIP_OK(hipEventSynchronize(event_));
    memcpy(counts_out, h_counts_, sizeof(int32_t) * (size_t)n_req);
    return true;
  }

  ~GpuAdmissionDirect() {
    if (!ready_) return;
    (void)hipStreamDestroy(stream_);
    (void)hipEventDestroy(event_);
    (void)hipHostFree(h_bytes_);
    (void)hipHostFree(h_off_);
    (void)hipHostFree(h_counts_);
    for (void* p : {(void*)d_htab_keys_, (void*)d_htab_rank_, (void*)d_bytes_,
                    (void*)d_off_, (void*)d_flags_, (void*)d_gflags_,
                    (void*)d_blk_, (void*)d_excl_, (void*)d_totals_,
                    (void*)d_seg_start_, (void*)d_seg_req_, (void*)d_ghead_,
                    (void*)d_out_ids_, (void*)d_counts_})
      (void)hipFree(p);
  }

  bool ready() const { return ready_; }

 private:
  bool ready_ = false;
  int device_ = 0;
  size_t max_bytes_ = 0;
  int max_req_ = 0;
  int htab_mask_ = 0;
  hipStream_t stream_{};
  hipEvent_t event_{};
  char* h_bytes_ = nullptr;
  int64_t* h_off_ = nullptr;
  int32_t* h_counts_ = nullptr;
  long long* d_htab_keys_ = nullptr;
  int32_t* d_htab_rank_ = nullptr;
  uint8_t* d_bytes_ = nullptr;
  int64_t* d_off_ = nullptr;
  uint8_t* d_flags_ = nullptr;
  uint8_t* d_gflags_ = nullptr;
  int32_t* d_blk_ = nullptr;
  int32_t* d_excl_ = nullptr;
  int32_t* d_totals_ = nullptr;
  int32_t* d_seg_start_ = nullptr;
  int32_t* d_seg_req_ = nullptr;
  int32_t* d_ghead_ = nullptr;
  int32_t* d_out_ids_ = nullptr;
  int32_t* d_counts_ = nullptr;
};

// C-style entry points used by fastpath.cpp (keeps fastpath free of HIP
// headers; admission.hip is the only HIP translation unit in aigw_fast).
GpuAdmissionDirect* admission_create(const long long* htab_keys,
                                     const int32_t* htab_rank, int htab_n,
                                     size_t max_bytes, int max_req, int device) {
  auto* a = new GpuAdmissionDirect();
  if (!a->init(htab_keys, htab_rank, htab_n, max_bytes, max_req, device)) {
    delete a;
    return nullptr;
  }
  return a;
}

bool admission_count(GpuAdmissionDirect* a, const char* bytes, size_t n,
                     const int64_t* offsets, int n_req, int32_t* counts_out) {
  return a->count(bytes, n, offsets, n_req, counts_out);
}

void admission_destroy(GpuAdmissionDirect* a) { delete a; }

}  // namespace aigw_fast
