// Byte-level BPE tokenizer kernels (gfx950 / CDNA4), shared between the
// torch extension (csrc/aigw_kernels.hip, Python data plane) and the
// torch-free native admission path (csrc/admission.hip, C++ fast front).
// Definitions live in this header because the two users are separate
// shared objects; each translation unit that includes it gets its own
// copy (no cross-.so linking).
#pragma once

#include <hip/hip_runtime.h>

#include <climits>
#include <cstdint>

// Byte classes for the segmentation rule (must match aigw/ops/bpe_ref.py):
// segment starts at i iff i==0, byte==' ', or class(b[i])!=class(b[i-1]) and
// b[i-1]!=' '.
__device__ __forceinline__ int byte_class(uint8_t b) {
  if (b == ' ') return 0;
  if (b == '\t' || b == '\n' || b == '\r' || b == 0x0b || b == 0x0c) return 1;
  if (b >= '0' && b <= '9') return 2;
  if ((b >= 'A' && b <= 'Z') || (b >= 'a' && b <= 'z') || b >= 0x80) return 3;
  return 4;  // punctuation / other
}

__global__ void seg_flags_kernel(const uint8_t* __restrict__ bytes, int n,
                                 uint8_t* __restrict__ flags) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint8_t f;
  if (i == 0) {
    f = 1;
  } else {
    uint8_t b = bytes[i], p = bytes[i - 1];
    f = (b == ' ') || (byte_class(b) != byte_class(p) && p != ' ');
  }
  flags[i] = f;
}

__global__ void seg_force_starts_kernel(const int64_t* __restrict__ req_off,
                                        int n_req, uint8_t* __restrict__ flags) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n_req) flags[req_off[i]] = 1;
}

// per-256-byte-block flag counts (for the ordered compaction scan)
__global__ void seg_block_count_kernel(const uint8_t* __restrict__ flags, int n,
                                       int32_t* __restrict__ blk_counts) {
  __shared__ int cnt;
  if (threadIdx.x == 0) cnt = 0;
  __syncthreads();
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t ballot = __ballot(i < n && flags[i]);
  if ((threadIdx.x & 63) == 0) atomicAdd(&cnt, __popcll(ballot));
  __syncthreads();
  if (threadIdx.x == 0) blk_counts[blockIdx.x] = cnt;
}

// write segment start offsets + owning request id (ordered compaction)
__global__ void seg_write_kernel(const uint8_t* __restrict__ flags, int n,
                                 const int32_t* __restrict__ blk_excl,  // exclusive scan
                                 const int64_t* __restrict__ req_off, int n_req,
                                 int32_t* __restrict__ seg_start,
                                 int32_t* __restrict__ seg_req) {
  __shared__ int wave_base[4];
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  bool f = (i < n) && flags[i];
  uint64_t ballot = __ballot(f);
  // block-local exclusive offsets: wave sums staged through LDS
  if (lane == 0) wave_base[wave] = __popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    int acc = 0;
    for (int w = 0; w < 4; ++w) {
      int c = wave_base[w];
      wave_base[w] = acc;
      acc += c;
    }
  }
  __syncthreads();
  if (!f) return;
  int pos_in_wave = __popcll(ballot & ((1ull << lane) - 1ull));
  int pos = blk_excl[blockIdx.x] + wave_base[wave] + pos_in_wave;
  seg_start[pos] = i;
  // owning request: upper_bound(req_off, i) - 1
  int lo = 0, hi = n_req;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (req_off[mid] <= i) lo = mid + 1; else hi = mid;
  }
  seg_req[pos] = lo - 1;
}

// The BPE merge loop, shared by both scheduling paths. Tokens live in lane
// registers; the merge-pair table is an open-addressing hash in HBM (hot
// entries L2-resident). Pair ranks are wave-min-reduced with __shfl_xor;
// occurrence selection and merging operate on wave-uniform 64-bit masks.
// `myseg` confines merges to a segment: a pair is valid only when both
// lanes carry the same segment id (constant 0 for the single-segment
// path), so packing several small segments into one wave changes nothing
// semantically — within each segment the lowest-rank pair present still
// merges first (cross-segment rank interleaving cannot reorder merges
// inside a segment).
__device__ __forceinline__ uint64_t bpe_merge_lanes(
    int& tok, int myseg, uint64_t active, int lane,
    const long long* __restrict__ htab_keys, const int32_t* __restrict__ htab_rank,
    int htab_mask) {
  // pair-rank memo: a merge round only changes the pairs adjacent to a
  // merge site, so most lanes can reuse last round's hash-probe result
  // (the probe's L2 loads dominate the VALU-bound loop otherwise)
  long long cached_key = -1;
  int cached_rank = INT_MAX;
  for (;;) {
    // next active lane above mine
    uint64_t above = (lane < 63) ? (active & (~0ull << (lane + 1))) : 0ull;
    int nxt = above ? (__ffsll((long long)above) - 1) : -1;
    int nxttok = __shfl(tok, nxt < 0 ? 0 : nxt);
    int nxtseg = __shfl(myseg, nxt < 0 ? 0 : nxt);
    int rank = INT_MAX;
    if (tok >= 0 && nxt >= 0 && nxtseg == myseg) {
      long long key = ((long long)tok << 32) | (unsigned)nxttok;
      if (key == cached_key) {
        rank = cached_rank;
      } else {
        uint64_t h = (uint64_t)key * 0x9E3779B97F4A7C15ull;
        int idx = (int)(h >> 40) & htab_mask;
        for (;;) {
          long long k = htab_keys[idx];
          if (k == key) { rank = htab_rank[idx]; break; }
          if (k == -1) break;
          idx = (idx + 1) & htab_mask;
        }
        cached_key = key;
        cached_rank = rank;
      }
    }
    // wave min-reduce of rank
    int minrank = rank;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      minrank = min(minrank, __shfl_xor(minrank, off));
    if (minrank == INT_MAX) return active;

    uint64_t occ = __ballot(rank == minrank);
    // greedy leftmost non-overlapping selection (wave-uniform scalar loop)
    uint64_t m = occ, sel = 0;
    while (m) {
      int i = __ffsll((long long)m) - 1;
      sel |= 1ull << i;
      int ni = __shfl(nxt, i);
      m &= ~(1ull << i);
      if (ni >= 0) m &= ~(1ull << ni);
    }
    int newid = 256 + minrank;
    // previous active lane (my potential merge head)
    uint64_t below = active & ((lane ? (1ull << lane) : 1ull) - 1ull);
    int prev = below ? (63 - __clzll((long long)below)) : -1;
    bool merged_into_prev = (prev >= 0) && ((sel >> prev) & 1);
    if ((sel >> lane) & 1) tok = newid;
    if (merged_into_prev) tok = -1;
    active = __ballot(tok >= 0);
  }
}

// Single-segment path: 64-byte chunks of one segment per wave (also used
// for group tail segments that do not fit the packed window). Returns the
// number of tokens written.
__device__ __forceinline__ int bpe_encode_one_segment(
    const uint8_t* __restrict__ bytes, int s, int e, int lane,
    const long long* __restrict__ htab_keys, const int32_t* __restrict__ htab_rank,
    int htab_mask, int32_t* __restrict__ out_ids) {
  int written = 0;
  for (int chunk = s; chunk < e; chunk += 64) {
    int len = min(e - chunk, 64);
    int tok = (lane < len) ? (int)bytes[chunk + lane] : -1;
    uint64_t active = __ballot(tok >= 0);
    active = bpe_merge_lanes(tok, 0, active, lane, htab_keys, htab_rank, htab_mask);
    int cnt = __popcll(active);
    if (tok >= 0) {
      int pos = __popcll(active & ((lane ? (1ull << lane) : 1ull) - 1ull));
      out_ids[s + written + pos] = tok;
    }
    written += cnt;
  }
  return written;
}

// Grouped scheduling: one wave per GROUP of consecutive segments (group =
// segments of one request whose starts share a 32-byte cell relative to
// the request start, computed by group_head_flags_kernel). Grouping is
// scheduling-only — token output is bit-identical to segment-per-wave.
// Typical English text has ~5-byte segments, so segment-per-wave leaves
// ~58/64 lanes idle; packing a whole group into the lanes recovers wave
// occupancy. The 32-byte cell rule bounds the packed window: all group
// segments start within one 32-byte cell, so the span up to the last
// segment's start is < 32 bytes, and the whole group is packed whenever
// it ends within 64 bytes of the group start (long tails fall back to the
// chunked path).
__device__ __forceinline__ void bpe_encode_grouped_body(
    const uint8_t* __restrict__ bytes, const int32_t* __restrict__ seg_start,
    const int32_t* __restrict__ seg_req, int n_segs, int n_bytes,
    const int32_t* __restrict__ ghead, int n_groups,
    const long long* __restrict__ htab_keys,
    const int32_t* __restrict__ htab_rank, int htab_mask,
    int32_t* __restrict__ out_ids, int32_t* __restrict__ req_counts) {
  int g = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  if (g >= n_groups) return;
  int lane = threadIdx.x & 63;
  int first = ghead[g];
  int next_first = (g + 1 < n_groups) ? ghead[g + 1] : n_segs;
  int lastseg = next_first - 1;
  int s0 = seg_start[first];
  int sE = (next_first < n_segs) ? seg_start[next_first] : n_bytes;
  int written = 0;

  if (lastseg == first) {
    written = bpe_encode_one_segment(bytes, s0, sE, lane, htab_keys, htab_rank,
                                     htab_mask, out_ids);
  } else {
    int sL = seg_start[lastseg];
    bool include_last = (sE - s0) <= 64;
    int win_end = include_last ? sE : sL;
    int pos = s0 + lane;
    int tok = (pos < win_end) ? (int)bytes[pos] : -1;
    // reconstruct per-lane segment-start flags (same rule as
    // seg_flags_kernel; no request boundary can occur inside a group)
    bool f = false;
    if (pos < win_end) {
      if (pos == s0) {
        f = true;
      } else {
        uint8_t b = bytes[pos], pb = bytes[pos - 1];
        f = (b == ' ') || (byte_class(b) != byte_class(pb) && pb != ' ');
      }
    }
    uint64_t startmask = __ballot(f);
    int myseg = __popcll(startmask & ((lane ? (1ull << lane) : 1ull) - 1ull)) + (f ? 1 : 0);
    uint64_t active = __ballot(tok >= 0);
    active = bpe_merge_lanes(tok, myseg, active, lane, htab_keys, htab_rank, htab_mask);
    written = __popcll(active);
    if (tok >= 0) {
      // my segment's first lane = highest start flag at or below me; its
      // lane offset equals the segment's byte offset within the window
      uint64_t at_or_below =
          startmask & ((lane < 63 ? (1ull << (lane + 1)) : 0ull) - 1ull);
      int start_lane = 63 - __clzll((long long)at_or_below);
      int pos_in_seg =
          __popcll(active & ((lane ? (1ull << lane) : 1ull) - 1ull) &
                   ~((start_lane ? (1ull << start_lane) : 1ull) - 1ull));
      out_ids[s0 + start_lane + pos_in_seg] = tok;
    }
    if (!include_last)
      written += bpe_encode_one_segment(bytes, sL, sE, lane, htab_keys, htab_rank,
                                        htab_mask, out_ids);
  }
  if (lane == 0 && written > 0) atomicAdd(&req_counts[seg_req[first]], written);
}

__global__ void __launch_bounds__(256)
bpe_encode_grouped_kernel(const uint8_t* __restrict__ bytes,
                          const int32_t* __restrict__ seg_start,
                          const int32_t* __restrict__ seg_req, int n_segs,
                          int n_bytes,
                          const int32_t* __restrict__ ghead, int n_groups,
                          const long long* __restrict__ htab_keys,
                          const int32_t* __restrict__ htab_rank, int htab_mask,
                          int32_t* __restrict__ out_ids,
                          int32_t* __restrict__ req_counts) {
  bpe_encode_grouped_body(bytes, seg_start, seg_req, n_segs, n_bytes, ghead,
                          n_groups, htab_keys, htab_rank, htab_mask, out_ids,
                          req_counts);
}

// Sync-free variant: segment/group totals live in DEVICE scalars (the scan
// tails), so the host never calls .item() — the whole pipeline is launched
// blind with upper-bound grids and each kernel self-bounds. This is what
// lets the serving path await a hipEvent cooperatively instead of
// busy-polling a host sync (ROCm host syncs spin a core).
__global__ void __launch_bounds__(256)
bpe_encode_grouped_dev_kernel(const uint8_t* __restrict__ bytes,
                              const int32_t* __restrict__ seg_start,
                              const int32_t* __restrict__ seg_req,
                              const int32_t* __restrict__ n_segs_dev,
                              int n_bytes,
                              const int32_t* __restrict__ ghead,
                              const int32_t* __restrict__ n_groups_dev,
                              const long long* __restrict__ htab_keys,
                              const int32_t* __restrict__ htab_rank, int htab_mask,
                              int32_t* __restrict__ out_ids,
                              int32_t* __restrict__ req_counts) {
  bpe_encode_grouped_body(bytes, seg_start, seg_req, *n_segs_dev, n_bytes, ghead,
                          *n_groups_dev, htab_keys, htab_rank, htab_mask, out_ids,
                          req_counts);
}

// group-head flags over the segment array: a group breaks on request
// change or a new 32-byte cell (relative to the request start)
__global__ void group_head_flags_kernel(const int32_t* __restrict__ seg_start,
                                        const int32_t* __restrict__ seg_req,
                                        int n_segs,
                                        const int64_t* __restrict__ req_off,
                                        uint8_t* __restrict__ flags) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_segs) return;
  uint8_t f;
  if (i == 0 || seg_req[i] != seg_req[i - 1]) {
    f = 1;
  } else {
    long long off = req_off[seg_req[i]];
    f = ((seg_start[i] - off) >> 5) != ((seg_start[i - 1] - off) >> 5);
  }
  flags[i] = f;
}

__global__ void group_head_flags_dev_kernel(const int32_t* __restrict__ seg_start,
                                            const int32_t* __restrict__ seg_req,
                                            const int32_t* __restrict__ n_segs_dev,
                                            const int64_t* __restrict__ req_off,
                                            uint8_t* __restrict__ flags) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= *n_segs_dev) return;
  uint8_t f;
  if (i == 0 || seg_req[i] != seg_req[i - 1]) {
    f = 1;
  } else {
    long long off = req_off[seg_req[i]];
    f = ((seg_start[i] - off) >> 5) != ((seg_start[i - 1] - off) >> 5);
  }
  flags[i] = f;
}

// generic ordered index-compaction write (flag positions -> indices)
__global__ void flag_compact_write_kernel(const uint8_t* __restrict__ flags, int n,
                                          const int32_t* __restrict__ blk_excl,
                                          int32_t* __restrict__ out_idx) {
  __shared__ int wave_base[4];
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  bool f = (i < n) && flags[i];
  uint64_t ballot = __ballot(f);
  if (lane == 0) wave_base[wave] = __popcll(ballot);
  __syncthreads();
  if (threadIdx.x == 0) {
    int acc = 0;
    for (int w = 0; w < 4; ++w) {
      int c = wave_base[w];
      wave_base[w] = acc;
      acc += c;
    }
  }
  __syncthreads();
  if (!f) return;
  int pos_in_wave = __popcll(ballot & ((1ull << lane) - 1ull));
  out_idx[blk_excl[blockIdx.x] + wave_base[wave] + pos_in_wave] = i;
}

