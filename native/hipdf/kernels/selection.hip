// Row-selection kernels: stream-compaction (filter), gather (take) for
// fixed-width + string columns + validity bitmasks, bitmask range-copy for
// concat. These are the highest-call-count kernels in the engine (reference
// counts: Table.filter 91 call sites, gather 14 — SURVEY.md §2.8).
//
// Stream compaction is the classic 3-phase scheme: per-block counts ->
// exclusive scan of counts (device-wide, see scan.hip; orchestrated from
// python) -> scatter with intra-block prefix from wave ballots. Phase 1 and
// phase 3 iterate rows in the same order so offsets line up.
#include "hipdf_common.h"

#define SEL_ITEMS 4  // rows per thread per block pass

// ---- phase 1: per-block count of selected rows ---------------------------
__global__ void k_mask_count(const uint8_t* __restrict__ mask,
                             const uint64_t* __restrict__ mvalid,
                             int64_t* __restrict__ block_counts, int64_t n) {
  int64_t base = (int64_t)blockIdx.x * blockDim.x * SEL_ITEMS;
  int count = 0;
  for (int it = 0; it < SEL_ITEMS; ++it) {
    int64_t row = base + it * blockDim.x + threadIdx.x;
    bool sel = row < n && mask[row] != 0 && valid_bit(mvalid, row);
    count += sel ? 1 : 0;
  }
  // wave reduce then LDS
  for (int off = WAVE / 2; off > 0; off >>= 1)
    count += __shfl_down(count, off);
  __shared__ int warp_sums[HIPDF_BLOCK / WAVE];
  if (lane_id() == 0) warp_sums[threadIdx.x / WAVE] = count;
  __syncthreads();
  if (threadIdx.x == 0) {
    int total = 0;
    for (int w = 0; w < HIPDF_BLOCK / WAVE; ++w) total += warp_sums[w];
    block_counts[blockIdx.x] = total;
  }
}

// ---- phase 3: scatter selected row indices -------------------------------
__global__ void k_mask_scatter(const uint8_t* __restrict__ mask,
                               const uint64_t* __restrict__ mvalid,
                               const int64_t* __restrict__ block_offsets,
                               int32_t* __restrict__ out_idx, int64_t n) {
  __shared__ int64_t running;
  if (threadIdx.x == 0) running = block_offsets[blockIdx.x];
  __syncthreads();
  int64_t base = (int64_t)blockIdx.x * blockDim.x * SEL_ITEMS;
  __shared__ int warp_base[HIPDF_BLOCK / WAVE];
  for (int it = 0; it < SEL_ITEMS; ++it) {
    int64_t row = base + it * blockDim.x + threadIdx.x;
    bool sel = row < n && mask[row] != 0 && valid_bit(mvalid, row);
    uint64_t ballot = __ballot(sel);
    int wid = threadIdx.x / WAVE;
    if (lane_id() == 0) warp_base[wid] = __popcll(ballot);
    __syncthreads();
    // exclusive scan of the (up to 4) per-wave counts by thread 0
    if (threadIdx.x == 0) {
      int acc = 0;
      for (int w = 0; w < HIPDF_BLOCK / WAVE; ++w) {
        int c = warp_base[w];
        warp_base[w] = acc;
        acc += c;
      }
      warp_base[0] |= acc << 16;  // stash total in high bits of slot 0
    }
    __syncthreads();
    int wbase = warp_base[wid] & 0xFFFF;
    int total = warp_base[0] >> 16;
    if (sel) {
      int prefix = __popcll(ballot & ((lane_id() == 0) ? 0ull
                            : (~0ull >> (64 - lane_id()))));
      out_idx[running + wbase + prefix] = (int32_t)row;
    }
    __syncthreads();
    if (threadIdx.x == 0) running += total;
    __syncthreads();
  }
}

// ---- gather --------------------------------------------------------------
template <typename T>
__global__ void k_gather_fixed(const T* __restrict__ in,
                               const int32_t* __restrict__ idx,
                               T* __restrict__ out, int64_t n_out) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n_out;
       j += (int64_t)gridDim.x * blockDim.x) {
    int32_t i = idx[j];
    T zero{};
    out[j] = i >= 0 ? in[i] : zero;
  }
}

__global__ void k_gather_validity(const uint64_t* __restrict__ in_valid,
                                  int in_has_valid,
                                  const int32_t* __restrict__ idx,
                                  uint64_t* __restrict__ out_valid,
                                  int64_t nstripe, int64_t n_out) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t j = s * WAVE + lane;
    bool ok = false;
    if (j < n_out) {
      int32_t i = idx[j];
      ok = i >= 0 && (!in_has_valid || valid_bit(in_valid, i));
    }
    uint64_t ballot = __ballot(ok);
    if (lane == 0) out_valid[s] = ballot;
  }
}

// fused whole-table gather for fixed-width columns: one launch gathers
// every column (the index load and ballot amortize across columns)
struct GatherCol {
  int esize;
  int in_has_valid;
  const void* in;
  const uint64_t* in_valid;
  void* out;
  uint64_t* out_valid;  // may be null
};

__global__ void k_gather_table(const GatherCol* __restrict__ cols, int ncols,
                               const int32_t* __restrict__ idx,
                               int64_t nstripe, int64_t n_out) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t j = s * WAVE + lane;
    int32_t i = j < n_out ? idx[j] : -1;
    for (int c = 0; c < ncols; ++c) {
      const GatherCol& g = cols[c];
      if (j < n_out) {
        int32_t src = i >= 0 ? i : 0;
        switch (g.esize) {
          case 1: ((uint8_t*)g.out)[j] = i >= 0 ? ((const uint8_t*)g.in)[src] : 0; break;
          case 2: ((uint16_t*)g.out)[j] = i >= 0 ? ((const uint16_t*)g.in)[src] : 0; break;
          case 4: ((uint32_t*)g.out)[j] = i >= 0 ? ((const uint32_t*)g.in)[src] : 0; break;
          case 16: {
            ulonglong2 z{0, 0};
            ((ulonglong2*)g.out)[j] = i >= 0 ? ((const ulonglong2*)g.in)[src] : z;
            break;
          }
          default: ((uint64_t*)g.out)[j] = i >= 0 ? ((const uint64_t*)g.in)[src] : 0; break;
        }
      }
      if (g.out_valid) {
        bool ok = j < n_out && i >= 0 &&
                  (!g.in_has_valid || valid_bit(g.in_valid, i));
        uint64_t ballot = __ballot(ok);
        if (lane == 0) g.out_valid[s] = ballot;
      }
    }
  }
}

// string gather phase 1: per-output-row byte length
__global__ void k_gather_str_lens(const int32_t* __restrict__ offsets,
                                  const int32_t* __restrict__ idx,
                                  int64_t* __restrict__ lens, int64_t n_out) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n_out;
       j += (int64_t)gridDim.x * blockDim.x) {
    int32_t i = idx[j];
    lens[j] = i >= 0 ? (int64_t)(offsets[i + 1] - offsets[i]) : 0;
  }
}

// string gather phase 2: wave-per-row byte copy using the scanned offsets
__global__ void k_gather_str_bytes(const uint8_t* __restrict__ in_bytes,
                                   const int32_t* __restrict__ in_offsets,
                                   const int32_t* __restrict__ idx,
                                   const int64_t* __restrict__ out_offsets,
                                   uint8_t* __restrict__ out_bytes,
                                   int64_t n_out) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t j = wave_global; j < n_out; j += wave_count) {
    int32_t i = idx[j];
    if (i < 0) continue;
    int32_t src = in_offsets[i];
    int32_t len = in_offsets[i + 1] - src;
    int64_t dst = out_offsets[j];
    for (int b = lane; b < len; b += WAVE) out_bytes[dst + b] = in_bytes[src + b];
  }
}

// thread-per-row variant for SHORT strings (dictionary keys, state codes,
// ids: avg len well under a wavefront). The wave-per-row kernel leaves
// (WAVE - len) lanes idle per row — at len 16 that is 75% of the machine;
// here every lane owns a row, so a wave moves 64 rows' bytes per
// iteration instead of one row's. Measured (r02 roofline): wave-per-row
// ran at 8% of HBM roof on the NDS string gathers.
__global__ void k_gather_str_bytes_tpr(const uint8_t* __restrict__ in_bytes,
                                       const int32_t* __restrict__ in_offsets,
                                       const int32_t* __restrict__ idx,
                                       const int64_t* __restrict__ out_offsets,
                                       uint8_t* __restrict__ out_bytes,
                                       int64_t n_out) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n_out;
       j += (int64_t)gridDim.x * blockDim.x) {
    int32_t i = idx[j];
    if (i < 0) continue;
    int32_t src = in_offsets[i];
    int32_t len = in_offsets[i + 1] - src;
    int64_t dst = out_offsets[j];
    int b = 0;
    for (; b + 4 <= len; b += 4) {
      out_bytes[dst + b] = in_bytes[src + b];
      out_bytes[dst + b + 1] = in_bytes[src + b + 1];
      out_bytes[dst + b + 2] = in_bytes[src + b + 2];
      out_bytes[dst + b + 3] = in_bytes[src + b + 3];
    }
    for (; b < len; ++b) out_bytes[dst + b] = in_bytes[src + b];
  }
}

// copy int64 lens -> int32 offsets tail (offsets[j+1]=scan[j]+len[j] handled
// in python by scanning; here: narrow an int64 array into int32)
__global__ void k_narrow_i64_i32(const int64_t* __restrict__ in,
                                 int32_t* __restrict__ out, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x)
    out[j] = (int32_t)in[j];
}

// ---- bitmask range copy for concat ---------------------------------------
// OR n bits of src (starting at src bit 0) into dst starting at dst_off.
// dst must be pre-zeroed; edge words may be shared across concat inputs so
// all writes are atomicOr (device-scope, XCD-safe).
__global__ void k_copy_valid_range(const uint64_t* __restrict__ src,
                                   int src_has_valid, int64_t dst_off,
                                   unsigned long long* __restrict__ dst,
                                   int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t src_row = s * WAVE + lane;
    bool ok = src_row < n && (!src_has_valid || valid_bit(src, src_row));
    uint64_t ballot = __ballot(ok);
    if (lane == 0 && ballot) {
      int64_t bit = dst_off + s * WAVE;
      int64_t w = bit >> 6;
      int sh = (int)(bit & 63);
      atomicOr(&dst[w], (unsigned long long)(ballot << sh));
      if (sh && (ballot >> (64 - sh)))
        atomicOr(&dst[w + 1], (unsigned long long)(ballot >> (64 - sh)));
    }
  }
}

// ---- host entry points ---------------------------------------------------
// explode: for each input row, its list span [off[i], off[i+1]) emits one
// output row per element; rowid/pos map built wave-per-row (long lists
// copy with full wave parallelism).
__global__ void k_expand_rows(const int32_t* __restrict__ offsets,
                              int32_t* __restrict__ rowid,
                              int32_t* __restrict__ pos, int64_t nrows) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t i = wave_global; i < nrows; i += wave_count) {
    int32_t a = offsets[i], b = offsets[i + 1];
    for (int32_t j = a + lane; j < b; j += WAVE) {
      rowid[j] = (int32_t)i;
      pos[j] = j - a;
    }
  }
}

extern "C" {

void hipdf_expand_rows(const void* offsets, void* rowid, void* pos,
                       int64_t nrows, hipStream_t stream) {
  hipLaunchKernelGGL(k_expand_rows, flat_grid(nrows), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)offsets, (int32_t*)rowid,
                     (int32_t*)pos, nrows);
}


int64_t sel_num_blocks(int64_t n) {
  return (n + (int64_t)HIPDF_BLOCK * SEL_ITEMS - 1) /
         ((int64_t)HIPDF_BLOCK * SEL_ITEMS);
}

void hipdf_mask_count(const void* mask, const void* mvalid,
                      void* block_counts, int64_t n, hipStream_t stream) {
  int64_t nb = sel_num_blocks(n);
  hipLaunchKernelGGL(k_mask_count, dim3((uint32_t)nb), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)mask, (const uint64_t*)mvalid,
                     (int64_t*)block_counts, n);
}

void hipdf_mask_scatter(const void* mask, const void* mvalid,
                        const void* block_offsets, void* out_idx, int64_t n,
                        hipStream_t stream) {
  int64_t nb = sel_num_blocks(n);
  hipLaunchKernelGGL(k_mask_scatter, dim3((uint32_t)nb), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)mask, (const uint64_t*)mvalid,
                     (const int64_t*)block_offsets, (int32_t*)out_idx, n);
}

void hipdf_gather_fixed(int esize, const void* in, const void* idx, void* out,
                        int64_t n_out, hipStream_t stream) {
  dim3 grid = flat_grid(n_out);
  switch (esize) {
    case 1:
      hipLaunchKernelGGL((k_gather_fixed<uint8_t>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, (const uint8_t*)in, (const int32_t*)idx,
                         (uint8_t*)out, n_out);
      break;
    case 2:
      hipLaunchKernelGGL((k_gather_fixed<uint16_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint16_t*)in, (const int32_t*)idx,
                         (uint16_t*)out, n_out);
      break;
    case 4:
      hipLaunchKernelGGL((k_gather_fixed<uint32_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint32_t*)in, (const int32_t*)idx,
                         (uint32_t*)out, n_out);
      break;
    case 8:
      hipLaunchKernelGGL((k_gather_fixed<uint64_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint64_t*)in, (const int32_t*)idx,
                         (uint64_t*)out, n_out);
      break;
    case 16:
      hipLaunchKernelGGL((k_gather_fixed<ulonglong2>), grid,
                         dim3(HIPDF_BLOCK), 0, stream, (const ulonglong2*)in,
                         (const int32_t*)idx, (ulonglong2*)out, n_out);
      break;
    default:
      throw std::runtime_error("gather: bad element size");
  }
}

void hipdf_gather_table(const void* cols, int ncols, const void* idx,
                        int64_t n_out, hipStream_t stream) {
  hipLaunchKernelGGL(k_gather_table, stripe_grid(n_out), dim3(HIPDF_BLOCK), 0,
                     stream, (const GatherCol*)cols, ncols,
                     (const int32_t*)idx, n_stripes(n_out), n_out);
}

void hipdf_gather_validity(const void* in_valid, int in_has_valid,
                           const void* idx, void* out_valid, int64_t n_out,
                           hipStream_t stream) {
  hipLaunchKernelGGL(k_gather_validity, stripe_grid(n_out), dim3(HIPDF_BLOCK),
                     0, stream, (const uint64_t*)in_valid, in_has_valid,
                     (const int32_t*)idx, (uint64_t*)out_valid,
                     n_stripes(n_out), n_out);
}

void hipdf_gather_str_lens(const void* offsets, const void* idx, void* lens,
                           int64_t n_out, hipStream_t stream) {
  hipLaunchKernelGGL(k_gather_str_lens, flat_grid(n_out), dim3(HIPDF_BLOCK),
                     0, stream, (const int32_t*)offsets, (const int32_t*)idx,
                     (int64_t*)lens, n_out);
}

void hipdf_gather_str_bytes(const void* in_bytes, const void* in_offsets,
                            const void* idx, const void* out_offsets,
                            void* out_bytes, int64_t n_out,
                            int64_t total_bytes, hipStream_t stream) {
  // short strings (avg <= 32 B): thread-per-row keeps all 64 lanes busy;
  // long strings: wave-per-row for coalesced within-row copies
  if (n_out > 0 && total_bytes >= 0 && total_bytes <= 32 * n_out) {
    hipLaunchKernelGGL(k_gather_str_bytes_tpr, flat_grid(n_out),
                       dim3(HIPDF_BLOCK), 0, stream,
                       (const uint8_t*)in_bytes, (const int32_t*)in_offsets,
                       (const int32_t*)idx, (const int64_t*)out_offsets,
                       (uint8_t*)out_bytes, n_out);
    return;
  }
  int64_t blocks = (n_out * WAVE + HIPDF_BLOCK - 1) / HIPDF_BLOCK;
  if (blocks > 4 * HIPDF_MAX_BLOCKS) blocks = 4 * HIPDF_MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_gather_str_bytes, dim3((uint32_t)blocks),
                     dim3(HIPDF_BLOCK), 0, stream, (const uint8_t*)in_bytes,
                     (const int32_t*)in_offsets, (const int32_t*)idx,
                     (const int64_t*)out_offsets, (uint8_t*)out_bytes, n_out);
}

void hipdf_narrow_i64_i32(const void* in, void* out, int64_t n,
                          hipStream_t stream) {
  hipLaunchKernelGGL(k_narrow_i64_i32, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)in, (int32_t*)out, n);
}

void hipdf_copy_valid_range(const void* src, int src_has_valid,
                            int64_t dst_off, void* dst, int64_t n,
                            hipStream_t stream) {
  hipLaunchKernelGGL(k_copy_valid_range, stripe_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint64_t*)src, src_has_valid, dst_off,
                     (unsigned long long*)dst, n_stripes(n), n);
}

}  // extern "C"
