// Window building blocks (reference analogue: the GpuWindowExec family's
// device pieces — SURVEY.md §2.4 window variants). The heavy lifting
// (sorting, scans, per-segment aggregation, gathers) reuses the sort /
// scan / groupby kernels; this file adds the segment machinery:
//  - change flags: row differs from the previous row on the given keys
//  - iota: row index column
//  - double-typed device-wide scan blocks (running sums over f64)
#include "hipdf_common.h"
#include "keys.h"

__global__ void k_change_flags(const KeyCol* __restrict__ keys, int nkeys,
                               uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = (i == 0 || !rows_equal(keys, keys, nkeys, i, i - 1)) ? 1 : 0;
  }
}

__global__ void k_iota_i32(int32_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int32_t)i;
}

#define SCAN_ITEMS_W 8

__global__ void k_scan_block_f64(const double* __restrict__ in,
                                 double* __restrict__ out,
                                 double* __restrict__ sums, int64_t n) {
  __shared__ double wave_sums[HIPDF_BLOCK / WAVE];
  int64_t base = (int64_t)blockIdx.x * HIPDF_BLOCK * SCAN_ITEMS_W;
  int tid = threadIdx.x;
  int lane = lane_id();
  int wid = tid / WAVE;
  double vals[SCAN_ITEMS_W];
  double tsum = 0;
  int64_t tbase = base + (int64_t)tid * SCAN_ITEMS_W;
#pragma unroll
  for (int k = 0; k < SCAN_ITEMS_W; ++k) {
    int64_t i = tbase + k;
    vals[k] = i < n ? in[i] : 0.0;
    tsum += vals[k];
  }
  double incl = tsum;
  for (int off = 1; off < WAVE; off <<= 1) {
    double up = __shfl_up(incl, off);
    if (lane >= off) incl += up;
  }
  if (lane == WAVE - 1) wave_sums[wid] = incl;
  __syncthreads();
  if (tid == 0) {
    double acc = 0;
    for (int w = 0; w < HIPDF_BLOCK / WAVE; ++w) {
      double c = wave_sums[w];
      wave_sums[w] = acc;
      acc += c;
    }
    sums[blockIdx.x] = acc;
  }
  __syncthreads();
  double excl = wave_sums[wid] + incl - tsum;
#pragma unroll
  for (int k = 0; k < SCAN_ITEMS_W; ++k) {
    int64_t i = tbase + k;
    if (i < n) out[i] = excl;
    excl += vals[k];
  }
}

__global__ void k_scan_add_offsets_f64(double* __restrict__ out,
                                       const double* __restrict__ sums,
                                       int64_t n) {
  double off = sums[blockIdx.x];
  int64_t base = (int64_t)blockIdx.x * HIPDF_BLOCK * SCAN_ITEMS_W;
  for (int k = 0; k < SCAN_ITEMS_W; ++k) {
    int64_t i = base + (int64_t)threadIdx.x + (int64_t)k * HIPDF_BLOCK;
    if (i < n) out[i] += off;
  }
}

// RANGE frame bounds: for each row, binary-search its segment (ascending
// order key, cast to double) for the first/last row whose key lies in
// [key_i + lo, key_i + hi]; unbounded ends snap to the segment edge.
__global__ void k_range_bounds(const double* __restrict__ vals,
                               const int32_t* __restrict__ seg_start,
                               const int32_t* __restrict__ seg_end,
                               double lo, double hi, int lo_unb, int hi_unb,
                               int32_t* __restrict__ a_idx,
                               int32_t* __restrict__ b_idx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = seg_start[i], e0 = seg_end[i];
    double v = vals[i];
    int32_t a = s0, b = e0;
    if (!lo_unb) {
      double target = v + lo;
      int32_t l = s0, r = e0 + 1;  // first idx with vals[idx] >= target
      while (l < r) {
        int32_t m = l + (r - l) / 2;
        if (vals[m] < target) l = m + 1;
        else r = m;
      }
      a = l;
    }
    if (!hi_unb) {
      double target = v + hi;
      int32_t l = s0, r = e0 + 1;  // first idx with vals[idx] > target
      while (l < r) {
        int32_t m = l + (r - l) / 2;
        if (vals[m] <= target) l = m + 1;
        else r = m;
      }
      b = l - 1;
    }
    a_idx[i] = a;
    b_idx[i] = b;
  }
}

// sparse-table range min/max query: levels[l][i] covers [i, i+2^l-1];
// answer = op(levels[k][a], levels[k][b-2^k+1]) with k = floor(log2(len)).
// Frames never cross partition boundaries, so the table builds globally.
template <typename T>
__global__ void k_win_minmax(const int64_t* __restrict__ level_ptrs,
                             int nlevels, const int32_t* __restrict__ a_idx,
                             const int32_t* __restrict__ b_idx, int is_min,
                             T* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t a = a_idx[i], b = b_idx[i];
    if (b < a) {
      out[i] = (T)0;
      continue;
    }
    int64_t len = (int64_t)b - a + 1;
    int k = 63 - __clzll((unsigned long long)len);
    if (k >= nlevels) k = nlevels - 1;
    const T* lv = (const T*)level_ptrs[k];
    T x = lv[a];
    T y = lv[b - (1ll << k) + 1];
    out[i] = is_min ? (x < y ? x : y) : (x > y ? x : y);
  }
}

extern "C" {

void hipdf_win_minmax(int is_double, const void* level_ptrs, int nlevels,
                      const void* a_idx, const void* b_idx, int is_min,
                      void* out, int64_t n, hipStream_t stream) {
  if (is_double)
    hipLaunchKernelGGL((k_win_minmax<double>), flat_grid(n),
                       dim3(HIPDF_BLOCK), 0, stream,
                       (const int64_t*)level_ptrs, nlevels,
                       (const int32_t*)a_idx, (const int32_t*)b_idx, is_min,
                       (double*)out, n);
  else
    hipLaunchKernelGGL((k_win_minmax<int64_t>), flat_grid(n),
                       dim3(HIPDF_BLOCK), 0, stream,
                       (const int64_t*)level_ptrs, nlevels,
                       (const int32_t*)a_idx, (const int32_t*)b_idx, is_min,
                       (int64_t*)out, n);
}


void hipdf_range_bounds(const void* vals, const void* seg_start,
                        const void* seg_end, double lo, double hi,
                        int lo_unb, int hi_unb, void* a_idx, void* b_idx,
                        int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_range_bounds, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const double*)vals, (const int32_t*)seg_start,
                     (const int32_t*)seg_end, lo, hi, lo_unb, hi_unb,
                     (int32_t*)a_idx, (int32_t*)b_idx, n);
}


void hipdf_change_flags(const void* keys, int nkeys, void* out, int64_t n,
                        hipStream_t stream) {
  hipLaunchKernelGGL(k_change_flags, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const KeyCol*)keys, nkeys, (uint8_t*)out, n);
}

void hipdf_iota_i32(void* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_iota_i32, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (int32_t*)out, n);
}

void hipdf_scan_block_f64(const void* in, void* out, void* sums, int64_t n,
                          hipStream_t stream) {
  int64_t per = (int64_t)HIPDF_BLOCK * SCAN_ITEMS_W;
  int64_t nb = (n + per - 1) / per;
  if (nb < 1) nb = 1;
  hipLaunchKernelGGL(k_scan_block_f64, dim3((uint32_t)nb), dim3(HIPDF_BLOCK),
                     0, stream, (const double*)in, (double*)out, (double*)sums,
                     n);
}

void hipdf_scan_add_offsets_f64(void* out, const void* sums, int64_t n,
                                hipStream_t stream) {
  int64_t per = (int64_t)HIPDF_BLOCK * SCAN_ITEMS_W;
  int64_t nb = (n + per - 1) / per;
  if (nb < 1) nb = 1;
  hipLaunchKernelGGL(k_scan_add_offsets_f64, dim3((uint32_t)nb),
                     dim3(HIPDF_BLOCK), 0, stream, (double*)out,
                     (const double*)sums, n);
}

}  // extern "C"
