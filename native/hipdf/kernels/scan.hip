// Device-wide exclusive scan (int64) + whole-column reductions.
// Scan is the 3-phase scheme: block-local scan emitting per-block sums,
// python recursively scans the (small) sums array, then an add-offsets pass.
// Used by: filter compaction, string gather offsets, join match offsets,
// partition offsets. Reductions: wave shfl -> LDS -> one device atomic per
// block (guide §6 G12).
#include "hipdf_common.h"

#define SCAN_ITEMS 8  // elements per thread

// exclusive block scan of in[base .. base+2048) -> out, block sum -> sums
__global__ void k_scan_block(const int64_t* __restrict__ in,
                             int64_t* __restrict__ out,
                             int64_t* __restrict__ sums, int64_t n) {
  __shared__ int64_t wave_sums[HIPDF_BLOCK / WAVE];
  int64_t base = (int64_t)blockIdx.x * HIPDF_BLOCK * SCAN_ITEMS;
  int tid = threadIdx.x;
  int lane = lane_id();
  int wid = tid / WAVE;

  // per-thread sequential chunk
  int64_t vals[SCAN_ITEMS];
  int64_t tsum = 0;
  int64_t tbase = base + (int64_t)tid * SCAN_ITEMS;
#pragma unroll
  for (int k = 0; k < SCAN_ITEMS; ++k) {
    int64_t i = tbase + k;
    vals[k] = i < n ? in[i] : 0;
    tsum += vals[k];
  }
  // wave inclusive scan of per-thread sums
  int64_t incl = tsum;
  for (int off = 1; off < WAVE; off <<= 1) {
    int64_t up = __shfl_up(incl, off);
    if (lane >= off) incl += up;
  }
  if (lane == WAVE - 1) wave_sums[wid] = incl;
  __syncthreads();
  if (tid == 0) {
    int64_t acc = 0;
    for (int w = 0; w < HIPDF_BLOCK / WAVE; ++w) {
      int64_t c = wave_sums[w];
      wave_sums[w] = acc;
      acc += c;
    }
    sums[blockIdx.x] = acc;
  }
  __syncthreads();
  int64_t excl = wave_sums[wid] + incl - tsum;
#pragma unroll
  for (int k = 0; k < SCAN_ITEMS; ++k) {
    int64_t i = tbase + k;
    if (i < n) out[i] = excl;
    excl += vals[k];
  }
}

__global__ void k_scan_add_offsets(int64_t* __restrict__ out,
                                   const int64_t* __restrict__ scanned_sums,
                                   int64_t n) {
  int64_t off = scanned_sums[blockIdx.x];
  int64_t base = (int64_t)blockIdx.x * HIPDF_BLOCK * SCAN_ITEMS;
  for (int k = 0; k < SCAN_ITEMS; ++k) {
    int64_t i = base + (int64_t)threadIdx.x + (int64_t)k * HIPDF_BLOCK;
    if (i < n) out[i] += off;
  }
}

// ---- reductions ----------------------------------------------------------
enum RedOp : int { RED_SUM = 0, RED_MIN, RED_MAX, RED_COUNT };

__device__ __forceinline__ void atomic_min_i64(int64_t* p, int64_t v) {
  atomicMin((long long*)p, (long long)v);
}
__device__ __forceinline__ void atomic_max_i64(int64_t* p, int64_t v) {
  atomicMax((long long*)p, (long long)v);
}
__device__ __forceinline__ void atomic_min_f64(double* p, double v) {
  unsigned long long* up = (unsigned long long*)p;
  unsigned long long old = *up, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    if (!(v < cur)) break;
    old = atomicCAS(up, assumed, (unsigned long long)__double_as_longlong(v));
  } while (old != assumed);
}
__device__ __forceinline__ void atomic_max_f64(double* p, double v) {
  unsigned long long* up = (unsigned long long*)p;
  unsigned long long old = *up, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    if (!(v > cur)) break;
    old = atomicCAS(up, assumed, (unsigned long long)__double_as_longlong(v));
  } while (old != assumed);
}

// accumulate into acc[0] (int64 accum for integral, double accum for float);
// count of valid rows into acc_count[0]
template <typename T, typename ACC>
__global__ void k_reduce(int op, const T* __restrict__ a,
                         const uint64_t* __restrict__ av, ACC* __restrict__ acc,
                         int64_t* __restrict__ acc_count, int64_t n) {
  ACC local_sum = 0;
  ACC local_min = std::numeric_limits<ACC>::max();
  ACC local_max = std::numeric_limits<ACC>::lowest();
  int64_t local_count = 0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (!valid_bit(av, i)) continue;
    ACC v = (ACC)a[i];
    local_sum += v;
    local_min = v < local_min ? v : local_min;
    local_max = v > local_max ? v : local_max;
    local_count += 1;
  }
  // wave reduce
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    local_sum += __shfl_down(local_sum, off);
    ACC m = __shfl_down(local_min, off);
    local_min = m < local_min ? m : local_min;
    ACC M = __shfl_down(local_max, off);
    local_max = M > local_max ? M : local_max;
    local_count += __shfl_down(local_count, off);
  }
  if (lane_id() == 0) {
    if constexpr (std::is_same_v<ACC, double>) {
      if (op == RED_SUM) atomicAdd((double*)&acc[0], local_sum);
      if (op == RED_MIN && local_count) atomic_min_f64((double*)&acc[0], local_min);
      if (op == RED_MAX && local_count) atomic_max_f64((double*)&acc[0], local_max);
    } else {
      if (op == RED_SUM)
        atomicAdd((unsigned long long*)&acc[0], (unsigned long long)local_sum);
      if (op == RED_MIN && local_count) atomic_min_i64((int64_t*)&acc[0], local_min);
      if (op == RED_MAX && local_count) atomic_max_i64((int64_t*)&acc[0], local_max);
    }
    atomicAdd((unsigned long long*)&acc_count[0],
              (unsigned long long)local_count);
  }
}

extern "C" {

int64_t scan_num_blocks(int64_t n) {
  int64_t per = (int64_t)HIPDF_BLOCK * SCAN_ITEMS;
  int64_t nb = (n + per - 1) / per;
  return nb < 1 ? 1 : nb;
}

void hipdf_scan_block(const void* in, void* out, void* sums, int64_t n,
                      hipStream_t stream) {
  int64_t nb = scan_num_blocks(n);
  hipLaunchKernelGGL(k_scan_block, dim3((uint32_t)nb), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)in, (int64_t*)out, (int64_t*)sums,
                     n);
}

void hipdf_scan_add_offsets(void* out, const void* scanned_sums, int64_t n,
                            hipStream_t stream) {
  int64_t nb = scan_num_blocks(n);
  hipLaunchKernelGGL(k_scan_add_offsets, dim3((uint32_t)nb),
                     dim3(HIPDF_BLOCK), 0, stream, (int64_t*)out,
                     (const int64_t*)scanned_sums, n);
}

void hipdf_reduce(int op, int t, const void* a, const void* av, void* acc,
                  void* acc_count, int64_t n, hipStream_t stream) {
  dim3 grid = flat_grid(n, 4);
  dispatch_type(t, [&]<typename T>() {
    if constexpr (std::is_same_v<T, float> || std::is_same_v<T, double>) {
      hipLaunchKernelGGL((k_reduce<T, double>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const uint64_t*)av,
                         (double*)acc, (int64_t*)acc_count, n);
    } else {
      hipLaunchKernelGGL((k_reduce<T, int64_t>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const uint64_t*)av,
                         (int64_t*)acc, (int64_t*)acc_count, n);
    }
  });
}

}  // extern "C"
