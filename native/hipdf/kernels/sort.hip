// Stable LSD radix sort producing a permutation (sort_order), multi-key via
// least-significant-key-first passes (reference analogue: cudf sortOrder /
// orderBy reached from GpuSortExec — SURVEY.md §2.8A).
//
// Per key column: an order-preserving u64 key transform (sign-flip ints,
// IEEE flip floats with Spark NaN/-0.0 normalization, descending = bit
// inversion, null byte above the value bytes encoding NULLS FIRST/LAST),
// then 8-bit-digit passes. Each pass: per-block 256-bin LDS histogram
// (digit-major global layout so one flat exclusive scan yields
// [digit][block] offsets), then a stable scatter where the within-block
// rank comes from a wave-level multi-split (8 ballots) + per-bin cross-wave
// scan in LDS — rows keep their relative order, which LSD correctness
// requires.
#include "hipdf_common.h"

#define SORT_BLOCK 256
#define RADIX_BINS 256
#define SORT_ITEMS 16  // rows per thread per pass (tile = 4096 rows/block)

// ---- key transform -------------------------------------------------------

template <typename T>
__device__ __forceinline__ uint64_t sort_key_of(T v);

template <> __device__ __forceinline__ uint64_t sort_key_of<uint8_t>(uint8_t v) {
  return v;
}
template <> __device__ __forceinline__ uint64_t sort_key_of<int8_t>(int8_t v) {
  return (uint8_t)(v ^ (int8_t)0x80);
}
template <> __device__ __forceinline__ uint64_t sort_key_of<int16_t>(int16_t v) {
  return (uint16_t)(v ^ (int16_t)0x8000);
}
template <> __device__ __forceinline__ uint64_t sort_key_of<int32_t>(int32_t v) {
  return (uint32_t)(v ^ (int32_t)0x80000000);
}
template <> __device__ __forceinline__ uint64_t sort_key_of<int64_t>(int64_t v) {
  return (uint64_t)v ^ 0x8000000000000000ull;
}
template <> __device__ __forceinline__ uint64_t sort_key_of<float>(float v) {
  if (isnan(v)) v = __uint_as_float(0x7FC00000u);  // canonical, greatest
  if (v == 0.0f) v = 0.0f;                          // -0.0 -> 0.0
  uint32_t b = __float_as_uint(v);
  b = (b & 0x80000000u) ? ~b : (b | 0x80000000u);
  return b;
}
template <> __device__ __forceinline__ uint64_t sort_key_of<double>(double v) {
  if (isnan(v)) v = __longlong_as_double(0x7FF8000000000000ll);
  if (v == 0.0) v = 0.0;
  uint64_t b = (uint64_t)__double_as_longlong(v);
  return (b & 0x8000000000000000ull) ? ~b : (b | 0x8000000000000000ull);
}

// key width in value bytes per type
static inline int sort_key_width(int t) {
  switch (t) {
    case HT_U8: case HT_I8: return 1;
    case HT_I16: return 2;
    case HT_I32: case HT_F32: return 4;
    default: return 8;
  }
}

// build u64 keys for rows in permutation order: key[i] = transform(col[perm[i]])
template <typename T>
__global__ void k_make_sort_keys(const T* __restrict__ data,
                                 const uint64_t* __restrict__ valid,
                                 const int32_t* __restrict__ perm, int desc,
                                 int width_bytes, int null_byte_null,
                                 int null_only, uint64_t* __restrict__ keys,
                                 int64_t n) {
  // width 8 leaves no room for the null byte in 64 bits: value-only keys
  // here, and the caller runs one extra null_only pass afterwards.
  uint64_t vmask = width_bytes >= 8 ? ~0ull : ((1ull << (8 * width_bytes)) - 1);
  bool embed_null = width_bytes < 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t r = perm ? perm[i] : (int32_t)i;
    bool ok = valid_bit(valid, r);
    uint64_t nb = ok ? (uint64_t)(1 - null_byte_null) : (uint64_t)null_byte_null;
    if (null_only) {
      keys[i] = nb;
      continue;
    }
    uint64_t k = 0;
    if (ok) {
      k = sort_key_of<T>(data[r]) & vmask;
      if (desc) k = (~k) & vmask;
    }
    keys[i] = embed_null ? (k | (nb << (8 * width_bytes))) : k;
  }
}

// string sort keys: big-endian 8-byte chunk `chunk` of each string
// (short strings pad with 0x00, which orders prefixes first like byte
// comparison). The host loops chunks last-to-first through the stable
// radix, LSD-style over chunks; the null-order pass reuses null_only.
__global__ void k_make_sort_keys_str(const int32_t* __restrict__ offsets,
                                     const uint8_t* __restrict__ bytes,
                                     const uint64_t* __restrict__ valid,
                                     const int32_t* __restrict__ perm,
                                     int desc, int chunk,
                                     uint64_t* __restrict__ keys,
                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t r = perm ? perm[i] : (int32_t)i;
    uint64_t k = 0;
    if (valid_bit(valid, r)) {
      int32_t a = offsets[r] + 8 * chunk, b = offsets[r + 1];
      for (int j = 0; j < 8; ++j) {
        uint8_t c = (a + j < b && a + j >= offsets[r]) ? bytes[a + j] : 0;
        k = (k << 8) | c;
      }
      if (desc) k = ~k;
    }
    keys[i] = k;
  }
}

// decimal128 sort keys: word `word` (0 = lo unsigned-biased, 1 = hi
// signed-biased) of the interleaved pairs; host runs lo then hi.
__global__ void k_make_sort_keys_i128(const int64_t* __restrict__ data,
                                      const uint64_t* __restrict__ valid,
                                      const int32_t* __restrict__ perm,
                                      int desc, int word,
                                      uint64_t* __restrict__ keys,
                                      int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t r = perm ? perm[i] : (int32_t)i;
    uint64_t k = 0;
    if (valid_bit(valid, r)) {
      if (word == 0) {
        k = (uint64_t)data[2 * r];  // lo is unsigned: raw order
      } else {
        k = (uint64_t)data[2 * r + 1] ^ 0x8000000000000000ull;  // sign bias
      }
      if (desc) k = ~k;
    }
    keys[i] = k;
  }
}

// ---- radix pass ----------------------------------------------------------

__global__ void k_radix_count(const uint64_t* __restrict__ keys, int shift,
                              int64_t* __restrict__ counts, int64_t nblocks,
                              int64_t n) {
  __shared__ int lcnt[RADIX_BINS];
  for (int b = threadIdx.x; b < RADIX_BINS; b += blockDim.x) lcnt[b] = 0;
  __syncthreads();
  int64_t base = (int64_t)blockIdx.x * SORT_BLOCK * SORT_ITEMS;
#pragma unroll 4
  for (int it = 0; it < SORT_ITEMS; ++it) {
    int64_t i = base + (int64_t)it * SORT_BLOCK + threadIdx.x;
    if (i < n) {
      int digit = (int)((keys[i] >> shift) & 255);
      atomicAdd(&lcnt[digit], 1);
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < RADIX_BINS; b += blockDim.x)
    counts[(int64_t)b * nblocks + blockIdx.x] = lcnt[b];
}

// Stable multi-round scatter: each round handles 256 rows in row order,
// maintaining a running per-digit offset in LDS across rounds so relative
// order is preserved within the block (LSD stability requirement).
__global__ void k_radix_scatter(const uint64_t* __restrict__ keys_in,
                                const int32_t* __restrict__ perm_in,
                                int shift,
                                const int64_t* __restrict__ offsets,
                                int64_t nblocks,
                                uint64_t* __restrict__ keys_out,
                                int32_t* __restrict__ perm_out, int64_t n) {
  __shared__ int wave_bin[SORT_BLOCK / WAVE][RADIX_BINS];
  __shared__ int running[RADIX_BINS];
  int tid = threadIdx.x;
  int wid = tid / WAVE;
  int lane = tid & (WAVE - 1);
  for (int b = tid; b < RADIX_BINS; b += blockDim.x) running[b] = 0;
  int64_t base0 = (int64_t)blockIdx.x * SORT_BLOCK * SORT_ITEMS;

  for (int it = 0; it < SORT_ITEMS; ++it) {
    int64_t i = base0 + (int64_t)it * SORT_BLOCK + tid;
    bool active = i < n;
    uint64_t key = active ? keys_in[i] : 0;
    int digit = (int)((key >> shift) & 255);

    uint64_t active_mask = __ballot(active);
    uint64_t peers = active_mask;
    for (int b = 0; b < 8; ++b) {
      uint64_t m = __ballot((digit >> b) & 1);
      peers &= ((digit >> b) & 1) ? m : ~m;
    }
    uint64_t lt = lane == 0 ? 0ull : (~0ull >> (64 - lane));
    int rank_in_wave = __popcll(peers & lt);

    for (int w = 0; w < SORT_BLOCK / WAVE; ++w)
      for (int b = tid; b < RADIX_BINS; b += blockDim.x) wave_bin[w][b] = 0;
    __syncthreads();
    if (active && rank_in_wave == 0) wave_bin[wid][digit] = __popcll(peers);
    __syncthreads();
    // per-bin exclusive scan across the 4 waves, on top of `running`
    for (int b = tid; b < RADIX_BINS; b += blockDim.x) {
      int acc = running[b];
      for (int w = 0; w < SORT_BLOCK / WAVE; ++w) {
        int c = wave_bin[w][b];
        wave_bin[w][b] = acc;
        acc += c;
      }
      running[b] = acc;
    }
    __syncthreads();
    if (active) {
      int64_t base = offsets[(int64_t)digit * nblocks + blockIdx.x];
      int64_t pos = base + wave_bin[wid][digit] + rank_in_wave;
      keys_out[pos] = key;
      perm_out[pos] = perm_in ? perm_in[i] : (int32_t)i;
    }
    __syncthreads();
  }
}

extern "C" {

int64_t sort_num_blocks(int64_t n) {
  int64_t per = (int64_t)SORT_BLOCK * SORT_ITEMS;
  int64_t nb = (n + per - 1) / per;
  return nb < 1 ? 1 : nb;
}

int hipdf_sort_key_width(int t) { return sort_key_width(t); }

void hipdf_make_sort_keys_str(const void* offsets, const void* bytes,
                              const void* valid, const void* perm, int desc,
                              int chunk, void* keys, int64_t n,
                              hipStream_t stream) {
  hipLaunchKernelGGL(k_make_sort_keys_str, flat_grid(n), dim3(HIPDF_BLOCK),
                     0, stream, (const int32_t*)offsets,
                     (const uint8_t*)bytes, (const uint64_t*)valid,
                     (const int32_t*)perm, desc, chunk, (uint64_t*)keys, n);
}

void hipdf_make_sort_keys_i128(const void* data, const void* valid,
                               const void* perm, int desc, int word,
                               void* keys, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_make_sort_keys_i128, flat_grid(n), dim3(HIPDF_BLOCK),
                     0, stream, (const int64_t*)data, (const uint64_t*)valid,
                     (const int32_t*)perm, desc, word, (uint64_t*)keys, n);
}

void hipdf_make_sort_keys(int t, const void* data, const void* valid,
                          const void* perm, int desc, int nulls_last,
                          int null_only, void* keys, int64_t n,
                          hipStream_t stream) {
  int w = sort_key_width(t);
  // null byte: NULLS LAST -> null rows get 1 (sort after valid rows);
  // NULLS FIRST -> null rows get 0 and valid rows 1
  int null_byte_null = nulls_last ? 1 : 0;
  dispatch_type(t, [&]<typename T>() {
    hipLaunchKernelGGL((k_make_sort_keys<T>), flat_grid(n), dim3(HIPDF_BLOCK),
                       0, stream, (const T*)data, (const uint64_t*)valid,
                       (const int32_t*)perm, desc, w, null_byte_null,
                       null_only, (uint64_t*)keys, n);
  });
}

void hipdf_radix_count(const void* keys, int shift, void* counts, int64_t n,
                       hipStream_t stream) {
  int64_t nb = sort_num_blocks(n);
  hipLaunchKernelGGL(k_radix_count, dim3((uint32_t)nb), dim3(SORT_BLOCK), 0,
                     stream, (const uint64_t*)keys, shift, (int64_t*)counts,
                     nb, n);
}

void hipdf_radix_scatter(const void* keys_in, const void* perm_in, int shift,
                         const void* offsets, void* keys_out, void* perm_out,
                         int64_t n, hipStream_t stream) {
  int64_t nb = sort_num_blocks(n);
  hipLaunchKernelGGL(k_radix_scatter, dim3((uint32_t)nb), dim3(SORT_BLOCK), 0,
                     stream, (const uint64_t*)keys_in, (const int32_t*)perm_in,
                     shift, (const int64_t*)offsets, nb, (uint64_t*)keys_out,
                     (int32_t*)perm_out, n);
}

}  // extern "C"
