// Spark-compatible murmur3_x86_32 row hashing + partition-id computation.
// Must produce bit-identical results to the CPU reference
// (spark_rapids_amd/ops/cpu_backend.py murmur3_hash) and to Spark's
// Murmur3Hash so GPU shuffle partitioning lines up with CPU Spark
// (reference analogue: spark-rapids-jni Hash.murmurHash32 — SURVEY.md §2.8B).
#include "hipdf_common.h"

__device__ __forceinline__ uint32_t rotl32(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}
__device__ __forceinline__ uint32_t mix_k1(uint32_t k1) {
  k1 *= 0xCC9E2D51u;
  k1 = rotl32(k1, 15);
  return k1 * 0x1B873593u;
}
__device__ __forceinline__ uint32_t mix_h1(uint32_t h1, uint32_t k1) {
  h1 ^= k1;
  h1 = rotl32(h1, 13);
  return h1 * 5u + 0xE6546B64u;
}
__device__ __forceinline__ uint32_t fmix(uint32_t h1, uint32_t len) {
  h1 ^= len;
  h1 ^= h1 >> 16;
  h1 *= 0x85EBCA6Bu;
  h1 ^= h1 >> 13;
  h1 *= 0xC2B2AE35u;
  h1 ^= h1 >> 16;
  return h1;
}
__device__ __forceinline__ uint32_t hash_int(uint32_t v, uint32_t seed) {
  return fmix(mix_h1(seed, mix_k1(v)), 4);
}
__device__ __forceinline__ uint32_t hash_long(uint64_t v, uint32_t seed) {
  uint32_t h1 = mix_h1(seed, mix_k1((uint32_t)v));
  h1 = mix_h1(h1, mix_k1((uint32_t)(v >> 32)));
  return fmix(h1, 8);
}

// hash kinds: how the column's value maps into murmur input
enum HashKind : int {
  HK_INT = 0,   // int8/16/32, bool, date: sign-extended to int32
  HK_LONG = 1,  // int64 / timestamp / decimal64 unscaled
  HK_FLOAT = 2,
  HK_DOUBLE = 3,
  HK_STRING = 4,
  HK_I128 = 5,  // decimal128: chained hash_long of (lo, hi)
};

template <typename T>
__global__ void k_murmur3_col(int kind, const T* __restrict__ a,
                              const uint64_t* __restrict__ av,
                              const int32_t* __restrict__ sel,
                              int32_t* __restrict__ seeds, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(av, i)) continue;  // null keeps previous hash
    uint32_t seed = (uint32_t)seeds[j];
    uint32_t h;
    if (kind == HK_I128) {
      // interleaved pairs; only instantiated with T = int64_t
      uint64_t lo = (uint64_t)(int64_t)a[2 * i];
      uint64_t hi = (uint64_t)(int64_t)a[2 * i + 1];
      h = hash_long(hi, hash_long(lo, seed));
      seeds[j] = (int32_t)h;
      continue;
    }
    T v = a[i];
    if (kind == HK_LONG) {
      h = hash_long((uint64_t)(int64_t)v, seed);
    } else if (kind == HK_FLOAT) {
      float f = (float)v;
      if (f == 0.0f) f = 0.0f;  // -0.0 -> 0.0
      h = hash_int(__float_as_uint(f), seed);
    } else if (kind == HK_DOUBLE) {
      double d = (double)v;
      if (d == 0.0) d = 0.0;
      h = hash_long((uint64_t)__double_as_longlong(d), seed);
    } else {
      h = hash_int((uint32_t)(int32_t)(int64_t)v, seed);
    }
    seeds[j] = (int32_t)h;
  }
}

__global__ void k_murmur3_str(const int32_t* __restrict__ offsets,
                              const uint8_t* __restrict__ bytes,
                              const uint64_t* __restrict__ av,
                              const int32_t* __restrict__ sel,
                              int32_t* __restrict__ seeds, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(av, i)) continue;
    uint32_t h1 = (uint32_t)seeds[j];
    int32_t s = offsets[i], e = offsets[i + 1];
    int32_t len = e - s, p = s;
    // Spark hashUnsafeBytes: 4-byte LE words then SIGNED per-byte tail
    for (; p + 4 <= e; p += 4) {
      uint32_t w = (uint32_t)bytes[p] | ((uint32_t)bytes[p + 1] << 8) |
                   ((uint32_t)bytes[p + 2] << 16) |
                   ((uint32_t)bytes[p + 3] << 24);
      h1 = mix_h1(h1, mix_k1(w));
    }
    for (; p < e; ++p) h1 = mix_h1(h1, mix_k1((uint32_t)(int32_t)(int8_t)bytes[p]));
    seeds[j] = (int32_t)fmix(h1, (uint32_t)len);
  }
}

// partition id = pmod(hash, nparts) as int32
__global__ void k_pmod_part(const int32_t* __restrict__ h, int32_t nparts,
                            int32_t* __restrict__ part, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t r = h[i] % nparts;
    part[i] = r < 0 ? r + nparts : r;
  }
}

// ---- xxHash64 (reference analogue: spark-rapids-jni Hash.xxhash64) ------
// Canonical XXH64. Fixed-width values hash as their 8-byte widened
// little-endian form; strings hash their UTF-8 bytes. Seeds chain across
// columns like the murmur path.

#define XXP1 0x9E3779B185EBCA87ULL
#define XXP2 0xC2B2AE3D27D4EB4FULL
#define XXP3 0x165667B19E3779F9ULL
#define XXP4 0x85EBCA77C2B2AE63ULL
#define XXP5 0x27D4EB2F165667C5ULL

__device__ __forceinline__ uint64_t xx_rotl(uint64_t x, int r) {
  return (x << r) | (x >> (64 - r));
}

__device__ __forceinline__ uint64_t xx_round(uint64_t acc, uint64_t input) {
  acc += input * XXP2;
  acc = xx_rotl(acc, 31);
  return acc * XXP1;
}

__device__ __forceinline__ uint64_t xx_merge(uint64_t h, uint64_t v) {
  h ^= xx_round(0, v);
  return h * XXP1 + XXP4;
}

__device__ __forceinline__ uint64_t xx_avalanche(uint64_t h) {
  h ^= h >> 33;
  h *= XXP2;
  h ^= h >> 29;
  h *= XXP3;
  h ^= h >> 32;
  return h;
}

__device__ uint64_t xxh64_bytes(const uint8_t* p, int64_t len,
                                uint64_t seed) {
  const uint8_t* end = p + len;
  uint64_t h;
  if (len >= 32) {
    uint64_t v1 = seed + XXP1 + XXP2, v2 = seed + XXP2, v3 = seed,
             v4 = seed - XXP1;
    const uint8_t* limit = end - 32;
    do {
      uint64_t k;
      memcpy(&k, p, 8);
      v1 = xx_round(v1, k);
      memcpy(&k, p + 8, 8);
      v2 = xx_round(v2, k);
      memcpy(&k, p + 16, 8);
      v3 = xx_round(v3, k);
      memcpy(&k, p + 24, 8);
      v4 = xx_round(v4, k);
      p += 32;
    } while (p <= limit);
    h = xx_rotl(v1, 1) + xx_rotl(v2, 7) + xx_rotl(v3, 12) +
        xx_rotl(v4, 18);
    h = xx_merge(h, v1);
    h = xx_merge(h, v2);
    h = xx_merge(h, v3);
    h = xx_merge(h, v4);
  } else {
    h = seed + XXP5;
  }
  h += (uint64_t)len;
  while (p + 8 <= end) {
    uint64_t k;
    memcpy(&k, p, 8);
    h ^= xx_round(0, k);
    h = xx_rotl(h, 27) * XXP1 + XXP4;
    p += 8;
  }
  if (p + 4 <= end) {
    uint32_t k;
    memcpy(&k, p, 4);
    h ^= (uint64_t)k * XXP1;
    h = xx_rotl(h, 23) * XXP2 + XXP3;
    p += 4;
  }
  while (p < end) {
    h ^= (uint64_t)(*p) * XXP5;
    h = xx_rotl(h, 11) * XXP1;
    ++p;
  }
  return xx_avalanche(h);
}

__device__ __forceinline__ uint64_t xxh64_long(uint64_t v, uint64_t seed) {
  uint64_t h = seed + XXP5 + 8;
  h ^= xx_round(0, v);
  h = xx_rotl(h, 27) * XXP1 + XXP4;
  return xx_avalanche(h);
}

// fixed-width column xxhash64: values widened to int64 (floats normalized
// like the murmur path: -0.0 -> 0.0, NaN -> canonical); NULL rows keep
// the incoming seed unchanged (Spark null semantics)
template <typename T>
__global__ void k_xxhash64_col(int kind, const T* __restrict__ a,
                               const uint64_t* __restrict__ av,
                               const int32_t* __restrict__ sel,
                               int64_t* __restrict__ seeds, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(av, i)) continue;
    uint64_t bits;
    if (kind == 2) {  // float
      float f = (float)a[i];
      if (isnan(f)) f = __int_as_float(0x7fc00000);
      if (f == 0.0f) f = 0.0f;
      double d = (double)f;
      memcpy(&bits, &d, 8);
    } else if (kind == 3) {  // double
      double d = (double)a[i];
      if (isnan(d)) d = __longlong_as_double(0x7ff8000000000000LL);
      if (d == 0.0) d = 0.0;
      memcpy(&bits, &d, 8);
    } else {
      bits = (uint64_t)(int64_t)a[i];
    }
    seeds[j] = (int64_t)xxh64_long(bits, (uint64_t)seeds[j]);
  }
}

__global__ void k_xxhash64_str(const int32_t* __restrict__ offsets,
                               const uint8_t* __restrict__ bytes,
                               const uint64_t* __restrict__ av,
                               const int32_t* __restrict__ sel,
                               int64_t* __restrict__ seeds, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(av, i)) continue;
    seeds[j] = (int64_t)xxh64_bytes(bytes + offsets[i],
                                    offsets[i + 1] - offsets[i],
                                    (uint64_t)seeds[j]);
  }
}

extern "C" {

void hipdf_xxhash64_col(int kind, int t, const void* a, const void* av,
                        const void* sel, void* seeds, int64_t n,
                        hipStream_t stream) {
  dim3 grid = flat_grid(n);
  dispatch_type(t, [&]<typename T>() {
    hipLaunchKernelGGL((k_xxhash64_col<T>), grid, dim3(HIPDF_BLOCK), 0,
                       stream, kind, (const T*)a, (const uint64_t*)av,
                       (const int32_t*)sel, (int64_t*)seeds, n);
  });
}

void hipdf_xxhash64_str(const void* offsets, const void* bytes,
                        const void* av, const void* sel, void* seeds,
                        int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_xxhash64_str, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)offsets,
                     (const uint8_t*)bytes, (const uint64_t*)av,
                     (const int32_t*)sel, (int64_t*)seeds, n);
}

void hipdf_murmur3_col(int kind, int t, const void* a, const void* av,
                       const void* sel, void* seeds, int64_t n,
                       hipStream_t stream) {
  dim3 grid = flat_grid(n);
  dispatch_type(t, [&]<typename T>() {
    hipLaunchKernelGGL((k_murmur3_col<T>), grid, dim3(HIPDF_BLOCK), 0, stream,
                       kind, (const T*)a, (const uint64_t*)av,
                       (const int32_t*)sel, (int32_t*)seeds, n);
  });
}

void hipdf_murmur3_str(const void* offsets, const void* bytes, const void* av,
                       const void* sel, void* seeds, int64_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_murmur3_str, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)offsets, (const uint8_t*)bytes,
                     (const uint64_t*)av, (const int32_t*)sel,
                     (int32_t*)seeds, n);
}

void hipdf_pmod_part(const void* h, int nparts, void* part, int64_t n,
                     hipStream_t stream) {
  hipLaunchKernelGGL(k_pmod_part, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)h, (int32_t)nparts, (int32_t*)part, n);
}

}  // extern "C"
