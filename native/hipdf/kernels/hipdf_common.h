// hipdf: hand-written CDNA4 (gfx950) columnar kernels for MI355X.
// Common device helpers: wave64 validity-bitmask idioms, grid sizing,
// type dispatch. See /root/repo/SURVEY.md §2.8 for the op surface this
// library implements (the reference reaches it via cudf JNI; here it is
// native HIP called from Python via pybind11).
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <limits>
#include <stdexcept>
#include <string>
#include <type_traits>

#define WAVE 64
#define HIPDF_BLOCK 256
// memory-bound launch shape: cap grid, grid-stride the rest (guide §6 G11)
#define HIPDF_MAX_BLOCKS 2048

// Physical type ids shared with python (spark_rapids_amd/ops/gpu_backend.py)
enum HType : int {
  HT_U8 = 0,   // bool
  HT_I8 = 1,
  HT_I16 = 2,
  HT_I32 = 3,
  HT_I64 = 4,
  HT_F32 = 5,
  HT_F64 = 6,
};

static inline int htype_size(int t) {
  switch (t) {
    case HT_U8: case HT_I8: return 1;
    case HT_I16: return 2;
    case HT_I32: case HT_F32: return 4;
    default: return 8;
  }
}

// ---------------- validity bitmask helpers (Arrow LSB-first) --------------
// Masks are padded to 64 B so 64-bit word loads are always in-bounds.
__device__ __forceinline__ uint64_t valid_word(const uint64_t* mask,
                                               int64_t word_idx) {
  return mask ? mask[word_idx] : ~0ull;
}

__device__ __forceinline__ bool valid_bit(const uint64_t* mask, int64_t row) {
  return !mask || ((mask[row >> 6] >> (row & 63)) & 1ull);
}

// Each wave owns one 64-row stripe: lane l handles row stripe*64+l, and the
// wave's __ballot() of per-lane validity IS the output bitmask word.
__device__ __forceinline__ void write_valid_word(uint64_t* mask,
                                                 int64_t word_idx,
                                                 uint64_t ballot, int lane) {
  if (mask && lane == 0) mask[word_idx] = ballot;
}

__device__ __forceinline__ int lane_id() { return threadIdx.x & (WAVE - 1); }

static inline int64_t n_stripes(int64_t n) { return (n + WAVE - 1) / WAVE; }

static inline dim3 stripe_grid(int64_t n) {
  int64_t waves_per_block = HIPDF_BLOCK / WAVE;
  int64_t blocks = (n_stripes(n) + waves_per_block - 1) / waves_per_block;
  if (blocks > HIPDF_MAX_BLOCKS) blocks = HIPDF_MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  return dim3((uint32_t)blocks);
}

static inline dim3 flat_grid(int64_t n, int per_thread = 1) {
  int64_t blocks = (n + (int64_t)HIPDF_BLOCK * per_thread - 1) /
                   ((int64_t)HIPDF_BLOCK * per_thread);
  if (blocks > HIPDF_MAX_BLOCKS) blocks = HIPDF_MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  return dim3((uint32_t)blocks);
}

#define HIPDF_CHECK(expr)                                              \
  do {                                                                 \
    hipError_t _e = (expr);                                            \
    if (_e != hipSuccess) {                                            \
      throw std::runtime_error(std::string("hipdf: ") +                \
                               hipGetErrorString(_e));                 \
    }                                                                  \
  } while (0)

// Runtime type dispatch to a functor templated on the C type.
template <typename F>
inline void dispatch_type(int t, F&& f) {
  switch (t) {
    case HT_U8: f.template operator()<uint8_t>(); break;
    case HT_I8: f.template operator()<int8_t>(); break;
    case HT_I16: f.template operator()<int16_t>(); break;
    case HT_I32: f.template operator()<int32_t>(); break;
    case HT_I64: f.template operator()<int64_t>(); break;
    case HT_F32: f.template operator()<float>(); break;
    case HT_F64: f.template operator()<double>(); break;
    default: throw std::runtime_error("hipdf: bad type id");
  }
}
