// Shared multi-column row-equality machinery for hash groupby / hash join.
// Key columns are described by device-readable descriptors; equality follows
// Spark semantics: NULL == NULL for grouping (nulls form one group),
// NaN == NaN, -0.0 == 0.0.
#pragma once
#include "hipdf_common.h"

struct KeyCol {
  int type;        // HType
  int is_string;   // 1 -> data=offsets(int32*), aux=bytes(uint8*)
  const void* data;
  const uint64_t* valid;
  const void* aux;
};

__device__ __forceinline__ bool keyval_equal(const KeyCol& c, int64_t ra,
                                             const KeyCol& cb, int64_t rb) {
  bool va = valid_bit(c.valid, ra);
  bool vb = valid_bit(cb.valid, rb);
  if (va != vb) return false;
  if (!va) return true;  // both null -> equal (grouping semantics)
  if (c.is_string) {
    const int32_t* oa = (const int32_t*)c.data;
    const int32_t* ob = (const int32_t*)cb.data;
    int32_t sa = oa[ra], la = oa[ra + 1] - sa;
    int32_t sb = ob[rb], lb = ob[rb + 1] - sb;
    if (la != lb) return false;
    const uint8_t* ba = (const uint8_t*)c.aux;
    const uint8_t* bb = (const uint8_t*)cb.aux;
    for (int32_t k = 0; k < la; ++k)
      if (ba[sa + k] != bb[sb + k]) return false;
    return true;
  }
  switch (c.type) {
    case HT_U8: case HT_I8:
      return ((const int8_t*)c.data)[ra] == ((const int8_t*)cb.data)[rb];
    case HT_I16:
      return ((const int16_t*)c.data)[ra] == ((const int16_t*)cb.data)[rb];
    case HT_I32:
      return ((const int32_t*)c.data)[ra] == ((const int32_t*)cb.data)[rb];
    case HT_I64:
      return ((const int64_t*)c.data)[ra] == ((const int64_t*)cb.data)[rb];
    case HT_F32: {
      float a = ((const float*)c.data)[ra], b = ((const float*)cb.data)[rb];
      if (isnan(a) && isnan(b)) return true;
      return a == b;  // covers -0.0 == 0.0
    }
    case HT_F64: {
      double a = ((const double*)c.data)[ra], b = ((const double*)cb.data)[rb];
      if (isnan(a) && isnan(b)) return true;
      return a == b;
    }
    case 7: {  // int128 pairs (decimal128)
      const int64_t* pa = (const int64_t*)c.data;
      const int64_t* pb = (const int64_t*)cb.data;
      return pa[2 * ra] == pb[2 * rb] && pa[2 * ra + 1] == pb[2 * rb + 1];
    }
  }
  return false;
}

__device__ __forceinline__ bool rows_equal(const KeyCol* ca, const KeyCol* cb,
                                           int ncols, int64_t ra, int64_t rb) {
  for (int k = 0; k < ncols; ++k)
    if (!keyval_equal(ca[k], ra, cb[k], rb)) return false;
  return true;
}

__device__ __forceinline__ bool row_has_null_key(const KeyCol* cols, int ncols,
                                                 int64_t r) {
  for (int k = 0; k < ncols; ++k)
    if (!valid_bit(cols[k].valid, r)) return true;
  return false;
}

// finalize a 32-bit row hash into a table slot (avalanche the low bits)
__device__ __forceinline__ uint32_t slot_of(uint32_t h, uint32_t mask) {
  h ^= h >> 16;
  h *= 0x85EBCA6Bu;
  h ^= h >> 13;
  return h & mask;
}
