// Hash equi-join gather maps: build chained hash table on the build side,
// probe from the stream side, emit (left_map, right_map) row indices
// (reference analogue: cudf innerJoinGatherMaps / leftJoinGatherMaps /
// leftSemi/AntiJoinGatherMap reached from GpuHashJoin — SURVEY.md §2.8A).
// Null join keys never match (Spark equi-join semantics): they are neither
// inserted nor probed.
//
// Chained table: head[slot] -> newest row, next[row] -> older row in bucket.
// Duplicates and collisions live on the chain; equality filters on probe.
#include "hipdf_common.h"
#include "keys.h"

enum JoinHow : int { J_INNER = 0, J_LEFT, J_SEMI, J_ANTI, J_FULL };

__global__ void k_join_build(const int32_t* __restrict__ hashes,
                             const KeyCol* __restrict__ keys, int nkeys,
                             int32_t* __restrict__ head,
                             int32_t* __restrict__ next, uint32_t slot_mask,
                             int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (row_has_null_key(keys, nkeys, i)) continue;
    uint32_t slot = slot_of((uint32_t)hashes[i], slot_mask);
    next[i] = atomicExch(&head[slot], (int32_t)i);
  }
}

// pass 1: per-probe-row output count
__global__ void k_join_count(int how, const int32_t* __restrict__ lhashes,
                             const KeyCol* __restrict__ lkeys,
                             const KeyCol* __restrict__ rkeys, int nkeys,
                             const int32_t* __restrict__ head,
                             const int32_t* __restrict__ next,
                             uint32_t slot_mask, int64_t* __restrict__ counts,
                             int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t matches = 0;
    if (!row_has_null_key(lkeys, nkeys, i)) {
      uint32_t slot = slot_of((uint32_t)lhashes[i], slot_mask);
      for (int32_t r = head[slot]; r != -1; r = next[r]) {
        if (rows_equal(lkeys, rkeys, nkeys, i, r)) {
          ++matches;
          if (how == J_SEMI || how == J_ANTI) break;
        }
      }
    }
    int64_t c;
    switch (how) {
      case J_INNER: c = matches; break;
      case J_LEFT: case J_FULL: c = matches ? matches : 1; break;
      case J_SEMI: c = matches ? 1 : 0; break;
      default: c = matches ? 0 : 1; break;  // anti
    }
    counts[i] = c;
  }
}

// pass 2: fill gather maps at scanned offsets
__global__ void k_join_fill(int how, const int32_t* __restrict__ lhashes,
                            const KeyCol* __restrict__ lkeys,
                            const KeyCol* __restrict__ rkeys, int nkeys,
                            const int32_t* __restrict__ head,
                            const int32_t* __restrict__ next,
                            uint32_t slot_mask,
                            const int64_t* __restrict__ offsets,
                            int32_t* __restrict__ lmap,
                            int32_t* __restrict__ rmap,
                            uint8_t* __restrict__ right_matched, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t off = offsets[i];
    int64_t matches = 0;
    if (!row_has_null_key(lkeys, nkeys, i)) {
      uint32_t slot = slot_of((uint32_t)lhashes[i], slot_mask);
      for (int32_t r = head[slot]; r != -1; r = next[r]) {
        if (rows_equal(lkeys, rkeys, nkeys, i, r)) {
          ++matches;
          if (how == J_INNER || how == J_LEFT || how == J_FULL) {
            lmap[off] = (int32_t)i;
            rmap[off] = r;
            if (right_matched) right_matched[r] = 1;
            ++off;
          } else {
            break;
          }
        }
      }
    }
    if ((how == J_LEFT || how == J_FULL) && matches == 0) {
      lmap[off] = (int32_t)i;
      rmap[off] = -1;  // null right row
    } else if (how == J_SEMI && matches) {
      lmap[off] = (int32_t)i;
    } else if (how == J_ANTI && matches == 0) {
      lmap[off] = (int32_t)i;
    }
  }
}

extern "C" {

void hipdf_join_build(const void* hashes, const void* keys, int nkeys,
                      void* head, void* next, int64_t cap, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_join_build, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)hashes, (const KeyCol*)keys, nkeys,
                     (int32_t*)head, (int32_t*)next, (uint32_t)(cap - 1), n);
}

void hipdf_join_count(int how, const void* lhashes, const void* lkeys,
                      const void* rkeys, int nkeys, const void* head,
                      const void* next, int64_t cap, void* counts, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_join_count, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     how, (const int32_t*)lhashes, (const KeyCol*)lkeys,
                     (const KeyCol*)rkeys, nkeys, (const int32_t*)head,
                     (const int32_t*)next, (uint32_t)(cap - 1),
                     (int64_t*)counts, n);
}

void hipdf_join_fill(int how, const void* lhashes, const void* lkeys,
                     const void* rkeys, int nkeys, const void* head,
                     const void* next, int64_t cap, const void* offsets,
                     void* lmap, void* rmap, void* right_matched, int64_t n,
                     hipStream_t stream) {
  hipLaunchKernelGGL(k_join_fill, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     how, (const int32_t*)lhashes, (const KeyCol*)lkeys,
                     (const KeyCol*)rkeys, nkeys, (const int32_t*)head,
                     (const int32_t*)next, (uint32_t)(cap - 1),
                     (const int64_t*)offsets, (int32_t*)lmap, (int32_t*)rmap,
                     (uint8_t*)right_matched, n);
}

}  // extern "C"
