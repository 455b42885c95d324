// Hash/radix partition: reorder rows so each partition's rows are
// contiguous, returning the permutation (reference analogue: cudf
// Table.partition feeding GpuShuffleExchangeExec — SURVEY.md §3.4).
// LDS-tiled per-block histograms; the per-(part, block) count matrix is
// laid out part-major so a single flat exclusive scan yields scatter
// offsets (classic radix-partition scheme). Also used as one digit pass
// of the LSD radix sort.
#include "hipdf_common.h"

#define PART_MAX_LDS 4096  // max partitions held in LDS (16 KiB of int32)

// counts[p * nblocks + b] = rows of partition p seen by block b
__global__ void k_part_hist(const int32_t* __restrict__ part, int32_t nparts,
                            int64_t rows_per_block,
                            int64_t* __restrict__ counts, int64_t nblocks,
                            int64_t n) {
  extern __shared__ int32_t lcounts[];
  for (int p = threadIdx.x; p < nparts; p += blockDim.x) lcounts[p] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * rows_per_block;
  int64_t end = min(start + rows_per_block, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    atomicAdd(&lcounts[part[i]], 1);
  __syncthreads();
  for (int p = threadIdx.x; p < nparts; p += blockDim.x)
    counts[(int64_t)p * nblocks + blockIdx.x] = lcounts[p];
}

// offsets = exclusive scan of counts (same layout); scatter rows to perm
__global__ void k_part_scatter(const int32_t* __restrict__ part,
                               int32_t nparts, int64_t rows_per_block,
                               const int64_t* __restrict__ offsets,
                               int64_t nblocks, int32_t* __restrict__ perm,
                               int64_t n) {
  extern __shared__ int32_t lofs[];  // running offset per partition
  for (int p = threadIdx.x; p < nparts; p += blockDim.x)
    lofs[p] = (int32_t)offsets[(int64_t)p * nblocks + blockIdx.x];
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * rows_per_block;
  int64_t end = min(start + rows_per_block, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    int32_t pos = atomicAdd(&lofs[part[i]], 1);
    perm[pos] = (int32_t)i;
  }
}

extern "C" {

int64_t part_num_blocks(int64_t n) {
  int64_t nb = (n + 4 * HIPDF_BLOCK - 1) / (4 * HIPDF_BLOCK);
  if (nb > HIPDF_MAX_BLOCKS) nb = HIPDF_MAX_BLOCKS;
  return nb < 1 ? 1 : nb;
}

void hipdf_part_hist(const void* part, int nparts, void* counts, int64_t n,
                     hipStream_t stream) {
  if (nparts > PART_MAX_LDS) throw std::runtime_error("too many partitions");
  int64_t nb = part_num_blocks(n);
  int64_t rpb = (n + nb - 1) / nb;
  hipLaunchKernelGGL(k_part_hist, dim3((uint32_t)nb), dim3(HIPDF_BLOCK),
                     (size_t)nparts * 4, stream, (const int32_t*)part,
                     (int32_t)nparts, rpb, (int64_t*)counts, nb, n);
}

void hipdf_part_scatter(const void* part, int nparts, const void* offsets,
                        void* perm, int64_t n, hipStream_t stream) {
  int64_t nb = part_num_blocks(n);
  int64_t rpb = (n + nb - 1) / nb;
  hipLaunchKernelGGL(k_part_scatter, dim3((uint32_t)nb), dim3(HIPDF_BLOCK),
                     (size_t)nparts * 4, stream, (const int32_t*)part,
                     (int32_t)nparts, rpb, (const int64_t*)offsets, nb,
                     (int32_t*)perm, n);
}

}  // extern "C"
