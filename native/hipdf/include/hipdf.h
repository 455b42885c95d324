/* hipdf C ABI — the kernel surface of the MI355X columnar engine as plain
 * extern "C" functions over raw device pointers (reference analogue: the
 * JNI contract the Spark plugin holds against cudf-java, SURVEY.md §2.8A;
 * pybind (hipdf_module.cpp) is ONE consumer of this ABI, this header and
 * libhipdf.so let a JVM/C/C++ host be another).
 *
 * Column model (Arrow-style):
 *   - fixed-width column: device array of `size` values (HipdfType)
 *   - validity: uint64 words, bit i = row i valid; NULL => all valid
 *   - string column: int32 offsets[size+1] + uint8 bytes
 * All functions enqueue onto the given HIP stream and return immediately.
 */
#ifndef HIPDF_H
#define HIPDF_H

#include <hip/hip_runtime.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum {
  HIPDF_U8 = 0,
  HIPDF_I8 = 1,
  HIPDF_I16 = 2,
  HIPDF_I32 = 3,
  HIPDF_I64 = 4,
  HIPDF_F32 = 5,
  HIPDF_F64 = 6,
} HipdfType;

/* Arrow-C-data-interface-shaped column descriptor for host bookkeeping.
 * Device kernels take the raw pointers below. */
typedef struct {
  int dtype;            /* HipdfType */
  int64_t size;         /* rows */
  const void* data;     /* device values (or string bytes) */
  const void* validity; /* device uint64 words or NULL */
  const void* offsets;  /* device int32[size+1] for strings, else NULL */
} HipdfColumnDesc;

/* multi-column row-equality descriptor for group-by / join builds
 * (layout must match kernels/keys.h KeyCol) */
typedef struct {
  int type;            /* HipdfType */
  int is_string;       /* 1 -> data=int32* offsets, aux=uint8* bytes */
  const void* data;
  const uint64_t* valid;
  const void* aux;
} HipdfKeyCol;

/* ---- elementwise ------------------------------------------------------ */
/* ops: 0 add 1 sub 2 mul 3 div ... (see hipdf_module.cpp _BIN_OPS) */
void hipdf_binary_arith(int op, int t, const void* a, const void* b,
                        double scalar_d, int64_t scalar_i, int scalar_rhs,
                        const void* av, const void* bv, void* out, void* ov,
                        int64_t n, hipStream_t stream);

/* ---- selection (filter) ----------------------------------------------- */
int64_t sel_num_blocks(int64_t n);
/* per-block popcounts of a uint8 boolean mask (mvalid may be NULL) */
void hipdf_mask_count(const void* mask, const void* mvalid,
                      void* block_counts /* int64[sel_num_blocks] */,
                      int64_t n, hipStream_t stream);
/* emit the selected row indices given exclusive per-block offsets */
void hipdf_mask_scatter(const void* mask, const void* mvalid,
                        const void* block_offsets, void* out_idx /* int32 */,
                        int64_t n, hipStream_t stream);
void hipdf_gather_fixed(int esize, const void* in, const void* idx,
                        void* out, int64_t n_out, hipStream_t stream);

/* ---- hashing / group-by ----------------------------------------------- */
/* kind: 0 int32-like, 1 int64-like, 2 float, 3 double (spark murmur3);
 * seeds: int32[n] initialized with the seed, updated in place */
void hipdf_murmur3_col(int kind, int t, const void* a, const void* av,
                       const void* sel, void* seeds, int64_t n,
                       hipStream_t stream);
/* CAS hash build over a power-of-two slot table (cap slots, slot_row
 * int32[cap] initialized to -1); claimed_slots int32[n]; ngroups int32[1]
 * zero-initialized */
void hipdf_gb_build(const void* hashes, const void* keys /* HipdfKeyCol[] */,
                    int nkeys, const void* sel, void* slot_row,
                    void* row_slot, void* claimed_slots, void* ngroups,
                    int64_t cap, int64_t n, hipStream_t stream);
void hipdf_gb_number(const void* claimed_slots, const void* slot_row,
                     void* slot_gid, void* ngroups, void* leaders,
                     int64_t n, hipStream_t stream);
void hipdf_gb_rowgid(const void* row_slot, const void* slot_gid,
                     void* row_gid, int64_t n, hipStream_t stream);
/* acc init for op: 0 sum 1 min 2 max 3 count */
void hipdf_gb_acc_init(int op, void* acc, int acc_is_double,
                       int32_t ngroups, hipStream_t stream);
void hipdf_gb_agg(int op, int t, const void* vals, const void* vvalid,
                  const void* row_gid, void* acc, void* cnt,
                  int acc_is_double, int32_t ngroups, int64_t n,
                  hipStream_t stream);

/* ---- device memory pool (pool.hip) ------------------------------------ */
int hipdf_pool_init(double fraction, size_t bytes);
void* hipdf_pool_alloc(size_t n);
void hipdf_pool_free(void* p);
size_t hipdf_pool_used(void);

#ifdef __cplusplus
}
#endif

#endif /* HIPDF_H */
