// Hash group-by aggregation (reference analogue: cudf groupBy().aggregate
// reached from GpuHashAggregateExec — SURVEY.md §3.5).
//
// Structure (race-free 3-pass build + per-agg accumulate):
//   1. k_gb_build: open-addressing CAS on a row-index slot table; every row
//      finds the slot whose leader row has equal keys (Spark semantics:
//      nulls group together, NaN==NaN) and records its slot id.
//   2. k_gb_number: slots with a leader get dense group ids (atomic counter).
//   3. k_gb_rowgid: row -> group id via its slot.
//   4. k_gb_agg: per (op, value column) accumulation. Low-cardinality groups
//      (the common SQL case) take the LDS-tiled path: per-block accumulators
//      in LDS, one device atomic per (block, group) at flush — this is what
//      makes 4-group q1-style aggregations run at HBM bandwidth instead of
//      serializing on 4 hot atomics.
#include "hipdf_common.h"
#include "keys.h"

#define GB_EMPTY (-1)
#define GB_LDS_GROUPS 2048

__global__ void k_gb_build(const int32_t* __restrict__ hashes,
                           const KeyCol* __restrict__ keys, int nkeys,
                           const int32_t* __restrict__ sel,
                           int32_t* __restrict__ slot_row,
                           int32_t* __restrict__ row_slot,
                           int32_t* __restrict__ claimed_slots,
                           int32_t* __restrict__ ngroups, uint32_t slot_mask,
                           int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int32_t i = sel ? sel[j] : (int32_t)j;
    uint32_t slot = slot_of((uint32_t)hashes[j], slot_mask);
    while (true) {
      // plain read first: slots are write-once (EMPTY -> row), so a stale
      // EMPTY just falls through to the CAS; this keeps the hot
      // low-cardinality case (every row hitting a few claimed slots) on
      // cached loads instead of serializing atomics across 256 CUs
      int32_t cur = slot_row[slot];
      if (cur == GB_EMPTY) {
        cur = atomicCAS(&slot_row[slot], GB_EMPTY, (int32_t)i);
        if (cur == GB_EMPTY) {
          // claimed a new group: record its slot so numbering touches only
          // live slots instead of sweeping the whole table
          claimed_slots[atomicAdd(ngroups, 1)] = (int32_t)slot;
          row_slot[j] = (int32_t)slot;
          break;
        }
      }
      if (cur == i || rows_equal(keys, keys, nkeys, i, cur)) {
        row_slot[j] = (int32_t)slot;
        break;
      }
      slot = (slot + 1) & slot_mask;
    }
  }
}

// ---- dense integer-key fast path -----------------------------------------
// When the (possibly multi-) integer group keys span a small value range,
// the group id is a direct radix index — no hash table, no probe, no
// leader gather. gid = sum_i code_i * stride_i with code_i = key_i - min_i
// (range_i for NULL, so nulls group together). Empty ids are compacted by
// the host afterwards.
struct DenseKey {
  int type;        // HType of the key column
  int pad;
  const void* vals;
  const uint64_t* valid;
  int64_t kmin;
  int64_t range;   // #distinct slots for values; NULL takes index `range`
  int64_t stride;
};

__device__ __forceinline__ int64_t dense_load_i64(const void* p, int t,
                                                  int64_t i) {
  switch (t) {
    case HT_U8: return ((const uint8_t*)p)[i];
    case HT_I8: return ((const int8_t*)p)[i];
    case HT_I16: return ((const int16_t*)p)[i];
    case HT_I32: return ((const int32_t*)p)[i];
    default: return ((const int64_t*)p)[i];
  }
}

__global__ void k_dense_gid(const DenseKey* __restrict__ keys, int nkeys,
                            const int32_t* __restrict__ sel,
                            int32_t* __restrict__ row_gid, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    int64_t gid = 0;
    for (int k = 0; k < nkeys; ++k) {
      const DenseKey& d = keys[k];
      int64_t code = valid_bit(d.valid, i)
                         ? dense_load_i64(d.vals, d.type, i) - d.kmin
                         : d.range;
      gid += code * d.stride;
    }
    row_gid[j] = (int32_t)gid;
  }
}

__global__ void k_gb_number(const int32_t* __restrict__ claimed_slots,
                            const int32_t* __restrict__ slot_row,
                            int32_t* __restrict__ slot_gid,
                            const int32_t* __restrict__ ngroups,
                            int32_t* __restrict__ leaders) {
  int32_t ng = *ngroups;
  for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; g < ng;
       g += (int64_t)gridDim.x * blockDim.x) {
    int32_t s = claimed_slots[g];
    slot_gid[s] = (int32_t)g;
    leaders[g] = slot_row[s];
  }
}

__global__ void k_gb_rowgid(const int32_t* __restrict__ row_slot,
                            const int32_t* __restrict__ slot_gid,
                            int32_t* __restrict__ row_gid, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    row_gid[i] = slot_gid[row_slot[i]];
}

// ---- aggregation ---------------------------------------------------------
enum GbOp : int { GB_SUM = 0, GB_MIN, GB_MAX, GB_COUNT, GB_COUNT_ALL,
                  GB_FIRST, GB_BITAND, GB_BITOR, GB_BITXOR };

template <typename ACC>
__device__ __forceinline__ void acc_atomic(int op, ACC* addr, ACC v);

template <>
__device__ __forceinline__ void acc_atomic<int64_t>(int op, int64_t* addr,
                                                    int64_t v) {
  if (op == GB_MIN) atomicMin((long long*)addr, (long long)v);
  else if (op == GB_MAX) atomicMax((long long*)addr, (long long)v);
  else if (op == GB_BITAND)
    atomicAnd((unsigned long long*)addr, (unsigned long long)v);
  else if (op == GB_BITOR)
    atomicOr((unsigned long long*)addr, (unsigned long long)v);
  else if (op == GB_BITXOR)
    atomicXor((unsigned long long*)addr, (unsigned long long)v);
  else if (op == GB_FIRST) *addr = v;  // any-value semantics (Spark first
  // without ordering is unspecified); aligned 8B store cannot tear
  else atomicAdd((unsigned long long*)addr, (unsigned long long)v);
}

template <>
__device__ __forceinline__ void acc_atomic<double>(int op, double* addr,
                                                   double v) {
  if (op == GB_SUM) {
    atomicAdd(addr, v);
    return;
  }
  if (op == GB_FIRST) {
    *addr = v;
    return;
  }
  unsigned long long* up = (unsigned long long*)addr;
  unsigned long long old = *up, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    bool better = op == GB_MIN
                      ? (!isnan(v) && (isnan(cur) || v < cur))
                      : ((v > cur && !isnan(cur)) || (isnan(v) && !isnan(cur)));
    // Spark ordering: NaN is the greatest value
    if (!better) break;
    old = atomicCAS(up, assumed, (unsigned long long)__double_as_longlong(v));
  } while (old != assumed);
}

template <typename ACC>
__device__ __forceinline__ ACC acc_init(int op) {
  // float identities must be ±infinity (not ±DBL_MAX): a group whose
  // real extreme IS ±inf must beat the init under plain compare
  if (op == GB_MIN)
    // float min identity is NaN: under Spark's total order NaN is the
    // GREATEST value (above +inf), so min{all NaN} must stay NaN while
    // any real value (incl. +inf) replaces it
    return std::numeric_limits<ACC>::has_infinity
               ? std::numeric_limits<ACC>::quiet_NaN()
               : std::numeric_limits<ACC>::max();
  if (op == GB_MAX)
    return std::numeric_limits<ACC>::has_infinity
               ? -std::numeric_limits<ACC>::infinity()
               : std::numeric_limits<ACC>::lowest();
  if (op == GB_BITAND) return (ACC)-1;  // all ones identity
  return (ACC)0;
}

// one value column, one op; acc/cnt indexed by group id
template <typename T, typename ACC, bool USE_LDS>
__global__ void k_gb_agg(int op, const T* __restrict__ vals,
                         const uint64_t* __restrict__ vvalid,
                         const int32_t* __restrict__ row_gid,
                         ACC* __restrict__ acc, int64_t* __restrict__ cnt,
                         int32_t ngroups, int64_t n) {
  extern __shared__ char lds_raw[];
  ACC* lacc = (ACC*)lds_raw;
  int64_t* lcnt = (int64_t*)(lds_raw + (USE_LDS ? sizeof(ACC) * ngroups : 0));
  if (USE_LDS) {
    for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
      lacc[g] = acc_init<ACC>(op);
      lcnt[g] = 0;
    }
    __syncthreads();
  }
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t g = row_gid[i];
    if (op == GB_COUNT_ALL) {
      if (USE_LDS) atomicAdd((unsigned long long*)&lcnt[g], 1ull);
      else atomicAdd((unsigned long long*)&cnt[g], 1ull);
      continue;
    }
    if (!valid_bit(vvalid, i)) continue;
    ACC v = (ACC)vals[i];
    if (USE_LDS) {
      if (op != GB_COUNT) acc_atomic<ACC>(op, &lacc[g], v);
      atomicAdd((unsigned long long*)&lcnt[g], 1ull);
    } else {
      if (op != GB_COUNT) acc_atomic<ACC>(op, &acc[g], v);
      atomicAdd((unsigned long long*)&cnt[g], 1ull);
    }
  }
  if (USE_LDS) {
    __syncthreads();
    for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
      if (lcnt[g]) {
        if (op != GB_COUNT && op != GB_COUNT_ALL)
          acc_atomic<ACC>(op, &acc[g], lacc[g]);
        atomicAdd((unsigned long long*)&cnt[g], (unsigned long long)lcnt[g]);
      }
    }
  }
}

// ---- fused multi-aggregate: one pass over row_gid and every value column
// (saves one full re-read of row_gid + one kernel launch per aggregate) ----
struct AggDesc {
  int op;             // GbOp
  int type;           // HType of the value column
  int acc_is_double;  // accumulator: double (1) or int64 (0)
  int skip_cnt;       // sum/min/max over a non-null column: the count
                      // atomic is pure overhead (group validity is
                      // implied by group existence) — skip it in the
                      // global-atomic path; the LDS path still counts
                      // (its flush is gated on lcnt)
  const void* vals;
  const uint64_t* valid;
  void* acc;          // [ngroups] double or int64
  int64_t* cnt;       // [ngroups]
};

__device__ __forceinline__ double load_as_double(const void* p, int t,
                                                 int64_t i) {
  switch (t) {
    case HT_U8: return ((const uint8_t*)p)[i];
    case HT_I8: return ((const int8_t*)p)[i];
    case HT_I16: return ((const int16_t*)p)[i];
    case HT_I32: return ((const int32_t*)p)[i];
    case HT_I64: return (double)((const int64_t*)p)[i];
    case HT_F32: return ((const float*)p)[i];
    default: return ((const double*)p)[i];
  }
}

__device__ __forceinline__ int64_t load_as_i64(const void* p, int t,
                                               int64_t i) {
  switch (t) {
    case HT_U8: return ((const uint8_t*)p)[i];
    case HT_I8: return ((const int8_t*)p)[i];
    case HT_I16: return ((const int16_t*)p)[i];
    case HT_I32: return ((const int32_t*)p)[i];
    case HT_I64: return ((const int64_t*)p)[i];
    case HT_F32: return (int64_t)((const float*)p)[i];
    default: return (int64_t)((const double*)p)[i];
  }
}

template <bool USE_LDS>
__global__ void k_gb_agg_multi(const AggDesc* __restrict__ aggs, int naggs,
                               const int32_t* __restrict__ row_gid,
                               const int32_t* __restrict__ sel,
                               int32_t ngroups, int nrep, int64_t n) {
  // nrep > 1 (non-LDS path): each wave accumulates into one of nrep
  // replica accumulator arrays, so a zipf-skewed hot group spreads its
  // atomics over nrep addresses instead of serializing on one; replicas
  // are folded by k_gb_reduce_reps afterwards.
  size_t rep_off = 0;
  if (!USE_LDS && nrep > 1) {
    int wave = (int)(((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64);
    rep_off = (size_t)(wave % nrep) * ngroups;
  }
  extern __shared__ char lds_raw[];
  // LDS layout per agg a: acc[a][ngroups] (8B each) then cnt[a][ngroups]
  if (USE_LDS) {
    for (int a = 0; a < naggs; ++a) {
      char* base = lds_raw + (size_t)a * ngroups * 16;
      int op = aggs[a].op;
      if (aggs[a].acc_is_double) {
        double* lacc = (double*)base;
        for (int g = threadIdx.x; g < ngroups; g += blockDim.x)
          lacc[g] = acc_init<double>(op);
      } else {
        int64_t* lacc = (int64_t*)base;
        for (int g = threadIdx.x; g < ngroups; g += blockDim.x)
          lacc[g] = acc_init<int64_t>(op);
      }
      int64_t* lcnt = (int64_t*)(base + (size_t)ngroups * 8);
      for (int g = threadIdx.x; g < ngroups; g += blockDim.x) lcnt[g] = 0;
    }
    __syncthreads();
  }
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int32_t g = row_gid[j];
    int64_t i = sel ? (int64_t)sel[j] : j;
    for (int a = 0; a < naggs; ++a) {
      const AggDesc& d = aggs[a];
      char* base = lds_raw + (size_t)a * ngroups * 16;
      int64_t* cnt_p = USE_LDS ? (int64_t*)(base + (size_t)ngroups * 8)
                               : d.cnt + rep_off;
      if (d.op == GB_COUNT_ALL) {
        atomicAdd((unsigned long long*)&cnt_p[g], 1ull);
        continue;
      }
      if (!valid_bit(d.valid, i)) continue;
      if (USE_LDS || !d.skip_cnt)
        atomicAdd((unsigned long long*)&cnt_p[g], 1ull);
      if (d.op == GB_COUNT) continue;
      if (d.acc_is_double) {
        double* acc_p = USE_LDS ? (double*)base : (double*)d.acc + rep_off;
        acc_atomic<double>(d.op, &acc_p[g], load_as_double(d.vals, d.type, i));
      } else {
        int64_t* acc_p = USE_LDS ? (int64_t*)base : (int64_t*)d.acc + rep_off;
        acc_atomic<int64_t>(d.op, &acc_p[g], load_as_i64(d.vals, d.type, i));
      }
    }
  }
  if (USE_LDS) {
    __syncthreads();
    for (int a = 0; a < naggs; ++a) {
      const AggDesc& d = aggs[a];
      char* base = lds_raw + (size_t)a * ngroups * 16;
      int64_t* lcnt = (int64_t*)(base + (size_t)ngroups * 8);
      for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
        if (!lcnt[g]) continue;
        atomicAdd((unsigned long long*)&d.cnt[g],
                  (unsigned long long)lcnt[g]);
        if (d.op == GB_COUNT || d.op == GB_COUNT_ALL) continue;
        if (d.acc_is_double)
          acc_atomic<double>(d.op, &((double*)d.acc)[g], ((double*)base)[g]);
        else
          acc_atomic<int64_t>(d.op, &((int64_t*)d.acc)[g],
                              ((int64_t*)base)[g]);
      }
    }
  }
}

// fold replica accumulators back into replica 0
template <typename ACC>
__global__ void k_gb_reduce_reps(int op, ACC* __restrict__ acc,
                                 int64_t* __restrict__ cnt, int32_t ngroups,
                                 int nrep) {
  for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
       g += gridDim.x * blockDim.x) {
    int64_t c = cnt[g];
    ACC v = acc[g];
    for (int r = 1; r < nrep; ++r) {
      int64_t cr = cnt[(size_t)r * ngroups + g];
      ACC vr = acc[(size_t)r * ngroups + g];
      if (op == GB_SUM) v = v + vr;
      else if (op == GB_MIN) {
        // integers: the init identity (INT64_MAX) can never win a min,
        // so the fold needs no cnt — which also makes it correct for
        // skip_cnt aggs whose cnt replicas are all zero. Floats keep
        // cnt (empty-replica +inf is ambiguous with a real +inf when
        // NaN values exist) and compare in Spark order (NaN greatest).
        bool take;
        if constexpr (std::is_floating_point<ACC>::value)
          take = cr && (!c || (!isnan((double)vr) &&
                               (isnan((double)v) || vr < v)));
        else
          take = vr < v;
        if (take) v = vr;
      }
      else if (op == GB_MAX) {
        bool take;
        if constexpr (std::is_floating_point<ACC>::value)
          take = cr && (!c || (vr > v && !isnan((double)v)) ||
                        (isnan((double)vr) && !isnan((double)v)));
        else
          take = vr > v;
        if (take) v = vr;
      }
      else if (op == GB_BITAND)
        v = (ACC)((int64_t)v & (int64_t)vr);
      else if (op == GB_BITOR)
        v = (ACC)((int64_t)v | (int64_t)vr);
      else if (op == GB_BITXOR)
        v = (ACC)((int64_t)v ^ (int64_t)vr);
      else if (op == GB_FIRST && c == 0 && cr > 0) v = vr;
      c += cr;
    }
    cnt[g] = c;
    acc[g] = v;
  }
}

// init global accumulators for min/max identities
template <typename ACC>
__global__ void k_gb_acc_init(int op, ACC* __restrict__ acc, int32_t ngroups) {
  for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
       g += gridDim.x * blockDim.x)
    acc[g] = acc_init<ACC>(op);
}

// bitmask from nonzero counts (output validity of aggregates)
__global__ void k_mask_from_nonzero(const int64_t* __restrict__ cnt,
                                    uint64_t* __restrict__ mask,
                                    int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t i = s * WAVE + lane;
    bool ok = i < n && cnt[i] != 0;
    uint64_t ballot = __ballot(ok);
    if (lane == 0) mask[s] = ballot;
  }
}

// ---- collect_list / collect_set ------------------------------------------
// Two-pass grouped gather into a LIST column: count valid values per group,
// host scans counts into offsets, then an atomic per-group cursor places
// each value (order within a group is unspecified, matching Spark).
__global__ void k_gb_collect_count(const uint64_t* __restrict__ vvalid,
                                   const int32_t* __restrict__ row_gid,
                                   const int32_t* __restrict__ sel,
                                   int64_t* __restrict__ counts, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(vvalid, i)) continue;
    atomicAdd((unsigned long long*)&counts[row_gid[j]], 1ull);
  }
}

__global__ void k_gb_collect_fill(int esize, const uint8_t* __restrict__ vals,
                                  const uint64_t* __restrict__ vvalid,
                                  const int32_t* __restrict__ row_gid,
                                  const int32_t* __restrict__ sel,
                                  const int64_t* __restrict__ offsets,
                                  int64_t* __restrict__ cursor,
                                  uint8_t* __restrict__ out, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(vvalid, i)) continue;
    int32_t g = row_gid[j];
    int64_t pos = offsets[g] +
        (int64_t)atomicAdd((unsigned long long*)&cursor[g], 1ull);
    switch (esize) {
      case 1: out[pos] = vals[i]; break;
      case 2: ((uint16_t*)out)[pos] = ((const uint16_t*)vals)[i]; break;
      case 4: ((uint32_t*)out)[pos] = ((const uint32_t*)vals)[i]; break;
      case 8: ((uint64_t*)out)[pos] = ((const uint64_t*)vals)[i]; break;
      case 16:
        ((ulonglong2*)out)[pos] = ((const ulonglong2*)vals)[i];
        break;
    }
  }
}

// exact percentile with linear interpolation over per-group sorted values.
// perm: row order sorted by (gid, value, nulls last); starts: first row of
// each group in that order (scan of total counts); vcnt: valid counts.
__global__ void k_gb_percentile(const double* __restrict__ vals,
                                const int32_t* __restrict__ perm,
                                const int64_t* __restrict__ starts,
                                const int64_t* __restrict__ vcnt, double p,
                                double* __restrict__ out, int32_t ngroups) {
  for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
       g += gridDim.x * blockDim.x) {
    int64_t c = vcnt[g];
    if (!c) {
      out[g] = 0.0;
      continue;
    }
    double pos = p * (double)(c - 1);
    int64_t lo = (int64_t)pos;
    int64_t hi = lo + 1 < c ? lo + 1 : c - 1;
    double frac = pos - (double)lo;
    double a = vals[perm[starts[g] + lo]];
    double b = vals[perm[starts[g] + hi]];
    out[g] = a + (b - a) * frac;
  }
}


// ---- HyperLogLog++ register update (approx_count_distinct) ---------------
// One uint8 register array of 2^p entries per group; rho = leading-zero
// count of the suffix of the 64-bit hash + 1, kept as a byte max
// (reference analogue: HyperLogLogPlusPlusHostUDF in spark-rapids-jni).

__device__ __forceinline__ void atomic_max_u8(uint8_t* addr, uint8_t val) {
  uint32_t* base = (uint32_t*)((uintptr_t)addr & ~(uintptr_t)3);
  int shift = (int)((uintptr_t)addr & 3) * 8;
  uint32_t old = *base;
  while (true) {
    uint8_t cur = (uint8_t)((old >> shift) & 0xFF);
    if (cur >= val) return;
    uint32_t updated = (old & ~(0xFFu << shift)) |
                       ((uint32_t)val << shift);
    uint32_t seen = atomicCAS(base, old, updated);
    if (seen == old) return;
    old = seen;
  }
}

__global__ void k_gb_hll(const int64_t* __restrict__ hashes,
                         const uint64_t* __restrict__ valid,
                         const int32_t* __restrict__ row_gid,
                         const int32_t* __restrict__ sel,
                         uint8_t* __restrict__ regs, int p, int64_t n) {
  int64_t m = (int64_t)1 << p;
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(valid, i)) continue;
    uint64_t h = (uint64_t)hashes[j];
    uint32_t idx = (uint32_t)(h >> (64 - p));
    uint64_t w = h << p;
    int rho = w == 0 ? (64 - p + 1) : (__clzll((long long)w) + 1);
    atomic_max_u8(&regs[(int64_t)row_gid[j] * m + idx], (uint8_t)rho);
  }
}

extern "C" {

void hipdf_gb_hll(const void* hashes, const void* valid,
                  const void* row_gid, const void* sel, void* regs, int p,
                  int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_hll, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int64_t*)hashes, (const uint64_t*)valid,
                     (const int32_t*)row_gid, (const int32_t*)sel,
                     (uint8_t*)regs, p, n);
}


void hipdf_gb_percentile(const void* vals, const void* perm,
                         const void* starts, const void* vcnt, double p,
                         void* out, int ngroups, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_percentile, flat_grid(ngroups), dim3(HIPDF_BLOCK),
                     0, stream, (const double*)vals, (const int32_t*)perm,
                     (const int64_t*)starts, (const int64_t*)vcnt, p,
                     (double*)out, ngroups);
}

void hipdf_dense_gid(const void* keys, int nkeys, const void* sel,
                     void* row_gid, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_dense_gid, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const DenseKey*)keys, nkeys, (const int32_t*)sel,
                     (int32_t*)row_gid, n);
}

void hipdf_gb_collect_count(const void* vvalid, const void* row_gid,
                            const void* sel, void* counts, int64_t n,
                            hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_collect_count, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint64_t*)vvalid, (const int32_t*)row_gid,
                     (const int32_t*)sel, (int64_t*)counts, n);
}

void hipdf_gb_collect_fill(int esize, const void* vals, const void* vvalid,
                           const void* row_gid, const void* sel,
                           const void* offsets, void* cursor, void* out,
                           int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_collect_fill, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, esize, (const uint8_t*)vals,
                     (const uint64_t*)vvalid, (const int32_t*)row_gid,
                     (const int32_t*)sel, (const int64_t*)offsets,
                     (int64_t*)cursor, (uint8_t*)out, n);
}


void hipdf_gb_build(const void* hashes, const void* keys, int nkeys,
                    const void* sel, void* slot_row, void* row_slot,
                    void* claimed_slots, void* ngroups, int64_t cap,
                    int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_build, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)hashes, (const KeyCol*)keys, nkeys,
                     (const int32_t*)sel, (int32_t*)slot_row,
                     (int32_t*)row_slot, (int32_t*)claimed_slots,
                     (int32_t*)ngroups, (uint32_t)(cap - 1), n);
}

void hipdf_gb_number(const void* claimed_slots, const void* slot_row,
                     void* slot_gid, const void* ngroups, void* leaders,
                     int64_t max_groups, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_number, flat_grid(max_groups), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)claimed_slots,
                     (const int32_t*)slot_row, (int32_t*)slot_gid,
                     (const int32_t*)ngroups, (int32_t*)leaders);
}

void hipdf_gb_rowgid(const void* row_slot, const void* slot_gid, void* row_gid,
                     int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_rowgid, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)row_slot, (const int32_t*)slot_gid,
                     (int32_t*)row_gid, n);
}

void hipdf_gb_agg(int op, int t, const void* vals, const void* vvalid,
                  const void* row_gid, void* acc, void* cnt, int acc_is_double,
                  int32_t ngroups, int64_t n, hipStream_t stream) {
  bool use_lds = ngroups <= GB_LDS_GROUPS;
  size_t lds = use_lds ? (size_t)ngroups * (8 + 8) : 0;
  dim3 grid = flat_grid(n, 4);
  auto launch = [&]<typename T, typename ACC>() {
    // init global accumulators to the op identity first
    hipLaunchKernelGGL((k_gb_acc_init<ACC>), flat_grid(ngroups),
                       dim3(HIPDF_BLOCK), 0, stream, op, (ACC*)acc, ngroups);
    if (use_lds)
      hipLaunchKernelGGL((k_gb_agg<T, ACC, true>), grid, dim3(HIPDF_BLOCK),
                         lds, stream, op, (const T*)vals,
                         (const uint64_t*)vvalid, (const int32_t*)row_gid,
                         (ACC*)acc, (int64_t*)cnt, ngroups, n);
    else
      hipLaunchKernelGGL((k_gb_agg<T, ACC, false>), grid, dim3(HIPDF_BLOCK),
                         0, stream, op, (const T*)vals,
                         (const uint64_t*)vvalid, (const int32_t*)row_gid,
                         (ACC*)acc, (int64_t*)cnt, ngroups, n);
  };
  dispatch_type(t, [&]<typename T>() {
    if (acc_is_double) launch.template operator()<T, double>();
    else launch.template operator()<T, int64_t>();
  });
}

void hipdf_gb_acc_init(int op, void* acc, int acc_is_double, int32_t ngroups,
                       hipStream_t stream) {
  if (acc_is_double)
    hipLaunchKernelGGL((k_gb_acc_init<double>), flat_grid(ngroups),
                       dim3(HIPDF_BLOCK), 0, stream, op, (double*)acc,
                       ngroups);
  else
    hipLaunchKernelGGL((k_gb_acc_init<int64_t>), flat_grid(ngroups),
                       dim3(HIPDF_BLOCK), 0, stream, op, (int64_t*)acc,
                       ngroups);
}

void hipdf_gb_agg_multi(const void* aggs, int naggs, const void* row_gid,
                        const void* sel, int32_t ngroups, int nrep,
                        int64_t n, hipStream_t stream) {
  size_t lds = (size_t)naggs * ngroups * 16;
  bool use_lds = lds > 0 && lds <= 64 * 1024;
  dim3 grid = flat_grid(n, 4);
  if (use_lds)
    hipLaunchKernelGGL((k_gb_agg_multi<true>), grid, dim3(HIPDF_BLOCK), lds,
                       stream, (const AggDesc*)aggs, naggs,
                       (const int32_t*)row_gid, (const int32_t*)sel, ngroups,
                       1, n);
  else
    hipLaunchKernelGGL((k_gb_agg_multi<false>), grid, dim3(HIPDF_BLOCK), 0,
                       stream, (const AggDesc*)aggs, naggs,
                       (const int32_t*)row_gid, (const int32_t*)sel, ngroups,
                       nrep, n);
}

void hipdf_gb_reduce_reps(int op, void* acc, int acc_is_double, void* cnt,
                          int32_t ngroups, int nrep, hipStream_t stream) {
  if (acc_is_double)
    hipLaunchKernelGGL((k_gb_reduce_reps<double>), flat_grid(ngroups),
                       dim3(HIPDF_BLOCK), 0, stream, op, (double*)acc,
                       (int64_t*)cnt, ngroups, nrep);
  else
    hipLaunchKernelGGL((k_gb_reduce_reps<int64_t>), flat_grid(ngroups),
                       dim3(HIPDF_BLOCK), 0, stream, op, (int64_t*)acc,
                       (int64_t*)cnt, ngroups, nrep);
}

void hipdf_mask_from_nonzero(const void* cnt, void* mask, int64_t n,
                             hipStream_t stream) {
  hipLaunchKernelGGL(k_mask_from_nonzero, stripe_grid(n), dim3(HIPDF_BLOCK),
                     0, stream, (const int64_t*)cnt, (uint64_t*)mask,
                     n_stripes(n), n);
}

}  // extern "C"
