// 128-bit decimal kernels (reference analogue: spark-rapids-jni Arithmetic
// — decimal128 add/sub/compare — SURVEY.md §2.8B). Layout: interleaved
// little-endian (lo, hi) int64 word pairs per row (arrow decimal128 LE).
#include "hipdf_common.h"

struct i128 {
  uint64_t lo;
  int64_t hi;
};

__device__ __forceinline__ i128 load128(const int64_t* p, int64_t i) {
  return {(uint64_t)p[2 * i], p[2 * i + 1]};
}

__device__ __forceinline__ void store128(int64_t* p, int64_t i, i128 v) {
  p[2 * i] = (int64_t)v.lo;
  p[2 * i + 1] = v.hi;
}

__device__ __forceinline__ i128 add128(i128 a, i128 b) {
  uint64_t lo = a.lo + b.lo;
  int64_t carry = lo < a.lo ? 1 : 0;
  return {lo, a.hi + b.hi + carry};
}

__device__ __forceinline__ i128 neg128(i128 a) {
  uint64_t lo = ~a.lo + 1;
  int64_t hi = ~a.hi + (lo == 0 ? 1 : 0);
  return {lo, hi};
}

__device__ __forceinline__ int cmp128(i128 a, i128 b) {
  if (a.hi != b.hi) return a.hi < b.hi ? -1 : 1;
  if (a.lo != b.lo) return a.lo < b.lo ? -1 : 1;
  return 0;
}

// ---- elementwise ---------------------------------------------------------

// op: 0 add, 1 sub, 22 min, 23 max (matches BinOp ids used by python)
__global__ void k_i128_arith(int op, const int64_t* __restrict__ a,
                             const int64_t* __restrict__ b,
                             const uint64_t* __restrict__ av,
                             const uint64_t* __restrict__ bv,
                             int64_t* __restrict__ out,
                             uint64_t* __restrict__ ov, int64_t nstripe,
                             int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row) && valid_bit(bv, row);
      i128 x = load128(a, row), y = load128(b, row);
      i128 r{0, 0};
      if (ok) {
        switch (op) {
          case 0: r = add128(x, y); break;
          case 1: r = add128(x, neg128(y)); break;
          case 22: r = cmp128(x, y) <= 0 ? x : y; break;
          case 23: r = cmp128(x, y) >= 0 ? x : y; break;
        }
      }
      store128(out, row, r);
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// cmp op ids match StrCmpOp: 0 eq 1 ne 2 lt 3 le 4 gt 5 ge
__global__ void k_i128_cmp(int op, const int64_t* __restrict__ a,
                           const int64_t* __restrict__ b,
                           const uint64_t* __restrict__ av,
                           const uint64_t* __restrict__ bv,
                           uint8_t* __restrict__ out,
                           uint64_t* __restrict__ ov, int64_t nstripe,
                           int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row) && valid_bit(bv, row);
      int c = cmp128(load128(a, row), load128(b, row));
      bool r;
      switch (op) {
        case 0: r = c == 0; break;
        case 1: r = c != 0; break;
        case 2: r = c < 0; break;
        case 3: r = c <= 0; break;
        case 4: r = c > 0; break;
        default: r = c >= 0; break;
      }
      out[row] = ok ? (uint8_t)r : 0;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// sign-extend int64 -> int128 pairs (decimal64 -> decimal128 widening)
__global__ void k_i64_to_i128(const int64_t* __restrict__ in,
                              int64_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t v = in[i];
    out[2 * i] = v;
    out[2 * i + 1] = v < 0 ? -1 : 0;
  }
}

// int128 pairs -> double (for casts / mean)
__global__ void k_i128_to_f64(const int64_t* __restrict__ in,
                              double* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    i128 v = load128(in, i);
    out[i] = (double)v.hi * 18446744073709551616.0 + (double)v.lo;
  }
}

// grouped sum of int64 values into int128 accumulators (carry-correct
// split atomics: lo add returns the old value, carry derived from wrap)
__global__ void k_gb_sum_i64_to_i128(const int64_t* __restrict__ vals,
                                     const uint64_t* __restrict__ vvalid,
                                     const int32_t* __restrict__ row_gid,
                                     const int32_t* __restrict__ sel,
                                     int64_t* __restrict__ acc,  // 2*ngroups
                                     int64_t* __restrict__ cnt,
                                     int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(vvalid, i)) continue;
    int32_t g = row_gid[j];
    int64_t v = vals[i];
    uint64_t lo = (uint64_t)v;
    int64_t hi = v < 0 ? -1 : 0;
    unsigned long long old = atomicAdd((unsigned long long*)&acc[2 * g],
                                       (unsigned long long)lo);
    if (old + lo < old)  // wrapped: carry into the high word
      atomicAdd((unsigned long long*)&acc[2 * g + 1], 1ull);
    if (hi)
      atomicAdd((unsigned long long*)&acc[2 * g + 1],
                (unsigned long long)hi);
    atomicAdd((unsigned long long*)&cnt[g], 1ull);
  }
}

// grouped sum of int128 values into int128 accumulators
__global__ void k_gb_sum_i128(const int64_t* __restrict__ vals,
                              const uint64_t* __restrict__ vvalid,
                              const int32_t* __restrict__ row_gid,
                              const int32_t* __restrict__ sel,
                              int64_t* __restrict__ acc,
                              int64_t* __restrict__ cnt, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(vvalid, i)) continue;
    int32_t g = row_gid[j];
    i128 v = load128(vals, i);
    unsigned long long old = atomicAdd((unsigned long long*)&acc[2 * g],
                                       (unsigned long long)v.lo);
    if (old + v.lo < old)
      atomicAdd((unsigned long long*)&acc[2 * g + 1], 1ull);
    if (v.hi)
      atomicAdd((unsigned long long*)&acc[2 * g + 1],
                (unsigned long long)v.hi);
    atomicAdd((unsigned long long*)&cnt[g], 1ull);
  }
}

// LDS-staged grouped i128 sum for small group counts: each block
// accumulates into shared memory (24 B/group: lo, hi, cnt) and flushes
// once, turning ~n global atomics on a few hot addresses into
// ngroups-per-block. in_is_64: values are int64 (widened) vs i128 pairs.
__global__ void k_gb_sum_i128_lds(int in_is_64,
                                  const int64_t* __restrict__ vals,
                                  const uint64_t* __restrict__ vvalid,
                                  const int32_t* __restrict__ row_gid,
                                  const int32_t* __restrict__ sel,
                                  int64_t* __restrict__ acc,
                                  int64_t* __restrict__ cnt, int ngroups,
                                  int64_t n) {
  extern __shared__ unsigned long long smem[];
  unsigned long long* s_lo = smem;
  unsigned long long* s_hi = smem + ngroups;
  unsigned long long* s_cnt = smem + 2 * ngroups;
  for (int g = threadIdx.x; g < 3 * ngroups; g += blockDim.x) smem[g] = 0;
  __syncthreads();
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = sel ? (int64_t)sel[j] : j;
    if (!valid_bit(vvalid, i)) continue;
    int32_t g = row_gid[j];
    uint64_t lo;
    int64_t hi;
    if (in_is_64) {
      int64_t v = vals[i];
      lo = (uint64_t)v;
      hi = v < 0 ? -1 : 0;
    } else {
      i128 v = load128(vals, i);
      lo = v.lo;
      hi = v.hi;
    }
    unsigned long long old = atomicAdd(&s_lo[g], (unsigned long long)lo);
    if (old + lo < old) atomicAdd(&s_hi[g], 1ull);
    if (hi) atomicAdd(&s_hi[g], (unsigned long long)hi);
    atomicAdd(&s_cnt[g], 1ull);
  }
  __syncthreads();
  for (int g = threadIdx.x; g < ngroups; g += blockDim.x) {
    unsigned long long lo = s_lo[g];
    if (lo) {
      unsigned long long old =
          atomicAdd((unsigned long long*)&acc[2 * g], lo);
      if (old + lo < old)
        atomicAdd((unsigned long long*)&acc[2 * g + 1], 1ull);
    }
    if (s_hi[g])
      atomicAdd((unsigned long long*)&acc[2 * g + 1], s_hi[g]);
    if (s_cnt[g]) atomicAdd((unsigned long long*)&cnt[g], s_cnt[g]);
  }
}

// ---- decimal multiply / divide -------------------------------------------
// Spark DecimalPrecision result scales (reference analogue: GpuMultiply /
// GpuDivide over cudf fixed-point). Operands are decimal64 (int64 backing,
// each at its own scale); native __int128 keeps the math exact:
//   mul: r = HALF_UP((x*y) / 10^shift)         shift = s1+s2-st >= 0
//   div: r = HALF_UP((x*10^shift) / y)         shift = st+s2-s1 (may be <0)
// The div numerator is built by chunked long division so x*10^shift never
// has to fit in 128 bits. NULL on divide-by-zero and on overflow of
// 10^out_prec (Spark non-ANSI overflow -> null).

typedef unsigned __int128 u128;

__device__ __forceinline__ u128 pow10_128(int p) {
  u128 r = 1;
  for (int i = 0; i < p; ++i) r *= 10;
  return r;
}

__global__ void k_dec64_mul_div(int is_div, const int64_t* __restrict__ a,
                                const int64_t* __restrict__ b,
                                const uint64_t* __restrict__ av,
                                const uint64_t* __restrict__ bv,
                                int64_t* __restrict__ out,
                                uint64_t* __restrict__ ov, int out_is_128,
                                int shift, int out_prec, int64_t nstripe,
                                int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  const u128 u128_max = ~(u128)0;
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row) && valid_bit(bv, row);
      u128 q = 0;
      bool neg = false;
      if (ok) {
        int64_t xs = a[row], ys = b[row];
        neg = (xs < 0) != (ys < 0);
        u128 x = (u128)(xs < 0 ? -(__int128)xs : (__int128)xs);
        u128 y = (u128)(ys < 0 ? -(__int128)ys : (__int128)ys);
        if (!is_div) {
          q = x * y;  // |x|,|y| < 10^19 -> fits
          if (shift > 0) {
            u128 d = pow10_128(shift);
            q = (2 * q + d) / (2 * d);
          }
        } else if (ys == 0) {
          ok = false;
        } else {
          u128 den = y;
          int m = shift;
          if (m < 0) {
            den = den * pow10_128(-m);
            m = 0;
          }
          q = x / den;
          u128 r = x % den;
          while (m > 0) {
            int c = m > 18 ? 18 : m;
            u128 p = pow10_128(c);
            if (q > u128_max / p) {
              q = u128_max;  // forces the precision-overflow null below
              break;
            }
            q = q * p + (r * p) / den;
            r = (r * p) % den;
            m -= c;
          }
          if (2 * r >= den) q += 1;  // HALF_UP on the magnitude
        }
        if (ok && q >= pow10_128(out_prec)) ok = false;
      }
      __int128 v = ok ? (neg ? -(__int128)q : (__int128)q) : 0;
      if (out_is_128) {
        out[2 * row] = (int64_t)(u128)v;
        out[2 * row + 1] = (int64_t)((u128)v >> 64);
      } else {
        out[row] = (int64_t)v;
      }
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// rescale int128 decimal values by 10^|shift|: shift>0 multiplies (null on
// u128 overflow), shift<0 divides with HALF_UP on the magnitude. Result is
// nulled when |v| >= 10^out_prec (Spark non-ANSI cast overflow -> null;
// out_prec<=38 so the bound fits u128). out_is_64 narrows to int64 words
// (decimal128 -> decimal64 cast), otherwise writes (lo,hi) pairs.
__global__ void k_i128_rescale(const int64_t* __restrict__ in,
                               const uint64_t* __restrict__ iv,
                               int64_t* __restrict__ out,
                               uint64_t* __restrict__ ov, int shift,
                               int out_prec, int out_is_64, int64_t nstripe,
                               int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  const u128 u128_max = ~(u128)0;
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(iv, row);
      u128 q = 0;
      bool neg = false;
      if (ok) {
        i128 v = load128(in, row);
        neg = v.hi < 0;
        i128 m = neg ? neg128(v) : v;
        q = ((u128)(uint64_t)m.hi << 64) | m.lo;
        if (shift > 0) {
          u128 p = pow10_128(shift);
          if (q > u128_max / p) ok = false;
          else q *= p;
        } else if (shift < 0) {
          u128 d = pow10_128(-shift);
          u128 r = q % d;
          q /= d;
          if (2 * r >= d) q += 1;
        }
        if (ok && out_prec <= 38 && q >= pow10_128(out_prec)) ok = false;
        if (ok && out_is_64 && q > (u128)0x7fffffffffffffffULL) ok = false;
      }
      __int128 w = ok ? (neg ? -(__int128)q : (__int128)q) : 0;
      if (out_is_64) {
        out[row] = (int64_t)w;
      } else {
        out[2 * row] = (int64_t)(u128)w;
        out[2 * row + 1] = (int64_t)((u128)w >> 64);
      }
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// ---- wide decimal multiply / divide (decimal128 operands) ----------------
// mul: 256-bit product of the u128 magnitudes, then HALF_UP removal of
// 10^shift one digit at a time (the last removed digit decides rounding).
// div: schoolbook long division producing one decimal digit per step with
// an overflow-safe u192 remainder (r*10 can exceed u128 when the divisor
// is close to 10^38).

struct u256 {
  uint64_t w[4];  // little-endian limbs
};

__device__ __forceinline__ u256 mul_u128(u128 a, u128 b) {
  uint64_t a0 = (uint64_t)a, a1 = (uint64_t)(a >> 64);
  uint64_t b0 = (uint64_t)b, b1 = (uint64_t)(b >> 64);
  u128 p00 = (u128)a0 * b0;
  u128 p01 = (u128)a0 * b1;
  u128 p10 = (u128)a1 * b0;
  u128 p11 = (u128)a1 * b1;
  u256 r;
  r.w[0] = (uint64_t)p00;
  u128 mid = (u128)(uint64_t)(p00 >> 64) + (uint64_t)p01 + (uint64_t)p10;
  r.w[1] = (uint64_t)mid;
  u128 hi = (u128)(uint64_t)(mid >> 64) + (uint64_t)(p01 >> 64) +
            (uint64_t)(p10 >> 64) + (uint64_t)p11;
  r.w[2] = (uint64_t)hi;
  r.w[3] = (uint64_t)(hi >> 64) + (uint64_t)(p11 >> 64);
  return r;
}

// v /= 10, returns the remainder digit
__device__ __forceinline__ int div10_u256(u256* v) {
  u128 rem = 0;
  for (int i = 3; i >= 0; --i) {
    u128 cur = (rem << 64) | v->w[i];
    v->w[i] = (uint64_t)(cur / 10);
    rem = cur % 10;
  }
  return (int)rem;
}

__global__ void k_dec_mul_div_wide(int is_div, const int64_t* __restrict__ a,
                                   const int64_t* __restrict__ b,
                                   const uint64_t* __restrict__ av,
                                   const uint64_t* __restrict__ bv,
                                   int a_is_128, int b_is_128,
                                   int64_t* __restrict__ out,
                                   uint64_t* __restrict__ ov, int out_is_128,
                                   int shift, int out_prec, int64_t nstripe,
                                   int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row) && valid_bit(bv, row);
      u128 q = 0;
      bool neg = false;
      if (ok) {
        u128 x, y;
        bool xn, yn;
        if (a_is_128) {
          i128 v = load128(a, row);
          xn = v.hi < 0;
          i128 m = xn ? neg128(v) : v;
          x = ((u128)(uint64_t)m.hi << 64) | m.lo;
        } else {
          int64_t v = a[row];
          xn = v < 0;
          x = (u128)(xn ? -(__int128)v : (__int128)v);
        }
        if (b_is_128) {
          i128 v = load128(b, row);
          yn = v.hi < 0;
          i128 m = yn ? neg128(v) : v;
          y = ((u128)(uint64_t)m.hi << 64) | m.lo;
        } else {
          int64_t v = b[row];
          yn = v < 0;
          y = (u128)(yn ? -(__int128)v : (__int128)v);
        }
        neg = xn != yn;
        if (!is_div) {
          u256 p = mul_u128(x, y);
          int last = 0;
          for (int k = 0; k < shift; ++k) last = div10_u256(&p);
          if (p.w[2] | p.w[3]) {
            ok = false;
          } else {
            q = ((u128)p.w[1] << 64) | p.w[0];
            if (last >= 5) q += 1;
          }
        } else if (y == 0) {
          ok = false;
        } else {
          int m = shift;
          if (m < 0) {
            // denominator gains the digits instead; y*10^-m must fit
            for (int k = 0; k < -m && ok; ++k) {
              if (y > ~(u128)0 / 10) ok = false;
              else y *= 10;
            }
            m = 0;
          }
          if (ok) {
            q = x / y;
            u128 r = x % y;
            const u128 q_lim = (~(u128)0 - 9) / 10;
            for (int k = 0; k < m && ok; ++k) {
              if (q > q_lim) { ok = false; break; }
              // u192 rem10 = r * 10 (can exceed u128)
              uint64_t rl = (uint64_t)r, rh = (uint64_t)(r >> 64);
              u128 lo10 = (u128)rl * 10;
              u128 hi10 = (u128)rh * 10 + (uint64_t)(lo10 >> 64);
              u128 lo = ((u128)(uint64_t)hi10 << 64) | (uint64_t)lo10;
              uint32_t hi = (uint32_t)(hi10 >> 64);
              int digit = 0;
              while (hi || lo >= y) {
                if (lo < y) --hi;
                lo -= y;
                ++digit;
              }
              q = q * 10 + digit;
              r = lo;
            }
            if (ok && 2 * r >= y) {
              // HALF_UP on the true remainder of the last digit
              if (q == ~(u128)0) ok = false;
              else q += 1;
            }
          }
        }
        if (ok && out_prec <= 38 && q >= pow10_128(out_prec)) ok = false;
        if (ok && !out_is_128 && q > (u128)0x7fffffffffffffffULL) ok = false;
      }
      __int128 w = ok ? (neg ? -(__int128)q : (__int128)q) : 0;
      if (out_is_128) {
        out[2 * row] = (int64_t)(u128)w;
        out[2 * row + 1] = (int64_t)((u128)w >> 64);
      } else {
        out[row] = (int64_t)w;
      }
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

extern "C" {

void hipdf_dec_mul_div_wide(int is_div, const void* a, const void* b,
                            const void* av, const void* bv, int a_is_128,
                            int b_is_128, void* out, void* ov,
                            int out_is_128, int shift, int out_prec,
                            int64_t n, hipStream_t stream) {
  int64_t nstripe = (n + WAVE - 1) / WAVE;
  hipLaunchKernelGGL(k_dec_mul_div_wide, stripe_grid(nstripe),
                     dim3(HIPDF_BLOCK), 0, stream, is_div,
                     (const int64_t*)a, (const int64_t*)b,
                     (const uint64_t*)av, (const uint64_t*)bv, a_is_128,
                     b_is_128, (int64_t*)out, (uint64_t*)ov, out_is_128,
                     shift, out_prec, nstripe, n);
}

void hipdf_i128_rescale(const void* in, const void* iv, void* out, void* ov,
                        int shift, int out_prec, int out_is_64, int64_t n,
                        hipStream_t stream) {
  int64_t nstripe = (n + WAVE - 1) / WAVE;
  hipLaunchKernelGGL(k_i128_rescale, stripe_grid(nstripe), dim3(HIPDF_BLOCK),
                     0, stream, (const int64_t*)in, (const uint64_t*)iv,
                     (int64_t*)out, (uint64_t*)ov, shift, out_prec, out_is_64,
                     nstripe, n);
}

void hipdf_dec64_mul_div(int is_div, const void* a, const void* b,
                         const void* av, const void* bv, void* out, void* ov,
                         int out_is_128, int shift, int out_prec, int64_t n,
                         hipStream_t stream) {
  int64_t nstripe = (n + WAVE - 1) / WAVE;
  hipLaunchKernelGGL(k_dec64_mul_div, stripe_grid(nstripe), dim3(HIPDF_BLOCK),
                     0, stream, is_div, (const int64_t*)a, (const int64_t*)b,
                     (const uint64_t*)av, (const uint64_t*)bv, (int64_t*)out,
                     (uint64_t*)ov, out_is_128, shift, out_prec, nstripe, n);
}

void hipdf_i128_arith(int op, const void* a, const void* b, const void* av,
                      const void* bv, void* out, void* ov, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_i128_arith, stripe_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, op, (const int64_t*)a, (const int64_t*)b,
                     (const uint64_t*)av, (const uint64_t*)bv, (int64_t*)out,
                     (uint64_t*)ov, n_stripes(n), n);
}

void hipdf_i128_cmp(int op, const void* a, const void* b, const void* av,
                    const void* bv, void* out, void* ov, int64_t n,
                    hipStream_t stream) {
  hipLaunchKernelGGL(k_i128_cmp, stripe_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     op, (const int64_t*)a, (const int64_t*)b,
                     (const uint64_t*)av, (const uint64_t*)bv, (uint8_t*)out,
                     (uint64_t*)ov, n_stripes(n), n);
}

void hipdf_i64_to_i128(const void* in, void* out, int64_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_i64_to_i128, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)in, (int64_t*)out, n);
}

void hipdf_i128_to_f64(const void* in, void* out, int64_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_i128_to_f64, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)in, (double*)out, n);
}

void hipdf_gb_sum_i128_lds(int in_is_64, const void* vals,
                           const void* vvalid, const void* row_gid,
                           const void* sel, void* acc, void* cnt,
                           int ngroups, int64_t n, hipStream_t stream) {
  size_t shmem = (size_t)ngroups * 24;
  int64_t blocks = (n + HIPDF_BLOCK - 1) / HIPDF_BLOCK;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(k_gb_sum_i128_lds, dim3((uint32_t)blocks),
                     dim3(HIPDF_BLOCK), shmem, stream, in_is_64,
                     (const int64_t*)vals, (const uint64_t*)vvalid,
                     (const int32_t*)row_gid, (const int32_t*)sel,
                     (int64_t*)acc, (int64_t*)cnt, ngroups, n);
}

void hipdf_gb_sum_i64_to_i128(const void* vals, const void* vvalid,
                              const void* row_gid, const void* sel, void* acc,
                              void* cnt, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_sum_i64_to_i128, flat_grid(n, 4),
                     dim3(HIPDF_BLOCK), 0, stream, (const int64_t*)vals,
                     (const uint64_t*)vvalid, (const int32_t*)row_gid,
                     (const int32_t*)sel, (int64_t*)acc, (int64_t*)cnt, n);
}

void hipdf_gb_sum_i128(const void* vals, const void* vvalid,
                       const void* row_gid, const void* sel, void* acc,
                       void* cnt, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_gb_sum_i128, flat_grid(n, 4), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)vals, (const uint64_t*)vvalid,
                     (const int32_t*)row_gid, (const int32_t*)sel,
                     (int64_t*)acc, (int64_t*)cnt, n);
}

}  // extern "C"
