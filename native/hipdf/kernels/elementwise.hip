// Elementwise binary/unary/cast kernels with Spark SQL semantics.
// Wave64 stripe structure: lane l of each wave handles row stripe*64+l and
// the wave __ballot()s per-row validity into the output bitmask word — the
// bitmask write is free (one store per 64 rows). Null semantics implemented
// in-kernel: div/mod by zero -> NULL, log(<=0) -> NULL, Kleene AND/OR,
// NaN==NaN true and NaN greatest (Spark ordering).
#include "hipdf_common.h"
#include <math.h>

enum BinOp : int {
  OP_ADD = 0, OP_SUB, OP_MUL, OP_DIV, OP_INT_DIV, OP_MOD, OP_PMOD, OP_POW,
  OP_EQ, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE, OP_EQ_NS, OP_AND, OP_OR,
  OP_BITAND, OP_BITOR, OP_BITXOR, OP_SHL, OP_SHR, OP_MIN, OP_MAX, OP_ROUND,
};

enum UnOp : int {
  UOP_NEG = 0, UOP_ABS, UOP_NOT, UOP_SQRT, UOP_EXP, UOP_LOG, UOP_FLOOR,
  UOP_CEIL, UOP_SIN, UOP_COS, UOP_TAN, UOP_ISNAN, UOP_YEAR, UOP_MONTH,
  UOP_DAY,
};

template <typename T> __device__ __forceinline__ bool is_nan_v(T) { return false; }
template <> __device__ __forceinline__ bool is_nan_v<float>(float v) { return isnan(v); }
template <> __device__ __forceinline__ bool is_nan_v<double>(double v) { return isnan(v); }

template <typename T> struct IsFloat { static constexpr bool value = false; };
template <> struct IsFloat<float> { static constexpr bool value = true; };
template <> struct IsFloat<double> { static constexpr bool value = true; };

// Spark total order comparisons: NaN == NaN, NaN greater than everything.
template <typename T>
__device__ __forceinline__ bool spark_eq(T a, T b) {
  if (IsFloat<T>::value && is_nan_v(a) && is_nan_v(b)) return true;
  return a == b;
}
template <typename T>
__device__ __forceinline__ bool spark_lt(T a, T b) {
  if (IsFloat<T>::value) {
    if (is_nan_v(a)) return false;
    if (is_nan_v(b)) return true;
  }
  return a < b;
}

template <typename T>
__device__ __forceinline__ T fmod_t(T a, T b) {
  return (T)fmod((double)a, (double)b);
}
template <> __device__ __forceinline__ float fmod_t<float>(float a, float b) {
  return fmodf(a, b);
}
template <> __device__ __forceinline__ int8_t fmod_t(int8_t a, int8_t b) { return (int8_t)(a % b); }
template <> __device__ __forceinline__ int16_t fmod_t(int16_t a, int16_t b) { return (int16_t)(a % b); }
template <> __device__ __forceinline__ int32_t fmod_t(int32_t a, int32_t b) { return a % b; }
template <> __device__ __forceinline__ int64_t fmod_t(int64_t a, int64_t b) { return a % b; }
template <> __device__ __forceinline__ uint8_t fmod_t(uint8_t a, uint8_t b) { return a % b; }

template <typename T>
__device__ __forceinline__ int64_t to_i64(T v) { return (int64_t)v; }

// Arithmetic: T x T -> T. Returns value; sets `ok=false` for null-producing
// ops (div/mod family with zero divisor).
template <typename T>
__device__ __forceinline__ T arith_one(int op, T a, T b, bool& ok) {
  switch (op) {
    case OP_ADD: return a + b;
    case OP_SUB: return a - b;
    case OP_MUL: return a * b;
    case OP_DIV: {
      if (b == (T)0) { ok = false; return (T)0; }
      return a / b;  // only instantiated for float types by the planner
    }
    case OP_INT_DIV: {
      if (b == (T)0) { ok = false; return (T)0; }
      if (IsFloat<T>::value) return (T)trunc((double)a / (double)b);
      // match Spark/Java: Long.MIN_VALUE / -1 wraps
      int64_t ia = to_i64(a), ib = to_i64(b);
      if (ib == -1) return (T)(-(uint64_t)ia);
      return (T)(ia / ib);
    }
    case OP_MOD: {
      if (b == (T)0) { ok = false; return (T)0; }
      if (!IsFloat<T>::value && to_i64(b) == -1) return (T)0;
      return fmod_t(a, b);
    }
    case OP_PMOD: {
      if (b == (T)0) { ok = false; return (T)0; }
      T r = (!IsFloat<T>::value && to_i64(b) == -1) ? (T)0 : fmod_t(a, b);
      if (r != (T)0 && ((r < (T)0) != (b < (T)0))) r = r + b;
      return r;
    }
    case OP_POW: return (T)pow((double)a, (double)b);
    case OP_BITAND: return (T)(to_i64(a) & to_i64(b));
    case OP_BITOR: return (T)(to_i64(a) | to_i64(b));
    case OP_BITXOR: return (T)(to_i64(a) ^ to_i64(b));
    case OP_SHL: return (T)(to_i64(a) << (to_i64(b) & (sizeof(T) == 8 ? 63 : 31)));
    case OP_SHR: return (T)(to_i64(a) >> (to_i64(b) & (sizeof(T) == 8 ? 63 : 31)));
    case OP_MIN: return spark_lt(a, b) ? a : b;
    case OP_MAX: return spark_lt(a, b) ? b : a;
    case OP_ROUND: {
      // HALF_UP at scale encoded as b = 10^scale (Spark round())
      double p = (double)b;
      double x = (double)a * p;
      double r = x >= 0 ? floor(x + 0.5) : ceil(x - 0.5);
      return (T)(r / p);
    }
    default: return (T)0;
  }
}

template <typename T>
__device__ __forceinline__ bool cmp_one(int op, T a, T b) {
  switch (op) {
    case OP_EQ: return spark_eq(a, b);
    case OP_NE: return !spark_eq(a, b);
    case OP_LT: return spark_lt(a, b);
    case OP_LE: return spark_lt(a, b) || spark_eq(a, b);
    case OP_GT: return spark_lt(b, a);
    case OP_GE: return spark_lt(b, a) || spark_eq(a, b);
    default: return false;
  }
}

// SCALAR_RHS: b comes from a broadcast scalar (sb) instead of a column.
template <typename T, bool SCALAR_RHS>
__global__ void k_binary_arith(int op, const T* __restrict__ a,
                               const T* __restrict__ b, T sb,
                               const uint64_t* __restrict__ av,
                               const uint64_t* __restrict__ bv,
                               T* __restrict__ out, uint64_t* __restrict__ ov,
                               int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    T r = (T)0;
    if (row < n) {
      ok = valid_bit(av, row) && (SCALAR_RHS || valid_bit(bv, row));
      T va = a[row];
      T vb = SCALAR_RHS ? sb : b[row];
      if (ok) r = arith_one<T>(op, va, vb, ok);
      out[row] = r;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

template <typename T, bool SCALAR_RHS>
__global__ void k_binary_cmp(int op, const T* __restrict__ a,
                             const T* __restrict__ b, T sb,
                             const uint64_t* __restrict__ av,
                             const uint64_t* __restrict__ bv,
                             uint8_t* __restrict__ out,
                             uint64_t* __restrict__ ov,
                             int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      bool va_ok = valid_bit(av, row);
      bool vb_ok = SCALAR_RHS || valid_bit(bv, row);
      T va = a[row];
      T vb = SCALAR_RHS ? sb : b[row];
      uint8_t r;
      if (op == OP_EQ_NS) {  // null-safe equal: never null
        ok = true;
        r = (va_ok && vb_ok) ? (uint8_t)spark_eq(va, vb)
                             : (uint8_t)(va_ok == vb_ok);
      } else {
        ok = va_ok && vb_ok;
        r = ok ? (uint8_t)cmp_one<T>(op, va, vb) : (uint8_t)0;
      }
      out[row] = r;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// Kleene AND/OR on bool(u8) columns.
template <bool SCALAR_RHS>
__global__ void k_binary_bool(int op, const uint8_t* __restrict__ a,
                              const uint8_t* __restrict__ b, uint8_t sb,
                              const uint64_t* __restrict__ av,
                              const uint64_t* __restrict__ bv,
                              uint8_t* __restrict__ out,
                              uint64_t* __restrict__ ov,
                              int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      bool va_ok = valid_bit(av, row);
      bool vb_ok = SCALAR_RHS || valid_bit(bv, row);
      bool va = a[row] != 0;
      bool vb = (SCALAR_RHS ? sb : b[row]) != 0;
      bool r;
      if (op == OP_AND) {
        r = va && vb;
        ok = (va_ok && vb_ok) || (va_ok && !va) || (vb_ok && !vb);
      } else {  // OP_OR
        r = va || vb;
        ok = (va_ok && vb_ok) || (va_ok && va) || (vb_ok && vb);
      }
      out[row] = (uint8_t)r;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// ---------------- unary --------------------------------------------------

// days-since-epoch -> civil (Howard Hinnant's algorithm, public domain)
__device__ __forceinline__ void civil_from_days(int32_t z, int32_t& y,
                                                int32_t& m, int32_t& d) {
  z += 719468;
  int32_t era = (z >= 0 ? z : z - 146096) / 146097;
  uint32_t doe = (uint32_t)(z - era * 146097);
  uint32_t yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int32_t yr = (int32_t)yoe + era * 400;
  uint32_t doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  uint32_t mp = (5 * doy + 2) / 153;
  d = (int32_t)(doy - (153 * mp + 2) / 5 + 1);
  m = (int32_t)(mp < 10 ? mp + 3 : mp - 9);
  y = yr + (m <= 2);
}

template <typename T>
__global__ void k_unary(int op, const T* __restrict__ a,
                        const uint64_t* __restrict__ av, T* __restrict__ out,
                        uint64_t* __restrict__ ov, int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row);
      T v = a[row];
      T r = (T)0;
      if (ok) {
        switch (op) {
          case UOP_NEG: r = (T)(-to_i64(v)); if (IsFloat<T>::value) r = (T)(-(double)v); break;
          case UOP_ABS: r = spark_lt(v, (T)0) ? (T)(-(double)v) : v;
                        if (!IsFloat<T>::value) r = v < (T)0 ? (T)(-to_i64(v)) : v; break;
          case UOP_SQRT: r = (T)sqrt((double)v); break;
          case UOP_EXP: r = (T)exp((double)v); break;
          case UOP_LOG:
            if ((double)v <= 0.0) { ok = false; }
            else r = (T)log((double)v);
            break;
          case UOP_FLOOR: r = (T)floor((double)v); break;
          case UOP_CEIL: r = (T)ceil((double)v); break;
          case UOP_SIN: r = (T)sin((double)v); break;
          case UOP_COS: r = (T)cos((double)v); break;
          case UOP_TAN: r = (T)tan((double)v); break;
          default: r = v; break;
        }
      }
      out[row] = r;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

__global__ void k_not(const uint8_t* __restrict__ a, uint8_t* __restrict__ out,
                      int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = a[i] == 0 ? 1 : 0;
}

template <typename T>
__global__ void k_isnan(const T* __restrict__ a,
                        const uint64_t* __restrict__ av,
                        uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (valid_bit(av, i) && is_nan_v(a[i])) ? 1 : 0;
}

__global__ void k_date_field(int op, const int32_t* __restrict__ days,
                             int32_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t y, m, d;
    civil_from_days(days[i], y, m, d);
    out[i] = op == UOP_YEAR ? y : (op == UOP_MONTH ? m : d);
  }
}

// ---------------- cast ----------------------------------------------------

template <typename To>
__device__ __forceinline__ To sat_from_double(double v) {
  // Spark (non-ANSI) double -> integral: NaN -> 0, saturate at bounds
  if (isnan(v)) return (To)0;
  constexpr double lo = (double)std::numeric_limits<To>::min();
  constexpr double hi = (double)std::numeric_limits<To>::max();
  if (v <= lo) return std::numeric_limits<To>::min();
  if (v >= hi) return std::numeric_limits<To>::max();
  return (To)v;  // truncation toward zero
}

template <typename From, typename To>
__global__ void k_cast(const From* __restrict__ a, To* __restrict__ out,
                       int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    From v = a[i];
    if constexpr (IsFloat<From>::value && !IsFloat<To>::value) {
      out[i] = sat_from_double<To>((double)v);
    } else {
      out[i] = (To)v;
    }
  }
}

// bool target: nonzero -> 1
template <typename From>
__global__ void k_cast_to_bool(const From* __restrict__ a,
                               uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = a[i] != (From)0 ? 1 : 0;
}

// f64 -> i64 with round-to-nearest-even (float -> decimal cast; matches
// the CPU backend's np.round; plain k_cast truncates like Spark int casts)
__global__ void k_f64_to_i64_rint(const double* __restrict__ a,
                                  int64_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    double v = rint(a[i]);
    if (v != v) out[i] = 0;
    else if (v >= 9.223372036854775807e18) out[i] = INT64_MAX;
    else if (v <= -9.223372036854775808e18) out[i] = INT64_MIN;
    else out[i] = (int64_t)v;
  }
}

// decimal rescale: out = in * 10^k (k>0) or round-half-up(in / 10^-k)
__global__ void k_decimal_rescale(const int64_t* __restrict__ a,
                                  int64_t* __restrict__ out, int64_t pow10,
                                  int up, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t v = a[i];
    if (up) {
      out[i] = v * pow10;
    } else {
      int64_t q = v / pow10;
      int64_t r = v - q * pow10;
      int64_t half = pow10 / 2;
      if (r >= half + (pow10 & 1)) q += 1;
      else if (-r >= half + (pow10 & 1)) q -= 1;
      else if (r == half && pow10 % 2 == 0 && v > 0) q += 1;   // half-up
      else if (-r == half && pow10 % 2 == 0 && v < 0) q -= 1;
      out[i] = q;
    }
  }
}

// if_else (CASE WHEN): rows where cond is true AND valid take a, else b
template <typename T>
__global__ void k_if_else(const uint8_t* __restrict__ cond,
                          const uint64_t* __restrict__ cv,
                          const T* __restrict__ a,
                          const uint64_t* __restrict__ av,
                          const T* __restrict__ b,
                          const uint64_t* __restrict__ bv, T* __restrict__ out,
                          uint64_t* __restrict__ ov, int64_t nstripe,
                          int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t row = s * WAVE + lane;
    bool ok = false;
    if (row < n) {
      bool take_a = cond[row] != 0 && valid_bit(cv, row);
      ok = take_a ? valid_bit(av, row) : valid_bit(bv, row);
      out[row] = take_a ? a[row] : b[row];
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s, ballot, lane);
  }
}

// expand a validity bitmask into one byte per row (invert for IS NULL)
__global__ void k_mask_expand(const uint64_t* __restrict__ mask,
                              uint8_t* __restrict__ out, int invert,
                              int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool v = valid_bit(mask, i);
    out[i] = (uint8_t)(invert ? !v : v);
  }
}

// ---------------- host entry points ---------------------------------------

extern "C" {

void hipdf_if_else(int t, const void* cond, const void* cv, const void* a,
                   const void* av, const void* b, const void* bv, void* out,
                   void* ov, int64_t n, hipStream_t stream) {
  dim3 grid = stripe_grid(n);
  dispatch_type(t, [&]<typename T>() {
    hipLaunchKernelGGL((k_if_else<T>), grid, dim3(HIPDF_BLOCK), 0, stream,
                       (const uint8_t*)cond, (const uint64_t*)cv, (const T*)a,
                       (const uint64_t*)av, (const T*)b, (const uint64_t*)bv,
                       (T*)out, (uint64_t*)ov, n_stripes(n), n);
  });
}

void hipdf_mask_expand(const void* mask, void* out, int invert, int64_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_mask_expand, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint64_t*)mask, (uint8_t*)out, invert, n);
}

void hipdf_binary_arith(int op, int t, const void* a, const void* b,
                        double sd, int64_t si, int scalar_rhs, const void* av,
                        const void* bv, void* out, void* ov, int64_t n,
                        hipStream_t stream) {
  dim3 grid = stripe_grid(n);
  dispatch_type(t, [&]<typename T>() {
    T sb = IsFloat<T>::value ? (T)sd : (T)si;
    if (scalar_rhs)
      hipLaunchKernelGGL((k_binary_arith<T, true>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const T*)b, sb,
                         (const uint64_t*)av, (const uint64_t*)bv, (T*)out,
                         (uint64_t*)ov, n_stripes(n), n);
    else
      hipLaunchKernelGGL((k_binary_arith<T, false>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const T*)b, sb,
                         (const uint64_t*)av, (const uint64_t*)bv, (T*)out,
                         (uint64_t*)ov, n_stripes(n), n);
  });
}

void hipdf_binary_cmp(int op, int t, const void* a, const void* b, double sd,
                      int64_t si, int scalar_rhs, const void* av,
                      const void* bv, void* out, void* ov, int64_t n,
                      hipStream_t stream) {
  dim3 grid = stripe_grid(n);
  dispatch_type(t, [&]<typename T>() {
    T sb = IsFloat<T>::value ? (T)sd : (T)si;
    if (scalar_rhs)
      hipLaunchKernelGGL((k_binary_cmp<T, true>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const T*)b, sb,
                         (const uint64_t*)av, (const uint64_t*)bv,
                         (uint8_t*)out, (uint64_t*)ov, n_stripes(n), n);
    else
      hipLaunchKernelGGL((k_binary_cmp<T, false>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, op, (const T*)a, (const T*)b, sb,
                         (const uint64_t*)av, (const uint64_t*)bv,
                         (uint8_t*)out, (uint64_t*)ov, n_stripes(n), n);
  });
}

void hipdf_binary_bool(int op, const void* a, const void* b, int sb,
                       int scalar_rhs, const void* av, const void* bv,
                       void* out, void* ov, int64_t n, hipStream_t stream) {
  dim3 grid = stripe_grid(n);
  if (scalar_rhs)
    hipLaunchKernelGGL((k_binary_bool<true>), grid, dim3(HIPDF_BLOCK), 0,
                       stream, op, (const uint8_t*)a, (const uint8_t*)b,
                       (uint8_t)sb, (const uint64_t*)av, (const uint64_t*)bv,
                       (uint8_t*)out, (uint64_t*)ov, n_stripes(n), n);
  else
    hipLaunchKernelGGL((k_binary_bool<false>), grid, dim3(HIPDF_BLOCK), 0,
                       stream, op, (const uint8_t*)a, (const uint8_t*)b,
                       (uint8_t)sb, (const uint64_t*)av, (const uint64_t*)bv,
                       (uint8_t*)out, (uint64_t*)ov, n_stripes(n), n);
}

void hipdf_unary(int op, int t, const void* a, const void* av, void* out,
                 void* ov, int64_t n, hipStream_t stream) {
  if (op == UOP_NOT) {
    hipLaunchKernelGGL(k_not, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                       (const uint8_t*)a, (uint8_t*)out, n);
    return;
  }
  if (op == UOP_ISNAN) {
    dispatch_type(t, [&]<typename T>() {
      hipLaunchKernelGGL((k_isnan<T>), flat_grid(n), dim3(HIPDF_BLOCK), 0,
                         stream, (const T*)a, (const uint64_t*)av,
                         (uint8_t*)out, n);
    });
    return;
  }
  if (op == UOP_YEAR || op == UOP_MONTH || op == UOP_DAY) {
    hipLaunchKernelGGL(k_date_field, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                       stream, op, (const int32_t*)a, (int32_t*)out, n);
    return;
  }
  dim3 grid = stripe_grid(n);
  dispatch_type(t, [&]<typename T>() {
    hipLaunchKernelGGL((k_unary<T>), grid, dim3(HIPDF_BLOCK), 0, stream, op,
                       (const T*)a, (const uint64_t*)av, (T*)out,
                       (uint64_t*)ov, n_stripes(n), n);
  });
}

void hipdf_cast(int from_t, int to_t, const void* a, void* out, int64_t n,
                hipStream_t stream) {
  dim3 grid = flat_grid(n);
  dispatch_type(from_t, [&]<typename From>() {
    if (to_t == HT_U8) {
      hipLaunchKernelGGL((k_cast_to_bool<From>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, (const From*)a, (uint8_t*)out, n);
      return;
    }
    dispatch_type(to_t, [&]<typename To>() {
      hipLaunchKernelGGL((k_cast<From, To>), grid, dim3(HIPDF_BLOCK), 0,
                         stream, (const From*)a, (To*)out, n);
    });
  });
}

void hipdf_f64_to_i64_rint(const void* a, void* out, int64_t n,
                           hipStream_t stream) {
  hipLaunchKernelGGL(k_f64_to_i64_rint, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const double*)a, (int64_t*)out, n);
}

void hipdf_decimal_rescale(const void* a, void* out, int64_t pow10, int up,
                           int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_decimal_rescale, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)a, (int64_t*)out, pow10, up, n);
}

}  // extern "C"
