// Datetime kernels: timezone conversion against a device transition table
// (reference analogue: spark-rapids-jni GpuTimeZoneDB), and fixed-width
// date_format / timestamp parse driven by a compiled token program
// (reference analogue: the datetimeExpressions format family).
#include "hipdf_common.h"

__device__ __forceinline__ int64_t floor_div_dt(int64_t a, int64_t b) {
  int64_t q = a / b;
  return q * b > a ? q - 1 : q;
}

// Howard Hinnant's civil-from-days / days-from-civil
__device__ __forceinline__ void civil_from_days(int64_t z, int* yy, int* mm,
                                                int* dd) {
  z += 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  unsigned doe = (unsigned)(z - era * 146097);
  unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t y = (int64_t)yoe + era * 400;
  unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  unsigned mp = (5 * doy + 2) / 153;
  unsigned d = doy - (153 * mp + 2) / 5 + 1;
  unsigned m = mp < 10 ? mp + 3 : mp - 9;
  *yy = (int)(y + (m <= 2));
  *mm = (int)m;
  *dd = (int)d;
}

__device__ __forceinline__ int64_t days_from_civil(int y, int m, int d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + (int64_t)doe - 719468;
}

__device__ __forceinline__ int32_t tz_offset_at(
    const int64_t* __restrict__ trans, const int32_t* __restrict__ offs,
    int n_trans, int64_t sec) {
  // last transition <= sec (trans[0] is an INT64_MIN sentinel)
  int lo = 0, hi = n_trans - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (trans[mid] <= sec) lo = mid;
    else hi = mid - 1;
  }
  return offs[lo];
}

// to_utc=0: utc->wall (from_utc_timestamp): + offset(utc)
// to_utc=1: wall->utc (to_utc_timestamp): two-step offset resolve
__global__ void k_tz_convert(const int64_t* __restrict__ ts,
                             const int64_t* __restrict__ trans,
                             const int32_t* __restrict__ offs, int n_trans,
                             int to_utc, int64_t* __restrict__ out,
                             int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t us = ts[i];
    int64_t sec = floor_div_dt(us, 1000000);
    if (!to_utc) {
      out[i] = us + (int64_t)tz_offset_at(trans, offs, n_trans, sec)
               * 1000000;
    } else {
      int32_t o0 = tz_offset_at(trans, offs, n_trans, sec);
      int32_t o1 = tz_offset_at(trans, offs, n_trans, sec - o0);
      out[i] = us - (int64_t)o1 * 1000000;
    }
  }
}

// token kinds for format/parse programs (tokens: 2 ints each)
#define DT_LIT 0   // arg = the literal byte
#define DT_YYYY 1  // width 4
#define DT_MM 2    // width 2
#define DT_DD 3
#define DT_HH 4
#define DT_MI 5
#define DT_SS 6

__device__ __forceinline__ void write2(uint8_t* w, int v) {
  w[0] = (uint8_t)('0' + v / 10);
  w[1] = (uint8_t)('0' + v % 10);
}

// fixed-width format: every row writes `width` bytes at i*width
__global__ void k_date_format(const int64_t* __restrict__ ts_us,
                              const int32_t* __restrict__ tokens, int ntok,
                              int width, uint8_t* __restrict__ out,
                              int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t us = ts_us[i];
    int64_t sec = floor_div_dt(us, 1000000);
    int64_t days = floor_div_dt(sec, 86400);
    int64_t tod = sec - days * 86400;
    int y, m, d;
    civil_from_days(days, &y, &m, &d);
    uint8_t* w = out + i * width;
    for (int t = 0; t < ntok; ++t) {
      int kind = tokens[2 * t], arg = tokens[2 * t + 1];
      switch (kind) {
        case DT_LIT: *w++ = (uint8_t)arg; break;
        case DT_YYYY: {
          int yy = y < 0 ? 0 : y;
          w[0] = (uint8_t)('0' + (yy / 1000) % 10);
          w[1] = (uint8_t)('0' + (yy / 100) % 10);
          w[2] = (uint8_t)('0' + (yy / 10) % 10);
          w[3] = (uint8_t)('0' + yy % 10);
          w += 4;
          break;
        }
        case DT_MM: write2(w, m); w += 2; break;
        case DT_DD: write2(w, d); w += 2; break;
        case DT_HH: write2(w, (int)(tod / 3600)); w += 2; break;
        case DT_MI: write2(w, (int)(tod / 60 % 60)); w += 2; break;
        case DT_SS: write2(w, (int)(tod % 60)); w += 2; break;
      }
    }
  }
}

__device__ __forceinline__ bool rd2(const uint8_t* s, int* v) {
  if (s[0] < '0' || s[0] > '9' || s[1] < '0' || s[1] > '9') return false;
  *v = (s[0] - '0') * 10 + (s[1] - '0');
  return true;
}

// fixed-position parse of the same token programs; NULL on any mismatch
// or out-of-range field (Spark non-ANSI to_timestamp)
__global__ void k_ts_parse(const int32_t* __restrict__ ao,
                           const uint8_t* __restrict__ ab,
                           const uint64_t* __restrict__ av,
                           const int32_t* __restrict__ tokens, int ntok,
                           int width, int64_t* __restrict__ out,
                           uint64_t* __restrict__ ov, int64_t nstripe,
                           int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t st = wave_global; st < nstripe; st += wave_count) {
    int64_t row = st * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row);
      int64_t us = 0;
      if (ok) {
        const uint8_t* s = ab + ao[row];
        int nb = ao[row + 1] - ao[row];
        if (nb != width) ok = false;
        int y = 1970, mo = 1, d = 1, hh = 0, mi = 0, ss = 0;
        int p = 0;
        for (int t = 0; ok && t < ntok; ++t) {
          int kind = tokens[2 * t], arg = tokens[2 * t + 1];
          switch (kind) {
            case DT_LIT:
              if (s[p++] != (uint8_t)arg) ok = false;
              break;
            case DT_YYYY: {
              y = 0;
              for (int k = 0; k < 4; ++k) {
                uint8_t c = s[p + k];
                if (c < '0' || c > '9') { ok = false; break; }
                y = y * 10 + (c - '0');
              }
              p += 4;
              break;
            }
            case DT_MM: ok = ok && rd2(s + p, &mo); p += 2; break;
            case DT_DD: ok = ok && rd2(s + p, &d); p += 2; break;
            case DT_HH: ok = ok && rd2(s + p, &hh); p += 2; break;
            case DT_MI: ok = ok && rd2(s + p, &mi); p += 2; break;
            case DT_SS: ok = ok && rd2(s + p, &ss); p += 2; break;
          }
        }
        if (ok && (mo < 1 || mo > 12 || d < 1 || d > 31 || hh > 23 ||
                   mi > 59 || ss > 59)) ok = false;
        if (ok) {
          // day must exist in the month (round-trip check)
          int64_t days = days_from_civil(y, mo, d);
          int cy, cm, cd;
          civil_from_days(days, &cy, &cm, &cd);
          if (cy != y || cm != mo || cd != d) ok = false;
          us = (days * 86400 + hh * 3600 + mi * 60 + ss) * 1000000LL;
        }
      }
      out[row] = ok ? us : 0;
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, st, ballot, lane);
  }
}

extern "C" {

void hipdf_tz_convert(const void* ts, const void* trans, const void* offs,
                      int n_trans, int to_utc, void* out, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_tz_convert, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)ts, (const int64_t*)trans,
                     (const int32_t*)offs, n_trans, to_utc, (int64_t*)out,
                     n);
}

void hipdf_date_format(const void* ts_us, const void* tokens, int ntok,
                       int width, void* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_date_format, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)ts_us, (const int32_t*)tokens,
                     ntok, width, (uint8_t*)out, n);
}

void hipdf_ts_parse(const void* ao, const void* ab, const void* av,
                    const void* tokens, int ntok, int width, void* out,
                    void* ov, int64_t n, hipStream_t stream) {
  int64_t nstripe = (n + WAVE - 1) / WAVE;
  hipLaunchKernelGGL(k_ts_parse, stripe_grid(nstripe), dim3(HIPDF_BLOCK),
                     0, stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint64_t*)av, (const int32_t*)tokens, ntok,
                     width, (int64_t*)out, (uint64_t*)ov, nstripe, n);
}

}  // extern "C"
