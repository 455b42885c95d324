// Exact string <-> integral/decimal casts with Spark semantics
// (reference analogue: spark-rapids-jni CastStrings — SURVEY.md §2.8B).
//
// string -> decimal/int: trim ASCII whitespace, optional sign, digits with
// one optional '.', optional e/E exponent. The unscaled value accumulates
// in unsigned __int128 (exact for every representable decimal(38));
// rescaling to the target scale rounds HALF_UP for decimal targets and
// truncates toward zero for integral targets (Spark: cast('12.9' as int)
// = 12). Unparsable / overflow -> NULL (non-ANSI).
//
// decimal -> string: fixed-scale formatting with trailing zeros
// ("1.50" for decimal(_,2)), two-pass lengths/write like k_i64_to_str.
#include "hipdf_common.h"

typedef unsigned __int128 u128;

__device__ __forceinline__ u128 cs_pow10(int p) {
  u128 r = 1;
  for (int i = 0; i < p; ++i) r *= 10;
  return r;
}

struct ParsedDec {
  u128 mag;       // significant digits (up to 38)
  int scale;      // digits after the point minus exponent
  bool neg;
  bool ok;
  int round_digit;  // first dropped digit (for HALF_UP), -1 if none
  bool sticky;      // nonzero beyond the dropped digit
  bool had_exp;     // scientific notation seen (Spark rejects for ints)
};

__device__ ParsedDec parse_decimal_str(const uint8_t* s, int32_t len) {
  ParsedDec r{0, 0, false, false, -1, false, false};
  int32_t i = 0, j = len;
  while (i < j && (s[i] == ' ' || s[i] == '\t' || s[i] == '\r' ||
                   s[i] == '\n')) ++i;
  while (j > i && (s[j - 1] == ' ' || s[j - 1] == '\t' ||
                   s[j - 1] == '\r' || s[j - 1] == '\n')) --j;
  if (i >= j) return r;
  if (s[i] == '+' || s[i] == '-') {
    r.neg = s[i] == '-';
    ++i;
  }
  int ndig = 0, frac = 0;
  bool saw_digit = false, saw_dot = false, dropping = false;
  for (; i < j; ++i) {
    uint8_t c = s[i];
    if (c >= '0' && c <= '9') {
      saw_digit = true;
      if (dropping || ndig >= 38) {
        // beyond 38 significant digits: integral overflow unless the
        // extra digits are fractional (they only affect rounding)
        if (!saw_dot) return r;  // > 10^38 integral part: overflow
        if (r.round_digit < 0) r.round_digit = c - '0';
        else if (c != '0') r.sticky = true;
        dropping = true;
        continue;  // scale stays at the retained precision
      }
      u128 nxt = r.mag * 10 + (c - '0');
      r.mag = nxt;
      if (r.mag != 0) ++ndig;
      if (saw_dot) ++frac;
    } else if (c == '.' && !saw_dot) {
      saw_dot = true;
    } else if ((c == 'e' || c == 'E') && saw_digit) {
      r.had_exp = true;
      ++i;
      bool eneg = false;
      if (i < j && (s[i] == '+' || s[i] == '-')) {
        eneg = s[i] == '-';
        ++i;
      }
      if (i >= j) return r;
      int ev = 0;
      for (; i < j; ++i) {
        if (s[i] < '0' || s[i] > '9') return r;
        ev = ev * 10 + (s[i] - '0');
        if (ev > 1000) return r;
      }
      frac += eneg ? ev : -ev;
      break;
    } else {
      return r;
    }
  }
  if (!saw_digit) return r;
  r.scale = frac;
  r.ok = true;
  return r;
}

// rescale parsed value to target scale; mode 0 = HALF_UP (decimal),
// mode 1 = truncate toward zero (integral). Returns ok=false on overflow
// of 10^out_prec.
__device__ bool rescale_to(ParsedDec p, int out_scale, int out_prec,
                           int trunc_mode, u128* out) {
  int shift = out_scale - p.scale;
  u128 v = p.mag;
  if (shift >= 0) {
    if (shift > 38) return false;
    // digits dropped at parse (>38 significant) round at the retained
    // scale before shifting
    if (!trunc_mode && p.round_digit >= 5) v += 1;
    u128 f = cs_pow10(shift);
    if (v != 0 && v > ~(u128)0 / f) return false;
    v *= f;
  } else {
    int d = -shift;
    if (d > 38) {
      v = 0;
      if (!trunc_mode && p.mag != 0) {
        // entire value below the scale: rounds to 0 (or 1 ulp)
        if (d == 39 && p.mag >= 5 * cs_pow10(38)) v = 1;
      }
    } else {
      u128 f = cs_pow10(d);
      u128 rem = v % f;
      v /= f;
      if (!trunc_mode && rem * 2 >= f) v += 1;
    }
  }
  if (out_prec <= 38 && v >= cs_pow10(out_prec)) return false;
  *out = v;
  return true;
}

// out_kind: 0 dec64 (int64), 1 dec128 (2xint64), 2..5 int8/16/32/64
__global__ void k_str_to_dec(const int32_t* __restrict__ ao,
                             const uint8_t* __restrict__ ab,
                             const uint64_t* __restrict__ av,
                             int out_kind, int out_scale, int out_prec,
                             int64_t* __restrict__ out,
                             uint64_t* __restrict__ ov, int64_t nstripe,
                             int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s0 = wave_global; s0 < nstripe; s0 += wave_count) {
    int64_t row = s0 * WAVE + lane;
    bool ok = false;
    if (row < n) {
      ok = valid_bit(av, row);
      u128 v = 0;
      bool neg = false;
      if (ok) {
        ParsedDec p = parse_decimal_str(ab + ao[row],
                                        ao[row + 1] - ao[row]);
        ok = p.ok;
        neg = p.neg;
        if (ok) {
          int trunc_mode = out_kind >= 2 ? 1 : 0;
          if (trunc_mode && p.had_exp)
            ok = false;  // Spark: '1e2' is not a valid int literal
          else
            ok = rescale_to(p, out_scale, out_prec, trunc_mode, &v);
        }
        if (ok && out_kind >= 2) {
          // integral bounds (int8/16/32/64)
          u128 lim = out_kind == 2 ? 128u
                     : out_kind == 3 ? 32768u
                     : out_kind == 4 ? 2147483648u
                     : (u128)1 << 63;
          if (neg ? v > lim : v >= lim) ok = false;
        }
        if (ok && out_kind == 0 && v > (u128)0x7fffffffffffffffULL)
          ok = false;
      }
      __int128 w = ok ? (neg ? -(__int128)v : (__int128)v) : 0;
      if (out_kind == 1) {
        out[2 * row] = (int64_t)(u128)w;
        out[2 * row + 1] = (int64_t)((u128)w >> 64);
      } else {
        out[row] = (int64_t)w;
      }
    }
    uint64_t ballot = __ballot(ok);
    write_valid_word(ov, s0, ballot, lane);
  }
}

// decimal -> string. in_is_128: (lo,hi) pairs vs int64. mode 0: write
// byte lengths to out_len; mode 1: write characters at out_off.
__global__ void k_dec_to_str(const int64_t* __restrict__ vals, int in_is_128,
                             int scale, const int64_t* __restrict__ out_off,
                             int64_t* __restrict__ out_len,
                             uint8_t* __restrict__ out, int mode,
                             int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool neg;
    u128 v;
    if (in_is_128) {
      int64_t hi = vals[2 * i + 1];
      uint64_t lo = (uint64_t)vals[2 * i];
      neg = hi < 0;
      if (neg) {
        lo = ~lo + 1;
        hi = ~hi + (lo == 0 ? 1 : 0);
      }
      v = ((u128)(uint64_t)hi << 64) | lo;
    } else {
      int64_t x = vals[i];
      neg = x < 0;
      v = (u128)(neg ? -(__int128)x : (__int128)x);
    }
    // extract digits (max 39)
    uint8_t dig[40];
    int nd = 0;
    u128 t = v;
    do {
      dig[nd++] = (uint8_t)('0' + (int)(t % 10));
      t /= 10;
    } while (t != 0);
    int int_digits = nd > scale ? nd - scale : 1;  // "0.xx" keeps one 0
    int64_t len = (neg ? 1 : 0) + int_digits + (scale ? 1 + scale : 0);
    if (mode == 0) {
      out_len[i] = len;
      continue;
    }
    uint8_t* w = out + out_off[i];
    if (neg) *w++ = '-';
    for (int k = int_digits - 1; k >= 0; --k) {
      int src = scale + k;
      *w++ = src < nd ? dig[src] : '0';
    }
    if (scale) {
      *w++ = '.';
      for (int k = scale - 1; k >= 0; --k)
        *w++ = k < nd ? dig[k] : '0';
    }
  }
}

extern "C" {

void hipdf_str_to_dec(const void* ao, const void* ab, const void* av,
                      int out_kind, int out_scale, int out_prec, void* out,
                      void* ov, int64_t n, hipStream_t stream) {
  int64_t nstripe = (n + WAVE - 1) / WAVE;
  hipLaunchKernelGGL(k_str_to_dec, stripe_grid(nstripe), dim3(HIPDF_BLOCK),
                     0, stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint64_t*)av, out_kind, out_scale, out_prec,
                     (int64_t*)out, (uint64_t*)ov, nstripe, n);
}

void hipdf_dec_to_str(const void* vals, int in_is_128, int scale,
                      const void* out_off, void* out_len, void* out,
                      int mode, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_dec_to_str, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)vals, in_is_128, scale,
                     (const int64_t*)out_off, (int64_t*)out_len,
                     (uint8_t*)out, mode, n);
}

}  // extern "C"
