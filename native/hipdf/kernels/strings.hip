// String kernels over Arrow offsets+bytes columns (reference analogue: the
// cudf strings family reached from stringFunctions.scala — SURVEY.md §2.4).
// Byte-wise UTF-8: comparisons are memcmp-order (== codepoint order),
// length/substring count codepoints, upper/lower transform ASCII bytes
// (non-ASCII passes through; gated by incompatibleOps like the reference's
// incompat ops).
#include "hipdf_common.h"

enum StrCmpOp : int { SC_EQ = 0, SC_NE, SC_LT, SC_LE, SC_GT, SC_GE };
enum StrFindOp : int { SF_CONTAINS = 0, SF_STARTS, SF_ENDS };

__device__ __forceinline__ int str_cmp(const uint8_t* a, int32_t la,
                                       const uint8_t* b, int32_t lb) {
  int32_t n = la < lb ? la : lb;
  for (int32_t i = 0; i < n; ++i) {
    if (a[i] != b[i]) return a[i] < b[i] ? -1 : 1;
  }
  return la == lb ? 0 : (la < lb ? -1 : 1);
}

__device__ __forceinline__ bool cmp_result(int op, int c) {
  switch (op) {
    case SC_EQ: return c == 0;
    case SC_NE: return c != 0;
    case SC_LT: return c < 0;
    case SC_LE: return c <= 0;
    case SC_GT: return c > 0;
    default: return c >= 0;
  }
}

// column vs column compare; null handling done by caller's validity AND
__global__ void k_str_cmp(int op, const int32_t* __restrict__ ao,
                          const uint8_t* __restrict__ ab,
                          const int32_t* __restrict__ bo,
                          const uint8_t* __restrict__ bb,
                          uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = str_cmp(ab + ao[i], ao[i + 1] - ao[i], bb + bo[i],
                    bo[i + 1] - bo[i]);
    out[i] = (uint8_t)cmp_result(op, c);
  }
}

// column vs scalar pattern
__global__ void k_str_cmp_scalar(int op, const int32_t* __restrict__ ao,
                                 const uint8_t* __restrict__ ab,
                                 const uint8_t* __restrict__ pat, int32_t plen,
                                 uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = str_cmp(ab + ao[i], ao[i + 1] - ao[i], pat, plen);
    out[i] = (uint8_t)cmp_result(op, c);
  }
}

__global__ void k_str_find(int mode, const int32_t* __restrict__ ao,
                           const uint8_t* __restrict__ ab,
                           const uint8_t* __restrict__ pat, int32_t plen,
                           uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* s = ab + ao[i];
    int32_t len = ao[i + 1] - ao[i];
    bool r = false;
    if (plen == 0) {
      r = true;
    } else if (mode == SF_STARTS) {
      r = len >= plen && str_cmp(s, plen, pat, plen) == 0;
    } else if (mode == SF_ENDS) {
      r = len >= plen && str_cmp(s + len - plen, plen, pat, plen) == 0;
    } else {
      for (int32_t p = 0; p + plen <= len; ++p) {
        if (str_cmp(s + p, plen, pat, plen) == 0) {
          r = true;
          break;
        }
      }
    }
    out[i] = (uint8_t)r;
  }
}

// SQL LIKE: % any-run, _ one char (byte-approx: one codepoint via lead
// byte); backslash escapes the next pattern char (Spark's default LIKE
// escape: \% and \_ match literally).
__device__ bool like_match(const uint8_t* s, int32_t sl, const uint8_t* p,
                           int32_t pl) {
  int32_t si = 0, pi = 0, star_p = -1, star_s = 0;
  while (si < sl) {
    if (pi < pl && p[pi] == '%') {
      star_p = ++pi;
      star_s = si;
      continue;
    }
    if (pi < pl && p[pi] == '\\' && pi + 1 < pl) {
      if (p[pi + 1] == s[si]) {
        ++si;
        pi += 2;
        continue;
      }
    } else if (pi < pl && p[pi] == '_') {
      // skip one UTF-8 codepoint
      ++si;
      while (si < sl && (s[si] & 0xC0) == 0x80) ++si;
      ++pi;
      continue;
    } else if (pi < pl && p[pi] == s[si]) {
      ++si;
      ++pi;
      continue;
    }
    if (star_p >= 0) {
      pi = star_p;
      si = ++star_s;
    } else {
      return false;
    }
  }
  while (pi < pl && p[pi] == '%') ++pi;
  return pi == pl;
}

__global__ void k_str_like(const int32_t* __restrict__ ao,
                           const uint8_t* __restrict__ ab,
                           const uint8_t* __restrict__ pat, int32_t plen,
                           uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (uint8_t)like_match(ab + ao[i], ao[i + 1] - ao[i], pat, plen);
}

// lpad/rpad to `width` codepoints, cycling the fill string (Spark
// lpad('hi',5,'xy') = 'xyxhi'); longer inputs truncate to width
// codepoints. mode 0: byte lengths; mode 1: write (GpuStringLPad/RPad).
__device__ __forceinline__ int32_t cp_count(const uint8_t* s, int32_t nb) {
  int32_t c = 0;
  for (int32_t i = 0; i < nb; ++i) c += (s[i] & 0xC0) != 0x80;
  return c;
}

// byte length of the first `cps` codepoints
__device__ __forceinline__ int32_t cp_prefix_bytes(const uint8_t* s,
                                                   int32_t nb,
                                                   int32_t cps) {
  int32_t seen = 0;
  for (int32_t i = 0; i < nb; ++i) {
    if ((s[i] & 0xC0) != 0x80) {
      if (seen == cps) return i;
      ++seen;
    }
  }
  return nb;
}

__global__ void k_str_pad(int left, const int32_t* __restrict__ ao,
                          const uint8_t* __restrict__ ab,
                          const uint8_t* __restrict__ fill, int32_t fill_nb,
                          int32_t fill_cps, int32_t width,
                          const int64_t* __restrict__ out_off,
                          int64_t* __restrict__ out_len,
                          uint8_t* __restrict__ out, int mode, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* s = ab + ao[i];
    int32_t nb = ao[i + 1] - ao[i];
    int32_t cps = cp_count(s, nb);
    int64_t len;
    int32_t pad_cps = 0;
    if (cps >= width || fill_nb == 0) {
      len = cp_prefix_bytes(s, nb, width);
    } else {
      pad_cps = width - cps;
      int32_t full = pad_cps / fill_cps;
      int32_t part = pad_cps % fill_cps;
      len = nb + (int64_t)full * fill_nb +
            cp_prefix_bytes(fill, fill_nb, part);
    }
    if (mode == 0) {
      out_len[i] = len;
      continue;
    }
    uint8_t* w = out + out_off[i];
    if (cps >= width || fill_nb == 0) {
      for (int64_t k = 0; k < len; ++k) w[k] = s[k];
      continue;
    }
    int32_t pad_bytes = (int32_t)(len - nb);
    uint8_t* dst = left ? w : w + nb;
    for (int32_t k = 0; k < pad_bytes; ++k) {
      // cycle the fill string byte-wise (whole codepoints by
      // construction of pad_bytes)
      dst[k] = fill[k % fill_nb];
    }
    uint8_t* sdst = left ? w + pad_bytes : w;
    for (int32_t k = 0; k < nb; ++k) sdst[k] = s[k];
  }
}

// locate(needle, s, pos): 1-based CODEPOINT index of the first match at or
// after codepoint `pos`; 0 when absent; empty needle -> pos (Spark).
__global__ void k_str_locate(const int32_t* __restrict__ ao,
                             const uint8_t* __restrict__ ab,
                             const uint8_t* __restrict__ needle,
                             int32_t needle_nb, int32_t pos,
                             int32_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* s = ab + ao[i];
    int32_t nb = ao[i + 1] - ao[i];
    if (needle_nb == 0) {
      out[i] = pos <= cp_count(s, nb) + 1 ? pos : 0;
      continue;
    }
    int32_t cp = 0;
    int32_t found = 0;
    for (int32_t b = 0; b + needle_nb <= nb && !found; ++b) {
      if ((s[b] & 0xC0) == 0x80) continue;
      ++cp;  // s[b] starts codepoint number `cp` (1-based)
      if (cp < pos) continue;
      bool eq = true;
      for (int32_t k = 0; k < needle_nb; ++k)
        if (s[b + k] != needle[k]) { eq = false; break; }
      if (eq) found = cp;
    }
    out[i] = found;
  }
}

// codepoint length
__global__ void k_str_length(const int32_t* __restrict__ ao,
                             const uint8_t* __restrict__ ab,
                             int32_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t cnt = 0;
    for (int32_t p = ao[i]; p < ao[i + 1]; ++p)
      cnt += (ab[p] & 0xC0) != 0x80;
    out[i] = cnt;
  }
}

// ASCII upper/lower (offsets unchanged)
__global__ void k_str_case(int upper, const uint8_t* __restrict__ in,
                           uint8_t* __restrict__ out, int64_t nbytes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nbytes;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t c = in[i];
    if (upper && c >= 'a' && c <= 'z') c -= 32;
    if (!upper && c >= 'A' && c <= 'Z') c += 32;
    out[i] = c;
  }
}

// substring (1-based start in codepoints, length in codepoints; Spark
// semantics: start 0 behaves like 1, negative counts from the end).
// pass 1: byte [start,len) per row
// int64 -> decimal string (cast long/int to string). mode 0: lengths,
// mode 1: write digits.
__global__ void k_i64_to_str(const int64_t* __restrict__ vals,
                             const int64_t* __restrict__ out_off,
                             int64_t* __restrict__ out_len,
                             uint8_t* __restrict__ out, int mode,
                             int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t v = vals[i];
    uint64_t u = v < 0 ? (uint64_t)(-(v + 1)) + 1 : (uint64_t)v;
    int digits = 1;
    for (uint64_t t = u; t >= 10; t /= 10) ++digits;
    int len = digits + (v < 0 ? 1 : 0);
    if (!mode) {
      out_len[i] = len;
      continue;
    }
    uint8_t* dst = out + out_off[i];
    if (v < 0) dst[0] = '-';
    for (int d = digits - 1; d >= 0; --d) {
      dst[(v < 0 ? 1 : 0) + d] = '0' + (uint8_t)(u % 10);
      u /= 10;
    }
  }
}

// concat_ws: join n string columns with a separator, skipping NULL
// values (result is never null — all-null rows give ""). Two-pass.
struct StrColDesc {
  const int32_t* off;
  const uint8_t* bytes;
  const uint64_t* valid;
};

__global__ void k_str_concat_ws(const StrColDesc* __restrict__ cols,
                                int ncols, const uint8_t* __restrict__ sep,
                                int seplen,
                                const int64_t* __restrict__ out_off,
                                int64_t* __restrict__ out_len,
                                uint8_t* __restrict__ out, int mode,
                                int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t w = mode ? out_off[i] : 0;
    int64_t len = 0;
    bool first = true;
    for (int c = 0; c < ncols; ++c) {
      const StrColDesc& d = cols[c];
      if (d.valid && !((d.valid[i >> 6] >> (i & 63)) & 1ull)) continue;
      if (!first) {
        if (mode)
          for (int k = 0; k < seplen; ++k) out[w + len + k] = sep[k];
        len += seplen;
      }
      int32_t a = d.off[i], b = d.off[i + 1];
      if (mode)
        for (int32_t k = a; k < b; ++k) out[w + len + (k - a)] = d.bytes[k];
      len += b - a;
      first = false;
    }
    if (!mode) out_len[i] = len;
  }
}

// initcap (ASCII): uppercase the first letter of each whitespace-split
// word, lowercase the rest. reverse: reverse UTF-8 codepoint order in
// place (same byte length).
__global__ void k_str_initcap(const int32_t* __restrict__ ao,
                              const uint8_t* __restrict__ ab,
                              uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = ao[i], s1 = ao[i + 1];
    bool word_start = true;
    for (int32_t p = s0; p < s1; ++p) {
      uint8_t c = ab[p];
      if (c == ' ') {
        out[p] = c;
        word_start = true;
      } else {
        if (word_start && c >= 'a' && c <= 'z') c -= 32;
        else if (!word_start && c >= 'A' && c <= 'Z') c += 32;
        out[p] = c;
        word_start = false;
      }
    }
  }
}

__global__ void k_str_reverse(const int32_t* __restrict__ ao,
                              const uint8_t* __restrict__ ab,
                              uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = ao[i], s1 = ao[i + 1];
    int32_t w = s1;
    int32_t p = s0;
    while (p < s1) {
      int32_t q = p + 1;  // span one UTF-8 codepoint
      while (q < s1 && (ab[q] & 0xC0) == 0x80) ++q;
      w -= q - p;
      for (int32_t k = 0; k < q - p; ++k) out[w + k] = ab[p + k];
      p = q;
    }
  }
}

// split by a literal delimiter -> per-part (start, len) spans.
// Java limit-0 semantics: trailing empty parts are dropped ("a,,".split
// -> ["a"], ",,".split -> [], "".split -> [""]).
__global__ void k_str_split_count(const int32_t* __restrict__ ao,
                                  const uint8_t* __restrict__ ab,
                                  const uint8_t* __restrict__ delim,
                                  int dlen, int64_t* __restrict__ counts,
                                  int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = ao[i], s1 = ao[i + 1];
    if (s1 == s0) {
      counts[i] = 1;  // "" -> [""]
      continue;
    }
    int part = 0;
    int last_nonempty = -1;
    int32_t start = s0;
    for (int32_t p = s0; p + dlen <= s1;) {
      bool m = true;
      for (int k = 0; k < dlen; ++k)
        if (ab[p + k] != delim[k]) { m = false; break; }
      if (m) {
        if (p > start) last_nonempty = part;
        ++part;
        p += dlen;
        start = p;
      } else {
        ++p;
      }
    }
    if (s1 > start) last_nonempty = part;
    counts[i] = last_nonempty + 1;
  }
}

__global__ void k_str_split_fill(const int32_t* __restrict__ ao,
                                 const uint8_t* __restrict__ ab,
                                 const uint8_t* __restrict__ delim, int dlen,
                                 const int64_t* __restrict__ part_off,
                                 const int64_t* __restrict__ counts,
                                 int32_t* __restrict__ out_ss,
                                 int64_t* __restrict__ out_sl, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = ao[i], s1 = ao[i + 1];
    int64_t want = counts[i];
    int64_t w = part_off[i];
    int64_t part = 0;
    int32_t start = s0;
    for (int32_t p = s0; part < want && p + dlen <= s1;) {
      bool m = true;
      for (int k = 0; k < dlen; ++k)
        if (ab[p + k] != delim[k]) { m = false; break; }
      if (m) {
        out_ss[w] = start;
        out_sl[w] = p - start;
        ++w;
        ++part;
        p += dlen;
        start = p;
      } else {
        ++p;
      }
    }
    if (part < want) {
      out_ss[w] = start;
      out_sl[w] = s1 - start;
    }
  }
}

// trim spans: mode 0 both, 1 leading, 2 trailing (ascii space like Spark
// trim's default)
__global__ void k_str_trim_ranges(int mode, const int32_t* __restrict__ ao,
                                  const uint8_t* __restrict__ ab,
                                  int32_t* __restrict__ bstart,
                                  int64_t* __restrict__ blen, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t a = ao[i], b = ao[i + 1];
    // modes 0..2: spark trim()/ltrim()/rtrim() strip only 0x20; mode 3:
    // full ASCII whitespace both sides (string->numeric cast semantics)
    if (mode == 3) {
      while (a < b && (ab[a] == ' ' || ab[a] == '\t' || ab[a] == '\r' ||
                       ab[a] == '\n' || ab[a] == '\f' || ab[a] == '\v'))
        ++a;
      while (b > a && (ab[b - 1] == ' ' || ab[b - 1] == '\t' ||
                       ab[b - 1] == '\r' || ab[b - 1] == '\n' ||
                       ab[b - 1] == '\f' || ab[b - 1] == '\v'))
        --b;
    } else {
      if (mode != 2)
        while (a < b && ab[a] == ' ') ++a;
      if (mode != 1)
        while (b > a && ab[b - 1] == ' ') --b;
    }
    bstart[i] = a;
    blen[i] = b - a;
  }
}

// concat two string columns: pass 0 lengths, pass 1 bytes
__global__ void k_str_concat2(const int32_t* __restrict__ ao,
                              const uint8_t* __restrict__ ab,
                              const int32_t* __restrict__ bo,
                              const uint8_t* __restrict__ bb,
                              const int64_t* __restrict__ out_off,
                              int64_t* __restrict__ out_len,
                              uint8_t* __restrict__ out_bytes, int mode,
                              int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t i = wave_global; i < n; i += wave_count) {
    int32_t la = ao[i + 1] - ao[i], lb = bo[i + 1] - bo[i];
    if (!mode) {
      if (lane == 0) out_len[i] = (int64_t)la + lb;
      continue;
    }
    uint8_t* dst = out_bytes + out_off[i];
    for (int32_t k = lane; k < la; k += WAVE) dst[k] = ab[ao[i] + k];
    for (int32_t k = lane; k < lb; k += WAVE) dst[la + k] = bb[bo[i] + k];
  }
}

__global__ void k_substr_ranges(const int32_t* __restrict__ ao,
                                const uint8_t* __restrict__ ab, int32_t start,
                                int32_t slen, int32_t* __restrict__ bstart,
                                int64_t* __restrict__ blen, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t s0 = ao[i], s1 = ao[i + 1];
    // count codepoints
    int32_t cps = 0;
    for (int32_t p = s0; p < s1; ++p) cps += (ab[p] & 0xC0) != 0x80;
    int32_t begin = start > 0 ? start - 1 : (start < 0 ? cps + start : 0);
    if (begin < 0) begin = 0;
    int32_t end = slen < 0 ? cps : begin + slen;
    if (end > cps) end = cps;
    if (begin >= cps || end <= begin) {
      bstart[i] = s0;
      blen[i] = 0;
      continue;
    }
    // walk to byte positions
    int32_t cp = 0, bs = s1, be = s1;
    for (int32_t p = s0; p < s1; ++p) {
      if ((ab[p] & 0xC0) == 0x80) continue;
      if (cp == begin) bs = p;
      if (cp == end) {
        be = p;
        break;
      }
      ++cp;
    }
    if (cp < end) be = s1;
    bstart[i] = bs;
    blen[i] = be - bs;
  }
}

// pass 2: copy using scanned output offsets
__global__ void k_substr_copy(const uint8_t* __restrict__ ab,
                              const int32_t* __restrict__ bstart,
                              const int64_t* __restrict__ blen,
                              const int64_t* __restrict__ out_off,
                              uint8_t* __restrict__ out_bytes, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t i = wave_global; i < n; i += wave_count) {
    int64_t len = blen[i];
    int64_t dst = out_off[i];
    int32_t src = bstart[i];
    for (int64_t b = lane; b < len; b += WAVE)
      out_bytes[dst + b] = ab[src + b];
  }
}

extern "C" {

void hipdf_str_cmp(int op, const void* ao, const void* ab, const void* bo,
                   const void* bb, void* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_cmp, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     op, (const int32_t*)ao, (const uint8_t*)ab,
                     (const int32_t*)bo, (const uint8_t*)bb, (uint8_t*)out, n);
}

void hipdf_str_cmp_scalar(int op, const void* ao, const void* ab,
                          const void* pat, int plen, void* out, int64_t n,
                          hipStream_t stream) {
  hipLaunchKernelGGL(k_str_cmp_scalar, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, op, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)pat, (int32_t)plen, (uint8_t*)out, n);
}

void hipdf_str_find(int mode, const void* ao, const void* ab, const void* pat,
                    int plen, void* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_find, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     mode, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)pat, (int32_t)plen, (uint8_t*)out, n);
}

void hipdf_str_like(const void* ao, const void* ab, const void* pat, int plen,
                    void* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_like, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)pat, (int32_t)plen, (uint8_t*)out, n);
}

void hipdf_str_pad(int left, const void* ao, const void* ab,
                   const void* fill, int32_t fill_nb, int32_t fill_cps,
                   int32_t width, const void* out_off, void* out_len,
                   void* out, int mode, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_pad, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     left, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)fill, fill_nb, fill_cps, width,
                     (const int64_t*)out_off, (int64_t*)out_len,
                     (uint8_t*)out, mode, n);
}

void hipdf_str_locate(const void* ao, const void* ab, const void* needle,
                      int32_t needle_nb, int32_t pos, void* out, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_str_locate, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)needle, needle_nb, pos, (int32_t*)out,
                     n);
}

void hipdf_str_length(const void* ao, const void* ab, void* out, int64_t n,
                      hipStream_t stream) {
  hipLaunchKernelGGL(k_str_length, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const int32_t*)ao, (const uint8_t*)ab, (int32_t*)out, n);
}

void hipdf_str_case(int upper, const void* in, void* out, int64_t nbytes,
                    hipStream_t stream) {
  hipLaunchKernelGGL(k_str_case, flat_grid(nbytes), dim3(HIPDF_BLOCK), 0,
                     stream, upper, (const uint8_t*)in, (uint8_t*)out, nbytes);
}

void hipdf_i64_to_str(const void* vals, const void* out_off, void* out_len,
                      void* out, int mode, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_i64_to_str, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int64_t*)vals, (const int64_t*)out_off,
                     (int64_t*)out_len, (uint8_t*)out, mode, n);
}

void hipdf_str_concat_ws(const void* cols, int ncols, const void* sep,
                         int seplen, const void* out_off, void* out_len,
                         void* out, int mode, int64_t n,
                         hipStream_t stream) {
  hipLaunchKernelGGL(k_str_concat_ws, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const StrColDesc*)cols, ncols,
                     (const uint8_t*)sep, seplen, (const int64_t*)out_off,
                     (int64_t*)out_len, (uint8_t*)out, mode, n);
}

void hipdf_str_initcap(const void* ao, const void* ab, void* out,
                       int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_initcap, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (uint8_t*)out, n);
}

void hipdf_str_reverse(const void* ao, const void* ab, void* out, int64_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_str_reverse, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (uint8_t*)out, n);
}

void hipdf_str_split_count(const void* ao, const void* ab,
                           const void* delim, int dlen, void* counts,
                           int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_split_count, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)delim, dlen, (int64_t*)counts, n);
}

void hipdf_str_split_fill(const void* ao, const void* ab, const void* delim,
                          int dlen, const void* part_off, const void* counts,
                          void* out_ss, void* out_sl, int64_t n,
                          hipStream_t stream) {
  hipLaunchKernelGGL(k_str_split_fill, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (const uint8_t*)delim, dlen, (const int64_t*)part_off,
                     (const int64_t*)counts, (int32_t*)out_ss,
                     (int64_t*)out_sl, n);
}

void hipdf_str_trim_ranges(int mode, const void* ao, const void* ab,
                           void* bstart, void* blen, int64_t n,
                           hipStream_t stream) {
  hipLaunchKernelGGL(k_str_trim_ranges, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, mode, (const int32_t*)ao, (const uint8_t*)ab,
                     (int32_t*)bstart, (int64_t*)blen, n);
}

void hipdf_str_concat2(const void* ao, const void* ab, const void* bo,
                       const void* bb, const void* out_off, void* out_len,
                       void* out_bytes, int mode, int64_t n,
                       hipStream_t stream) {
  int64_t waves_needed = n;
  hipLaunchKernelGGL(k_str_concat2, flat_grid(waves_needed),
                     dim3(HIPDF_BLOCK), 0, stream, (const int32_t*)ao,
                     (const uint8_t*)ab, (const int32_t*)bo,
                     (const uint8_t*)bb, (const int64_t*)out_off,
                     (int64_t*)out_len, (uint8_t*)out_bytes, mode, n);
}

void hipdf_substr_ranges(const void* ao, const void* ab, int start, int slen,
                         void* bstart, void* blen, int64_t n,
                         hipStream_t stream) {
  hipLaunchKernelGGL(k_substr_ranges, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)ao, (const uint8_t*)ab,
                     (int32_t)start, (int32_t)slen, (int32_t*)bstart,
                     (int64_t*)blen, n);
}

void hipdf_substr_copy(const void* ab, const void* bstart, const void* blen,
                       const void* out_off, void* out_bytes, int64_t n,
                       hipStream_t stream) {
  int64_t blocks = (n * WAVE + HIPDF_BLOCK - 1) / HIPDF_BLOCK;
  if (blocks > 4 * HIPDF_MAX_BLOCKS) blocks = 4 * HIPDF_MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_substr_copy, dim3((uint32_t)blocks), dim3(HIPDF_BLOCK),
                     0, stream, (const uint8_t*)ab, (const int32_t*)bstart,
                     (const int64_t*)blen, (const int64_t*)out_off,
                     (uint8_t*)out_bytes, n);
}

}  // extern "C"
