// GPU CSV decode (reference analogue: GpuCsvScan over cudf's CSV reader —
// SURVEY.md §2.3). One thread per row walks to its field, parses in
// place: int64 / float64 / date days / raw string spans. Quoted fields
// bump `unsupported` and the host falls back to the CPU reader for the
// file (exactness over a half-implemented dialect).
#include "hipdf_common.h"

__global__ void k_byte_eq(const uint8_t* __restrict__ bytes, uint8_t target,
                          uint8_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = bytes[i] == target;
}

// locate the field span [fs, fe) of column `field_idx` inside a row
__device__ __forceinline__ bool csv_field_span(
    const uint8_t* b, int32_t rs, int32_t re, uint8_t delim, int field_idx,
    int32_t* fs, int32_t* fe, int* unsupported) {
  int32_t p = rs;
  for (int k = 0; k < field_idx; ++k) {
    while (p < re && b[p] != delim) ++p;
    if (p >= re) return false;  // missing field -> null
    ++p;
  }
  int32_t q = p;
  while (q < re && b[q] != delim) ++q;
  if (q > p && b[q - 1] == '\r') --q;  // CRLF on the last field
  if (p < q && b[p] == '"') {
    atomicAdd(unsupported, 1);
    return false;
  }
  *fs = p;
  *fe = q;
  return true;
}

// type: 0 = int64, 1 = float64, 2 = string spans
__global__ void k_csv_parse(const uint8_t* __restrict__ bytes,
                            const int32_t* __restrict__ row_start,
                            const int32_t* __restrict__ row_end,
                            uint8_t delim, int field_idx, int type,
                            int64_t* __restrict__ out_i64,
                            double* __restrict__ out_f64,
                            int32_t* __restrict__ out_ss,
                            int64_t* __restrict__ out_sl,
                            uint8_t* __restrict__ valid,
                            int* __restrict__ unsupported, int64_t nrows) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nrows; i += (int64_t)gridDim.x * blockDim.x) {
    int32_t fs, fe;
    bool ok = csv_field_span(bytes, row_start[i], row_end[i], delim,
                             field_idx, &fs, &fe, unsupported) && fe > fs;
    if (type == 2) {
      out_ss[i] = ok ? fs : 0;
      out_sl[i] = ok ? fe - fs : 0;
      valid[i] = ok;
      continue;
    }
    if (!ok) {
      valid[i] = 0;
      if (type == 0) out_i64[i] = 0;
      else out_f64[i] = 0.0;
      continue;
    }
    int32_t p = fs;
    bool neg = false;
    if (bytes[p] == '-' || bytes[p] == '+') {
      neg = bytes[p] == '-';
      ++p;
    }
    if (type == 0) {
      int64_t v = 0;
      bool any = false, bad = false;
      for (; p < fe; ++p) {
        uint8_t c = bytes[p];
        if (c < '0' || c > '9') {
          bad = true;
          break;
        }
        v = v * 10 + (c - '0');
        any = true;
      }
      valid[i] = any && !bad;
      out_i64[i] = neg ? -v : v;
      continue;
    }
    // float64: mantissa as integer digits (exact to 18 digits), then
    // decimal exponent applied once
    double mant = 0.0;
    int exp10 = 0;
    bool any = false, bad = false, seen_dot = false;
    for (; p < fe; ++p) {
      uint8_t c = bytes[p];
      if (c >= '0' && c <= '9') {
        mant = mant * 10.0 + (c - '0');
        if (seen_dot) --exp10;
        any = true;
      } else if (c == '.' && !seen_dot) {
        seen_dot = true;
      } else if ((c == 'e' || c == 'E') && any) {
        ++p;
        bool eneg = false;
        if (p < fe && (bytes[p] == '-' || bytes[p] == '+')) {
          eneg = bytes[p] == '-';
          ++p;
        }
        int ev = 0;
        bool eany = false;
        for (; p < fe; ++p) {
          uint8_t ec = bytes[p];
          if (ec < '0' || ec > '9') {
            bad = true;
            break;
          }
          ev = ev * 10 + (ec - '0');
          eany = true;
        }
        if (!eany) bad = true;
        exp10 += eneg ? -ev : ev;
        break;
      } else {
        bad = true;
        break;
      }
    }
    if (!bad && any) {
      double v = mant;
      if (exp10 > 0)
        for (int k = 0; k < exp10; ++k) v *= 10.0;
      else
        for (int k = 0; k < -exp10; ++k) v /= 10.0;
      out_f64[i] = neg ? -v : v;
      valid[i] = 1;
    } else {
      out_f64[i] = 0.0;
      valid[i] = 0;
    }
  }
}

// JSON-lines field extraction: find `"name":` at object level (previous
// non-space char is '{' or ',') inside the row, parse the value.
// type: 0 int64, 1 float64, 2 string span, 3 bool
__global__ void k_json_field(const uint8_t* __restrict__ bytes,
                             const int32_t* __restrict__ row_start,
                             const int32_t* __restrict__ row_end,
                             const uint8_t* __restrict__ name, int name_len,
                             int type, int64_t* __restrict__ out_i64,
                             double* __restrict__ out_f64,
                             int32_t* __restrict__ out_ss,
                             int64_t* __restrict__ out_sl,
                             uint8_t* __restrict__ valid,
                             int* __restrict__ unsupported, int64_t nrows) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nrows; i += (int64_t)gridDim.x * blockDim.x) {
    int32_t rs = row_start[i], re = row_end[i];
    if (re > rs && bytes[re - 1] == '\r') --re;
    int32_t vpos = -1;
    for (int32_t p = rs; p + name_len + 2 < re; ++p) {
      if (bytes[p] != '"') continue;
      bool m = true;
      for (int k = 0; k < name_len; ++k)
        if (bytes[p + 1 + k] != name[k]) { m = false; break; }
      if (!m || bytes[p + 1 + name_len] != '"') continue;
      // previous non-space must open an object member
      int32_t q = p - 1;
      while (q >= rs && (bytes[q] == ' ' || bytes[q] == '\t')) --q;
      if (q < rs || (bytes[q] != '{' && bytes[q] != ',')) continue;
      q = p + 2 + name_len;
      while (q < re && (bytes[q] == ' ' || bytes[q] == '\t')) ++q;
      if (q < re && bytes[q] == ':') {
        ++q;
        while (q < re && (bytes[q] == ' ' || bytes[q] == '\t')) ++q;
        vpos = q;
        break;
      }
    }
    bool ok = false;
    if (type == 2 || type == 4) {
      out_ss[i] = 0;
      out_sl[i] = 0;
    } else if (type == 0) {
      out_i64[i] = 0;
    } else if (type == 1) {
      out_f64[i] = 0.0;
    } else {
      out_i64[i] = 0;
    }
    if (vpos >= 0 && vpos < re && !(bytes[vpos] == 'n')) {  // null -> null
      uint8_t c = bytes[vpos];
      if (type == 4) {  // any scalar as raw text (get_json_object)
        if (c == '"') {
          int32_t e = vpos + 1;
          bool esc = false;
          while (e < re && bytes[e] != '"') {
            if (bytes[e] == '\\') { esc = true; break; }
            ++e;
          }
          if (esc) atomicAdd(unsupported, 1);
          else if (e < re) {
            out_ss[i] = vpos + 1;
            out_sl[i] = e - vpos - 1;
            ok = true;
          }
        } else if (c == '{' || c == '[') {
          atomicAdd(unsupported, 1);  // nested -> CPU fallback
        } else {
          int32_t e = vpos;
          while (e < re && bytes[e] != ',' && bytes[e] != '}' &&
                 bytes[e] != ' ' && bytes[e] != '\t')
            ++e;
          out_ss[i] = vpos;
          out_sl[i] = e - vpos;
          ok = e > vpos;
        }
      } else if (type == 2) {
        if (c == '"') {
          int32_t e = vpos + 1;
          bool esc = false;
          while (e < re && bytes[e] != '"') {
            if (bytes[e] == '\\') { esc = true; break; }
            ++e;
          }
          if (esc) {
            atomicAdd(unsupported, 1);
          } else if (e < re) {
            out_ss[i] = vpos + 1;
            out_sl[i] = e - vpos - 1;
            ok = true;
          }
        }
      } else if (type == 3) {
        if (c == 't') { out_i64[i] = 1; ok = true; }
        else if (c == 'f') { out_i64[i] = 0; ok = true; }
      } else {
        // numeric: find the value end (",", "}", space)
        int32_t e = vpos;
        while (e < re && bytes[e] != ',' && bytes[e] != '}' &&
               bytes[e] != ' ')
          ++e;
        int32_t p = vpos;
        bool neg = false;
        if (p < e && (bytes[p] == '-' || bytes[p] == '+')) {
          neg = bytes[p] == '-';
          ++p;
        }
        if (type == 0) {
          int64_t v = 0;
          bool any = false, bad = false;
          for (; p < e; ++p) {
            uint8_t d = bytes[p];
            if (d < '0' || d > '9') { bad = true; break; }
            v = v * 10 + (d - '0');
            any = true;
          }
          if (any && !bad) { out_i64[i] = neg ? -v : v; ok = true; }
        } else {
          double mant = 0.0;
          int exp10 = 0;
          bool any = false, bad = false, dot = false;
          for (; p < e; ++p) {
            uint8_t d = bytes[p];
            if (d >= '0' && d <= '9') {
              mant = mant * 10.0 + (d - '0');
              if (dot) --exp10;
              any = true;
            } else if (d == '.' && !dot) {
              dot = true;
            } else if ((d == 'e' || d == 'E') && any) {
              ++p;
              bool en = false;
              if (p < e && (bytes[p] == '-' || bytes[p] == '+')) {
                en = bytes[p] == '-';
                ++p;
              }
              int ev = 0;
              for (; p < e; ++p) {
                if (bytes[p] < '0' || bytes[p] > '9') { bad = true; break; }
                ev = ev * 10 + (bytes[p] - '0');
              }
              exp10 += en ? -ev : ev;
              break;
            } else {
              bad = true;
              break;
            }
          }
          if (any && !bad) {
            double v = mant;
            if (exp10 > 0)
              for (int k = 0; k < exp10; ++k) v *= 10.0;
            else
              for (int k = 0; k < -exp10; ++k) v /= 10.0;
            out_f64[i] = neg ? -v : v;
            ok = true;
          }
        }
      }
    }
    valid[i] = ok;
  }
}

extern "C" {

void hipdf_json_field(const void* bytes, const void* row_start,
                      const void* row_end, const void* name, int name_len,
                      int type, void* out_i64, void* out_f64, void* out_ss,
                      void* out_sl, void* valid, void* unsupported,
                      int64_t nrows, hipStream_t stream) {
  hipLaunchKernelGGL(k_json_field, flat_grid(nrows), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)bytes,
                     (const int32_t*)row_start, (const int32_t*)row_end,
                     (const uint8_t*)name, name_len, type,
                     (int64_t*)out_i64, (double*)out_f64, (int32_t*)out_ss,
                     (int64_t*)out_sl, (uint8_t*)valid, (int*)unsupported,
                     nrows);
}

void hipdf_byte_eq(const void* bytes, int target, void* out, int64_t n,
                   hipStream_t stream) {
  hipLaunchKernelGGL(k_byte_eq, flat_grid(n), dim3(HIPDF_BLOCK), 0, stream,
                     (const uint8_t*)bytes, (uint8_t)target, (uint8_t*)out,
                     n);
}

void hipdf_csv_parse(const void* bytes, const void* row_start,
                     const void* row_end, int delim, int field_idx, int type,
                     void* out_i64, void* out_f64, void* out_ss,
                     void* out_sl, void* valid, void* unsupported,
                     int64_t nrows, hipStream_t stream) {
  hipLaunchKernelGGL(k_csv_parse, flat_grid(nrows), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)bytes,
                     (const int32_t*)row_start, (const int32_t*)row_end,
                     (uint8_t)delim, field_idx, type, (int64_t*)out_i64,
                     (double*)out_f64, (int32_t*)out_ss, (int64_t*)out_sl,
                     (uint8_t*)valid, (int*)unsupported, nrows);
}

}  // extern "C"
