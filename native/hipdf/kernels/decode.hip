// Parquet decode kernels: RLE/bit-packed hybrid runs (definition levels and
// dictionary indices), dense->row scatter, and def-level -> validity bitmask
// (reference analogue: libcudf's parquet decode kernels reached via
// Table.readParquet — SURVEY.md §2.8A; here: flat schemas, PLAIN +
// *_DICTIONARY encodings, host-decompressed pages).
//
// One workgroup decodes one page's RLE stream: lane 0 walks run headers
// (varint), the whole block expands runs in parallel (RLE fill / bit-packed
// unpack, LSB-first as parquet specifies).
#include "hipdf_common.h"

// decode an RLE/bit-packed hybrid stream of n_values values of bit_width
// bits into out[0..n_values); one workgroup per stream (function-scope
// __shared__ is per block, so the batched kernel reuses this body)
__device__ void rle_decode_one(const uint8_t* __restrict__ data,
                               int64_t nbytes, int bit_width,
                               int32_t* __restrict__ out,
                               int64_t n_values) {
  __shared__ int64_t s_pos;       // byte position in stream
  __shared__ int64_t s_out;       // values emitted
  __shared__ int64_t s_run_len;   // current run length (values)
  __shared__ int32_t s_run_val;   // RLE value
  __shared__ int s_is_packed;

  if (threadIdx.x == 0) {
    s_pos = 0;
    s_out = 0;
  }
  __syncthreads();
  int byte_per_val = (bit_width + 7) / 8;

  while (true) {
    __syncthreads();
    if (s_out >= n_values || s_pos >= nbytes) break;
    if (threadIdx.x == 0) {
      // varint header
      int64_t p = s_pos;
      uint64_t h = 0;
      int shift = 0;
      while (p < nbytes) {
        uint8_t b = data[p++];
        h |= (uint64_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) break;
        shift += 7;
      }
      if ((h & 1) == 0) {
        // RLE run: value stored in ceil(bw/8) LE bytes
        int64_t cnt = (int64_t)(h >> 1);
        int32_t v = 0;
        for (int k = 0; k < byte_per_val && p < nbytes; ++k)
          v |= (int32_t)data[p++] << (8 * k);
        s_is_packed = 0;
        s_run_len = cnt;
        s_run_val = v;
      } else {
        int64_t groups = (int64_t)(h >> 1);
        s_is_packed = 1;
        s_run_len = groups * 8;
      }
      s_pos = p;
    }
    __syncthreads();
    int64_t base = s_out;
    int64_t len = s_run_len;
    int64_t emit = min(len, n_values - base);
    int is_packed = s_is_packed;
    int64_t run_bytes_start = s_pos;
    int32_t run_val = s_run_val;
    __syncthreads();  // snapshot shared state before thread 0 mutates it
    if (is_packed) {
      for (int64_t j = threadIdx.x; j < emit; j += blockDim.x) {
        int64_t bitpos = j * bit_width;
        int64_t bytep = run_bytes_start + (bitpos >> 3);
        int sh = (int)(bitpos & 7);
        uint64_t w = 0;
        for (int k = 0; k < 8; ++k) {
          int64_t idx = bytep + k;
          if (idx < nbytes) w |= (uint64_t)data[idx] << (8 * k);
        }
        uint32_t mask = bit_width >= 32 ? 0xFFFFFFFFu
                                        : ((1u << bit_width) - 1u);
        out[base + j] = (int32_t)((w >> sh) & mask);
      }
      if (threadIdx.x == 0) {
        s_pos = run_bytes_start + (len / 8) * bit_width;  // groups*bw bytes
        s_out = base + emit;
      }
    } else {
      for (int64_t j = threadIdx.x; j < emit; j += blockDim.x)
        out[base + j] = run_val;
      if (threadIdx.x == 0) s_out = base + emit;
    }
  }
}

__global__ void k_rle_hybrid_decode(const uint8_t* __restrict__ data,
                                    int64_t nbytes, int bit_width,
                                    int32_t* __restrict__ out,
                                    int64_t n_values) {
  rle_decode_one(data, nbytes, bit_width, out, n_values);
}

// batched variant: one workgroup per RLE stream (per parquet page), so a
// whole column chunk's pages decode in one launch instead of serial
// per-page launches. descs: 5 int64 per stream
// [src_off, nbytes, out_off, n_values, bit_width] into a single base
// buffer (the decompressed chunk).
__global__ void k_rle_hybrid_batch(const uint8_t* __restrict__ base,
                                   const int64_t* __restrict__ descs,
                                   int32_t* __restrict__ out) {
  const int64_t* d = descs + 5 * (int64_t)blockIdx.x;
  rle_decode_one(base + d[0], d[1], (int)d[4], out + d[2], d[3]);
}

// fully parallel RLE expansion: the host walks the run HEADERS
// (hipdf_rle_walk_host — cheap, one varint per run) and ships a run table
// [kind, src_off, out_off, count] x int64; every output value then finds
// its run by binary search and decodes independently. This removes the
// single-workgroup-per-stream serialization that dominated large-page
// chunks (a 2.5M-value stream decoded by one block was 93% of window-
// query kernel time).
__global__ void k_rle_expand(const uint8_t* __restrict__ base,
                             const int64_t* __restrict__ runs,
                             int64_t nruns, int bit_width,
                             int32_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    // largest r with out_off[r] <= i
    int64_t lo = 0, hi = nruns - 1;
    while (lo < hi) {
      int64_t mid = (lo + hi + 1) >> 1;
      if (runs[4 * mid + 2] <= i) lo = mid;
      else hi = mid - 1;
    }
    const int64_t* r = runs + 4 * lo;
    int64_t local = i - r[2];
    // r[0] packs (bit_width << 1) | kind so streams with different bit
    // widths (dictionary pages) expand in one launch
    int kind = (int)(r[0] & 1);
    int bw = (int)(r[0] >> 1);
    if (kind == 0) {
      // RLE run: little-endian value at src_off
      int byte_per_val = (bw + 7) / 8;
      int32_t v = 0;
      for (int k = 0; k < byte_per_val; ++k)
        v |= (int32_t)base[r[1] + k] << (8 * k);
      out[i] = v;
    } else {
      int64_t bitpos = local * bw;
      int64_t bytep = r[1] + (bitpos >> 3);
      int sh = (int)(bitpos & 7);
      uint64_t w = 0;
      for (int k = 0; k < 8; ++k) w |= (uint64_t)base[bytep + k] << (8 * k);
      uint32_t mask = bw >= 32 ? 0xFFFFFFFFu : ((1u << bw) - 1u);
      out[i] = (int32_t)((w >> sh) & mask);
    }
  }
}

// out[idx[j]] = vals[j]
template <typename T>
__global__ void k_scatter_fixed(const T* __restrict__ vals,
                                const int32_t* __restrict__ idx,
                                T* __restrict__ out, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x)
    out[idx[j]] = vals[j];
}

// validity bitmask from a byte-per-row levels array (level==max -> valid)
__global__ void k_levels_to_mask(const int32_t* __restrict__ levels,
                                 int max_level, uint64_t* __restrict__ mask,
                                 int64_t nstripe, int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / WAVE;
  int lane = lane_id();
  for (int64_t s = wave_global; s < nstripe; s += wave_count) {
    int64_t i = s * WAVE + lane;
    bool ok = i < n && levels[i] == max_level;
    uint64_t ballot = __ballot(ok);
    if (lane == 0) mask[s] = ballot;
  }
}

// parquet DELTA_BINARY_PACKED (encoding 5): header(block_size,
// miniblocks/block, total, first zigzag) then per block min_delta +
// per-miniblock bit widths + LSB-first packed deltas. Sequential walker,
// one launch per page (pages decode concurrently on the stream).
__device__ __forceinline__ uint64_t pq_uvarint(const uint8_t* b,
                                               int64_t* p) {
  uint64_t out = 0;
  int sh = 0;
  while (true) {
    uint8_t v = b[(*p)++];
    out |= (uint64_t)(v & 0x7F) << sh;
    if (!(v & 0x80)) return out;
    sh += 7;
  }
}

__global__ void k_pq_delta_i64(const uint8_t* __restrict__ b,
                               int64_t nbytes, int64_t n,
                               int64_t* __restrict__ out) {
  if (blockIdx.x || threadIdx.x) return;
  int64_t p = 0;
  uint64_t block_size = pq_uvarint(b, &p);
  uint64_t mini_per_block = pq_uvarint(b, &p);
  uint64_t total = pq_uvarint(b, &p);
  (void)total;
  uint64_t fz = pq_uvarint(b, &p);
  int64_t cur = (int64_t)(fz >> 1) ^ -(int64_t)(fz & 1);
  uint64_t per_mini = block_size / mini_per_block;
  int64_t k = 0;
  if (k < n) out[k++] = cur;
  while (k < n && p < nbytes) {
    uint64_t mdz = pq_uvarint(b, &p);
    int64_t min_delta = (int64_t)(mdz >> 1) ^ -(int64_t)(mdz & 1);
    int64_t widths_at = p;
    p += mini_per_block;
    for (uint64_t m = 0; m < mini_per_block && k < n; ++m) {
      int w = b[widths_at + m];
      // LSB-first bit unpacking (parquet packing order)
      int64_t bit = 0;
      for (uint64_t i = 0; i < per_mini && k < n; ++i) {
        uint64_t v = 0;
        for (int got = 0; got < w; ++got, ++bit)
          v |= (uint64_t)((b[p + (bit >> 3)] >> (bit & 7)) & 1) << got;
        cur += min_delta + (int64_t)v;
        out[k++] = cur;
      }
      p += ((int64_t)w * per_mini + 7) / 8;
    }
  }
}

extern "C" {

void hipdf_pq_delta_i64(const void* b, int64_t nbytes, int64_t n, void* out,
                        hipStream_t stream) {
  hipLaunchKernelGGL(k_pq_delta_i64, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)b, nbytes, n, (int64_t*)out);
}

}  // extern "C"

// ---- ORC stream decoders (reference analogue: libcudf ORC reader fed by
// GpuOrcScan; SURVEY.md §2.3 ORC row). Sequential single-thread walkers:
// one launch per (stripe, column, stream) so stripes and columns decode
// concurrently on independent streams/blocks; within-run parallelism is a
// later optimization (ORC scan is not on the bench hot path).

// ORC boolean byte-RLE (PRESENT / bool DATA): emits one u8 per row
// (bits are MSB-first within each payload byte)
__global__ void k_orc_bool_rle(const uint8_t* __restrict__ b,
                               int64_t nbytes, int64_t n,
                               uint8_t* __restrict__ out) {
  if (blockIdx.x || threadIdx.x) return;
  int64_t p = 0, k = 0;
  while (k < n && p < nbytes) {
    int h = b[p++];
    int64_t count;
    bool run;
    uint8_t v = 0;
    if (h < 128) {
      count = h + 3;
      run = true;
      v = b[p++];
    } else {
      count = 256 - h;
      run = false;
    }
    for (int64_t c = 0; c < count && k < n; ++c) {
      uint8_t byte = run ? v : b[p + c];
      for (int bit = 7; bit >= 0 && k < n; --bit)
        out[k++] = (byte >> bit) & 1;
    }
    if (!run) p += count;
  }
}

__device__ __forceinline__ int64_t orc_varint(const uint8_t* b, int64_t* p) {
  uint64_t out = 0;
  int sh = 0;
  while (true) {
    uint8_t v = b[(*p)++];
    out |= (uint64_t)(v & 0x7F) << sh;
    if (!(v & 0x80)) return (int64_t)out;
    sh += 7;
  }
}

__device__ __forceinline__ int64_t orc_zigzag(int64_t v) {
  return (int64_t)((uint64_t)v >> 1) ^ -(v & 1);
}

__constant__ int ORC_WIDTHS[32] = {1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12,
                                   13, 14, 15, 16, 17, 18, 19, 20, 21, 22,
                                   23, 24, 26, 28, 30, 32, 40, 48, 56, 64};

__device__ __forceinline__ uint64_t orc_bits(const uint8_t* b, int64_t base,
                                             int64_t idx, int w) {
  // big-endian bit packing, element idx of width w starting at byte base
  uint64_t v = 0;
  int64_t bit = idx * (int64_t)w;
  for (int got = 0; got < w; ++got, ++bit)
    v = (v << 1) | ((b[base + (bit >> 3)] >> (7 - (bit & 7))) & 1);
  return v;
}

// RLEv2 integer decoding: short-repeat / direct / delta / patched-base
__global__ void k_orc_rle_v2(const uint8_t* __restrict__ b, int64_t nbytes,
                             int64_t n, int is_signed,
                             int64_t* __restrict__ out) {
  if (blockIdx.x || threadIdx.x) return;
  int64_t p = 0, k = 0;
  while (k < n && p < nbytes) {
    int h = b[p];
    int enc = h >> 6;
    if (enc == 0) {  // short repeat
      int width = ((h >> 3) & 7) + 1;
      int rep = (h & 7) + 3;
      ++p;
      uint64_t u = 0;
      for (int i = 0; i < width; ++i) u = (u << 8) | b[p++];
      int64_t v = is_signed ? orc_zigzag((int64_t)u) : (int64_t)u;
      for (int i = 0; i < rep && k < n; ++i) out[k++] = v;
    } else if (enc == 1) {  // direct
      int w = ORC_WIDTHS[(h >> 1) & 31];
      int ln = (((h & 1) << 8) | b[p + 1]) + 1;
      p += 2;
      for (int i = 0; i < ln && k < n; ++i) {
        uint64_t u = orc_bits(b, p, i, w);
        out[k++] = is_signed ? orc_zigzag((int64_t)u) : (int64_t)u;
      }
      p += ((int64_t)w * ln + 7) / 8;
    } else if (enc == 3) {  // delta
      int wcode = (h >> 1) & 31;
      int w = wcode == 0 ? 0 : ORC_WIDTHS[wcode];
      int ln = (((h & 1) << 8) | b[p + 1]) + 1;
      p += 2;
      int64_t base = orc_varint(b, &p);
      if (is_signed) base = orc_zigzag(base);
      int64_t d0 = orc_zigzag(orc_varint(b, &p));
      int64_t cur = base;
      if (k < n) out[k++] = cur;
      int emitted = 1;
      if (emitted < ln && k < n) {
        cur += d0;
        out[k++] = cur;
        ++emitted;
      }
      int64_t sign = d0 < 0 ? -1 : 1;
      for (int i = 0; emitted < ln && k < n; ++i, ++emitted) {
        int64_t d = w ? sign * (int64_t)orc_bits(b, p, i, w) : d0;
        cur += d;
        out[k++] = cur;
      }
      if (w) p += ((int64_t)w * (ln - 2) + 7) / 8;
    } else {  // patched base
      int w = ORC_WIDTHS[(h >> 1) & 31];
      int ln = (((h & 1) << 8) | b[p + 1]) + 1;
      int bw = ((b[p + 2] >> 5) & 7) + 1;
      int pw = ORC_WIDTHS[b[p + 2] & 31];
      int pgw = ((b[p + 3] >> 5) & 7) + 1;
      int pll = b[p + 3] & 31;
      p += 4;
      int64_t base = 0;
      for (int i = 0; i < bw; ++i) base = (base << 8) | b[p++];
      int64_t smask = 1ll << (bw * 8 - 1);
      if (base & smask) base = -(base & (smask - 1));
      int64_t vals_base = p;
      p += ((int64_t)w * ln + 7) / 8;
      int64_t patch_base = p;
      int pew = pw + pgw * 8;
      p += ((int64_t)pew * pll + 7) / 8;
      // apply patches while emitting: precompute into a small loop —
      // patches are sorted by gap, walk alongside
      int pi = 0;
      int64_t patched_idx = -1;
      uint64_t patch_val = 0;
      int64_t gap_acc = 0;
      if (pll > 0) {
        uint64_t pv = orc_bits(b, patch_base, 0, pew);
        gap_acc = (int64_t)(pv >> pw);
        patch_val = pv & ((1ull << pw) - 1);
        patched_idx = gap_acc;
        pi = 1;
      }
      for (int i = 0; i < ln && k < n; ++i) {
        uint64_t u = orc_bits(b, vals_base, i, w);
        while (patched_idx >= 0 && i == patched_idx) {
          u |= patch_val << w;
          if (pi < pll) {
            uint64_t pv = orc_bits(b, patch_base, pi, pew);
            gap_acc += (int64_t)(pv >> pw);
            patch_val = pv & ((1ull << pw) - 1);
            patched_idx = gap_acc;
            ++pi;
          } else {
            patched_idx = -1;
          }
        }
        out[k++] = base + (int64_t)u;
      }
    }
  }
}

extern "C" {

void hipdf_orc_bool_rle(const void* b, int64_t nbytes, int64_t n, void* out,
                        hipStream_t stream) {
  hipLaunchKernelGGL(k_orc_bool_rle, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)b, nbytes, n, (uint8_t*)out);
}

void hipdf_orc_rle_v2(const void* b, int64_t nbytes, int64_t n,
                      int is_signed, void* out, hipStream_t stream) {
  hipLaunchKernelGGL(k_orc_rle_v2, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)b, nbytes, n, is_signed,
                     (int64_t*)out);
}

}  // extern "C"

// PLAIN byte-array page: [u32 len][bytes]... — walk the lengths (single
// block, sequential dependency) emitting per-value (start, len); the
// byte compaction reuses the substr copy kernel.
__global__ void k_str_plain_offsets(const uint8_t* __restrict__ data,
                                    int64_t nbytes, int64_t n_values,
                                    int32_t* __restrict__ starts,
                                    int64_t* __restrict__ lens,
                                    int32_t* __restrict__ error) {
  if (blockIdx.x != 0 || threadIdx.x != 0) return;
  int64_t p = 0;
  for (int64_t i = 0; i < n_values; ++i) {
    if (p + 4 > nbytes) {
      atomicAdd(error, 1);
      return;
    }
    uint32_t ln = (uint32_t)data[p] | ((uint32_t)data[p + 1] << 8) |
                  ((uint32_t)data[p + 2] << 16) |
                  ((uint32_t)data[p + 3] << 24);
    p += 4;
    if (p + ln > (uint64_t)nbytes) {
      atomicAdd(error, 1);
      return;
    }
    starts[i] = (int32_t)p;
    lens[i] = ln;
    p += ln;
  }
}

// inverse of the PLAIN byte-array decode: emit [u32 len][bytes] per row
// (GPU parquet writer). mode 0: per-row output sizes; mode 1: write.
__global__ void k_str_plain_encode(const int32_t* __restrict__ offsets,
                                   const uint8_t* __restrict__ bytes,
                                   const int64_t* __restrict__ out_off,
                                   int64_t* __restrict__ out_len,
                                   uint8_t* __restrict__ out, int mode,
                                   int64_t n) {
  int64_t wave_global = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  int64_t wave_count = ((int64_t)gridDim.x * blockDim.x) / 64;
  int lane = (int)(threadIdx.x & 63);
  for (int64_t i = wave_global; i < n; i += wave_count) {
    int32_t a = offsets[i], b = offsets[i + 1];
    int32_t len = b - a;
    if (!mode) {
      if (lane == 0) out_len[i] = 4 + (int64_t)len;
      continue;
    }
    uint8_t* dst = out + out_off[i];
    if (lane == 0) {
      dst[0] = (uint8_t)len;
      dst[1] = (uint8_t)(len >> 8);
      dst[2] = (uint8_t)(len >> 16);
      dst[3] = (uint8_t)(len >> 24);
    }
    for (int32_t k = lane; k < len; k += 64) dst[4 + k] = bytes[a + k];
  }
}

extern "C" {

void hipdf_str_plain_encode(const void* offsets, const void* bytes,
                            const void* out_off, void* out_len, void* out,
                            int mode, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_plain_encode, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)offsets, (const uint8_t*)bytes,
                     (const int64_t*)out_off, (int64_t*)out_len,
                     (uint8_t*)out, mode, n);
}

void hipdf_str_plain_offsets(const void* data, int64_t nbytes,
                             int64_t n_values, void* starts, void* lens,
                             void* error, hipStream_t stream) {
  hipLaunchKernelGGL(k_str_plain_offsets, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)data, nbytes, n_values,
                     (int32_t*)starts, (int64_t*)lens, (int32_t*)error);
}

void hipdf_rle_hybrid_decode(const void* data, int64_t nbytes, int bit_width,
                             void* out, int64_t n_values, hipStream_t stream) {
  hipLaunchKernelGGL(k_rle_hybrid_decode, dim3(1), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)data, nbytes, bit_width,
                     (int32_t*)out, n_values);
}

void hipdf_rle_expand(const void* base, const void* runs, int64_t nruns,
                      int bit_width, void* out, int64_t n,
                      hipStream_t stream) {
  if (n <= 0 || nruns <= 0) return;
  hipLaunchKernelGGL(k_rle_expand, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const uint8_t*)base, (const int64_t*)runs,
                     nruns, bit_width, (int32_t*)out, n);
}

void hipdf_rle_hybrid_batch(const void* base, const void* descs,
                            int nstreams, void* out, hipStream_t stream) {
  if (nstreams <= 0) return;
  hipLaunchKernelGGL(k_rle_hybrid_batch, dim3((uint32_t)nstreams),
                     dim3(HIPDF_BLOCK), 0, stream, (const uint8_t*)base,
                     (const int64_t*)descs, (int32_t*)out);
}

void hipdf_scatter_fixed(int esize, const void* vals, const void* idx,
                         void* out, int64_t n, hipStream_t stream) {
  dim3 grid = flat_grid(n);
  switch (esize) {
    case 1:
      hipLaunchKernelGGL((k_scatter_fixed<uint8_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint8_t*)vals, (const int32_t*)idx,
                         (uint8_t*)out, n);
      break;
    case 2:
      hipLaunchKernelGGL((k_scatter_fixed<uint16_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint16_t*)vals,
                         (const int32_t*)idx, (uint16_t*)out, n);
      break;
    case 4:
      hipLaunchKernelGGL((k_scatter_fixed<uint32_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint32_t*)vals,
                         (const int32_t*)idx, (uint32_t*)out, n);
      break;
    case 8:
      hipLaunchKernelGGL((k_scatter_fixed<uint64_t>), grid, dim3(HIPDF_BLOCK),
                         0, stream, (const uint64_t*)vals,
                         (const int32_t*)idx, (uint64_t*)out, n);
      break;
    case 16:
      hipLaunchKernelGGL((k_scatter_fixed<ulonglong2>), grid,
                         dim3(HIPDF_BLOCK), 0, stream, (const ulonglong2*)vals,
                         (const int32_t*)idx, (ulonglong2*)out, n);
      break;
    default:
      throw std::runtime_error("scatter: bad element size");
  }
}

void hipdf_levels_to_mask(const void* levels, int max_level, void* mask,
                          int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_levels_to_mask, stripe_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)levels, max_level,
                     (uint64_t*)mask, n_stripes(n), n);
}

}  // extern "C"
