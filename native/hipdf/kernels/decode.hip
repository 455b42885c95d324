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
// bits into out[0..n_values)
__global__ void k_rle_hybrid_decode(const uint8_t* __restrict__ data,
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
