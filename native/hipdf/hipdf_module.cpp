// pybind11 bindings for the hipdf CDNA4 kernel library.
// Python (spark_rapids_amd/ops/gpu_backend.py) passes torch tensor
// data_ptr()s, sizes and the current torch HIP stream; all device buffers are
// allocated by the PyTorch-ROCm caching allocator (the RMM-style pool on
// 288 GB HBM3E) so hipdf itself never calls hipMalloc.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <string>

namespace py = pybind11;

extern "C" {
void hipdf_binary_arith(int, int, const void*, const void*, double, int64_t,
                        int, const void*, const void*, void*, void*, int64_t,
                        hipStream_t);
void hipdf_binary_cmp(int, int, const void*, const void*, double, int64_t, int,
                      const void*, const void*, void*, void*, int64_t,
                      hipStream_t);
void hipdf_binary_bool(int, const void*, const void*, int, int, const void*,
                       const void*, void*, void*, int64_t, hipStream_t);
void hipdf_unary(int, int, const void*, const void*, void*, void*, int64_t,
                 hipStream_t);
void hipdf_cast(int, int, const void*, void*, int64_t, hipStream_t);
void hipdf_f64_to_i64_rint(const void*, void*, int64_t, hipStream_t);
void hipdf_decimal_rescale(const void*, void*, int64_t, int, int64_t,
                           hipStream_t);
void hipdf_if_else(int, const void*, const void*, const void*, const void*,
                   const void*, const void*, void*, void*, int64_t,
                   hipStream_t);
void hipdf_mask_expand(const void*, void*, int, int64_t, hipStream_t);
int64_t sel_num_blocks(int64_t);
void hipdf_mask_count(const void*, const void*, void*, int64_t, hipStream_t);
void hipdf_mask_scatter(const void*, const void*, const void*, void*, int64_t,
                        hipStream_t);
void hipdf_gather_fixed(int, const void*, const void*, void*, int64_t,
                        hipStream_t);
void hipdf_gather_validity(const void*, int, const void*, void*, int64_t,
                           hipStream_t);
void hipdf_gather_table(const void*, int, const void*, int64_t, hipStream_t);
void hipdf_gather_str_lens(const void*, const void*, void*, int64_t,
                           hipStream_t);
void hipdf_gather_str_bytes(const void*, const void*, const void*,
                            const void*, void*, int64_t, int64_t,
                            hipStream_t);
void hipdf_narrow_i64_i32(const void*, void*, int64_t, hipStream_t);
void hipdf_copy_valid_range(const void*, int, int64_t, void*, int64_t,
                            hipStream_t);
int64_t scan_num_blocks(int64_t);
void hipdf_scan_block(const void*, void*, void*, int64_t, hipStream_t);
void hipdf_scan_add_offsets(void*, const void*, int64_t, hipStream_t);
void hipdf_reduce(int, int, const void*, const void*, void*, void*, int64_t,
                  hipStream_t);
void hipdf_murmur3_col(int, int, const void*, const void*, const void*,
                       void*, int64_t, hipStream_t);
void hipdf_murmur3_str(const void*, const void*, const void*, const void*,
                       void*, int64_t, hipStream_t);
void hipdf_pmod_part(const void*, int, void*, int64_t, hipStream_t);
void hipdf_gb_hll(const void*, const void*, const void*, const void*,
                  void*, int, int64_t, hipStream_t);
void hipdf_xxhash64_col(int, int, const void*, const void*, const void*,
                        void*, int64_t, hipStream_t);
void hipdf_xxhash64_str(const void*, const void*, const void*, const void*,
                        void*, int64_t, hipStream_t);
void hipdf_gb_build(const void*, const void*, int, const void*, void*,
                    void*, void*, void*, int64_t, int64_t, hipStream_t);
void hipdf_gb_number(const void*, const void*, void*, const void*, void*,
                     int64_t, hipStream_t);
void hipdf_gb_rowgid(const void*, const void*, void*, int64_t, hipStream_t);
void hipdf_gb_agg(int, int, const void*, const void*, const void*, void*,
                  void*, int, int32_t, int64_t, hipStream_t);
void hipdf_gb_agg_multi(const void*, int, const void*, const void*,
                        int32_t, int, int64_t, hipStream_t);
void hipdf_gb_reduce_reps(int, void*, int, void*, int32_t, int,
                          hipStream_t);
void hipdf_gb_acc_init(int, void*, int, int32_t, hipStream_t);
void hipdf_mask_from_nonzero(const void*, void*, int64_t, hipStream_t);
void hipdf_join_build(const void*, const void*, int, void*, void*, int64_t,
                      int64_t, hipStream_t);
void hipdf_join_count(int, const void*, const void*, const void*, int,
                      const void*, const void*, int64_t, void*, int64_t,
                      hipStream_t);
void hipdf_join_fill(int, const void*, const void*, const void*, int,
                     const void*, const void*, int64_t, const void*, void*,
                     void*, void*, int64_t, hipStream_t);
int64_t part_num_blocks(int64_t);
int64_t sort_num_blocks(int64_t);
void hipdf_dec64_mul_div(int, const void*, const void*, const void*,
                         const void*, void*, void*, int, int, int, int64_t,
                         hipStream_t);
void hipdf_i128_arith(int, const void*, const void*, const void*, const void*,
                      void*, void*, int64_t, hipStream_t);
void hipdf_i128_cmp(int, const void*, const void*, const void*, const void*,
                    void*, void*, int64_t, hipStream_t);
void hipdf_i64_to_i128(const void*, void*, int64_t, hipStream_t);
void hipdf_i128_rescale(const void*, const void*, void*, void*, int, int,
                        int, int64_t, hipStream_t);
void hipdf_dec_mul_div_wide(int, const void*, const void*, const void*,
                            const void*, int, int, void*, void*, int, int,
                            int, int64_t, hipStream_t);
void hipdf_i128_to_f64(const void*, void*, int64_t, hipStream_t);
void hipdf_gb_percentile(const void*, const void*, const void*, const void*,
                         double, void*, int, hipStream_t);
void hipdf_dense_gid(const void*, int, const void*, void*, int64_t,
                     hipStream_t);
void hipdf_expand_rows(const void*, void*, void*, int64_t, hipStream_t);
void hipdf_gb_collect_count(const void*, const void*, const void*, void*,
                            int64_t, hipStream_t);
void hipdf_gb_collect_fill(int, const void*, const void*, const void*,
                           const void*, const void*, void*, void*, int64_t,
                           hipStream_t);
void hipdf_gb_sum_i128_lds(int, const void*, const void*, const void*,
                           const void*, void*, void*, int, int64_t,
                           hipStream_t);
void hipdf_gb_sum_i64_to_i128(const void*, const void*, const void*,
                              const void*, void*, void*, int64_t, hipStream_t);
void hipdf_gb_sum_i128(const void*, const void*, const void*, const void*,
                       void*, void*, int64_t, hipStream_t);
void hipdf_win_minmax(int, const void*, int, const void*, const void*,
                      int, void*, int64_t, hipStream_t);
void hipdf_range_bounds(const void*, const void*, const void*, double,
                        double, int, int, void*, void*, int64_t,
                        hipStream_t);
void hipdf_change_flags(const void*, int, void*, int64_t, hipStream_t);
void hipdf_iota_i32(void*, int64_t, hipStream_t);
void hipdf_scan_block_f64(const void*, void*, void*, int64_t, hipStream_t);
void hipdf_scan_add_offsets_f64(void*, const void*, int64_t, hipStream_t);
void hipdf_rle_expand(const void*, const void*, int64_t, int, void*,
                      int64_t, hipStream_t);
void hipdf_rle_hybrid_batch(const void*, const void*, int, void*,
                            hipStream_t);
void hipdf_rle_hybrid_decode(const void*, int64_t, int, void*, int64_t,
                             hipStream_t);
void hipdf_pq_delta_i64(const void*, int64_t, int64_t, void*, hipStream_t);
void hipdf_orc_bool_rle(const void*, int64_t, int64_t, void*, hipStream_t);
void hipdf_orc_rle_v2(const void*, int64_t, int64_t, int, void*,
                      hipStream_t);
void hipdf_json_field(const void*, const void*, const void*, const void*,
                      int, int, void*, void*, void*, void*, void*, void*,
                      int64_t, hipStream_t);
void hipdf_byte_eq(const void*, int, void*, int64_t, hipStream_t);
void hipdf_csv_parse(const void*, const void*, const void*, int, int, int,
                     void*, void*, void*, void*, void*, void*, int64_t,
                     hipStream_t);
void hipdf_str_plain_encode(const void*, const void*, const void*, void*,
                            void*, int, int64_t, hipStream_t);
void hipdf_str_plain_offsets(const void*, int64_t, int64_t, void*, void*,
                             void*, hipStream_t);
void hipdf_scatter_fixed(int, const void*, const void*, void*, int64_t,
                         hipStream_t);
void hipdf_levels_to_mask(const void*, int, void*, int64_t, hipStream_t);
void hipdf_str_cmp(int, const void*, const void*, const void*, const void*,
                   void*, int64_t, hipStream_t);
void hipdf_regex_extract(const void*, int, const void*, const void*,
                         const void*, int, void*, void*, void*, int64_t,
                         hipStream_t);
void hipdf_regex_extract_all(const void*, int, const void*, const void*,
                             const void*, int, const void*, void*, void*,
                             void*, int, void*, int64_t, hipStream_t);
void hipdf_regex_replace(const void*, int, const void*, const void*,
                         const void*, const void*, int, const void*,
                         const void*, void*, void*, int, void*, int64_t,
                         hipStream_t);
void hipdf_regex_match(const void*, int, const void*, const void*,
                       const void*, void*, void*, int64_t, hipStream_t);
void hipdf_str_cmp_scalar(int, const void*, const void*, const void*, int,
                          void*, int64_t, hipStream_t);
void hipdf_str_find(int, const void*, const void*, const void*, int, void*,
                    int64_t, hipStream_t);
void hipdf_str_like(const void*, const void*, const void*, int, void*,
                    int64_t, hipStream_t);
void hipdf_str_length(const void*, const void*, void*, int64_t, hipStream_t);
void hipdf_str_case(int, const void*, void*, int64_t, hipStream_t);
void hipdf_i64_to_str(const void*, const void*, void*, void*, int,
                      int64_t, hipStream_t);
void hipdf_str_concat_ws(const void*, int, const void*, int, const void*,
                         void*, void*, int, int64_t, hipStream_t);
void hipdf_str_initcap(const void*, const void*, void*, int64_t,
                       hipStream_t);
void hipdf_str_reverse(const void*, const void*, void*, int64_t,
                       hipStream_t);
void hipdf_str_split_count(const void*, const void*, const void*, int,
                           void*, int64_t, hipStream_t);
void hipdf_str_split_fill(const void*, const void*, const void*, int,
                          const void*, const void*, void*, void*, int64_t,
                          hipStream_t);
void hipdf_str_trim_ranges(int, const void*, const void*, void*, void*,
                           int64_t, hipStream_t);
void hipdf_str_concat2(const void*, const void*, const void*, const void*,
                       const void*, void*, void*, int, int64_t, hipStream_t);
void hipdf_substr_ranges(const void*, const void*, int, int, void*, void*,
                         int64_t, hipStream_t);
void hipdf_substr_copy(const void*, const void*, const void*, const void*,
                       void*, int64_t, hipStream_t);
int hipdf_sort_key_width(int);
void hipdf_make_sort_keys(int, const void*, const void*, const void*, int,
                          int, int, void*, int64_t, hipStream_t);
void hipdf_make_sort_keys_str(const void*, const void*, const void*,
                              const void*, int, int, void*, int64_t,
                              hipStream_t);
void hipdf_make_sort_keys_i128(const void*, const void*, const void*, int,
                               int, void*, int64_t, hipStream_t);
void hipdf_radix_count(const void*, int, void*, int64_t, hipStream_t);
void hipdf_radix_scatter(const void*, const void*, int, const void*, void*,
                         void*, int64_t, hipStream_t);
void hipdf_part_hist(const void*, int, void*, int64_t, hipStream_t);
void hipdf_part_scatter(const void*, int, const void*, void*, int64_t,
                        hipStream_t);
}

static void check_async() {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("hipdf kernel launch: ") +
                             hipGetErrorString(e));
}

#define P(x) reinterpret_cast<const void*>(x)
#define PM(x) reinterpret_cast<void*>(x)
#define S(x) reinterpret_cast<hipStream_t>(x)

extern "C" {
void hipdf_str_pad(int, const void*, const void*, const void*, int32_t,
                   int32_t, int32_t, const void*, void*, void*, int,
                   int64_t, hipStream_t);
void hipdf_str_locate(const void*, const void*, const void*, int32_t,
                      int32_t, void*, int64_t, hipStream_t);
}

extern "C" {
void hipdf_tz_convert(const void*, const void*, const void*, int, int,
                      void*, int64_t, hipStream_t);
void hipdf_date_format(const void*, const void*, int, int, void*, int64_t,
                       hipStream_t);
void hipdf_ts_parse(const void*, const void*, const void*, const void*, int,
                    int, void*, void*, int64_t, hipStream_t);
}

// ---- exact string<->decimal casts (kernels/cast_str.hip) -----------------
extern "C" {
void hipdf_str_to_dec(const void*, const void*, const void*, int, int, int,
                      void*, void*, int64_t, hipStream_t);
void hipdf_dec_to_str(const void*, int, int, const void*, void*, void*, int,
                      int64_t, hipStream_t);
}

// ---- device memory pool (pool.hip) --------------------------------------
extern "C" {
int hipdf_pool_init(double, size_t);
int hipdf_pool_active();
typedef int (*hipdf_failure_cb)(size_t, int);
void hipdf_pool_set_failure_cb(hipdf_failure_cb);
size_t hipdf_pool_used();
size_t hipdf_pool_reserved();
size_t hipdf_pool_high_watermark();
size_t hipdf_pool_overflow();
int hipdf_pool_selftest();
}

// heap-allocated and intentionally leaked: a static py::object would run
// its destructor at .so unload AFTER Py_Finalize and crash the exit
static py::object* g_spill_cb = nullptr;

static int spill_cb_trampoline(size_t needed, int retry) {
  if (!g_spill_cb || !Py_IsInitialized()) return 0;
  py::gil_scoped_acquire gil;
  try {
    return (*g_spill_cb)((size_t)needed, retry).cast<int>();
  } catch (const std::exception& e) {
    fprintf(stderr, "[hipdf pool] spill callback raised: %s\n", e.what());
    PyErr_Clear();
    return 0;
  } catch (...) {
    PyErr_Clear();
    return 0;
  }
}

// Host-side walk of an RLE/bit-packed hybrid stream's RUN HEADERS: one
// varint per run (cheap), emitting [kind, src_off_abs, out_off, count]
// int64 records for the fully parallel k_rle_expand kernel. Returns the
// number of runs, or -1 when max_runs would overflow / stream malformed.
static int64_t rle_walk(const uint8_t* data, int64_t nbytes, int bit_width,
                        int64_t n_values, int64_t src_base, int64_t out_base,
                        int64_t* runs, int64_t max_runs) {
  int64_t pos = 0, emitted = 0, nruns = 0;
  int byte_per_val = (bit_width + 7) / 8;
  while (emitted < n_values) {
    if (pos >= nbytes) return -1;
    uint64_t h = 0;
    int shift = 0;
    while (pos < nbytes) {
      uint8_t b = data[pos++];
      h |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (nruns >= max_runs) return -1;
    int64_t* r = runs + 4 * nruns;
    if ((h & 1) == 0) {
      int64_t cnt = (int64_t)(h >> 1);
      if (cnt <= 0) return -1;
      r[0] = (int64_t)bit_width << 1;  // kind 0 | bw
      r[1] = src_base + pos;
      r[2] = out_base + emitted;
      r[3] = cnt;
      pos += byte_per_val;
      emitted += cnt;
    } else {
      int64_t groups = (int64_t)(h >> 1);
      if (groups <= 0) return -1;
      r[0] = ((int64_t)bit_width << 1) | 1;  // kind 1 | bw
      r[1] = src_base + pos;
      r[2] = out_base + emitted;
      r[3] = groups * 8;
      pos += groups * bit_width;
      emitted += groups * 8;
    }
    ++nruns;
  }
  return nruns;
}

// Host-side walk of parquet length-prefixed BYTE_ARRAY records. The chain
// pos -> len -> pos is inherently serial, so it runs on the CPU (one
// dependent L1 load per record, ~2ns) instead of a single GPU thread
// (~70ns per dependent global load); the GIL is released so the
// multi-file prefetch pool overlaps walks across column chunks.
static int64_t byte_array_offsets_walk(const uint8_t* p, int64_t nbytes,
                                       int64_t count, int32_t* starts,
                                       int64_t* lens) {
  int64_t pos = 0, total = 0;
  for (int64_t i = 0; i < count; ++i) {
    if (pos + 4 > nbytes) return -1;
    uint32_t l = (uint32_t)p[pos] | ((uint32_t)p[pos + 1] << 8) |
                 ((uint32_t)p[pos + 2] << 16) | ((uint32_t)p[pos + 3] << 24);
    pos += 4;
    if (pos + (int64_t)l > nbytes) return -1;
    starts[i] = (int32_t)pos;
    lens[i] = l;
    total += l;
    pos += l;
  }
  return total;
}

// Host decode of a parquet DELTA_BINARY_PACKED stream (header varints +
// per-block zigzag min_delta + bit-packed miniblocks). The format is a
// serial chain of varints, so like the walks above it runs on the CPU;
// the decoded ints are only per-page length vectors (DELTA_LENGTH_/
// DELTA_BYTE_ARRAY string pages), never bulk column data. Returns bytes
// consumed (so the caller can locate the payload that follows) or -1.
static int64_t pq_delta_walk(const uint8_t* p, int64_t nbytes, int64_t* out,
                             int64_t n) {
  int64_t pos = 0;
  auto varint = [&](uint64_t* v) -> bool {
    uint64_t r = 0;
    int shift = 0;
    while (pos < nbytes && shift < 64) {
      uint8_t b = p[pos++];
      r |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) {
        *v = r;
        return true;
      }
      shift += 7;
    }
    return false;
  };
  auto zigzag = [&](int64_t* v) -> bool {
    uint64_t u;
    if (!varint(&u)) return false;
    *v = (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
    return true;
  };
  uint64_t block_size, mb_per_block, total_count;
  int64_t first;
  if (!varint(&block_size) || !varint(&mb_per_block) ||
      !varint(&total_count) || !zigzag(&first))
    return -1;
  if (mb_per_block == 0 || block_size == 0 ||
      block_size % mb_per_block != 0)
    return -1;
  int64_t per_mb = (int64_t)(block_size / mb_per_block);
  if (per_mb % 8 != 0) return -1;
  int64_t want = (int64_t)total_count < n ? (int64_t)total_count : n;
  int64_t emitted = 0;
  int64_t cur = first;
  if (want > 0) out[emitted++] = cur;
  while (emitted < want) {
    int64_t min_delta;
    if (!zigzag(&min_delta)) return -1;
    if (pos + (int64_t)mb_per_block > nbytes) return -1;
    const uint8_t* bws = p + pos;
    pos += mb_per_block;
    for (uint64_t mb = 0; mb < mb_per_block; ++mb) {
      if (emitted >= want) break;  // trailing miniblocks are omitted
      int bw = bws[mb];
      if (bw > 64) return -1;
      int64_t mb_bytes = per_mb * bw / 8;
      if (pos + mb_bytes > nbytes) return -1;
      const uint8_t* src = p + pos;
      for (int64_t i = 0; i < per_mb && emitted < want; ++i) {
        uint64_t v = 0;
        if (bw > 0) {
          int64_t bit = i * bw;
          // miniblock is zero-padded to a full 8-byte-safe read window?
          // no: read byte-by-byte to stay in bounds
          for (int b = 0; b < bw; ++b) {
            int64_t bb = bit + b;
            if (src[bb >> 3] & (1 << (bb & 7))) v |= (uint64_t)1 << b;
          }
        }
        cur += min_delta + (int64_t)v;
        out[emitted++] = cur;
      }
      pos += mb_bytes;  // full (padded) miniblock is always present
    }
  }
  return pos;
}

// Host reconstruction of DELTA_BYTE_ARRAY strings: each value is
// prefix_len bytes of the PREVIOUS value + its suffix — a serial chain,
// so it runs on the CPU in one memcpy pass; the result uploads as a
// dense string column. Returns 0 or -1.
static int64_t delta_ba_concat(const int64_t* pre, const int64_t* suf,
                               const uint8_t* sufbytes, int64_t suf_nbytes,
                               int64_t n, uint8_t* out,
                               const int64_t* out_offs) {
  int64_t spos = 0;
  for (int64_t i = 0; i < n; ++i) {
    int64_t plen = pre[i], slen = suf[i];
    if (plen < 0 || slen < 0 || spos + slen > suf_nbytes) return -1;
    if (i == 0 ? plen != 0
               : plen > out_offs[i] - out_offs[i - 1])
      return -1;
    uint8_t* dst = out + out_offs[i];
    if (plen) memcpy(dst, out + out_offs[i - 1], (size_t)plen);
    if (slen) memcpy(dst + plen, sufbytes + spos, (size_t)slen);
    spos += slen;
  }
  return 0;
}

PYBIND11_MODULE(hipdf, m) {
  m.def("tz_convert", [](int64_t ts, int64_t trans, int64_t offs,
                         int n_trans, int to_utc, int64_t out, int64_t n,
                         int64_t stream) {
    hipdf_tz_convert(P(ts), P(trans), P(offs), n_trans, to_utc, PM(out), n,
                     S(stream));
  });
  m.def("date_format", [](int64_t ts, int64_t tokens, int ntok, int width,
                          int64_t out, int64_t n, int64_t stream) {
    hipdf_date_format(P(ts), P(tokens), ntok, width, PM(out), n, S(stream));
  });
  m.def("ts_parse", [](int64_t ao, int64_t ab, int64_t av, int64_t tokens,
                       int ntok, int width, int64_t out, int64_t ov,
                       int64_t n, int64_t stream) {
    hipdf_ts_parse(P(ao), P(ab), P(av), P(tokens), ntok, width, PM(out),
                   PM(ov), n, S(stream));
  });
  m.def("str_pad", [](int left, int64_t ao, int64_t ab, int64_t fill,
                      int fill_nb, int fill_cps, int width, int64_t out_off,
                      int64_t out_len, int64_t out, int mode, int64_t n,
                      int64_t stream) {
    hipdf_str_pad(left, P(ao), P(ab), P(fill), fill_nb, fill_cps, width,
                  P(out_off), PM(out_len), PM(out), mode, n, S(stream));
  });
  m.def("str_locate", [](int64_t ao, int64_t ab, int64_t needle,
                         int needle_nb, int pos, int64_t out, int64_t n,
                         int64_t stream) {
    hipdf_str_locate(P(ao), P(ab), P(needle), needle_nb, pos, PM(out), n,
                     S(stream));
  });
  m.def("str_to_dec", [](int64_t ao, int64_t ab, int64_t av, int out_kind,
                         int out_scale, int out_prec, int64_t out,
                         int64_t ov, int64_t n, int64_t stream) {
    hipdf_str_to_dec(P(ao), P(ab), P(av), out_kind, out_scale, out_prec,
                     PM(out), PM(ov), n, S(stream));
  });
  m.def("dec_to_str", [](int64_t vals, int in_is_128, int scale,
                         int64_t out_off, int64_t out_len, int64_t out,
                         int mode, int64_t n, int64_t stream) {
    hipdf_dec_to_str(P(vals), in_is_128, scale, P(out_off), PM(out_len),
                     PM(out), mode, n, S(stream));
  });
  m.def("gb_hll", [](int64_t hashes, int64_t valid, int64_t row_gid,
                     int64_t sel, int64_t regs, int p, int64_t n,
                     int64_t stream) {
    hipdf_gb_hll(P(hashes), P(valid), P(row_gid), P(sel), PM(regs), p, n,
                 S(stream));
  });
  m.def("xxhash64_col", [](int kind, int t, int64_t a, int64_t av,
                           int64_t sel, int64_t seeds, int64_t n,
                           int64_t stream) {
    hipdf_xxhash64_col(kind, t, P(a), P(av), P(sel), PM(seeds), n,
                       S(stream));
  });
  m.def("xxhash64_str", [](int64_t offsets, int64_t bytes, int64_t av,
                           int64_t sel, int64_t seeds, int64_t n,
                           int64_t stream) {
    hipdf_xxhash64_str(P(offsets), P(bytes), P(av), P(sel), PM(seeds), n,
                       S(stream));
  });
  m.def("pool_init", [](double fraction, size_t bytes) {
    return hipdf_pool_init(fraction, bytes);
  });
  m.def("pool_active", []() { return hipdf_pool_active() != 0; });
  m.def("pool_used", []() { return hipdf_pool_used(); });
  m.def("pool_reserved", []() { return hipdf_pool_reserved(); });
  m.def("pool_high_watermark", []() { return hipdf_pool_high_watermark(); });
  m.def("pool_overflow", []() { return hipdf_pool_overflow(); });
  m.def("pool_selftest", []() { return hipdf_pool_selftest(); });
  m.def("pool_set_spill_cb", [](py::object f) {
    if (!g_spill_cb) g_spill_cb = new py::object();
    *g_spill_cb = f;
    hipdf_pool_set_failure_cb(f.is_none() ? nullptr : &spill_cb_trampoline);
  });
  m.def("host_register", [](int64_t ptr, int64_t nbytes) -> int {
    // pin an existing host range (the parquet file mmap) so H2D copies
    // from it are direct DMA instead of a staged bounce
    return (int)hipHostRegister((void*)ptr, (size_t)nbytes,
                                hipHostRegisterDefault);
  });
  m.def("host_unregister", [](int64_t ptr) {
    hipHostUnregister((void*)ptr);
  });
  m.def("memcpy_h2d", [](int64_t dst, int64_t src, int64_t nbytes,
                         int64_t stream) -> int {
    return (int)hipMemcpyAsync((void*)dst, (const void*)src,
                               (size_t)nbytes, hipMemcpyHostToDevice,
                               S(stream));
  });
  m.def("rle_walk_host",
        [](int64_t data, int64_t nbytes, int bw, int64_t n_values,
           int64_t src_base, int64_t out_base, int64_t runs,
           int64_t max_runs) -> int64_t {
          return rle_walk((const uint8_t*)data, nbytes, bw, n_values,
                          src_base, out_base, (int64_t*)runs, max_runs);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("rle_expand", [](int64_t base, int64_t runs, int64_t nruns, int bw,
                         int64_t out, int64_t n, int64_t stream) {
    hipdf_rle_expand(P(base), P(runs), nruns, bw, PM(out), n, S(stream));
  });
  m.def("pq_delta_walk_host",
        [](int64_t data, int64_t nbytes, int64_t out,
           int64_t n) -> int64_t {
          return pq_delta_walk((const uint8_t*)data, nbytes, (int64_t*)out,
                               n);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("delta_ba_concat_host",
        [](int64_t pre, int64_t suf, int64_t sufbytes, int64_t suf_nbytes,
           int64_t n, int64_t out, int64_t out_offs) -> int64_t {
          return delta_ba_concat((const int64_t*)pre, (const int64_t*)suf,
                                 (const uint8_t*)sufbytes, suf_nbytes, n,
                                 (uint8_t*)out, (const int64_t*)out_offs);
        },
        py::call_guard<py::gil_scoped_release>());
  m.def("byte_array_offsets_host",
        [](int64_t data, int64_t nbytes, int64_t count, int64_t starts,
           int64_t lens) -> int64_t {
          return byte_array_offsets_walk((const uint8_t*)data, nbytes, count,
                                         (int32_t*)starts, (int64_t*)lens);
        },
        py::call_guard<py::gil_scoped_release>());
  m.doc() = "hand-written CDNA4 (gfx950) columnar kernels for MI355X";

  m.def("build_arch", []() { return std::string("gfx950"); });
  m.def("device_count", []() {
    int n = 0;
    hipGetDeviceCount(&n);
    return n;
  });
  m.def("synchronize", []() {
    hipError_t e = hipDeviceSynchronize();
    if (e != hipSuccess)
      throw std::runtime_error(hipGetErrorString(e));
  });

  m.def("binary_arith",
        [](int op, int t, int64_t a, int64_t b, double sd, int64_t si,
           bool scalar_rhs, int64_t av, int64_t bv, int64_t out, int64_t ov,
           int64_t n, int64_t stream) {
          hipdf_binary_arith(op, t, P(a), P(b), sd, si, scalar_rhs, P(av),
                             P(bv), PM(out), PM(ov), n, S(stream));
          check_async();
        });
  m.def("binary_cmp",
        [](int op, int t, int64_t a, int64_t b, double sd, int64_t si,
           bool scalar_rhs, int64_t av, int64_t bv, int64_t out, int64_t ov,
           int64_t n, int64_t stream) {
          hipdf_binary_cmp(op, t, P(a), P(b), sd, si, scalar_rhs, P(av), P(bv),
                           PM(out), PM(ov), n, S(stream));
          check_async();
        });
  m.def("binary_bool",
        [](int op, int64_t a, int64_t b, int sb, bool scalar_rhs, int64_t av,
           int64_t bv, int64_t out, int64_t ov, int64_t n, int64_t stream) {
          hipdf_binary_bool(op, P(a), P(b), sb, scalar_rhs, P(av), P(bv),
                            PM(out), PM(ov), n, S(stream));
          check_async();
        });
  m.def("unary", [](int op, int t, int64_t a, int64_t av, int64_t out,
                    int64_t ov, int64_t n, int64_t stream) {
    hipdf_unary(op, t, P(a), P(av), PM(out), PM(ov), n, S(stream));
    check_async();
  });
  m.def("cast", [](int ft, int tt, int64_t a, int64_t out, int64_t n,
                   int64_t stream) {
    hipdf_cast(ft, tt, P(a), PM(out), n, S(stream));
    check_async();
  });
  m.def("f64_to_i64_rint", [](int64_t a, int64_t out, int64_t n,
                              int64_t stream) {
    hipdf_f64_to_i64_rint(P(a), PM(out), n, S(stream));
    check_async();
  });
  m.def("decimal_rescale", [](int64_t a, int64_t out, int64_t pow10, bool up,
                              int64_t n, int64_t stream) {
    hipdf_decimal_rescale(P(a), PM(out), pow10, up, n, S(stream));
    check_async();
  });

  m.def("if_else", [](int t, int64_t cond, int64_t cv, int64_t a, int64_t av,
                      int64_t b, int64_t bv, int64_t out, int64_t ov,
                      int64_t n, int64_t stream) {
    hipdf_if_else(t, P(cond), P(cv), P(a), P(av), P(b), P(bv), PM(out),
                  PM(ov), n, S(stream));
    check_async();
  });
  m.def("mask_expand", [](int64_t mask, int64_t out, bool invert, int64_t n,
                          int64_t stream) {
    hipdf_mask_expand(P(mask), PM(out), invert, n, S(stream));
    check_async();
  });

  m.def("sel_num_blocks", &sel_num_blocks);
  m.def("mask_count", [](int64_t mask, int64_t mv, int64_t counts, int64_t n,
                         int64_t stream) {
    hipdf_mask_count(P(mask), P(mv), PM(counts), n, S(stream));
    check_async();
  });
  m.def("mask_scatter", [](int64_t mask, int64_t mv, int64_t offsets,
                           int64_t out_idx, int64_t n, int64_t stream) {
    hipdf_mask_scatter(P(mask), P(mv), P(offsets), PM(out_idx), n, S(stream));
    check_async();
  });
  m.def("gather_fixed", [](int esize, int64_t in, int64_t idx, int64_t out,
                           int64_t n_out, int64_t stream) {
    hipdf_gather_fixed(esize, P(in), P(idx), PM(out), n_out, S(stream));
    check_async();
  });
  m.def("gather_table", [](int64_t cols, int ncols, int64_t idx,
                           int64_t n_out, int64_t stream) {
    hipdf_gather_table(P(cols), ncols, P(idx), n_out, S(stream));
    check_async();
  });
  m.def("gather_validity", [](int64_t in_valid, bool in_has, int64_t idx,
                              int64_t out_valid, int64_t n_out,
                              int64_t stream) {
    hipdf_gather_validity(P(in_valid), in_has, P(idx), PM(out_valid), n_out,
                          S(stream));
    check_async();
  });
  m.def("gather_str_lens", [](int64_t offs, int64_t idx, int64_t lens,
                              int64_t n_out, int64_t stream) {
    hipdf_gather_str_lens(P(offs), P(idx), PM(lens), n_out, S(stream));
    check_async();
  });
  m.def("gather_str_bytes", [](int64_t in_bytes, int64_t in_offs, int64_t idx,
                               int64_t out_offs, int64_t out_bytes,
                               int64_t n_out, int64_t total_bytes,
                               int64_t stream) {
    hipdf_gather_str_bytes(P(in_bytes), P(in_offs), P(idx), P(out_offs),
                           PM(out_bytes), n_out, total_bytes, S(stream));
    check_async();
  });
  m.def("narrow_i64_i32", [](int64_t in, int64_t out, int64_t n,
                             int64_t stream) {
    hipdf_narrow_i64_i32(P(in), PM(out), n, S(stream));
    check_async();
  });
  m.def("copy_valid_range", [](int64_t src, bool src_has, int64_t dst_off,
                               int64_t dst, int64_t n, int64_t stream) {
    hipdf_copy_valid_range(P(src), src_has, dst_off, PM(dst), n, S(stream));
    check_async();
  });

  m.def("scan_num_blocks", &scan_num_blocks);
  m.def("scan_block", [](int64_t in, int64_t out, int64_t sums, int64_t n,
                         int64_t stream) {
    hipdf_scan_block(P(in), PM(out), PM(sums), n, S(stream));
    check_async();
  });
  m.def("scan_add_offsets", [](int64_t out, int64_t sums, int64_t n,
                               int64_t stream) {
    hipdf_scan_add_offsets(PM(out), P(sums), n, S(stream));
    check_async();
  });
  m.def("reduce", [](int op, int t, int64_t a, int64_t av, int64_t acc,
                     int64_t cnt, int64_t n, int64_t stream) {
    hipdf_reduce(op, t, P(a), P(av), PM(acc), PM(cnt), n, S(stream));
    check_async();
  });

  m.def("murmur3_col", [](int kind, int t, int64_t a, int64_t av,
                          int64_t sel, int64_t seeds, int64_t n,
                          int64_t stream) {
    hipdf_murmur3_col(kind, t, P(a), P(av), P(sel), PM(seeds), n, S(stream));
    check_async();
  });
  m.def("murmur3_str", [](int64_t offs, int64_t bytes, int64_t av,
                          int64_t sel, int64_t seeds, int64_t n,
                          int64_t stream) {
    hipdf_murmur3_str(P(offs), P(bytes), P(av), P(sel), PM(seeds), n,
                      S(stream));
    check_async();
  });
  m.def("pmod_part", [](int64_t h, int nparts, int64_t part, int64_t n,
                        int64_t stream) {
    hipdf_pmod_part(P(h), nparts, PM(part), n, S(stream));
    check_async();
  });

  m.def("gb_build", [](int64_t hashes, int64_t keys, int nkeys, int64_t sel,
                       int64_t slot_row, int64_t row_slot,
                       int64_t claimed_slots, int64_t ngroups, int64_t cap,
                       int64_t n, int64_t stream) {
    hipdf_gb_build(P(hashes), P(keys), nkeys, P(sel), PM(slot_row),
                   PM(row_slot), PM(claimed_slots), PM(ngroups), cap, n,
                   S(stream));
    check_async();
  });
  m.def("gb_number", [](int64_t claimed_slots, int64_t slot_row,
                        int64_t slot_gid, int64_t ngroups, int64_t leaders,
                        int64_t max_groups, int64_t stream) {
    hipdf_gb_number(P(claimed_slots), P(slot_row), PM(slot_gid), P(ngroups),
                    PM(leaders), max_groups, S(stream));
    check_async();
  });
  m.def("gb_rowgid", [](int64_t row_slot, int64_t slot_gid, int64_t row_gid,
                        int64_t n, int64_t stream) {
    hipdf_gb_rowgid(P(row_slot), P(slot_gid), PM(row_gid), n, S(stream));
    check_async();
  });
  m.def("gb_agg", [](int op, int t, int64_t vals, int64_t vvalid,
                     int64_t row_gid, int64_t acc, int64_t cnt,
                     bool acc_is_double, int ngroups, int64_t n,
                     int64_t stream) {
    hipdf_gb_agg(op, t, P(vals), P(vvalid), P(row_gid), PM(acc), PM(cnt),
                 acc_is_double, ngroups, n, S(stream));
    check_async();
  });
  m.def("gb_acc_init", [](int op, int64_t acc, bool is_double, int ngroups,
                          int64_t stream) {
    hipdf_gb_acc_init(op, PM(acc), is_double, ngroups, S(stream));
    check_async();
  });
  m.def("gb_agg_multi", [](int64_t aggs, int naggs, int64_t row_gid,
                           int64_t sel, int ngroups, int nrep, int64_t n,
                           int64_t stream) {
    hipdf_gb_agg_multi(P(aggs), naggs, P(row_gid), P(sel), ngroups, nrep, n,
                       S(stream));
    check_async();
  });
  m.def("gb_reduce_reps", [](int op, int64_t acc, int acc_is_double,
                             int64_t cnt, int ngroups, int nrep,
                             int64_t stream) {
    hipdf_gb_reduce_reps(op, PM(acc), acc_is_double, PM(cnt), ngroups, nrep,
                         S(stream));
    check_async();
  });
  m.def("mask_from_nonzero", [](int64_t cnt, int64_t mask, int64_t n,
                                int64_t stream) {
    hipdf_mask_from_nonzero(P(cnt), PM(mask), n, S(stream));
    check_async();
  });

  m.def("join_build", [](int64_t hashes, int64_t keys, int nkeys,
                         int64_t head, int64_t next, int64_t cap, int64_t n,
                         int64_t stream) {
    hipdf_join_build(P(hashes), P(keys), nkeys, PM(head), PM(next), cap, n,
                     S(stream));
    check_async();
  });
  m.def("join_count", [](int how, int64_t lh, int64_t lk, int64_t rk,
                         int nkeys, int64_t head, int64_t next, int64_t cap,
                         int64_t counts, int64_t n, int64_t stream) {
    hipdf_join_count(how, P(lh), P(lk), P(rk), nkeys, P(head), P(next), cap,
                     PM(counts), n, S(stream));
    check_async();
  });
  m.def("join_fill", [](int how, int64_t lh, int64_t lk, int64_t rk,
                        int nkeys, int64_t head, int64_t next, int64_t cap,
                        int64_t offsets, int64_t lmap, int64_t rmap,
                        int64_t right_matched, int64_t n, int64_t stream) {
    hipdf_join_fill(how, P(lh), P(lk), P(rk), nkeys, P(head), P(next), cap,
                    P(offsets), PM(lmap), PM(rmap), PM(right_matched), n,
                    S(stream));
    check_async();
  });

  m.def("win_minmax", [](int is_double, int64_t level_ptrs, int nlevels,
                         int64_t a_idx, int64_t b_idx, int is_min,
                         int64_t out, int64_t n, int64_t stream) {
    hipdf_win_minmax(is_double, P(level_ptrs), nlevels, P(a_idx), P(b_idx),
                     is_min, PM(out), n, S(stream));
    check_async();
  });
  m.def("range_bounds", [](int64_t vals, int64_t seg_start, int64_t seg_end,
                           double lo, double hi, int lo_unb, int hi_unb,
                           int64_t a_idx, int64_t b_idx, int64_t n,
                           int64_t stream) {
    hipdf_range_bounds(P(vals), P(seg_start), P(seg_end), lo, hi, lo_unb,
                       hi_unb, PM(a_idx), PM(b_idx), n, S(stream));
    check_async();
  });
  m.def("change_flags", [](int64_t keys, int nkeys, int64_t out, int64_t n,
                           int64_t stream) {
    hipdf_change_flags(P(keys), nkeys, PM(out), n, S(stream));
    check_async();
  });
  m.def("iota_i32", [](int64_t out, int64_t n, int64_t stream) {
    hipdf_iota_i32(PM(out), n, S(stream));
    check_async();
  });
  m.def("scan_block_f64", [](int64_t in, int64_t out, int64_t sums, int64_t n,
                             int64_t stream) {
    hipdf_scan_block_f64(P(in), PM(out), PM(sums), n, S(stream));
    check_async();
  });
  m.def("scan_add_offsets_f64", [](int64_t out, int64_t sums, int64_t n,
                                   int64_t stream) {
    hipdf_scan_add_offsets_f64(PM(out), P(sums), n, S(stream));
    check_async();
  });
  m.def("pq_delta_i64", [](int64_t b, int64_t nbytes, int64_t n,
                           int64_t out, int64_t stream) {
    hipdf_pq_delta_i64(P(b), nbytes, n, PM(out), S(stream));
    check_async();
  });
  m.def("orc_bool_rle", [](int64_t b, int64_t nbytes, int64_t n,
                           int64_t out, int64_t stream) {
    hipdf_orc_bool_rle(P(b), nbytes, n, PM(out), S(stream));
    check_async();
  });
  m.def("orc_rle_v2", [](int64_t b, int64_t nbytes, int64_t n, int is_signed,
                         int64_t out, int64_t stream) {
    hipdf_orc_rle_v2(P(b), nbytes, n, is_signed, PM(out), S(stream));
    check_async();
  });
  m.def("json_field", [](int64_t bytes, int64_t row_start, int64_t row_end,
                         int64_t name, int name_len, int type,
                         int64_t out_i64, int64_t out_f64, int64_t out_ss,
                         int64_t out_sl, int64_t valid, int64_t unsupported,
                         int64_t nrows, int64_t stream) {
    hipdf_json_field(P(bytes), P(row_start), P(row_end), P(name), name_len,
                     type, PM(out_i64), PM(out_f64), PM(out_ss), PM(out_sl),
                     PM(valid), PM(unsupported), nrows, S(stream));
    check_async();
  });
  m.def("byte_eq", [](int64_t bytes, int target, int64_t out, int64_t n,
                      int64_t stream) {
    hipdf_byte_eq(P(bytes), target, PM(out), n, S(stream));
    check_async();
  });
  m.def("csv_parse", [](int64_t bytes, int64_t row_start, int64_t row_end,
                        int delim, int field_idx, int type, int64_t out_i64,
                        int64_t out_f64, int64_t out_ss, int64_t out_sl,
                        int64_t valid, int64_t unsupported, int64_t nrows,
                        int64_t stream) {
    hipdf_csv_parse(P(bytes), P(row_start), P(row_end), delim, field_idx,
                    type, PM(out_i64), PM(out_f64), PM(out_ss), PM(out_sl),
                    PM(valid), PM(unsupported), nrows, S(stream));
    check_async();
  });
  m.def("str_plain_encode", [](int64_t offsets, int64_t bytes,
                               int64_t out_off, int64_t out_len, int64_t out,
                               int mode, int64_t n, int64_t stream) {
    hipdf_str_plain_encode(P(offsets), P(bytes), P(out_off), PM(out_len),
                           PM(out), mode, n, S(stream));
    check_async();
  });
  m.def("str_plain_offsets", [](int64_t data, int64_t nbytes,
                                int64_t n_values, int64_t starts,
                                int64_t lens, int64_t error, int64_t stream) {
    hipdf_str_plain_offsets(P(data), nbytes, n_values, PM(starts), PM(lens),
                            PM(error), S(stream));
    check_async();
  });
  m.def("rle_hybrid_batch", [](int64_t base, int64_t descs, int nstreams,
                               int64_t out, int64_t stream) {
    hipdf_rle_hybrid_batch(P(base), P(descs), nstreams, PM(out), S(stream));
  });
  m.def("rle_hybrid_decode", [](int64_t data, int64_t nbytes, int bw,
                                int64_t out, int64_t n, int64_t stream) {
    hipdf_rle_hybrid_decode(P(data), nbytes, bw, PM(out), n, S(stream));
    check_async();
  });
  m.def("scatter_fixed", [](int esize, int64_t vals, int64_t idx, int64_t out,
                            int64_t n, int64_t stream) {
    hipdf_scatter_fixed(esize, P(vals), P(idx), PM(out), n, S(stream));
    check_async();
  });
  m.def("levels_to_mask", [](int64_t levels, int max_level, int64_t mask,
                             int64_t n, int64_t stream) {
    hipdf_levels_to_mask(P(levels), max_level, PM(mask), n, S(stream));
    check_async();
  });

  m.def("regex_extract", [](int64_t prog, int nops, int64_t classes,
                            int64_t offsets, int64_t bytes, int group,
                            int64_t out_start, int64_t out_len,
                            int64_t overflow, int64_t n, int64_t stream) {
    hipdf_regex_extract(P(prog), nops, P(classes), P(offsets), P(bytes),
                        group, PM(out_start), PM(out_len), PM(overflow), n,
                        S(stream));
    check_async();
  });
  m.def("regex_extract_all", [](int64_t prog, int nops, int64_t classes,
                                int64_t offsets, int64_t bytes, int group,
                                int64_t part_off, int64_t counts,
                                int64_t out_ss, int64_t out_sl, int mode,
                                int64_t overflow, int64_t n,
                                int64_t stream) {
    hipdf_regex_extract_all(P(prog), nops, P(classes), P(offsets), P(bytes),
                            group, P(part_off), PM(counts), PM(out_ss),
                            PM(out_sl), mode, PM(overflow), n, S(stream));
    check_async();
  });
  m.def("regex_replace", [](int64_t prog, int nops, int64_t classes,
                            int64_t offsets, int64_t bytes, int64_t repl_ops,
                            int nrepl, int64_t lit, int64_t out_off,
                            int64_t out_len, int64_t out_bytes, int mode,
                            int64_t overflow, int64_t n, int64_t stream) {
    hipdf_regex_replace(P(prog), nops, P(classes), P(offsets), P(bytes),
                        P(repl_ops), nrepl, P(lit), P(out_off), PM(out_len),
                        PM(out_bytes), mode, PM(overflow), n, S(stream));
    check_async();
  });
  m.def("regex_match", [](int64_t prog, int nops, int64_t classes,
                          int64_t offsets, int64_t bytes, int64_t out,
                          int64_t overflow, int64_t n, int64_t stream) {
    hipdf_regex_match(P(prog), nops, P(classes), P(offsets), P(bytes),
                      PM(out), PM(overflow), n, S(stream));
    check_async();
  });
  m.def("str_cmp", [](int op, int64_t ao, int64_t ab, int64_t bo, int64_t bb,
                      int64_t out, int64_t n, int64_t stream) {
    hipdf_str_cmp(op, P(ao), P(ab), P(bo), P(bb), PM(out), n, S(stream));
    check_async();
  });
  m.def("str_cmp_scalar", [](int op, int64_t ao, int64_t ab, int64_t pat,
                             int plen, int64_t out, int64_t n, int64_t stream) {
    hipdf_str_cmp_scalar(op, P(ao), P(ab), P(pat), plen, PM(out), n, S(stream));
    check_async();
  });
  m.def("str_find", [](int mode, int64_t ao, int64_t ab, int64_t pat, int plen,
                       int64_t out, int64_t n, int64_t stream) {
    hipdf_str_find(mode, P(ao), P(ab), P(pat), plen, PM(out), n, S(stream));
    check_async();
  });
  m.def("str_like", [](int64_t ao, int64_t ab, int64_t pat, int plen,
                       int64_t out, int64_t n, int64_t stream) {
    hipdf_str_like(P(ao), P(ab), P(pat), plen, PM(out), n, S(stream));
    check_async();
  });
  m.def("str_length", [](int64_t ao, int64_t ab, int64_t out, int64_t n,
                         int64_t stream) {
    hipdf_str_length(P(ao), P(ab), PM(out), n, S(stream));
    check_async();
  });
  m.def("str_case", [](bool upper, int64_t in, int64_t out, int64_t nbytes,
                       int64_t stream) {
    hipdf_str_case(upper, P(in), PM(out), nbytes, S(stream));
    check_async();
  });
  m.def("i64_to_str", [](int64_t vals, int64_t out_off, int64_t out_len,
                         int64_t out, int mode, int64_t n, int64_t stream) {
    hipdf_i64_to_str(P(vals), P(out_off), PM(out_len), PM(out), mode, n,
                     S(stream));
    check_async();
  });
  m.def("str_concat_ws", [](int64_t cols, int ncols, int64_t sep,
                            int seplen, int64_t out_off, int64_t out_len,
                            int64_t out, int mode, int64_t n,
                            int64_t stream) {
    hipdf_str_concat_ws(P(cols), ncols, P(sep), seplen, P(out_off),
                        PM(out_len), PM(out), mode, n, S(stream));
    check_async();
  });
  m.def("str_initcap", [](int64_t ao, int64_t ab, int64_t out, int64_t n,
                          int64_t stream) {
    hipdf_str_initcap(P(ao), P(ab), PM(out), n, S(stream));
    check_async();
  });
  m.def("str_reverse", [](int64_t ao, int64_t ab, int64_t out, int64_t n,
                          int64_t stream) {
    hipdf_str_reverse(P(ao), P(ab), PM(out), n, S(stream));
    check_async();
  });
  m.def("str_split_count", [](int64_t ao, int64_t ab, int64_t delim,
                              int dlen, int64_t counts, int64_t n,
                              int64_t stream) {
    hipdf_str_split_count(P(ao), P(ab), P(delim), dlen, PM(counts), n,
                          S(stream));
    check_async();
  });
  m.def("str_split_fill", [](int64_t ao, int64_t ab, int64_t delim,
                             int dlen, int64_t part_off, int64_t counts,
                             int64_t out_ss, int64_t out_sl, int64_t n,
                             int64_t stream) {
    hipdf_str_split_fill(P(ao), P(ab), P(delim), dlen, P(part_off),
                         P(counts), PM(out_ss), PM(out_sl), n, S(stream));
    check_async();
  });
  m.def("str_trim_ranges", [](int mode, int64_t ao, int64_t ab,
                              int64_t bstart, int64_t blen, int64_t n,
                              int64_t stream) {
    hipdf_str_trim_ranges(mode, P(ao), P(ab), PM(bstart), PM(blen), n,
                          S(stream));
    check_async();
  });
  m.def("str_concat2", [](int64_t ao, int64_t ab, int64_t bo, int64_t bb,
                          int64_t out_off, int64_t out_len,
                          int64_t out_bytes, int mode, int64_t n,
                          int64_t stream) {
    hipdf_str_concat2(P(ao), P(ab), P(bo), P(bb), P(out_off), PM(out_len),
                      PM(out_bytes), mode, n, S(stream));
    check_async();
  });
  m.def("substr_ranges", [](int64_t ao, int64_t ab, int start, int slen,
                            int64_t bstart, int64_t blen, int64_t n,
                            int64_t stream) {
    hipdf_substr_ranges(P(ao), P(ab), start, slen, PM(bstart), PM(blen), n,
                        S(stream));
    check_async();
  });
  m.def("substr_copy", [](int64_t ab, int64_t bstart, int64_t blen,
                          int64_t out_off, int64_t out_bytes, int64_t n,
                          int64_t stream) {
    hipdf_substr_copy(P(ab), P(bstart), P(blen), P(out_off), PM(out_bytes), n,
                      S(stream));
    check_async();
  });

  m.def("dec64_mul_div", [](int is_div, int64_t a, int64_t b, int64_t av,
                            int64_t bv, int64_t out, int64_t ov,
                            int out_is_128, int shift, int out_prec,
                            int64_t n, int64_t stream) {
    hipdf_dec64_mul_div(is_div, P(a), P(b), P(av), P(bv), PM(out), PM(ov),
                        out_is_128, shift, out_prec, n, S(stream));
    check_async();
  });
  m.def("i128_arith", [](int op, int64_t a, int64_t b, int64_t av, int64_t bv,
                         int64_t out, int64_t ov, int64_t n, int64_t stream) {
    hipdf_i128_arith(op, P(a), P(b), P(av), P(bv), PM(out), PM(ov), n,
                     S(stream));
    check_async();
  });
  m.def("i128_cmp", [](int op, int64_t a, int64_t b, int64_t av, int64_t bv,
                       int64_t out, int64_t ov, int64_t n, int64_t stream) {
    hipdf_i128_cmp(op, P(a), P(b), P(av), P(bv), PM(out), PM(ov), n,
                   S(stream));
    check_async();
  });
  m.def("dec_mul_div_wide", [](int is_div, int64_t a, int64_t b, int64_t av,
                               int64_t bv, int a128, int b128, int64_t out,
                               int64_t ov, int out128, int shift,
                               int out_prec, int64_t n, int64_t stream) {
    hipdf_dec_mul_div_wide(is_div, P(a), P(b), P(av), P(bv), a128, b128,
                           PM(out), PM(ov), out128, shift, out_prec, n,
                           S(stream));
  });
  m.def("i128_rescale", [](int64_t in, int64_t iv, int64_t out, int64_t ov,
                           int shift, int out_prec, int out_is_64, int64_t n,
                           int64_t stream) {
    hipdf_i128_rescale(P(in), P(iv), PM(out), PM(ov), shift, out_prec,
                       out_is_64, n, S(stream));
  });
  m.def("i64_to_i128", [](int64_t in, int64_t out, int64_t n, int64_t stream) {
    hipdf_i64_to_i128(P(in), PM(out), n, S(stream));
    check_async();
  });
  m.def("i128_to_f64", [](int64_t in, int64_t out, int64_t n, int64_t stream) {
    hipdf_i128_to_f64(P(in), PM(out), n, S(stream));
    check_async();
  });
  m.def("gb_percentile", [](int64_t vals, int64_t perm, int64_t starts,
                            int64_t vcnt, double p, int64_t out, int ngroups,
                            int64_t stream) {
    hipdf_gb_percentile(P(vals), P(perm), P(starts), P(vcnt), p, PM(out),
                        ngroups, S(stream));
    check_async();
  });
  m.def("dense_gid", [](int64_t keys, int nkeys, int64_t sel,
                        int64_t row_gid, int64_t n, int64_t stream) {
    hipdf_dense_gid(P(keys), nkeys, P(sel), PM(row_gid), n, S(stream));
    check_async();
  });
  m.def("expand_rows", [](int64_t offsets, int64_t rowid, int64_t pos,
                          int64_t nrows, int64_t stream) {
    hipdf_expand_rows(P(offsets), PM(rowid), PM(pos), nrows, S(stream));
    check_async();
  });
  m.def("gb_collect_count", [](int64_t vvalid, int64_t row_gid, int64_t sel,
                               int64_t counts, int64_t n, int64_t stream) {
    hipdf_gb_collect_count(P(vvalid), P(row_gid), P(sel), PM(counts), n,
                           S(stream));
    check_async();
  });
  m.def("gb_collect_fill", [](int esize, int64_t vals, int64_t vvalid,
                              int64_t row_gid, int64_t sel, int64_t offsets,
                              int64_t cursor, int64_t out, int64_t n,
                              int64_t stream) {
    hipdf_gb_collect_fill(esize, P(vals), P(vvalid), P(row_gid), P(sel),
                          P(offsets), PM(cursor), PM(out), n, S(stream));
    check_async();
  });
  m.def("gb_sum_i128_lds", [](int in_is_64, int64_t vals, int64_t vvalid,
                              int64_t row_gid, int64_t sel, int64_t acc,
                              int64_t cnt, int ngroups, int64_t n,
                              int64_t stream) {
    hipdf_gb_sum_i128_lds(in_is_64, P(vals), P(vvalid), P(row_gid), P(sel),
                          PM(acc), PM(cnt), ngroups, n, S(stream));
    check_async();
  });
  m.def("gb_sum_i64_to_i128", [](int64_t vals, int64_t vvalid, int64_t row_gid,
                                 int64_t sel, int64_t acc, int64_t cnt,
                                 int64_t n, int64_t stream) {
    hipdf_gb_sum_i64_to_i128(P(vals), P(vvalid), P(row_gid), P(sel), PM(acc),
                             PM(cnt), n, S(stream));
    check_async();
  });
  m.def("gb_sum_i128", [](int64_t vals, int64_t vvalid, int64_t row_gid,
                          int64_t sel, int64_t acc, int64_t cnt, int64_t n,
                          int64_t stream) {
    hipdf_gb_sum_i128(P(vals), P(vvalid), P(row_gid), P(sel), PM(acc),
                      PM(cnt), n, S(stream));
    check_async();
  });

  m.def("part_num_blocks", &part_num_blocks);
  m.def("sort_num_blocks", &sort_num_blocks);
  m.def("sort_key_width", &hipdf_sort_key_width);
  m.def("make_sort_keys_str", [](int64_t offsets, int64_t bytes,
                                 int64_t valid, int64_t perm, int desc,
                                 int chunk, int64_t keys, int64_t n,
                                 int64_t stream) {
    hipdf_make_sort_keys_str(P(offsets), P(bytes), P(valid), P(perm), desc,
                             chunk, PM(keys), n, S(stream));
    check_async();
  });
  m.def("make_sort_keys_i128", [](int64_t data, int64_t valid, int64_t perm,
                                  int desc, int word, int64_t keys,
                                  int64_t n, int64_t stream) {
    hipdf_make_sort_keys_i128(P(data), P(valid), P(perm), desc, word,
                              PM(keys), n, S(stream));
    check_async();
  });
  m.def("make_sort_keys", [](int t, int64_t data, int64_t valid, int64_t perm,
                             bool desc, bool nulls_last, bool null_only,
                             int64_t keys, int64_t n, int64_t stream) {
    hipdf_make_sort_keys(t, P(data), P(valid), P(perm), desc, nulls_last,
                         null_only, PM(keys), n, S(stream));
    check_async();
  });
  m.def("radix_count", [](int64_t keys, int shift, int64_t counts, int64_t n,
                          int64_t stream) {
    hipdf_radix_count(P(keys), shift, PM(counts), n, S(stream));
    check_async();
  });
  m.def("radix_scatter", [](int64_t keys_in, int64_t perm_in, int shift,
                            int64_t offsets, int64_t keys_out,
                            int64_t perm_out, int64_t n, int64_t stream) {
    hipdf_radix_scatter(P(keys_in), P(perm_in), shift, P(offsets),
                        PM(keys_out), PM(perm_out), n, S(stream));
    check_async();
  });
  m.def("part_hist", [](int64_t part, int nparts, int64_t counts, int64_t n,
                        int64_t stream) {
    hipdf_part_hist(P(part), nparts, PM(counts), n, S(stream));
    check_async();
  });
  m.def("part_scatter", [](int64_t part, int nparts, int64_t offsets,
                           int64_t perm, int64_t n, int64_t stream) {
    hipdf_part_scatter(P(part), nparts, P(offsets), PM(perm), n, S(stream));
    check_async();
  });
}
