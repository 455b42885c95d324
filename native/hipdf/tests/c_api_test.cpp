// Non-python consumer of the hipdf C ABI: links
// against libhipdf.so and runs filter -> gather -> hash group-by on
// device, verifying results on the host. Build:
//   hipcc -I native/hipdf/include c_api_test.cpp -L. -lhipdf -o c_api_test
#include <hipdf.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define CHECK(x)                                                        \
  do {                                                                  \
    if (!(x)) {                                                         \
      fprintf(stderr, "FAILED at %s:%d: %s\n", __FILE__, __LINE__, #x); \
      return 1;                                                         \
    }                                                                   \
  } while (0)

#define HIP_OK(x) CHECK((x) == hipSuccess)

template <typename T>
T* dev(const std::vector<T>& h) {
  void* p;
  if (hipMalloc(&p, h.size() * sizeof(T) + 8) != hipSuccess) return nullptr;
  hipMemcpy(p, h.data(), h.size() * sizeof(T), hipMemcpyHostToDevice);
  return (T*)p;
}

int main() {
  const int64_t N = 100000;
  std::vector<int64_t> vals(N);
  std::vector<int32_t> keys(N);
  std::vector<uint8_t> mask(N);
  for (int64_t i = 0; i < N; ++i) {
    vals[i] = i % 1000;
    keys[i] = (int32_t)(i % 37);
    mask[i] = (i % 3) != 0;  // keep 2/3 of rows
  }
  int64_t* d_vals = dev(vals);
  int32_t* d_keys = dev(keys);
  uint8_t* d_mask = dev(mask);
  CHECK(d_vals && d_keys && d_mask);
  hipStream_t s = 0;

  // ---- filter: mask -> selected indices --------------------------------
  int64_t nb = sel_num_blocks(N);
  std::vector<int64_t> h_counts(nb);
  int64_t* d_counts = dev(h_counts);
  hipdf_mask_count(d_mask, nullptr, d_counts, N, s);
  HIP_OK(hipMemcpy(h_counts.data(), d_counts, nb * 8,
                   hipMemcpyDeviceToHost));
  int64_t n_sel = 0;
  std::vector<int64_t> h_offs(nb);
  for (int64_t b = 0; b < nb; ++b) {
    h_offs[b] = n_sel;
    n_sel += h_counts[b];
  }
  int64_t exp_sel = 0;
  for (int64_t i = 0; i < N; ++i) exp_sel += mask[i] ? 1 : 0;
  CHECK(n_sel == exp_sel);
  int64_t* d_offs = dev(h_offs);
  void* d_idx;
  HIP_OK(hipMalloc(&d_idx, n_sel * 4));
  hipdf_mask_scatter(d_mask, nullptr, d_offs, d_idx, N, s);

  // ---- gather the selected rows ---------------------------------------
  void* d_gvals;
  void* d_gkeys;
  HIP_OK(hipMalloc(&d_gvals, n_sel * 8));
  HIP_OK(hipMalloc(&d_gkeys, n_sel * 4));
  hipdf_gather_fixed(8, d_vals, d_idx, d_gvals, n_sel, s);
  hipdf_gather_fixed(4, d_keys, d_idx, d_gkeys, n_sel, s);

  // ---- elementwise: doubled = gvals + gvals ----------------------------
  void* d_doubled;
  HIP_OK(hipMalloc(&d_doubled, n_sel * 8));
  hipdf_binary_arith(/*add*/ 0, HIPDF_I64, d_gvals, d_gvals, 0.0, 0, 0,
                     nullptr, nullptr, d_doubled, nullptr, n_sel, s);

  // ---- hash group-by: sum(doubled) by key ------------------------------
  std::vector<int32_t> seeds(n_sel, 42);
  int32_t* d_seeds = dev(seeds);
  hipdf_murmur3_col(/*int kind*/ 0, HIPDF_I32, d_gkeys, nullptr, nullptr,
                    d_seeds, n_sel, s);
  int64_t cap = 1;
  while (cap < 2 * n_sel) cap <<= 1;
  std::vector<int32_t> slot_init(cap, -1);
  int32_t* d_slot_row = dev(slot_init);
  void *d_row_slot, *d_claimed, *d_ngroups, *d_slot_gid, *d_leaders;
  HIP_OK(hipMalloc(&d_row_slot, n_sel * 4));
  HIP_OK(hipMalloc(&d_claimed, n_sel * 4));
  HIP_OK(hipMalloc(&d_ngroups, 4));
  HIP_OK(hipMemset(d_ngroups, 0, 4));
  HIP_OK(hipMalloc(&d_slot_gid, cap * 4));
  HIP_OK(hipMalloc(&d_leaders, n_sel * 4));
  HipdfKeyCol kc = {HIPDF_I32, 0, d_gkeys, nullptr, nullptr};
  HipdfKeyCol* d_kc;
  HIP_OK(hipMalloc((void**)&d_kc, sizeof(kc)));
  HIP_OK(hipMemcpy(d_kc, &kc, sizeof(kc), hipMemcpyHostToDevice));
  hipdf_gb_build(d_seeds, d_kc, 1, nullptr, d_slot_row, d_row_slot,
                 d_claimed, d_ngroups, cap, n_sel, s);
  int32_t ngroups = 0;
  HIP_OK(hipMemcpy(&ngroups, d_ngroups, 4, hipMemcpyDeviceToHost));
  CHECK(ngroups == 37);
  hipdf_gb_number(d_claimed, d_slot_row, d_slot_gid, d_ngroups, d_leaders,
                  n_sel, s);
  void* d_rowgid;
  HIP_OK(hipMalloc(&d_rowgid, n_sel * 4));
  hipdf_gb_rowgid(d_row_slot, d_slot_gid, d_rowgid, n_sel, s);
  void *d_acc, *d_cnt;
  HIP_OK(hipMalloc(&d_acc, ngroups * 8));
  HIP_OK(hipMalloc(&d_cnt, ngroups * 8));
  HIP_OK(hipMemset(d_cnt, 0, ngroups * 8));
  hipdf_gb_acc_init(/*sum*/ 0, d_acc, 0, ngroups, s);
  hipdf_gb_agg(/*sum*/ 0, HIPDF_I64, d_doubled, nullptr, d_rowgid, d_acc,
               d_cnt, 0, ngroups, n_sel, s);
  HIP_OK(hipDeviceSynchronize());

  // ---- verify against a host reference ---------------------------------
  std::vector<int32_t> h_leaders(ngroups);
  HIP_OK(hipMemcpy(h_leaders.data(), d_leaders, ngroups * 4,
                   hipMemcpyDeviceToHost));
  std::vector<int64_t> h_acc(ngroups);
  HIP_OK(hipMemcpy(h_acc.data(), d_acc, ngroups * 8,
                   hipMemcpyDeviceToHost));
  std::vector<int32_t> h_gkeys(n_sel);
  HIP_OK(hipMemcpy(h_gkeys.data(), d_gkeys, n_sel * 4,
                   hipMemcpyDeviceToHost));
  // host reference sums
  std::vector<int64_t> ref(37, 0);
  for (int64_t i = 0; i < N; ++i)
    if (mask[i]) ref[keys[i]] += 2 * vals[i];
  for (int32_t g = 0; g < ngroups; ++g) {
    int32_t leader_row = h_leaders[g];
    int32_t key = h_gkeys[leader_row];
    CHECK(key >= 0 && key < 37);
    if (h_acc[g] != ref[key]) {
      fprintf(stderr, "group key %d: got %lld want %lld\n", key,
              (long long)h_acc[g], (long long)ref[key]);
      return 1;
    }
  }
  printf("C_API_OK filter=%lld/%lld groups=%d\n", (long long)n_sel,
         (long long)N, ngroups);
  return 0;
}
