// hipdf device memory pool: RMM-style sub-allocator on a hipMalloc slab,
// pluggable into PyTorch (torch.cuda.memory.CUDAPluggableAllocator), with
// an allocation-failure callback that lets the Python spill framework free
// memory BEFORE the allocation fails (reference analogue: RMM pool +
// DeviceMemoryEventHandler -> SpillFramework, GpuDeviceManager.scala:352,
// SpillFramework.scala:1363 — SURVEY.md §2.5).
//
// The arena logic is hardware-agnostic (address-ordered first-fit with
// immediate coalescing) so hipdf_pool_selftest() exercises it over plain
// host memory in the CPU test suite.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <map>
#include <mutex>

namespace {

struct Arena {
  uintptr_t base = 0;
  size_t capacity = 0;
  // free blocks by address -> size (address order enables O(log n)
  // neighbor coalescing); allocated blocks by address -> size
  std::map<uintptr_t, size_t> free_blocks;
  std::map<uintptr_t, size_t> used_blocks;
  size_t used = 0;
  size_t high_watermark = 0;

  static constexpr size_t ALIGN = 256;

  void init(uintptr_t b, size_t cap) {
    base = b;
    capacity = cap;
    free_blocks.clear();
    used_blocks.clear();
    free_blocks[b] = cap;
    used = 0;
    high_watermark = 0;
  }

  void* alloc(size_t n) {
    if (n == 0) n = ALIGN;
    n = (n + ALIGN - 1) & ~(ALIGN - 1);
    // first fit in address order: keeps the tail contiguous for the big
    // late allocations a columnar pipeline makes
    for (auto it = free_blocks.begin(); it != free_blocks.end(); ++it) {
      if (it->second >= n) {
        uintptr_t addr = it->first;
        size_t rest = it->second - n;
        free_blocks.erase(it);
        if (rest) free_blocks[addr + n] = rest;
        used_blocks[addr] = n;
        used += n;
        if (used > high_watermark) high_watermark = used;
        return (void*)addr;
      }
    }
    return nullptr;
  }

  // returns false if the pointer is not from this arena
  bool dealloc(void* p) {
    auto it = used_blocks.find((uintptr_t)p);
    if (it == used_blocks.end()) return false;
    uintptr_t addr = it->first;
    size_t n = it->second;
    used_blocks.erase(it);
    used -= n;
    // coalesce with next
    auto nxt = free_blocks.lower_bound(addr);
    if (nxt != free_blocks.end() && addr + n == nxt->first) {
      n += nxt->second;
      nxt = free_blocks.erase(nxt);
    }
    // coalesce with prev
    if (nxt != free_blocks.begin()) {
      auto prv = std::prev(nxt);
      if (prv->first + prv->second == addr) {
        prv->second += n;
        return true;
      }
    }
    free_blocks[addr] = n;
    return true;
  }

  bool owns(void* p) const {
    return (uintptr_t)p >= base && (uintptr_t)p < base + capacity;
  }
};

std::mutex g_mu;
Arena g_device;
void* g_slab = nullptr;
// overflow allocations served by raw hipMalloc when the slab is full and
// spilling could not help (tracked so free routes correctly)
std::map<void*, size_t> g_overflow;
size_t g_overflow_bytes = 0;
// failure callback: returns nonzero if it freed memory and the alloc
// should be retried (Python side: spill device->host)
typedef int (*hipdf_failure_cb)(size_t needed, int retry);
hipdf_failure_cb g_cb = nullptr;

}  // namespace

extern "C" {

// reserve `bytes` (or fraction of free memory if bytes==0) as the pool
// slab; returns 0 on success
int hipdf_pool_init(double fraction, size_t bytes) {
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_slab) return 0;  // idempotent
  if (bytes == 0) {
    size_t free_b = 0, total_b = 0;
    if (hipMemGetInfo(&free_b, &total_b) != hipSuccess) return 1;
    bytes = (size_t)((double)free_b * fraction);
    bytes &= ~((size_t)(1 << 21) - 1);  // trim to 2 MiB granularity
  }
  if (hipMalloc(&g_slab, bytes) != hipSuccess) return 2;
  g_device.init((uintptr_t)g_slab, bytes);
  return 0;
}

int hipdf_pool_active() { return g_slab != nullptr; }

void hipdf_pool_set_failure_cb(hipdf_failure_cb cb) { g_cb = cb; }

size_t hipdf_pool_used() {
  std::lock_guard<std::mutex> lk(g_mu);
  return g_device.used + g_overflow_bytes;
}

size_t hipdf_pool_reserved() {
  std::lock_guard<std::mutex> lk(g_mu);
  return g_device.capacity;
}

size_t hipdf_pool_high_watermark() {
  std::lock_guard<std::mutex> lk(g_mu);
  return g_device.high_watermark;
}

size_t hipdf_pool_overflow() {
  std::lock_guard<std::mutex> lk(g_mu);
  return g_overflow_bytes;
}

void* hipdf_pool_alloc(size_t n) {
  {
    std::lock_guard<std::mutex> lk(g_mu);
    if (g_slab) {
      void* p = g_device.alloc(n);
      if (p) return p;
    }
  }
  // slab exhausted: let the spill framework make room, then retry
  for (int retry = 0; g_cb && retry < 8; ++retry) {
    if (!g_cb(n, retry)) break;
    std::lock_guard<std::mutex> lk(g_mu);
    void* p = g_device.alloc(n);
    if (p) return p;
  }
  // last resort: raw hipMalloc outside the slab (keeps the engine alive
  // when the pool fraction was conservative); nullptr propagates as OOM
  void* raw = nullptr;
  if (hipMalloc(&raw, n) != hipSuccess) return nullptr;
  std::lock_guard<std::mutex> lk(g_mu);
  g_overflow[raw] = n;
  g_overflow_bytes += n;
  return raw;
}

void hipdf_pool_free(void* p) {
  if (!p) return;
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_device.owns(p)) {
    g_device.dealloc(p);
    return;
  }
  auto it = g_overflow.find(p);
  if (it != g_overflow.end()) {
    g_overflow_bytes -= it->second;
    g_overflow.erase(it);
    hipFree(p);
  }
}

// ---- torch CUDAPluggableAllocator entry points ---------------------------
void* hipdf_torch_malloc(size_t size, int device, hipStream_t stream) {
  (void)device;
  (void)stream;
  return hipdf_pool_alloc(size);
}

void hipdf_torch_free(void* ptr, size_t size, int device,
                      hipStream_t stream) {
  (void)size;
  (void)device;
  (void)stream;
  hipdf_pool_free(ptr);
}

// ---- host-memory selftest of the arena logic (runs in CPU test suite) ---
// returns 0 on success, a nonzero step id on the first failed invariant
int hipdf_pool_selftest() {
  const size_t CAP = 1 << 20;
  void* mem = malloc(CAP);
  Arena a;
  a.init((uintptr_t)mem, CAP);
  void* p1 = a.alloc(1000);
  void* p2 = a.alloc(5000);
  void* p3 = a.alloc(100000);
  if (!p1 || !p2 || !p3) { free(mem); return 1; }
  if (a.used != 1024 + 5120 + 100096) { free(mem); return 2; }
  if (!a.dealloc(p2)) { free(mem); return 3; }
  // reuse of the freed gap
  void* p4 = a.alloc(4000);
  if (p4 != p2) { free(mem); return 4; }
  a.dealloc(p1);
  a.dealloc(p4);
  a.dealloc(p3);
  if (a.used != 0) { free(mem); return 5; }
  // full coalescing: the arena must be one block again
  if (a.free_blocks.size() != 1 ||
      a.free_blocks.begin()->second != CAP) { free(mem); return 6; }
  // exhaustion returns nullptr, does not corrupt
  void* big = a.alloc(CAP + 1);
  if (big != nullptr) { free(mem); return 7; }
  // fill completely with mixed sizes then free in random-ish order
  void* ps[64];
  for (int i = 0; i < 64; ++i) {
    ps[i] = a.alloc(1024 * (1 + (i * 7) % 13));
    if (!ps[i]) { free(mem); return 8; }
  }
  for (int i = 0; i < 64; i += 2) a.dealloc(ps[i]);
  for (int i = 1; i < 64; i += 2) a.dealloc(ps[i]);
  if (a.used != 0 || a.free_blocks.size() != 1) { free(mem); return 9; }
  if (a.high_watermark == 0) { free(mem); return 10; }
  free(mem);
  return 0;
}

}  // extern "C"
