// trtlab_amd — native memory layer for MI355X (reference: trtlab/memory +
// trtlab/cuda allocators, redesigned for HIP/HBM3E).
//
//   - raw device / pinned-host allocation (hipMalloc / hipHostMalloc),
//     256-B min alignment on HBM3E (reference device_memory.h:36-47)
//   - BlockPool: fixed-size block pool with O(1) acquire/release free list
//     (reference memory_pool.h:65)
//   - byte accounting per memory type (reference tracking.h:183)
#include <sys/mman.h>

#include <cstring>

#include "runtime.h"

namespace trtlab {

static std::atomic<int64_t> g_device_bytes{0};
static std::atomic<int64_t> g_pinned_bytes{0};

void* device_malloc(size_t bytes, int device) {
  TRT_HIP_CHECK(hipSetDevice(device));
  void* p = nullptr;
  TRT_HIP_CHECK(hipMalloc(&p, bytes));
  g_device_bytes += (int64_t)bytes;
  return p;
}
void device_free(void* p, size_t bytes) {
  if (!p) return;
  TRT_HIP_CHECK(hipFree(p));
  g_device_bytes -= (int64_t)bytes;
}
void* pinned_malloc(size_t bytes) {
  void* p = nullptr;
  TRT_HIP_CHECK(hipHostMalloc(&p, bytes, hipHostMallocDefault));
  g_pinned_bytes += (int64_t)bytes;
  return p;
}
void pinned_free(void* p, size_t bytes) {
  if (!p) return;
  TRT_HIP_CHECK(hipHostFree(p));
  g_pinned_bytes -= (int64_t)bytes;
}
int64_t device_bytes_in_use() { return g_device_bytes.load(); }
int64_t pinned_bytes_in_use() { return g_pinned_bytes.load(); }

// ---- huge-page host allocator ----
// (reference trtlab/memory raw allocator set included a huge-page
// allocator; here: explicit 2 MiB hugetlb pages when the system pool has
// them, THP madvise fallback otherwise. The memset commits pages on the
// calling thread -> first-touch NUMA placement; allocate from an
// affinity-pinned thread for locality. pin=true hipHostRegisters the range
// so the GPU can DMA it like hipHostMalloc memory.)
static std::atomic<int64_t> g_huge_bytes{0};
static constexpr size_t kHugePageBytes = size_t(2) << 20;

void* huge_malloc(size_t bytes, bool pin, bool* hugetlb_out) {
  size_t sz = round_up(bytes, kHugePageBytes);
  void* p = mmap(nullptr, sz, PROT_READ | PROT_WRITE,
                 MAP_PRIVATE | MAP_ANONYMOUS | MAP_HUGETLB, -1, 0);
  bool hugetlb = (p != MAP_FAILED);
  if (!hugetlb) {
    p = mmap(nullptr, sz, PROT_READ | PROT_WRITE,
             MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
    if (p == MAP_FAILED)
      throw std::runtime_error("huge_malloc: mmap failed");
    madvise(p, sz, MADV_HUGEPAGE);
  }
  memset(p, 0, sz);  // first-touch commit
  if (pin) TRT_HIP_CHECK(hipHostRegister(p, sz, hipHostRegisterDefault));
  g_huge_bytes += (int64_t)sz;
  if (hugetlb_out) *hugetlb_out = hugetlb;
  return p;
}

void huge_free(void* p, size_t bytes, bool pin) {
  if (!p) return;
  size_t sz = round_up(bytes, kHugePageBytes);
  if (pin) (void)hipHostUnregister(p);
  munmap(p, sz);
  g_huge_bytes -= (int64_t)sz;
}

int64_t huge_bytes_in_use() { return g_huge_bytes.load(); }

BlockPool::BlockPool(size_t block_bytes, int count, int device)
    : block_bytes_(round_up(block_bytes, 256)), device_(device) {
  base_ = (char*)device_malloc(block_bytes_ * count, device);
  for (int i = count - 1; i >= 0; --i)
    free_.push_back(base_ + (size_t)i * block_bytes_);
  total_ = count;
}
BlockPool::~BlockPool() { device_free(base_, block_bytes_ * total_); }

void* BlockPool::acquire() {
  HybridLock g(mu_);
  if (free_.empty()) return nullptr;  // caller blocks at the Python Pool level
  void* p = free_.back();
  free_.pop_back();
  return p;
}
void BlockPool::release(void* p) {
  HybridLock g(mu_);
  free_.push_back((char*)p);
}
int BlockPool::available() {
  HybridLock g(mu_);
  return (int)free_.size();
}

}  // namespace trtlab
