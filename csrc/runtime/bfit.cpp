// trtlab_amd — runtime general-purpose device allocator.
//
// The reference serves dynamic workloads through a best-fit allocator over
// growing block arenas (trtlab/memory/include/trtlab/memory/bfit_allocator.h:121,
// block_arena.h:177, growing_block_allocator block_allocators.h:186). This is
// the MI355X-native equivalent: hipMalloc'd slabs that grow on demand, a
// best-fit free list with address-ordered coalescing, and a log2 allocation-
// size histogram (reference histogram_tracker, trackers.h:37).
//
// Key reference insight kept: device memory cannot hold free-list nodes, so
// ALL bookkeeping lives host-side, out-of-band (the reference's
// block_list_oob, detail/block_list.h:87).
#include "runtime.h"

#include <map>
#include <set>

namespace trtlab {

DeviceArena::DeviceArena(int device, size_t initial_bytes, size_t max_bytes,
                         size_t growth_bytes)
    : device_(device),
      max_bytes_(max_bytes ? max_bytes : (size_t)-1),
      growth_bytes_(growth_bytes ? growth_bytes : (64ull << 20)) {
  if (initial_bytes) grow(initial_bytes);
}

DeviceArena::~DeviceArena() {
  for (auto& s : slabs_) device_free(s.base, s.bytes);
}

void DeviceArena::grow(size_t need) {
  // grow by at least growth_bytes_ (fewer, larger slabs: HBM3E is 288 GB —
  // fragmentation from tiny slabs costs more than headroom)
  size_t want = need > growth_bytes_ ? need : growth_bytes_;
  if (capacity_ + want > max_bytes_) {
    want = max_bytes_ - capacity_;
    if (want < need) throw std::bad_alloc();
  }
  void* base = device_malloc(want, device_);
  slabs_.push_back({(char*)base, want});
  capacity_ += want;
  insert_free((char*)base, want);
}

void DeviceArena::insert_free(char* p, size_t bytes) {
  // coalesce with address-adjacent free neighbors (within the same slab —
  // slabs are disjoint hipMalloc regions so cross-slab adjacency can't
  // occur by construction)
  auto next = by_addr_.lower_bound(p);
  if (next != by_addr_.begin()) {
    auto prev = std::prev(next);
    if (prev->first + prev->second == p) {
      p = prev->first;
      bytes += prev->second;
      erase_size(prev->first, prev->second);
      by_addr_.erase(prev);
    }
  }
  if (next != by_addr_.end() && p + bytes == next->first) {
    bytes += next->second;
    erase_size(next->first, next->second);
    by_addr_.erase(next);
  }
  by_addr_[p] = bytes;
  by_size_.insert({bytes, p});
}

void DeviceArena::erase_size(char* p, size_t bytes) {
  auto it = by_size_.find({bytes, p});
  if (it != by_size_.end()) by_size_.erase(it);
}

void* DeviceArena::allocate(size_t bytes, size_t align) {
  if (bytes == 0) bytes = 1;
  if (align < 256) align = 256;  // HBM3E-friendly minimum
  std::lock_guard<std::mutex> lk(mu_);
  size_t need = round_up(bytes, align);

  auto fit = [&]() -> std::pair<char*, size_t> {
    // best-fit: smallest free node that can satisfy (size-ordered map);
    // nodes are 256-B aligned by construction so align<=256 always fits,
    // larger alignments may need slack
    for (auto it = by_size_.lower_bound({need, nullptr});
         it != by_size_.end(); ++it) {
      char* p = it->second;
      char* ap = (char*)round_up((int64_t)(uintptr_t)p, align);
      size_t slack = ap - p;
      if (it->first >= need + slack) return {p, it->first};
    }
    return {nullptr, 0};
  };

  auto node = fit();
  if (!node.first) {
    grow(need + align);
    node = fit();
    if (!node.first) throw std::bad_alloc();
  }
  char* p = node.first;
  size_t nb = node.second;
  erase_size(p, nb);
  by_addr_.erase(p);
  char* ap = (char*)round_up((int64_t)(uintptr_t)p, align);
  size_t head = ap - p;
  if (head) insert_free(p, head);
  size_t used = need;
  size_t tail = nb - head - used;
  if (tail) insert_free(ap + used, tail);

  live_[ap] = used;
  in_use_ += used;
  if (in_use_ > high_water_) high_water_ = in_use_;
  // log2 histogram (reference histogram_tracker): bucket = ceil(log2(b))
  int b = 0;
  while ((1ull << b) < bytes && b < 47) ++b;
  hist_[b]++;
  return ap;
}

void DeviceArena::deallocate(void* p) {
  if (!p) return;
  std::lock_guard<std::mutex> lk(mu_);
  auto it = live_.find((char*)p);
  if (it == live_.end())
    throw std::runtime_error("DeviceArena: free of unknown pointer");
  in_use_ -= it->second;
  insert_free(it->first, it->second);
  live_.erase(it);
}

DeviceArena::Stats DeviceArena::stats() const {
  std::lock_guard<std::mutex> lk(mu_);
  Stats s;
  s.capacity = capacity_;
  s.in_use = in_use_;
  s.high_water = high_water_;
  s.free_nodes = by_addr_.size();
  s.live_allocs = live_.size();
  size_t largest = 0;
  for (auto& kv : by_addr_) largest = std::max(largest, kv.second);
  s.largest_free = largest;
  for (int i = 0; i < 48; ++i) s.histogram[i] = hist_[i];
  return s;
}

}  // namespace trtlab
