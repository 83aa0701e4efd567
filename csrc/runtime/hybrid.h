// trtlab_amd — hybrid spin-then-futex mutex + condition variable.
//
// Role of the reference's hybrid_mutex/hybrid_condition
// (trtlab/core/hybrid_mutex.h:52, hybrid_condition.h:49): low-latency
// dispatch-path locking — spin a bounded number of exchanges first (the
// uncontended/short-hold fast path never enters the kernel), then park on
// a futex. Original implementation: a 2-state word (1 = locked,
// 2 = locked-with-waiters) with FUTEX_WAIT/WAKE on contention.
#pragma once
#include <linux/futex.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>

namespace trtlab {

inline long futex_call(std::atomic<uint32_t>* addr, int op, uint32_t val) {
  return syscall(SYS_futex, reinterpret_cast<uint32_t*>(addr), op, val,
                 nullptr, nullptr, 0);
}

class HybridMutex {
 public:
  explicit HybridMutex(uint32_t spins = 64) : spins_(spins) {}
  HybridMutex(const HybridMutex&) = delete;
  HybridMutex& operator=(const HybridMutex&) = delete;

  void lock() {
    // fast path: bounded spinning (short critical sections release the
    // lock within a few exchanges — no syscall)
    for (uint32_t i = 0; i < spins_; ++i) {
      uint32_t expect = 0;
      if (state_.compare_exchange_weak(expect, 1, std::memory_order_acquire,
                                       std::memory_order_relaxed))
        return;
#if defined(__x86_64__)
      __builtin_ia32_pause();
#endif
    }
    // slow path: mark contended and park until woken
    while (state_.exchange(2, std::memory_order_acquire) != 0)
      futex_call(&state_, FUTEX_WAIT_PRIVATE, 2);
  }

  bool try_lock() {
    uint32_t expect = 0;
    return state_.compare_exchange_strong(expect, 1,
                                          std::memory_order_acquire,
                                          std::memory_order_relaxed);
  }

  void unlock() {
    if (state_.exchange(0, std::memory_order_release) == 2)
      futex_call(&state_, FUTEX_WAKE_PRIVATE, 1);  // someone is parked
  }

 private:
  friend class HybridCondition;
  std::atomic<uint32_t> state_{0};
  uint32_t spins_;
};

class HybridCondition {
 public:
  void wait(HybridMutex& mu) {
    uint32_t seq = seq_.load(std::memory_order_relaxed);
    mu.unlock();
    futex_call(&seq_, FUTEX_WAIT_PRIVATE, seq);  // no-op if seq moved on
    mu.lock();
  }

  void notify_one() {
    seq_.fetch_add(1, std::memory_order_release);
    futex_call(&seq_, FUTEX_WAKE_PRIVATE, 1);
  }

  void notify_all() {
    seq_.fetch_add(1, std::memory_order_release);
    futex_call(&seq_, FUTEX_WAKE_PRIVATE, INT32_MAX);
  }

 private:
  std::atomic<uint32_t> seq_{0};
};

// RAII guard
class HybridLock {
 public:
  explicit HybridLock(HybridMutex& m) : m_(m) { m_.lock(); }
  ~HybridLock() { m_.unlock(); }

 private:
  HybridMutex& m_;
};

}  // namespace trtlab
