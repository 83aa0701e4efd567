// Host-side sanitizer target (adopt ASAN/TSAN jobs).
// Exercises the pure-host concurrency/bookkeeping primitives under
// -fsanitize=thread and -fsanitize=address,undefined — built and run by
// tests/test_sanitizers.py with plain g++ (no GPU, no HIP headers).
#include "../runtime/hybrid.h"

#include <cassert>
#include <cstdio>
#include <thread>
#include <vector>

using trtlab::HybridCondition;
using trtlab::HybridLock;
using trtlab::HybridMutex;

int main() {
  // 1) mutual exclusion: racy counter must come out exact
  {
    HybridMutex mu;
    long counter = 0;
    std::vector<std::thread> ts;
    for (int t = 0; t < 8; ++t)
      ts.emplace_back([&] {
        for (int i = 0; i < 50000; ++i) {
          HybridLock g(mu);
          ++counter;
        }
      });
    for (auto& th : ts) th.join();
    assert(counter == 8L * 50000);
  }
  // 2) condition variable: producer/consumer handoff, no lost wakeups
  {
    HybridMutex mu;
    HybridCondition cv;
    int ready = 0, consumed = 0;
    std::thread prod([&] {
      for (int i = 0; i < 2000; ++i) {
        {
          HybridLock g(mu);
          ++ready;
        }
        cv.notify_one();
      }
    });
    std::thread cons([&] {
      while (true) {
        HybridLock g(mu);
        while (ready == consumed && consumed < 2000) cv.wait(mu);
        if (consumed >= 2000) break;
        ++consumed;
        if (consumed >= 2000) break;
      }
    });
    prod.join();
    {
      HybridLock g(mu);
      cv.notify_all();
    }
    cons.join();
    assert(consumed >= 2000);
  }
  std::printf("SANITIZE_HOST_OK\n");
  return 0;
}
