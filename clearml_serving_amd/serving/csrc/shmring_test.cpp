// Sanitizer harness for the SHM ring (race/sanitizer posture
// for the native tier). Built with -fsanitize=address,undefined by
// tests/test_shm_front.py::test_shmring_asan_ubsan and exercises:
//   - single-thread push/drain round-trips across many wrap points
//   - a true SPSC producer/consumer pair on two threads (the production
//     topology: front pushes, owner drains) under ThreadSanitizer-visible
//     load when built with -fsanitize=thread instead
// Exit code 0 = all checks passed and no sanitizer report fired.
#define CMLS_SHMRING_NO_PYBIND 1
#include "shmring_core.h"

#include <atomic>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

using cmls::ShmRingCore;

static int fails = 0;
#define CHECK(cond)                                                       \
  do {                                                                    \
    if (!(cond)) {                                                        \
      fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__, __LINE__,  \
              #cond);                                                     \
      ++fails;                                                            \
    }                                                                     \
  } while (0)

int main() {
  std::string name = "/cmls_asan_" + std::to_string(getpid());
  {
    ShmRingCore w(name, 1 << 14, true);
    ShmRingCore r(name, 0, false);

    // wrap torture: variable-size records across the ring boundary
    for (int i = 0; i < 5000; ++i) {
      std::string payload((i % 900) + 1, char('a' + i % 26));
      CHECK(w.push(payload.data(), payload.size()));
      auto got = r.drain(4);
      CHECK(got.size() == 1);
      CHECK(got[0] == payload);
    }

    // backpressure + full drain
    int n = 0;
    std::string big(1000, 'z');
    while (w.push(big.data(), big.size())) ++n;
    CHECK(n >= 10);
    int drained = 0;
    for (;;) {
      auto batch = r.drain(64);
      if (batch.empty()) break;
      for (auto& b : batch) CHECK(b == big);
      drained += (int)batch.size();
    }
    CHECK(drained == n);

    // SPSC across threads (the production shape)
    std::atomic<bool> stop{false};
    std::atomic<long> consumed{0};
    long produced = 20000;
    std::thread consumer([&] {
      long seen = 0;
      while (!stop.load(std::memory_order_acquire) || seen < produced) {
        auto batch = r.drain(128);
        for (auto& b : batch) {
          long v;
          memcpy(&v, b.data(), sizeof(v));
          CHECK(v == seen);
          ++seen;
        }
      }
      consumed.store(seen);
    });
    for (long i = 0; i < produced; ++i) {
      while (!w.push(&i, sizeof(i))) {
      }
    }
    stop.store(true, std::memory_order_release);
    consumer.join();
    CHECK(consumed.load() == produced);
  }
  ShmRingCore::unlink(name);
  if (fails) {
    fprintf(stderr, "FAILED (%d checks)\n", fails);
    return 1;
  }
  printf("shmring sanitizer harness OK\n");
  return 0;
}
