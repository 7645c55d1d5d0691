// SPSC byte ring over POSIX shared memory -- core (pybind-free so the
// sanitizer harness shmring_test.cpp can build it standalone).
// Layout/semantics documented in shmring.cpp.
#pragma once

#include <atomic>
#include <cerrno>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

namespace cmls {

constexpr uint64_t kRingMagic = 0x434d4c53524e4721ull;  // "CMLSRNG!"
constexpr uint32_t kWrapMarker = 0xFFFFFFFFu;
constexpr size_t kRingHeaderSize = 192;

struct RingHeader {
  uint64_t magic;
  uint64_t capacity;
  char pad0[48];
  std::atomic<uint64_t> head;  // own cache line
  char pad1[56];
  std::atomic<uint64_t> tail;  // own cache line
  char pad2[56];
};
static_assert(sizeof(RingHeader) <= kRingHeaderSize, "header layout");

inline uint64_t ring_align8(uint64_t n) { return (n + 7) & ~7ull; }

class ShmRingCore {
 public:
  ShmRingCore(const std::string& name, uint64_t capacity, bool create)
      : name_(name) {
    capacity = ring_align8(capacity);
    int flags = create ? (O_RDWR | O_CREAT | O_EXCL) : O_RDWR;
    int fd = shm_open(name.c_str(), flags, 0600);
    if (fd < 0 && create && errno == EEXIST) {
      shm_unlink(name.c_str());
      fd = shm_open(name.c_str(), flags, 0600);
    }
    if (fd < 0) {
      throw std::runtime_error("shm_open('" + name + "') failed: " +
                               std::string(strerror(errno)));
    }
    size_t total = kRingHeaderSize + (create ? capacity : 0);
    if (create) {
      if (ftruncate(fd, (off_t)total) != 0) {
        int e = errno;
        ::close(fd);
        shm_unlink(name.c_str());
        throw std::runtime_error("ftruncate failed: " +
                                 std::string(strerror(e)));
      }
    } else {
      struct stat st;
      if (fstat(fd, &st) != 0 || (size_t)st.st_size < kRingHeaderSize) {
        ::close(fd);
        throw std::runtime_error("ring '" + name + "' not initialized");
      }
      total = (size_t)st.st_size;
    }
    void* mem =
        mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (mem == MAP_FAILED) {
      throw std::runtime_error("mmap failed: " +
                               std::string(strerror(errno)));
    }
    map_ = static_cast<char*>(mem);
    map_size_ = total;
    hdr_ = reinterpret_cast<RingHeader*>(map_);
    data_ = map_ + kRingHeaderSize;
    if (create) {
      hdr_->capacity = capacity;
      hdr_->head.store(0, std::memory_order_relaxed);
      hdr_->tail.store(0, std::memory_order_relaxed);
      std::atomic_thread_fence(std::memory_order_release);
      hdr_->magic = kRingMagic;
    } else {
      if (hdr_->magic != kRingMagic) {
        munmap(map_, map_size_);
        throw std::runtime_error("ring '" + name + "' bad magic");
      }
    }
    cap_ = hdr_->capacity;
  }

  ~ShmRingCore() { close(); }

  void close() {
    if (map_) {
      munmap(map_, map_size_);
      map_ = nullptr;
    }
  }

  static void unlink(const std::string& name) { shm_unlink(name.c_str()); }

  // producer side ------------------------------------------------------
  bool push(const void* src_v, uint64_t len) {
    const char* src = static_cast<const char*>(src_v);
    uint64_t need = ring_align8(4 + len);
    // > cap/2 records can DEADLOCK, not just stall: when the write head
    // sits so that at_end < need, the wrapped footprint (at_end + need)
    // can exceed the whole capacity -- no amount of draining frees
    // enough. Reject loudly; callers raise --ring-mb instead.
    if (need > cap_ / 2) {
      throw std::runtime_error(
          "record larger than half the ring capacity (raise --ring-mb)");
    }
    uint64_t head = hdr_->head.load(std::memory_order_relaxed);
    uint64_t tail = hdr_->tail.load(std::memory_order_acquire);
    uint64_t pos = head % cap_;
    uint64_t at_end = cap_ - pos;
    uint64_t total_need = need;
    bool wrap = false;
    if (at_end < need) {
      wrap = true;
      total_need = at_end + need;
    }
    if (cap_ - (head - tail) < total_need) {
      return false;  // full: caller retries (backpressure)
    }
    if (wrap) {
      if (at_end >= 4) {
        uint32_t m = kWrapMarker;
        memcpy(data_ + pos, &m, 4);
      }
      head += at_end;
      pos = 0;
    }
    uint32_t len32 = (uint32_t)len;
    memcpy(data_ + pos, &len32, 4);
    memcpy(data_ + pos + 4, src, len);
    hdr_->head.store(head + need, std::memory_order_release);
    return true;
  }

  // consumer side ------------------------------------------------------
  std::vector<std::string> drain(size_t max_n) {
    std::vector<std::string> out;
    uint64_t head = hdr_->head.load(std::memory_order_acquire);
    uint64_t tail = hdr_->tail.load(std::memory_order_relaxed);
    while (tail < head && out.size() < max_n) {
      uint64_t pos = tail % cap_;
      uint64_t at_end = cap_ - pos;
      uint32_t len32;
      if (at_end < 4) {
        tail += at_end;  // implicit wrap (marker didn't fit)
        continue;
      }
      memcpy(&len32, data_ + pos, 4);
      if (len32 == kWrapMarker) {
        tail += at_end;
        continue;
      }
      out.emplace_back(data_ + pos + 4, (size_t)len32);
      tail += ring_align8(4 + (uint64_t)len32);
    }
    hdr_->tail.store(tail, std::memory_order_release);
    return out;
  }

  uint64_t pending() const {
    return hdr_->head.load(std::memory_order_acquire) -
           hdr_->tail.load(std::memory_order_acquire);
  }

  uint64_t capacity() const { return cap_; }

 private:
  std::string name_;
  char* map_ = nullptr;
  size_t map_size_ = 0;
  RingHeader* hdr_ = nullptr;
  char* data_ = nullptr;
  uint64_t cap_ = 0;
};

}  // namespace cmls
