// SPSC byte ring over POSIX shared memory: the request/response transport
// between the multi-process HTTP front and the single engine-owner process
// per GPU (SURVEY.md §7 hard-part 5: one engine owner per GPU keeps dynamic
// batches whole; N uvicorn workers with N engine copies split them --
// measured in profiles/README.md §5).
//
// Layout:
//   [ header 192 B: magic u64 | capacity u64 | pad | head u64 (own line) |
//     pad | tail u64 (own line) ]
//   [ data region, `capacity` bytes ]
// Records: u32 length, payload, padded to 8 B. A length of 0xFFFFFFFF is a
// wrap marker (skip to offset 0). head/tail are monotonically increasing
// byte cursors; single producer writes head (release), single consumer
// writes tail (release).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

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

namespace py = pybind11;

namespace {

constexpr uint64_t kMagic = 0x434d4c53524e4721ull;  // "CMLSRNG!"
constexpr uint32_t kWrapMarker = 0xFFFFFFFFu;
constexpr size_t kHeaderSize = 192;

struct Header {
    uint64_t magic;
    uint64_t capacity;
    char pad0[48];
    std::atomic<uint64_t> head;  // own cache line
    char pad1[56];
    std::atomic<uint64_t> tail;  // own cache line
    char pad2[56];
};
static_assert(sizeof(Header) <= kHeaderSize, "header layout");

inline uint64_t align8(uint64_t n) { return (n + 7) & ~7ull; }

class ShmRing {
 public:
    ShmRing(const std::string& name, uint64_t capacity, bool create)
        : name_(name) {
        capacity = align8(capacity);
        int flags = create ? (O_RDWR | O_CREAT | O_EXCL) : O_RDWR;
        int fd = shm_open(name.c_str(), flags, 0600);
        if (fd < 0 && create && errno == EEXIST) {
            // stale ring from a crashed run: replace it
            shm_unlink(name.c_str());
            fd = shm_open(name.c_str(), flags, 0600);
        }
        if (fd < 0) {
            throw std::runtime_error("shm_open('" + name + "') failed: " +
                                     std::string(strerror(errno)));
        }
        size_t total = kHeaderSize + (create ? capacity : 0);
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
            if (fstat(fd, &st) != 0 || (size_t)st.st_size < kHeaderSize) {
                ::close(fd);
                throw std::runtime_error("ring '" + name + "' not initialized");
            }
            total = (size_t)st.st_size;
        }
        void* mem = mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED,
                         fd, 0);
        ::close(fd);
        if (mem == MAP_FAILED) {
            throw std::runtime_error("mmap failed: " +
                                     std::string(strerror(errno)));
        }
        map_ = static_cast<char*>(mem);
        map_size_ = total;
        hdr_ = reinterpret_cast<Header*>(map_);
        data_ = map_ + kHeaderSize;
        if (create) {
            hdr_->capacity = capacity;
            hdr_->head.store(0, std::memory_order_relaxed);
            hdr_->tail.store(0, std::memory_order_relaxed);
            std::atomic_thread_fence(std::memory_order_release);
            hdr_->magic = kMagic;
        } else {
            if (hdr_->magic != kMagic) {
                munmap(map_, map_size_);
                throw std::runtime_error("ring '" + name + "' bad magic");
            }
        }
        cap_ = hdr_->capacity;
    }

    ~ShmRing() { close(); }

    void close() {
        if (map_) {
            munmap(map_, map_size_);
            map_ = nullptr;
        }
    }

    static void unlink(const std::string& name) { shm_unlink(name.c_str()); }

    // producer side ------------------------------------------------------
    bool push(py::buffer buf) {
        py::buffer_info info = buf.request();
        const char* src = static_cast<const char*>(info.ptr);
        uint64_t len = (uint64_t)info.size * (uint64_t)info.itemsize;
        uint64_t need = align8(4 + len);
        if (need + 8 > cap_) {
            throw std::runtime_error("record larger than ring capacity");
        }
        uint64_t head = hdr_->head.load(std::memory_order_relaxed);
        uint64_t tail = hdr_->tail.load(std::memory_order_acquire);
        uint64_t pos = head % cap_;
        uint64_t at_end = cap_ - pos;
        uint64_t total_need = need;
        bool wrap = false;
        if (at_end < need) {
            // wrap marker consumes the remainder of the buffer
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
    std::vector<py::bytes> drain(size_t max_n) {
        std::vector<py::bytes> out;
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
            tail += align8(4 + (uint64_t)len32);
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
    Header* hdr_ = nullptr;
    char* data_ = nullptr;
    uint64_t cap_ = 0;
};

}  // namespace

PYBIND11_MODULE(_shmring, m) {
    m.doc() = "SPSC shared-memory byte ring (front <-> engine-owner transport)";
    py::class_<ShmRing>(m, "ShmRing")
        .def(py::init<const std::string&, uint64_t, bool>(),
             py::arg("name"), py::arg("capacity"), py::arg("create"))
        .def("push", &ShmRing::push, py::arg("data"))
        .def("drain", &ShmRing::drain, py::arg("max_n") = 1024)
        .def("pending", &ShmRing::pending)
        .def("capacity", &ShmRing::capacity)
        .def("close", &ShmRing::close)
        .def_static("unlink", &ShmRing::unlink, py::arg("name"));
}
