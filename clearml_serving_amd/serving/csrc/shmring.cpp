// pybind11 bindings over the SPSC shared-memory ring (shmring_core.h):
// the request/response transport between the multi-process HTTP front and
// the single engine-owner process per GPU (SURVEY.md §7 hard-part 5: one
// engine owner per GPU keeps dynamic batches whole -- N uvicorn workers
// with N engine copies split them, measured in profiles/README.md §5).
//
// Layout:
//   [ header 192 B: magic u64 | capacity u64 | pad | head u64 (own line) |
//     pad | tail u64 (own line) ]
//   [ data region, `capacity` bytes ]
// Records: u32 length, payload, padded to 8 B. A length of 0xFFFFFFFF is a
// wrap marker (skip to offset 0). head/tail are monotonically increasing
// byte cursors; single producer writes head (release), single consumer
// writes tail (release). Sanitizer harness: shmring_test.cpp (built with
// -fsanitize=address,undefined by the test suite).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "shmring_core.h"

namespace py = pybind11;

namespace {

class ShmRing {
 public:
  ShmRing(const std::string& name, uint64_t capacity, bool create)
      : core_(name, capacity, create) {}

  bool push(py::buffer buf) {
    py::buffer_info info = buf.request();
    return core_.push(info.ptr, (uint64_t)info.size * (uint64_t)info.itemsize);
  }

  std::vector<py::bytes> drain(size_t max_n) {
    std::vector<py::bytes> out;
    for (auto& s : core_.drain(max_n)) out.emplace_back(s);
    return out;
  }

  uint64_t pending() const { return core_.pending(); }
  uint64_t capacity() const { return core_.capacity(); }
  void close() { core_.close(); }
  static void unlink(const std::string& name) { cmls::ShmRingCore::unlink(name); }

 private:
  cmls::ShmRingCore core_;
};

}  // namespace

PYBIND11_MODULE(_shmring, m) {
  m.doc() = "SPSC shared-memory byte ring (front <-> engine-owner transport)";
  py::class_<ShmRing>(m, "ShmRing")
      .def(py::init<const std::string&, uint64_t, bool>(), py::arg("name"),
           py::arg("capacity"), py::arg("create"))
      .def("push", &ShmRing::push, py::arg("data"))
      .def("drain", &ShmRing::drain, py::arg("max_n") = 1024)
      .def("pending", &ShmRing::pending)
      .def("capacity", &ShmRing::capacity)
      .def("close", &ShmRing::close)
      .def_static("unlink", &ShmRing::unlink, py::arg("name"));
}
