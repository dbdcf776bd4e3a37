// FanoutSink — pybind wrapper over FanoutCore (fanout_core.h).
//
// The GPU crawl engine produces one pinned host buffer per chunk with
// per-channel byte ranges (gpu_runner.py); this module fans those
// ranges out to per-channel JSONL files through FanoutCore's worker
// pool with the GIL released. The wrapper ONLY converts buffers,
// validates slices and maps error counts to exceptions — all
// concurrency lives in fanout_core.h, which is stress-tested under
// ThreadSanitizer (tools/tsan_sink_stress.cc, tests/test_native_sink.py
// ::test_tsan_stress_harness).
//
// This is the runtime-native counterpart of the reference's Go writer
// goroutines (state/daprstate.go:1106-1248 post/file writes; the Dapr
// binding is replaced by direct local files here).
//
// Build: crawler_amd/ops/build.py (plain g++, no GPU dependency).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "fanout_core.h"

namespace py = pybind11;
using crawl_native::FanoutCore;
using crawl_native::Task;

namespace {

class FanoutSink {
 public:
  explicit FanoutSink(int n_threads = 4, int max_open = 256)
      : core_(n_threads, max_open) {}

  void write_batch(const std::vector<std::string>& paths, py::buffer buf,
                   const std::vector<size_t>& lo,
                   const std::vector<size_t>& hi) {
    auto tasks = make_tasks(paths, buf, lo, hi, 0);
    size_t errs;
    {
      py::gil_scoped_release rel;
      core_.enqueue(std::move(tasks), 0);
      errs = core_.drain();  // also evicts fds beyond the cap
    }
    if (errs != 0)
      throw std::runtime_error(
          "fanout sink: write(2) failures in batch (errno=" +
          std::to_string(core_.last_errno()) + "): " +
          std::to_string(errs));
  }

  void write_batch_nowait(const std::vector<std::string>& paths,
                          py::buffer buf, const std::vector<size_t>& lo,
                          const std::vector<size_t>& hi) {
    auto tasks = make_tasks(paths, buf, lo, hi, 0);
    core_.enqueue(std::move(tasks), 0);
  }

  uint64_t write_batch_ticket(const std::vector<std::string>& paths,
                              py::buffer buf,
                              const std::vector<size_t>& lo,
                              const std::vector<size_t>& hi) {
    uint64_t ticket = core_.next_ticket();
    auto tasks = make_tasks(paths, buf, lo, hi, ticket);
    core_.enqueue(std::move(tasks), ticket);
    return ticket;
  }

  void wait_ticket(uint64_t ticket) {
    size_t errs;
    {
      py::gil_scoped_release rel;
      errs = core_.wait_ticket(ticket);
    }
    if (errs != 0)
      throw std::runtime_error(
          "fanout sink: write(2) failures (errno=" +
          std::to_string(core_.last_errno()) + "): " +
                               std::to_string(errs));
  }

  void drain() {
    size_t errs;
    {
      py::gil_scoped_release rel;
      errs = core_.drain();
    }
    if (errs != 0)
      throw std::runtime_error(
          "fanout sink: write(2) failures (errno=" +
          std::to_string(core_.last_errno()) + "): " +
                               std::to_string(errs));
  }

  void flush() {
    py::gil_scoped_release rel;
    core_.flush();
  }

  void close() {
    py::gil_scoped_release rel;
    core_.close();
  }

  size_t bytes_written() const { return core_.bytes_written(); }
  size_t open_files() { return core_.open_files(); }

 private:
  std::vector<Task> make_tasks(const std::vector<std::string>& paths,
                               py::buffer& buf,
                               const std::vector<size_t>& lo,
                               const std::vector<size_t>& hi,
                               uint64_t ticket) {
    if (paths.size() != lo.size() || paths.size() != hi.size())
      throw std::invalid_argument("paths/lo/hi length mismatch");
    py::buffer_info info = buf.request();
    if (info.ndim != 1 || info.itemsize != 1)
      throw std::invalid_argument("buffer must be 1-D bytes");
    const char* base = static_cast<const char*>(info.ptr);
    size_t n_bytes = static_cast<size_t>(info.size);
    std::vector<Task> tasks;
    tasks.reserve(paths.size());
    for (size_t i = 0; i < paths.size(); ++i) {
      if (hi[i] < lo[i] || hi[i] > n_bytes)
        throw std::out_of_range("slice outside buffer");
      if (hi[i] == lo[i]) continue;
      tasks.push_back(Task{paths[i], base + lo[i], hi[i] - lo[i], ticket});
    }
    return tasks;
  }

  FanoutCore core_;
};

}  // namespace

PYBIND11_MODULE(fanout_native, m) {
  m.doc() = "native per-channel JSONL spill writer (thread-pool fan-out)";
  py::class_<FanoutSink>(m, "FanoutSink")
      .def(py::init<int, int>(), py::arg("n_threads") = 4,
           py::arg("max_open") = 256)
      .def("write_batch", &FanoutSink::write_batch)
      .def("write_batch_nowait", &FanoutSink::write_batch_nowait)
      .def("write_batch_ticket", &FanoutSink::write_batch_ticket)
      .def("wait_ticket", &FanoutSink::wait_ticket)
      .def("drain", &FanoutSink::drain)
      .def("flush", &FanoutSink::flush)
      .def("close", &FanoutSink::close)
      .def_property_readonly("bytes_written", &FanoutSink::bytes_written)
      .def_property_readonly("open_files", &FanoutSink::open_files);
}
