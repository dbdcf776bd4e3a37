// FanoutSink — native per-channel JSONL spill writer.
//
// The GPU crawl engine produces one pinned host buffer per chunk with
// per-channel byte ranges (gpu_runner.py). The Python path writes those
// ranges through per-channel file handles one at a time; this module
// does the same fan-out with a worker-thread pool, an fd cache and the
// GIL released, so 256 channels' appends hit the filesystem in parallel.
// This is the runtime-native counterpart of the reference's Go writer
// goroutines (state/daprstate.go:1106-1248 post/file writes; the Dapr
// binding is replaced by direct local files here).
//
// Build: crawler_amd/ops/build.py (plain g++, no GPU dependency).
#include <fcntl.h>
#include <pthread.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <mutex>
#include <queue>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

namespace py = pybind11;

namespace {

void make_parent_dirs(const std::string& path) {
  std::string dir = path.substr(0, path.find_last_of('/'));
  if (dir.empty()) return;
  std::string cur;
  size_t pos = 0;
  while (pos != std::string::npos) {
    pos = dir.find('/', pos + 1);
    cur = dir.substr(0, pos);
    if (!cur.empty()) ::mkdir(cur.c_str(), 0755);  // EEXIST is fine
  }
}

struct Task {
  std::string path;  // resolved to an fd by the WORKER (parallel opens)
  const char* data;
  size_t len;
  uint64_t ticket;  // 0 = untracked (write_batch/write_batch_nowait)
};

class FanoutSink {
 public:
  explicit FanoutSink(int n_threads = 4, int max_open = 256)
      : max_open_(max_open), stop_(false), pending_(0), errors_(0),
        bytes_(0) {
    if (n_threads < 1) n_threads = 1;
    for (int i = 0; i < n_threads; ++i)
      workers_.emplace_back([this] { this->worker(); });
  }

  ~FanoutSink() { close(); }

  // Append [lo, hi) slices of `buf` to each path. Blocks (GIL released)
  // until every slice of THIS batch is on its way through write(2) —
  // the buffer may be reused by the caller afterwards.
  void write_batch(const std::vector<std::string>& paths, py::buffer buf,
                   const std::vector<size_t>& lo,
                   const std::vector<size_t>& hi) {
    if (paths.size() != lo.size() || paths.size() != hi.size())
      throw std::invalid_argument("paths/lo/hi length mismatch");
    py::buffer_info info = buf.request();
    if (info.ndim != 1 || info.itemsize != 1)
      throw std::invalid_argument("buffer must be 1-D bytes");
    const char* base = static_cast<const char*>(info.ptr);
    size_t n_bytes = static_cast<size_t>(info.size);
    size_t errors_before = errors_.load();

    std::vector<Task> tasks;
    tasks.reserve(paths.size());
    for (size_t i = 0; i < paths.size(); ++i) {
      if (hi[i] < lo[i] || hi[i] > n_bytes)
        throw std::out_of_range("slice outside buffer");
      if (hi[i] == lo[i]) continue;
      tasks.push_back(Task{paths[i], base + lo[i], hi[i] - lo[i], 0});
    }
    {
      py::gil_scoped_release rel;
      {
        std::unique_lock<std::mutex> lk(mu_);
        for (auto& t : tasks) q_.push(std::move(t));
        pending_ += tasks.size();
      }
      cv_.notify_all();
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [this] { return pending_ == 0; });
    }
    evict_to_cap();  // safe: no writes in flight touch any fd now
    if (errors_.load() != errors_before)
      throw std::runtime_error(
          "fanout sink: write(2) failures in batch: " +
          std::to_string(errors_.load() - errors_before));
  }

  // Fire-and-forget variant: enqueue and return immediately. The CALLER
  // must keep `buf` alive until drain() returns (the GPU engine holds
  // the pinned host buffers of in-flight chunks and drains every few
  // chunks, overlapping disk writes with the next chunk's kernels).
  void write_batch_nowait(const std::vector<std::string>& paths,
                          py::buffer buf, const std::vector<size_t>& lo,
                          const std::vector<size_t>& hi) {
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
      tasks.push_back(Task{paths[i], base + lo[i], hi[i] - lo[i], 0});
    }
    {
      std::unique_lock<std::mutex> lk(mu_);
      for (auto& t : tasks) q_.push(std::move(t));
      pending_ += tasks.size();
    }
    cv_.notify_all();
    // NOTE: no eviction here — fds stay open until the next drain()
  }

  // Ticketed variant: like write_batch_nowait but returns a ticket that
  // wait_ticket() blocks on — lets a caller pipeline SEVERAL batches and
  // wait for the oldest only (the GPU engine runs the spill 2-deep).
  uint64_t write_batch_ticket(const std::vector<std::string>& paths,
                              py::buffer buf,
                              const std::vector<size_t>& lo,
                              const std::vector<size_t>& hi) {
    if (paths.size() != lo.size() || paths.size() != hi.size())
      throw std::invalid_argument("paths/lo/hi length mismatch");
    py::buffer_info info = buf.request();
    if (info.ndim != 1 || info.itemsize != 1)
      throw std::invalid_argument("buffer must be 1-D bytes");
    const char* base = static_cast<const char*>(info.ptr);
    size_t n_bytes = static_cast<size_t>(info.size);
    uint64_t ticket = ++ticket_seq_;
    std::vector<Task> tasks;
    tasks.reserve(paths.size());
    for (size_t i = 0; i < paths.size(); ++i) {
      if (hi[i] < lo[i] || hi[i] > n_bytes)
        throw std::out_of_range("slice outside buffer");
      if (hi[i] == lo[i]) continue;
      tasks.push_back(Task{paths[i], base + lo[i], hi[i] - lo[i], ticket});
    }
    {
      std::unique_lock<std::mutex> lk(mu_);
      ticket_pending_[ticket] = tasks.size();
      ticket_errors_[ticket] = 0;
      for (auto& t : tasks) q_.push(std::move(t));
      pending_ += tasks.size();
    }
    cv_.notify_all();
    return ticket;
  }

  void wait_ticket(uint64_t ticket) {
    size_t errs;
    {
      py::gil_scoped_release rel;
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [this, ticket] {
        auto it = ticket_pending_.find(ticket);
        return it == ticket_pending_.end() || it->second == 0;
      });
      errs = ticket_errors_[ticket];
      ticket_pending_.erase(ticket);
      ticket_errors_.erase(ticket);
    }
    // NOTE: no eviction — other tickets may still be in flight; the
    // periodic drain() (layer barrier) evicts.
    if (errs != 0)
      throw std::runtime_error("fanout sink: write(2) failures: " +
                               std::to_string(errs));
  }

  // Wait for all queued writes; then evict down to the fd cap. Raises if
  // any write since the last drain failed.
  void drain() {
    size_t errs;
    {
      py::gil_scoped_release rel;
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [this] { return pending_ == 0; });
      errs = errors_.exchange(0);
    }
    evict_to_cap();
    if (errs != 0)
      throw std::runtime_error("fanout sink: write(2) failures: " +
                               std::to_string(errs));
  }

  void flush() {
    py::gil_scoped_release rel;
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return pending_ == 0; });
    std::unique_lock<std::mutex> flk(fd_mu_);
    for (auto& kv : fds_) ::fsync(kv.second);
  }

  void close() {
    {
      std::unique_lock<std::mutex> lk(mu_);
      if (stop_) return;
      done_cv_.wait(lk, [this] { return pending_ == 0; });
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& t : workers_) t.join();
    workers_.clear();
    std::unique_lock<std::mutex> flk(fd_mu_);
    for (auto& kv : fds_) ::close(kv.second);
    fds_.clear();
    lru_.clear();
  }

  size_t bytes_written() const { return bytes_.load(); }
  size_t open_files() {
    std::unique_lock<std::mutex> lk(fd_mu_);
    return fds_.size();
  }

 private:
  // Worker-side open with its own lock: 1500 O_CREAT opens + mkdirs per
  // chunk are metadata-heavy; resolving in the pool parallelizes them
  // instead of serializing the enqueue thread. Returns -1 on failure.
  int fd_for(const std::string& path) {
    {
      std::unique_lock<std::mutex> lk(fd_mu_);
      auto it = fds_.find(path);
      if (it != fds_.end()) return it->second;
    }
    int fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
    if (fd < 0 && errno == ENOENT) {
      make_parent_dirs(path);
      fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
    }
    if (fd < 0) return -1;
    std::unique_lock<std::mutex> lk(fd_mu_);
    auto it = fds_.find(path);
    if (it != fds_.end()) {  // raced: another worker opened it first
      ::close(fd);
      return it->second;
    }
    fds_[path] = fd;
    lru_.push_back(path);
    return fd;
  }

  // Eviction runs ONLY between batches (no task may reference an fd):
  // a same-batch eviction could close an fd a queued write still holds.
  void evict_to_cap() {
    std::unique_lock<std::mutex> lk(fd_mu_);
    while (fds_.size() > static_cast<size_t>(max_open_) &&
           !lru_.empty()) {
      std::string old = lru_.front();          // oldest first; O_APPEND
      lru_.erase(lru_.begin());                // makes reopen safe
      auto oit = fds_.find(old);
      if (oit != fds_.end()) {
        ::close(oit->second);
        fds_.erase(oit);
      }
    }
  }

  void worker() {
    for (;;) {
      Task t;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return stop_ || !q_.empty(); });
        if (q_.empty()) {
          if (stop_) return;
          continue;
        }
        t = q_.front();
        q_.pop();
      }
      int fd = fd_for(t.path);
      const char* p = t.data;
      size_t left = t.len;
      bool ok = (fd >= 0);
      while (ok && left > 0) {
        ssize_t w = ::write(fd, p, left);
        if (w < 0) {
          if (errno == EINTR) continue;
          ok = false;
          break;
        }
        p += w;
        left -= static_cast<size_t>(w);
      }
      if (ok)
        bytes_.fetch_add(t.len);
      else if (t.ticket == 0)
        errors_.fetch_add(1);  // ticketed failures live in ticket_errors_
      {
        std::unique_lock<std::mutex> lk(mu_);
        if (t.ticket != 0) {
          auto it = ticket_pending_.find(t.ticket);
          if (it != ticket_pending_.end() && it->second > 0) --it->second;
          if (!ok) ++ticket_errors_[t.ticket];
        }
        if (--pending_ == 0 ||
            (t.ticket != 0 && ticket_pending_[t.ticket] == 0))
          done_cv_.notify_all();
      }
    }
  }

  int max_open_;
  bool stop_;
  size_t pending_;
  std::atomic<uint64_t> ticket_seq_{0};
  std::unordered_map<uint64_t, size_t> ticket_pending_;   // under mu_
  std::unordered_map<uint64_t, size_t> ticket_errors_;    // under mu_
  std::atomic<size_t> errors_;
  std::atomic<size_t> bytes_;
  std::mutex fd_mu_;  // guards fds_/lru_ (workers open in parallel)
  std::unordered_map<std::string, int> fds_;
  std::vector<std::string> lru_;
  std::queue<Task> q_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  std::vector<std::thread> workers_;
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
