// FanoutCore — the pybind-free heart of the native spill sink.
//
// Split out of fanout_sink.cc so the SAME concurrency logic can be
// stress-tested standalone under ThreadSanitizer
// (tools/tsan_sink_stress.cc; SURVEY §5.2 — the reference relies on
// convention, we add a sanitizer gate). The pybind wrapper
// (fanout_sink.cc) only converts buffers and releases the GIL around
// the blocking calls.
//
// Concurrency contract (locks documented per structure, reference
// convention):
//   mu_     guards q_, pending_, ticket_pending_, ticket_errors_, stop_
//   fd_mu_  guards fds_, lru_ (workers open files in parallel)
//   errors_/bytes_ are atomics; ticket_seq_ is an atomic counter.
// Eviction is refcount-guarded: fds a worker holds mid-write are
// pinned (FdEntry.refs) and skipped, so drain()/eviction is safe even
// with concurrent producers (a TSAN-found hole; see fd_acquire).
#pragma once

#include <fcntl.h>
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

namespace crawl_native {

inline void make_parent_dirs(const std::string& path) {
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
  uint64_t ticket;  // 0 = untracked
};

class FanoutCore {
 public:
  explicit FanoutCore(int n_threads = 4, int max_open = 256)
      : max_open_(max_open), stop_(false), pending_(0), errors_(0),
        bytes_(0) {
    if (n_threads < 1) n_threads = 1;
    for (int i = 0; i < n_threads; ++i)
      workers_.emplace_back([this] { this->worker(); });
  }

  ~FanoutCore() { close(); }

  // Enqueue one batch; ticket==0 -> untracked. Caller guarantees the
  // data pointers stay valid until the matching wait/drain.
  void enqueue(std::vector<Task>&& tasks, uint64_t ticket) {
    std::unique_lock<std::mutex> lk(mu_);
    if (ticket != 0) {
      ticket_pending_[ticket] = tasks.size();
      ticket_errors_[ticket] = 0;
    }
    for (auto& t : tasks) q_.push(std::move(t));
    pending_ += tasks.size();
    lk.unlock();
    cv_.notify_all();
  }

  uint64_t next_ticket() { return ++ticket_seq_; }

  // Returns the error count for the ticket (0 = clean).
  size_t wait_ticket(uint64_t ticket) {
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [this, ticket] {
        auto it = ticket_pending_.find(ticket);
        return it == ticket_pending_.end() || it->second == 0;
      });
    }
    // Evict surplus fds HERE too: a long ticketed pipeline (the GPU
    // random-walk touches ~1k new channel files per hop and only
    // drains at crawl end) otherwise accumulates open fds until the
    // process hits EMFILE. Pinned fds (mid-write) are skipped.
    evict_to_cap();
    std::unique_lock<std::mutex> lk(mu_);
    size_t errs = ticket_errors_[ticket];
    ticket_pending_.erase(ticket);
    ticket_errors_.erase(ticket);
    return errs;
  }

  // Wait until EVERYTHING queued has been written. Returns untracked
  // error count since the last drain and evicts fds beyond the cap.
  size_t drain() {
    size_t errs;
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [this] { return pending_ == 0; });
      errs = errors_.exchange(0);
    }
    evict_to_cap();
    return errs;
  }

  void flush() {
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return pending_ == 0; });
    std::unique_lock<std::mutex> flk(fd_mu_);
    for (auto& kv : fds_) ::fsync(kv.second.fd);
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
    for (auto& kv : fds_) ::close(kv.second.fd);
    fds_.clear();
    lru_.clear();
  }

  size_t bytes_written() const { return bytes_.load(); }
  int last_errno() const { return last_errno_.load(); }
  size_t open_files() {
    std::unique_lock<std::mutex> lk(fd_mu_);
    return fds_.size();
  }

 private:
  // Worker-side open with its own lock: O_CREAT opens + mkdirs are
  // metadata-heavy; resolving in the pool parallelizes them. The entry
  // is REFCOUNTED: acquire pins it against eviction until fd_release
  // (TSAN found the hole: a concurrent producer's drain() could evict
  // an fd another worker was mid-write(2) on). -1 = open failure.
  int fd_acquire(const std::string& path) {
    {
      std::unique_lock<std::mutex> lk(fd_mu_);
      auto it = fds_.find(path);
      if (it != fds_.end()) {
        ++it->second.refs;
        return it->second.fd;
      }
    }
    int fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
    if (fd < 0 && errno == ENOENT) {
      make_parent_dirs(path);
      fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
    }
    if (fd < 0) {
      last_errno_.store(errno);
      return -1;
    }
    std::unique_lock<std::mutex> lk(fd_mu_);
    auto it = fds_.find(path);
    if (it != fds_.end()) {  // raced: another worker opened it first
      ::close(fd);
      ++it->second.refs;
      return it->second.fd;
    }
    fds_[path] = FdEntry{fd, 1};
    lru_.push_back(path);
    return fd;
  }

  void fd_release(const std::string& path) {
    std::unique_lock<std::mutex> lk(fd_mu_);
    auto it = fds_.find(path);
    if (it != fds_.end() && it->second.refs > 0) --it->second.refs;
  }

  // Evict down to the cap, skipping fds a worker currently holds
  // (refs > 0). Safe to call concurrently with writes.
  void evict_to_cap() {
    std::unique_lock<std::mutex> lk(fd_mu_);
    size_t i = 0;
    while (fds_.size() > static_cast<size_t>(max_open_) &&
           i < lru_.size()) {
      const std::string& old = lru_[i];        // oldest first; O_APPEND
      auto oit = fds_.find(old);               // makes reopen safe
      if (oit == fds_.end()) {
        lru_.erase(lru_.begin() + i);
        continue;
      }
      if (oit->second.refs > 0) {              // pinned: skip
        ++i;
        continue;
      }
      ::close(oit->second.fd);
      fds_.erase(oit);
      lru_.erase(lru_.begin() + i);
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
        t = std::move(q_.front());
        q_.pop();
      }
      int fd = fd_acquire(t.path);
      const char* p = t.data;
      size_t left = t.len;
      bool ok = (fd >= 0);
      while (ok && left > 0) {
        ssize_t w = ::write(fd, p, left);
        if (w < 0) {
          if (errno == EINTR) continue;
          last_errno_.store(errno);
          ok = false;
          break;
        }
        p += w;
        left -= static_cast<size_t>(w);
      }
      if (fd >= 0) fd_release(t.path);
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
  std::atomic<int> last_errno_{0};
  struct FdEntry {
    int fd;
    int refs;  // workers mid-write pin the entry against eviction
  };
  std::mutex fd_mu_;  // guards fds_/lru_ (workers open in parallel)
  std::unordered_map<std::string, FdEntry> fds_;
  std::vector<std::string> lru_;
  std::queue<Task> q_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  std::vector<std::thread> workers_;
};

}  // namespace crawl_native
