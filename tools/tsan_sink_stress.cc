// ThreadSanitizer stress harness for FanoutCore (crawler_amd/native/
// fanout_core.h) — the sanitizer gate SURVEY §5.2 asks for (the
// reference has none; its concurrency safety is by convention).
//
// Build+run (tests/test_native_sink.py::test_tsan_stress_harness):
//   g++ -fsanitize=thread -O1 -g -std=c++17 -pthread \
//       tools/tsan_sink_stress.cc -o /tmp/tsan_sink && /tmp/tsan_sink DIR
// Exit 0 + no TSAN report = clean. Any data race aborts with a report
// (TSAN_OPTIONS=halt_on_error=1 is set by the test).
#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "../crawler_amd/native/fanout_core.h"

using crawl_native::FanoutCore;
using crawl_native::Task;

int main(int argc, char** argv) {
  std::string dir = argc > 1 ? argv[1] : "/tmp/tsan_sink_out";
  ::mkdir(dir.c_str(), 0755);

  FanoutCore core(4, 8);  // tiny fd cap -> eviction churn
  const int kProducers = 3;
  const int kRounds = 40;
  const int kFilesPerBatch = 12;
  const size_t kSlice = 512;

  std::vector<std::thread> producers;
  std::vector<std::vector<char>> bufs(kProducers,
                                      std::vector<char>(kSlice * 16));
  for (int p = 0; p < kProducers; ++p) {
    for (size_t i = 0; i < bufs[p].size(); ++i)
      bufs[p][i] = static_cast<char>('a' + p);
    producers.emplace_back([&, p] {
      for (int r = 0; r < kRounds; ++r) {
        uint64_t ticket = core.next_ticket();
        std::vector<Task> tasks;
        for (int f = 0; f < kFilesPerBatch; ++f) {
          // overlapping file set across producers -> fd-cache races
          std::string path = dir + "/f" + std::to_string((p + f) % 16) +
                             ".bin";
          tasks.push_back(Task{path, bufs[p].data() + (f % 8) * kSlice,
                               kSlice, ticket});
        }
        core.enqueue(std::move(tasks), ticket);
        if (r % 3 == 2) {
          size_t errs = core.wait_ticket(ticket);
          assert(errs == 0);
        } else if (r % 7 == 6) {
          core.drain();
        } else {
          core.wait_ticket(ticket);
        }
      }
    });
  }
  // a concurrent reader of the stats surface
  std::thread poller([&] {
    for (int i = 0; i < 200; ++i) {
      (void)core.bytes_written();
      (void)core.open_files();
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
  });
  for (auto& t : producers) t.join();
  poller.join();
  core.drain();
  core.flush();
  core.close();
  std::printf("tsan-stress OK bytes=%zu\n", core.bytes_written());
  return 0;
}
