// Per-channel reservoir sampling ("Fisher-Yates sampling
// (messages & posts) -> device RNG + reservoir/shuffle kernel").
//
// The reference shuffles a channel's fetched messages and keeps the first
// sample_size when --date-between + --sample-size are set
// (telegramutils.go:124-154). Uniform-without-replacement is preserved
// here with a per-channel reservoir driven by splitmix64(seed, channel,
// i) — deterministic, so the Python oracle replays it exactly.
//
// One wave per channel; the reservoir walk is lane-0-serial (P <= tens of
// thousands) while channels run in parallel across the grid.

#include "common.h"

namespace crawl {

DEV unsigned long long splitmix64_s(unsigned long long x) {
  unsigned long long z = x + 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

__global__ void __launch_bounds__(256)
reservoir_sample_kernel(long seed, int posts_per_channel, int n_channels,
                        int k, int* out_idx /* [K,k] global row ids */) {
  const int wave = wave_id();
  for (int c = blockIdx.x * 4 + wave; c < n_channels;
       c += gridDim.x * 4) {
    if (lane_id() != 0) continue;
    int* slot = out_idx + (long)c * k;
    const long base = (long)c * posts_per_channel;
    for (int i = 0; i < posts_per_channel; ++i) {
      if (i < k) {
        slot[i] = (int)(base + i);
      } else {
        unsigned long long r = splitmix64_s(
            (unsigned long long)seed
            ^ ((unsigned long long)c * 0x9E3779B1ULL + (unsigned)i));
        int j = (int)(r % (unsigned long long)(i + 1));
        if (j < k) slot[j] = (int)(base + i);
      }
    }
  }
}

}  // namespace crawl

extern "C" {

int crawl_reservoir_sample(long seed, int posts_per_channel, int n_channels,
                           int k, void* out_idx, int grid, void* stream) {
  hipLaunchKernelGGL(crawl::reservoir_sample_kernel, dim3(grid), dim3(256),
                     0, (hipStream_t)stream, seed, posts_per_channel,
                     n_channels, k, (int*)out_idx);
  return (int)hipGetLastError();
}

}  // extern "C"
