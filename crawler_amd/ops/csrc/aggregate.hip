// Segmented per-channel engagement aggregation (SURVEY §2.6:
// GetTotalChannelViews / message counts, telegramhelper/telegramutils.go:
// 230-296 -> "segmented reduction per channel batch").
//
// One workgroup per channel segment (messages are grouped K x P): block
// reduction of views/forwards/reply counts + post count, plus the global
// batch totals via one atomic per block. Feeds the layer-statistics
// metrics (standalone/runner.go:862-882 counterpart).

#include "common.h"

namespace crawl {

__global__ void __launch_bounds__(256)
channel_stats_kernel(const int* views, const int* forwards,
                     const int* replies, const int* line_len,
                     int posts_per_channel, int n_channels,
                     long* ch_views, long* ch_forwards, long* ch_replies,
                     int* ch_posts, unsigned long long* totals) {
  const int c = blockIdx.x;
  if (c >= n_channels) return;
  const long base = (long)c * posts_per_channel;
  long v = 0, f = 0, r = 0;
  int p = 0;
  for (int i = threadIdx.x; i < posts_per_channel; i += blockDim.x) {
    v += views[base + i];
    f += forwards[base + i];
    r += replies[base + i];
    p += (line_len == nullptr || line_len[base + i] > 0) ? 1 : 0;
  }
  // wave reduce then LDS cross-wave
  __shared__ long sv[4], sf[4], sr[4];
  __shared__ int sp[4];
  for (int d = WAVE / 2; d > 0; d >>= 1) {
    v += __shfl_down(v, d);
    f += __shfl_down(f, d);
    r += __shfl_down(r, d);
    p += __shfl_down(p, d);
  }
  const int wave = wave_id();
  if (lane_id() == 0) {
    sv[wave] = v; sf[wave] = f; sr[wave] = r; sp[wave] = p;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    long tv = 0, tf = 0, tr = 0;
    int tp = 0;
    const int nw = (blockDim.x + WAVE - 1) / WAVE;
    for (int w = 0; w < nw; ++w) {
      tv += sv[w]; tf += sf[w]; tr += sr[w]; tp += sp[w];
    }
    ch_views[c] = tv;
    ch_forwards[c] = tf;
    ch_replies[c] = tr;
    ch_posts[c] = tp;
    if (totals != nullptr) {
      atomicAdd(&totals[0], (unsigned long long)tv);
      atomicAdd(&totals[1], (unsigned long long)tf);
      atomicAdd(&totals[2], (unsigned long long)tr);
      atomicAdd(&totals[3], (unsigned long long)tp);
    }
  }
}

}  // namespace crawl

extern "C" {

int crawl_channel_stats(const void* views, const void* forwards,
                        const void* replies, const void* line_len,
                        int posts_per_channel, int n_channels,
                        void* ch_views, void* ch_forwards, void* ch_replies,
                        void* ch_posts, void* totals, void* stream) {
  hipLaunchKernelGGL(crawl::channel_stats_kernel, dim3(n_channels),
                     dim3(256), 0, (hipStream_t)stream,
                     (const int*)views, (const int*)forwards,
                     (const int*)replies, (const int*)line_len,
                     posts_per_channel, n_channels, (long*)ch_views,
                     (long*)ch_forwards, (long*)ch_replies, (int*)ch_posts,
                     (unsigned long long*)totals);
  return (int)hipGetLastError();
}

}  // extern "C"
