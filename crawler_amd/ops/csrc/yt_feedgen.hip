// Device-side YouTube synthetic-index generation — the YouTube half of
// the C++/HIP synthetic-API engine (the Telegram half is
// feedgen.hip). Generates the packed YouTubeBatch pools directly in
// HBM, byte-identical to the numpy reference builder
// (youtube/batch.py build_corpus_fast, itself pinned to the per-video
// python path by tests).
//
// Phases (torch cumsums/unique between them):
//   1. meta kernel: per-video hash chains -> video id bytes, numeric
//      fields, channel index source, desc digit values
//   2. video fill kernel: vid/title/desc pool bytes (incl. the linked
//      channel id, recomputed on the fly)
//   3. channel fill kernel: channel table pool bytes
#include "common.h"

namespace crawl {

DEV unsigned long long yt_smx(unsigned long long x) {
  unsigned long long z = x + 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

__device__ const char YT_ALPH[64 + 1] =
    "abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789-_";

// scalar _h prefix chain over a constant string
DEV unsigned long long yt_chain(unsigned long long a, const char* s) {
  for (; *s; ++s) a = yt_smx(a ^ (unsigned long long)(unsigned char)*s);
  return a;
}

// channel_id_of (youtube/synth.py:82-91): 'UC' + 22 chars
DEV void yt_cid(unsigned long long seed_base, unsigned long long a_chan,
                long long n, unsigned char* out) {
  unsigned long long v = yt_smx(a_chan ^ (unsigned long long)n);
  out[0] = 'U';
  out[1] = 'C';
  for (int j = 0; j < 22; ++j) {
    out[2 + j] = (unsigned char)YT_ALPH[v % 62ULL];
    v = (v >> 5) | ((v & 31ULL) << 58);
    if (v < 62ULL) v = yt_smx(seed_base ^ v);
  }
}

DEV int yt_itoa(unsigned char* dst, long long v) {
  // non-negative itoa; returns digits written
  char tmp[20];
  int k = 0;
  do {
    tmp[k++] = (char)('0' + (v % 10));
    v /= 10;
  } while (v);
  for (int j = 0; j < k; ++j) dst[j] = (unsigned char)tmp[k - 1 - j];
  return k;
}

struct YtGenParams {
  long long seed;
  long long universe;
  long long base_date;
  long long n;
  long long i0;  // first video index of this chunk
};

// per-video hashes, identical to build_corpus_fast
DEV void yt_video_hash(const YtGenParams& P, long long i,
                       unsigned char* vid /*11B*/,
                       unsigned long long* h_out) {
  unsigned long long base = (unsigned long long)P.seed ^ 0xC0FFEEULL;
  unsigned long long a = yt_chain(base, "vid");
  long long v = i;
  unsigned char pc[5];
  for (int j = 4; j >= 0; --j) {
    pc[j] = (unsigned char)('a' + (v % 26));
    v /= 26;
  }
  for (int j = 0; j < 5; ++j) a = yt_smx(a ^ (unsigned long long)pc[j]);
  unsigned long long h1 = yt_smx(a ^ (unsigned long long)(i % 7));
  for (int j = 0; j < 5; ++j) vid[j] = pc[j];
  vid[5] = '-';
  for (int j = 0; j < 5; ++j)
    vid[6 + j] = (unsigned char)YT_ALPH[(h1 >> (6 * j)) % 64ULL];
  unsigned long long a2 = yt_chain(base, "vidmeta");
  for (int j = 0; j < 11; ++j)
    a2 = yt_smx(a2 ^ (unsigned long long)vid[j]);
  *h_out = a2;
}

__global__ void yt_gen_meta_kernel(
    YtGenParams P, unsigned char* vid_pool /*[n,11]*/,
    long long* published, long long* views, int* likes, int* comments,
    int* duration_s, int* lang, long long* n_chan, int* topic,
    int* tnum, int* desc_len) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P.n) return;
  unsigned char vid[11];
  unsigned long long h;
  yt_video_hash(P, P.i0 + i, vid, &h);
  for (int j = 0; j < 11; ++j) vid_pool[i * 11 + j] = vid[j];
  n_chan[i] = (long long)(h % (unsigned long long)P.universe);
  int secs = (int)((h >> 8) % 7200ULL);
  duration_s[i] = secs ? secs : -1;
  published[i] = P.base_date + (long long)(h % 10000000ULL);
  views[i] = (long long)(h % 1000000ULL);
  likes[i] = (int)((h >> 12) % 50000ULL);
  comments[i] = (int)((h >> 22) % 5000ULL);
  lang[i] = (h % 4ULL) == 0 ? 1 : 0;  // LANGS: 0=en, 1=ru
  int tp = (int)(h % 1000ULL);
  int tn = (int)(h % 97ULL);
  topic[i] = tp;
  tnum[i] = tn;
  int dtp = tp >= 100 ? 3 : (tp >= 10 ? 2 : 1);
  int dtn = tn >= 10 ? 2 : 1;
  // "Video about topic " (18) + d(tp) + ". More: https://example.com/t"
  // (29) + d(tn) + " and channel https://www.youtube.com/channel/" (45)
  // + 24-char channel id
  desc_len[i] = 18 + dtp + 29 + dtn + 45 + 24;
}

__global__ void yt_gen_fill_kernel(
    YtGenParams P, const unsigned char* vid_pool,
    const long long* n_chan_unused, const int* topic, const int* tnum,
    const long long* title_off, const long long* desc_off,
    unsigned char* pool) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P.n) return;
  // vid region: pool[0 .. n*11) with stride 11 (host sets vid_off)
  for (int j = 0; j < 11; ++j) pool[i * 11 + j] = vid_pool[i * 11 + j];
  // title: "Synthetic video " + vid (27 bytes)
  const char* tp = "Synthetic video ";
  unsigned char* t = pool + title_off[i];
  for (int j = 0; j < 16; ++j) t[j] = (unsigned char)tp[j];
  for (int j = 0; j < 11; ++j) t[16 + j] = vid_pool[i * 11 + j];
  // desc
  unsigned char* d = pool + desc_off[i];
  const char* s1 = "Video about topic ";
  const char* s2 = ". More: https://example.com/t";
  const char* s3 = " and channel https://www.youtube.com/channel/";
  int c = 0;
  for (const char* p = s1; *p; ++p) d[c++] = (unsigned char)*p;
  c += yt_itoa(d + c, topic[i]);
  for (const char* p = s2; *p; ++p) d[c++] = (unsigned char)*p;
  c += yt_itoa(d + c, tnum[i]);
  for (const char* p = s3; *p; ++p) d[c++] = (unsigned char)*p;
  // link channel id: (h >> 13) % universe — recompute h
  unsigned char vid[11];
  unsigned long long h;
  yt_video_hash(P, P.i0 + i, vid, &h);
  unsigned long long base = (unsigned long long)P.seed ^ 0xC0FFEEULL;
  unsigned long long a_chan = yt_chain(base, "chan");
  yt_cid(base, a_chan,
         (long long)((h >> 13) % (unsigned long long)P.universe), d + c);
}

__global__ void yt_gen_channels_kernel(
    YtGenParams P, const long long* chan_ns, long long K,
    const long long* id_off, const long long* title_off,
    const long long* desc_off, const long long* country_off,
    long long* subs, int* videos, long long* ch_views,
    long long* ch_published, unsigned char* pool) {
  long long c = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= K) return;
  long long n = chan_ns[c];
  unsigned long long base = (unsigned long long)P.seed ^ 0xC0FFEEULL;
  unsigned long long a_meta = yt_chain(base, "chanmeta");
  unsigned long long hc = yt_smx(a_meta ^ (unsigned long long)n);
  subs[c] = (long long)(hc % 1000000ULL);
  videos[c] = (int)((hc >> 20) % 30ULL);
  ch_views[c] = (long long)((hc >> 25) % 50000000ULL);
  ch_published[c] = P.base_date - (long long)(hc % 100000000ULL);
  unsigned long long a_chan = yt_chain(base, "chan");
  yt_cid(base, a_chan, n, pool + id_off[c]);
  // title: "Synthetic YT Channel " + n
  const char* t1 = "Synthetic YT Channel ";
  unsigned char* t = pool + title_off[c];
  int k = 0;
  for (const char* p = t1; *p; ++p) t[k++] = (unsigned char)*p;
  yt_itoa(t + k, n);
  // desc: "Channel " + n + " description \xe2\x80\x94 see also UC friends"
  const char* d1 = "Channel ";
  const char* d2 = " description \xe2\x80\x94 see also UC friends";
  unsigned char* d = pool + desc_off[c];
  k = 0;
  for (const char* p = d1; *p; ++p) d[k++] = (unsigned char)*p;
  k += yt_itoa(d + k, n);
  for (const char* p = d2; *p; ++p) d[k++] = (unsigned char)*p;
  // country: "US" when hc % 3 == 0 (host computed the offsets/lens)
  if (hc % 3ULL == 0ULL) {
    unsigned char* u = pool + country_off[c];
    u[0] = 'U';
    u[1] = 'S';
  }
}

}  // namespace crawl

extern "C" {

int crawl_yt_gen_meta(long long seed, long long universe,
                      long long base_date, long long n, long long i0,
                      void* vid_pool, void* published, void* views,
                      void* likes, void* comments, void* duration_s,
                      void* lang, void* n_chan, void* topic, void* tnum,
                      void* desc_len, void* stream) {
  crawl::YtGenParams P{seed, universe, base_date, n, i0};
  int grid = (int)((n + 255) / 256);
  hipLaunchKernelGGL(crawl::yt_gen_meta_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, P, (unsigned char*)vid_pool,
                     (long long*)published, (long long*)views,
                     (int*)likes, (int*)comments, (int*)duration_s,
                     (int*)lang, (long long*)n_chan, (int*)topic,
                     (int*)tnum, (int*)desc_len);
  return (int)hipGetLastError();
}

int crawl_yt_gen_fill(long long seed, long long universe,
                      long long base_date, long long n, long long i0,
                      const void* vid_pool, const void* topic,
                      const void* tnum, const void* title_off,
                      const void* desc_off, void* pool, void* stream) {
  crawl::YtGenParams P{seed, universe, base_date, n, i0};
  int grid = (int)((n + 255) / 256);
  hipLaunchKernelGGL(crawl::yt_gen_fill_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, P,
                     (const unsigned char*)vid_pool,
                     (const long long*)nullptr, (const int*)topic,
                     (const int*)tnum, (const long long*)title_off,
                     (const long long*)desc_off, (unsigned char*)pool);
  return (int)hipGetLastError();
}

int crawl_yt_gen_channels(long long seed, long long universe,
                          long long base_date, const void* chan_ns,
                          long long K, const void* id_off,
                          const void* title_off, const void* desc_off,
                          const void* country_off, void* subs,
                          void* videos, void* ch_views,
                          void* ch_published, void* pool, void* stream) {
  crawl::YtGenParams P{seed, universe, base_date, 0, 0};
  int grid = (int)((K + 255) / 256);
  hipLaunchKernelGGL(crawl::yt_gen_channels_kernel, dim3(grid), dim3(256),
                     0, (hipStream_t)stream, P,
                     (const long long*)chan_ns, K,
                     (const long long*)id_off,
                     (const long long*)title_off,
                     (const long long*)desc_off,
                     (const long long*)country_off, (long long*)subs,
                     (int*)videos, (long long*)ch_views,
                     (long long*)ch_published, (unsigned char*)pool);
  return (int)hipGetLastError();
}

}  // extern "C"
