// Device-side synthetic feed generation — the C++/HIP synthetic-API engine
// (the TDLib stand-in). Generates the packed message batch
// directly in HBM, bit-identical to the numpy reference implementation in
// crawler_amd/feed/synth.py (which remains the test oracle).
//
// Three phases (cumsums run between them in torch):
//   1. meta kernel: per-message hashes -> template pick, scalar fields,
//      per-message block/entity/reaction/comment COUNTS
//   2. fill kernel: template bytes + aux + handle + digit-slot patches,
//      entity rows, reaction rows
//   3. comment kernel: per-comment text/handle pool + rows

#include "common.h"

namespace crawl {

DEV unsigned long long splitmix64(unsigned long long x) {
  unsigned long long z = x + 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

struct TemplateTables {
  const unsigned char* pool;   // concatenated template pool bytes
  const int* pool_off;         // [T]
  const int* pool_len;         // [T]
  const int* text_len;         // [T]
  const int* ctype;            // [T]
  const int* tflags;           // [T]
  const unsigned char* aux;    // concatenated aux bytes
  const int* aux_off;          // [T]
  const int* aux_len;          // [T]
  const int* ents;             // concatenated [E_t,5] rows
  const int* ent_off;          // [T] row offset
  const int* ent_cnt;          // [T]
  const int* slots;            // concatenated slot byte positions
  const int* slot_off;         // [T]
  const int* slot_cnt;         // [T]
  const double* cdf;           // [T]
  int n_templates;
  // comment text table
  const unsigned char* ctext;  // concatenated comment texts
  const int* ctext_off;        // [NC]
  const int* ctext_len;        // [NC]
  int n_ctexts;
};

struct FeedParams {
  long seed;
  long universe;
  long base_date;
  long date_step;
  long comment_rate_permille;
  long max_comments;
  int K;           // channels
  int P;           // posts per channel
  const long* channel_ids;  // [K]
};

#define NEMOJI 10
#define HANDLE_W 9

DEV void msg_hashes(const FeedParams& F, long cid, long pidx,
                    unsigned long long* h) {
  unsigned long long g =
      (unsigned long long)cid * 1000003ULL + (unsigned long long)pidx;
  h[0] = splitmix64((unsigned long long)F.seed ^ g);
  h[1] = splitmix64(h[0]);
  h[2] = splitmix64(h[1]);
  h[3] = splitmix64(h[2]);
}

DEV int pick_template(const TemplateTables& T, unsigned long long h0) {
  double u = (double)(h0 >> 11) / 9007199254740992.0;  // / 2^53
  // np.searchsorted(cdf, u, side='right'), clamped
  int idx = T.n_templates - 1;
  for (int i = 0; i < T.n_templates; ++i) {
    if (T.cdf[i] > u) { idx = i; break; }
  }
  return idx;
}

// ---- phase 1: meta ----
__global__ void __launch_bounds__(256)
feed_meta_kernel(TemplateTables T, FeedParams F,
                 long* chat_id, long* msg_id,
                 int* date, int* content_type, int* views, int* forwards,
                 int* media_album_id, int* channel_idx, int* flags,
                 int* text_len, int* aux_len, int* ent_cnt, int* react_cnt,
                 int* com_cnt, int* poster_len, int* reply_count,
                 long* block_len, int* tidx_out) {
  long n = (long)F.K * F.P;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (long)blockDim.x) {
    long k = i / F.P;
    long pidx = i % F.P;
    long cid = F.channel_ids[k];
    unsigned long long h[4];
    msg_hashes(F, cid, pidx, h);
    int t = pick_template(T, h[0]);
    tidx_out[i] = t;
    chat_id[i] = -1001000000000L - cid;
    msg_id[i] = (pidx + 1) << 20;
    date[i] = (int)(F.base_date + pidx * F.date_step + (long)(h[2] % 30ULL));
    content_type[i] = T.ctype[t];
    flags[i] = T.tflags[t];
    views[i] = (int)(h[1] % 100000ULL);
    forwards[i] = (int)((h[1] >> 17) % 1000ULL);
    media_album_id[i] =
        ((h[2] % 16ULL) == 0) ? (int)(h[2] % 2147483648ULL) : 0;
    channel_idx[i] = (int)k;
    text_len[i] = T.text_len[t];
    aux_len[i] = T.aux_len[t];
    ent_cnt[i] = T.ent_cnt[t];
    // reactions: popcount of (h2 & h2>>13 & mask)
    unsigned long long rmask = h[2] & (h[2] >> 13) & ((1ULL << NEMOJI) - 1);
    react_cnt[i] = __popcll(rmask);
    // comments
    bool has_c = (double)(h[3] % 1000ULL) < (double)F.comment_rate_permille;
    com_cnt[i] = has_c
        ? (int)((h[3] >> 32) % (unsigned long long)F.max_comments + 1ULL)
        : 0;
    reply_count[i] = com_cnt[i];
    poster_len[i] = HANDLE_W;
    block_len[i] = T.pool_len[t] + T.aux_len[t] + HANDLE_W;
  }
}

DEV void write_digits10(unsigned char* dst, long v) {
  for (int d = 9; d >= 0; --d) { dst[d] = '0' + (v % 10); v /= 10; }
}
DEV void write_digits8(unsigned char* dst, long v) {
  for (int d = 7; d >= 0; --d) { dst[d] = '0' + (v % 10); v /= 10; }
}

// ---- phase 2: fill (one wave per message) ----
__global__ void __launch_bounds__(256)
feed_fill_kernel(TemplateTables T, FeedParams F, const int* tidx,
                 const long* block_off, const long* ent_off_l,
                 const long* react_off_l, unsigned char* pool,
                 long* text_off, int* aux_off, int* poster_off,
                 int* ent_off, int* react_off, int* entities,
                 int* react_emoji, int* react_count) {
  long n = (long)F.K * F.P;
  const int lane = lane_id();
  const int wave = wave_id();
  for (long i = blockIdx.x * 4L + wave; i < n; i += gridDim.x * 4L) {
    long k = i / F.P;
    long pidx = i % F.P;
    long cid = F.channel_ids[k];
    unsigned long long h[4];
    msg_hashes(F, cid, pidx, h);
    int t = tidx[i];
    long boff = block_off[i];
    unsigned char* dst = pool + boff;
    const unsigned char* src = T.pool + T.pool_off[t];
    const int plen = T.pool_len[t];
    for (int j = lane; j < plen; j += WAVE) dst[j] = src[j];
    // aux
    const unsigned char* asrc = T.aux + T.aux_off[t];
    const int alen = T.aux_len[t];
    for (int j = lane; j < alen; j += WAVE) dst[plen + j] = asrc[j];
    if (lane == 0) {
      // handle "u" + 8 digits
      unsigned char* hd = dst + plen + alen;
      hd[0] = 'u';
      write_digits8(hd + 1, (long)(h[3] % 100000000ULL));
      // slot patches
      const int s0 = T.slot_off[t], sc = T.slot_cnt[t];
      for (int s = 0; s < sc; ++s) {
        long tgt = (long)(splitmix64(h[0] + 7919ULL * (s + 1)) %
                          (unsigned long long)F.universe);
        write_digits10(dst + T.slots[s0 + s], tgt);
      }
    }
    // offsets
    if (lane == 0) {
      text_off[i] = boff;
      aux_off[i] = (int)(boff + plen);
      poster_off[i] = (int)(boff + plen + alen);
      ent_off[i] = (int)ent_off_l[i];
      react_off[i] = (int)react_off_l[i];
    }
    // entity rows
    const int te0 = T.ent_off[t], tec = T.ent_cnt[t];
    for (int e = lane; e < tec; e += WAVE) {
      const int* row = T.ents + (long)(te0 + e) * 5;
      int* orow = entities + (ent_off_l[i] + e) * 5;
      orow[0] = row[0];
      orow[1] = row[1];
      orow[2] = row[2];
      orow[3] = row[4] > 0 ? row[3] + (int)boff : row[3];
      orow[4] = row[4];
    }
    // reaction rows (emoji ascending)
    if (lane == 0) {
      unsigned long long rmask = h[2] & (h[2] >> 13) & ((1ULL << NEMOJI) - 1);
      long r = react_off_l[i];
      for (int em = 0; em < NEMOJI; ++em) {
        if (rmask & (1ULL << em)) {
          react_emoji[r] = em;
          unsigned long long rh =
              splitmix64(h[2] + (unsigned long long)em);
          react_count[r] = (int)(rh % 500ULL + 1ULL);
          ++r;
        }
      }
    }
  }
}

// ---- phase 3: comments (one thread per comment) ----
__global__ void __launch_bounds__(256)
feed_comment_kernel(TemplateTables T, FeedParams F, const long* com_off_l,
                    long cblock, const long* cmsg_of,
                    long total_msg_pool, long n_comments,
                    unsigned char* pool, int* com_text_off, int* com_text_len,
                    int* com_handle_off, int* com_handle_len, int* com_views,
                    int* com_replies, int* com_react_off, int* com_react_cnt,
                    int* com_off_i32) {
  for (long c = blockIdx.x * (long)blockDim.x + threadIdx.x; c < n_comments;
       c += gridDim.x * (long)blockDim.x) {
    long i = cmsg_of[c];
    long k = i / F.P;
    long pidx = i % F.P;
    long cid = F.channel_ids[k];
    unsigned long long h[4];
    msg_hashes(F, cid, pidx, h);
    long cslot = c - com_off_l[i];
    unsigned long long ch =
        splitmix64(h[3] + (unsigned long long)cslot * 104729ULL);
    int ti = (int)(ch % (unsigned long long)T.n_ctexts);
    int tlen = T.ctext_len[ti];
    long dst = total_msg_pool + c * cblock;
    for (int j = 0; j < tlen; ++j) pool[dst + j] = T.ctext[T.ctext_off[ti] + j];
    pool[dst + tlen] = 'u';
    write_digits8((unsigned char*)pool + dst + tlen + 1,
                  (long)((ch >> 8) % 100000000ULL));
    com_text_off[c] = (int)dst;
    com_text_len[c] = tlen;
    com_handle_off[c] = (int)(dst + tlen);
    com_handle_len[c] = HANDLE_W;
    com_views[c] = (int)((ch >> 16) % 10000ULL);
    com_replies[c] = (int)((ch >> 24) % 50ULL);
    com_react_off[c] = 0;
    com_react_cnt[c] = 0;
    (void)com_off_i32;
  }
}

static TemplateTables make_tables(void** p, const long* s) {
  TemplateTables T;
  int k = 0;
  T.pool = (const unsigned char*)p[k++];
  T.pool_off = (const int*)p[k++];
  T.pool_len = (const int*)p[k++];
  T.text_len = (const int*)p[k++];
  T.ctype = (const int*)p[k++];
  T.tflags = (const int*)p[k++];
  T.aux = (const unsigned char*)p[k++];
  T.aux_off = (const int*)p[k++];
  T.aux_len = (const int*)p[k++];
  T.ents = (const int*)p[k++];
  T.ent_off = (const int*)p[k++];
  T.ent_cnt = (const int*)p[k++];
  T.slots = (const int*)p[k++];
  T.slot_off = (const int*)p[k++];
  T.slot_cnt = (const int*)p[k++];
  T.cdf = (const double*)p[k++];
  T.ctext = (const unsigned char*)p[k++];
  T.ctext_off = (const int*)p[k++];
  T.ctext_len = (const int*)p[k++];
  T.n_templates = (int)s[0];
  T.n_ctexts = (int)s[1];
  return T;
}

static FeedParams make_params(const long* s, const void* channel_ids) {
  FeedParams F;
  F.seed = s[2];
  F.universe = s[3];
  F.base_date = s[4];
  F.date_step = s[5];
  F.comment_rate_permille = s[6];
  F.max_comments = s[7];
  F.K = (int)s[8];
  F.P = (int)s[9];
  F.channel_ids = (const long*)channel_ids;
  return F;
}

}  // namespace crawl

extern "C" {

int crawl_feed_meta(void** tbl_ptrs, const long* scalars,
                    const void* channel_ids, void** out_ptrs, int grid,
                    void* stream) {
  crawl::TemplateTables T = crawl::make_tables(tbl_ptrs, scalars);
  crawl::FeedParams F = crawl::make_params(scalars, channel_ids);
  int k = 0;
  long* chat_id = (long*)out_ptrs[k++];
  long* msg_id = (long*)out_ptrs[k++];
  int* date = (int*)out_ptrs[k++];
  int* content_type = (int*)out_ptrs[k++];
  int* views = (int*)out_ptrs[k++];
  int* forwards = (int*)out_ptrs[k++];
  int* media_album_id = (int*)out_ptrs[k++];
  int* channel_idx = (int*)out_ptrs[k++];
  int* flags = (int*)out_ptrs[k++];
  int* text_len = (int*)out_ptrs[k++];
  int* aux_len = (int*)out_ptrs[k++];
  int* ent_cnt = (int*)out_ptrs[k++];
  int* react_cnt = (int*)out_ptrs[k++];
  int* com_cnt = (int*)out_ptrs[k++];
  int* poster_len = (int*)out_ptrs[k++];
  int* reply_count = (int*)out_ptrs[k++];
  long* block_len = (long*)out_ptrs[k++];
  int* tidx = (int*)out_ptrs[k++];
  hipLaunchKernelGGL(crawl::feed_meta_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, T, F, chat_id, msg_id, date,
                     content_type, views, forwards, media_album_id,
                     channel_idx, flags, text_len, aux_len, ent_cnt,
                     react_cnt, com_cnt, poster_len, reply_count, block_len,
                     tidx);
  return (int)hipGetLastError();
}

int crawl_feed_fill(void** tbl_ptrs, const long* scalars,
                    const void* channel_ids, const void* tidx,
                    const void* block_off, const void* ent_off_l,
                    const void* react_off_l, void** out_ptrs, int grid,
                    void* stream) {
  crawl::TemplateTables T = crawl::make_tables(tbl_ptrs, scalars);
  crawl::FeedParams F = crawl::make_params(scalars, channel_ids);
  int k = 0;
  unsigned char* pool = (unsigned char*)out_ptrs[k++];
  long* text_off = (long*)out_ptrs[k++];
  int* aux_off = (int*)out_ptrs[k++];
  int* poster_off = (int*)out_ptrs[k++];
  int* ent_off = (int*)out_ptrs[k++];
  int* react_off = (int*)out_ptrs[k++];
  int* entities = (int*)out_ptrs[k++];
  int* react_emoji = (int*)out_ptrs[k++];
  int* react_count = (int*)out_ptrs[k++];
  hipLaunchKernelGGL(crawl::feed_fill_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, T, F, (const int*)tidx,
                     (const long*)block_off, (const long*)ent_off_l,
                     (const long*)react_off_l, pool, text_off, aux_off,
                     poster_off, ent_off, react_off, entities, react_emoji,
                     react_count);
  return (int)hipGetLastError();
}

int crawl_feed_comments(void** tbl_ptrs, const long* scalars,
                        const void* channel_ids, const void* com_off_l,
                        long cblock, const void* cmsg_of,
                        long total_msg_pool, long n_comments, void** out_ptrs,
                        int grid, void* stream) {
  crawl::TemplateTables T = crawl::make_tables(tbl_ptrs, scalars);
  crawl::FeedParams F = crawl::make_params(scalars, channel_ids);
  int k = 0;
  unsigned char* pool = (unsigned char*)out_ptrs[k++];
  int* com_text_off = (int*)out_ptrs[k++];
  int* com_text_len = (int*)out_ptrs[k++];
  int* com_handle_off = (int*)out_ptrs[k++];
  int* com_handle_len = (int*)out_ptrs[k++];
  int* com_views = (int*)out_ptrs[k++];
  int* com_replies = (int*)out_ptrs[k++];
  int* com_react_off = (int*)out_ptrs[k++];
  int* com_react_cnt = (int*)out_ptrs[k++];
  hipLaunchKernelGGL(crawl::feed_comment_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, T, F, (const long*)com_off_l,
                     cblock, (const long*)cmsg_of,
                     total_msg_pool, n_comments, pool, com_text_off,
                     com_text_len, com_handle_off, com_handle_len, com_views,
                     com_replies, com_react_off, com_react_cnt, nullptr);
  return (int)hipGetLastError();
}

}  // extern "C"
