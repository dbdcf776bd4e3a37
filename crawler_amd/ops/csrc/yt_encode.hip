// CDNA4 YouTube video -> Post JSONL encoder (BASELINE config #4).
//
// MI355X-native equivalent of the reference's convertVideoToPost
// (crawler/youtube/youtube_crawler.go:530-836): engagement formula,
// thumbnail preference, ISO-duration-null handling, OCR rows, performance
// scores, description URL extraction with trailing-punctuation trim +
// dedup (youtube_crawler.go:489-510), sanitized media filename, and the
// same Go-JSON encoding rules. One wave per video, grid-stride; two
// passes share the templated emitter (oracle:
// crawler_amd/youtube/batch.py encode_yt_batch).

#include "common.h"

namespace crawl {

struct YtView {
  const int* vid_off;
  const int* channel_idx;
  const long* published;
  const long* views;
  const int* likes;
  const int* comments;
  const int* duration_s;
  const int* lang;
  const long* title_off;
  const int* title_len;
  const long* desc_off;
  const int* desc_len;
  const unsigned char* pool;
  const int* ch_id_off;
  const int* ch_title_off;
  const int* ch_title_len;
  const int* ch_desc_off;
  const int* ch_desc_len;
  const long* ch_subs;
  const int* ch_videos;
  const long* ch_views;
  const int* ch_country_off;
  const int* ch_country_len;
  const long* ch_published;
  const unsigned char* label;
  int label_len;
  const unsigned char* lang_pool;  // "enru"
  const unsigned char* created_str;
  int created_len;
  const unsigned char* capture_str;
  int capture_len;
  int n;
};

#define YT_MAX_URLS 8

DEV bool url_stop_char(unsigned char c) {
  return c <= ' ' || c == '<' || c == '>' || c == '"';
}
DEV bool url_trim_char(unsigned char c) {
  return c == ',' || c == '.' || c == ';' || c == ':' || c == '!' ||
         c == '?' || c == '(' || c == ')' || c == '\'' || c == '"';
}

// Extract up to YT_MAX_URLS http(s) URLs from text; stores (start,len)
// pairs referencing the text itself (no per-thread URL buffers -> no
// scratch spill). Oracle: convert.extract_urls.
DEV int extract_urls(const unsigned char* s, int n, int* starts,
                     int* lens) {
  int cnt = 0;
  int p = 0;
  while (p + 7 <= n && cnt < YT_MAX_URLS) {
    bool http = s[p] == 'h' && s[p + 1] == 't' && s[p + 2] == 't' &&
                s[p + 3] == 'p';
    if (!http) { ++p; continue; }
    int q = p + 4;
    if (q < n && s[q] == 's') ++q;
    if (!(q + 2 < n && s[q] == ':' && s[q + 1] == '/' && s[q + 2] == '/')) {
      ++p;
      continue;
    }
    int end = p;
    while (end < n && !url_stop_char(s[end])) ++end;
    int e2 = end;
    while (e2 > p && url_trim_char(s[e2 - 1])) --e2;
    int len = e2 - p;
    if (len > 0) {
      bool dup = false;
      for (int k = 0; k < cnt && !dup; ++k) {
        if (lens[k] != len) continue;
        bool eq = true;
        for (int j = 0; j < len; ++j) eq &= (s[starts[k] + j] == s[p + j]);
        dup = eq;
      }
      if (!dup) {
        starts[cnt] = p;
        lens[cnt] = len;
        ++cnt;
      }
    }
    p = end > p ? end : p + 1;
  }
  return cnt;
}

// YEmit extends the shared JsonEmit (common.h) with the sanitized
// filename writer (runs of non-[A-Za-z0-9._-] -> '_', cap 80;
// convert.sanitize_filename).

template <bool W>
struct YEmit : JsonEmit<W> {
  DEV void rfc(long secs) { this->rfc3339(secs); }
  DEV void sanitized(const unsigned char* s, int n) {
    int o = 0;
    int p = 0;
    while (p < n && o < 80) {
      unsigned char c = s[p];
      bool ok = is_word_char(c) || c == '.' || c == '-';
      if (ok) {
        if (W && lane_id() == 0) this->out[this->cur + o] = c;
        ++o;
        ++p;
      } else {
        if (W && lane_id() == 0) this->out[this->cur + o] = '_';
        ++o;
        while (p < n) {
          unsigned char d = s[p];
          if (is_word_char(d) || d == '.' || d == '-') break;
          ++p;
        }
      }
    }
    this->cur += o;
  }
};

template <bool W>
DEV int emit_yt(const YtView& B, int i, unsigned char* out) {
  YEmit<W> e{};
  e.out = out;
  e.cur = 0;
  const int c = B.channel_idx[i];
  const unsigned char* vid = B.pool + B.vid_off[i];
  const unsigned char* cid = B.pool + B.ch_id_off[c];
  const unsigned char* title = B.pool + B.title_off[i];
  const int title_n = B.title_len[i];
  const unsigned char* desc = B.pool + B.desc_off[i];
  const int desc_n = B.desc_len[i];
  const unsigned char* ctitle = B.pool + B.ch_title_off[c];
  const int ctitle_n = B.ch_title_len[c];
  const long views = B.views[i];
  const int likes = B.likes[i];
  const int comments = B.comments[i];
  const long long engagement =
      (long long)likes + comments + views / 100;

  int url_starts[YT_MAX_URLS];
  int url_lens[YT_MAX_URLS];
  const int n_urls = extract_urls(desc, desc_n, url_starts, url_lens);

  auto video_url = [&]() {
    LIT(e, "https://www.youtube.com/watch?v=");
    e.raw(vid, 11);
  };
  auto channel_url = [&]() {
    LIT(e, "https://www.youtube.com/channel/");
    e.raw(cid, 24);
  };

  LIT(e, "{\"post_link\":\"");
  video_url();
  LIT(e, "\",\"channel_id\":\"");
  e.raw(cid, 24);
  LIT(e, "\",\"post_uid\":\"");
  e.raw(vid, 11);
  LIT(e, "\",\"url\":\"");
  video_url();
  LIT(e, "\",\"published_at\":\"");
  e.rfc(B.published[i]);
  LIT(e, "\",\"created_at\":\"");
  e.raw(B.created_str, B.created_len);
  LIT(e, "\",\"language_code\":\"");
  e.raw(B.lang_pool + B.lang[i] * 2, 2);
  LIT(e, "\",\"engagement\":");
  e.i64(engagement);
  LIT(e, ",\"view_count\":");
  e.i64(views);
  LIT(e, ",\"like_count\":");
  e.i64(likes);
  LIT(e, ",\"share_count\":0,\"comment_count\":");
  e.i64(comments);
  LIT(e, ",\"crawl_label\":\"");
  e.esc(B.label, B.label_len);
  LIT(e, "\",\"list_ids\":null,\"channel_name\":\"");
  e.esc(ctitle, ctitle_n);
  LIT(e, "\",\"search_terms\":null,\"search_term_ids\":null,"
         "\"project_ids\":null,\"exercise_ids\":null,\"label_data\":null,"
         "\"labels_metadata\":null,\"project_labeled_post_ids\":null,"
         "\"labeler_ids\":null,\"all_labels\":null,\"label_ids\":null,"
         "\"is_ad\":false,\"transcript_text\":\"\",\"image_text\":\"\","
         "\"video_length\":");
  if (B.duration_s[i] < 0) LIT(e, "null");
  else e.i64(B.duration_s[i]);
  LIT(e, ",\"is_verified\":null,\"channel_data\":{\"channel_id\":\"");
  e.raw(cid, 24);
  LIT(e, "\",\"channel_name\":\"");
  e.esc(ctitle, ctitle_n);
  LIT(e, "\",\"channel_description\":\"");
  e.esc(B.pool + B.ch_desc_off[c], B.ch_desc_len[c]);
  LIT(e, "\",\"channel_profile_image\":\"https://i.ytimg.com/ch/");
  e.raw(cid, 24);
  LIT(e, "/default.jpg\",\"channel_engagement_data\":{\"follower_count\":");
  e.i64(B.ch_subs[c]);
  LIT(e, ",\"following_count\":0,\"like_count\":0,\"post_count\":");
  e.i64(B.ch_videos[c]);
  LIT(e, ",\"views_count\":");
  e.i64(B.ch_views[c]);
  LIT(e, ",\"comment_count\":0,\"share_count\":0},"
         "\"channel_url_external\":\"");
  channel_url();
  LIT(e, "\",\"channel_url\":\"");
  channel_url();
  LIT(e, "\",\"country_code\":\"");
  e.esc(B.pool + B.ch_country_off[c], B.ch_country_len[c]);
  LIT(e, "\",\"published_at\":\"");
  e.rfc(B.ch_published[c]);
  LIT(e, "\"},\"platform_name\":\"youtube\",\"shared_id\":null,"
         "\"quoted_id\":null,\"replied_id\":null,\"ai_label\":null,"
         "\"root_post_id\":null,\"engagement_steps_count\":0,"
         "\"ocr_data\":[{\"ocr_text\":\"YouTube thumbnail: default "
         "quality\",\"thumb_url\":\"https://i.ytimg.com/vi/");
  e.raw(vid, 11);
  LIT(e, "/default.jpg\"},{\"ocr_text\":\"YouTube thumbnail: high "
         "quality\",\"thumb_url\":\"https://i.ytimg.com/vi/");
  e.raw(vid, 11);
  LIT(e, "/hq.jpg\"}],\"performance_scores\":{\"likes\":");
  e.i64(likes);
  LIT(e, ",\"shares\":null,\"comments\":");
  e.i64(comments);
  LIT(e, ",\"views\":");
  e.i64(views);
  LIT(e, "},\"has_embed_media\":true,\"description\":\"");
  e.esc(desc, desc_n);
  LIT(e, "\",\"repost_channel_data\":null,\"post_type\":[\"video\"],"
         "\"inner_link\":{},\"post_title\":\"");
  e.esc(title, title_n);
  LIT(e, "\",\"media_data\":{\"document_name\":\"");
  e.raw(vid, 11);
  LIT(e, "-");
  e.sanitized(title, title_n);
  LIT(e, ".mp4\"},\"is_reply\":null,\"ad_fields\":null,\"likes_count\":");
  e.i64(likes);
  LIT(e, ",\"shares_count\":0,\"comments_count\":");
  e.i64(comments);
  LIT(e, ",\"views_count\":");
  e.i64(views);
  LIT(e, ",\"searchable_text\":\"");
  e.esc(title, title_n);
  LIT(e, " ");
  e.esc(desc, desc_n);
  LIT(e, "\",\"all_text\":\"");
  e.esc(title, title_n);
  LIT(e, " ");
  e.esc(desc, desc_n);
  LIT(e, "\",\"contrast_agent_project_ids\":null,\"agent_ids\":null,"
         "\"segment_ids\":null,\"thumb_url\":\"https://i.ytimg.com/vi/");
  e.raw(vid, 11);
  LIT(e, "/hq.jpg\",\"media_url\":\"\",\"comments\":null,"
         "\"reactions\":null,\"outlinks\":[");
  for (int k = 0; k < n_urls; ++k) {
    if (k) LIT(e, ",");
    LIT(e, "\"");
    e.esc(desc + url_starts[k], url_lens[k]);
    LIT(e, "\"");
  }
  LIT(e, "],\"capture_time\":\"");
  e.raw(B.capture_str, B.capture_len);
  LIT(e, "\",\"handle\":\"");
  e.esc(ctitle, ctitle_n);
  LIT(e, "\"}\n");
  return e.cur;
}

__global__ void __launch_bounds__(256, 4)
yt_measure_kernel(YtView B, int* line_len) {
  const int wave = wave_id();
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += gridDim.x * 4) {
    int len = emit_yt<false>(B, i, nullptr);
    if (lane_id() == 0) line_len[i] = len;
  }
}

__global__ void __launch_bounds__(256, 4)
yt_write_kernel(YtView B, const long* line_off, unsigned char* out) {
  const int wave = wave_id();
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += gridDim.x * 4) {
    emit_yt<true>(B, i, out + line_off[i]);
  }
}

static YtView yt_make_view(void** p, const long* s) {
  YtView B;
  int k = 0;
  B.vid_off = (const int*)p[k++];
  B.channel_idx = (const int*)p[k++];
  B.published = (const long*)p[k++];
  B.views = (const long*)p[k++];
  B.likes = (const int*)p[k++];
  B.comments = (const int*)p[k++];
  B.duration_s = (const int*)p[k++];
  B.lang = (const int*)p[k++];
  B.title_off = (const long*)p[k++];
  B.title_len = (const int*)p[k++];
  B.desc_off = (const long*)p[k++];
  B.desc_len = (const int*)p[k++];
  B.pool = (const unsigned char*)p[k++];
  B.ch_id_off = (const int*)p[k++];
  B.ch_title_off = (const int*)p[k++];
  B.ch_title_len = (const int*)p[k++];
  B.ch_desc_off = (const int*)p[k++];
  B.ch_desc_len = (const int*)p[k++];
  B.ch_subs = (const long*)p[k++];
  B.ch_videos = (const int*)p[k++];
  B.ch_views = (const long*)p[k++];
  B.ch_country_off = (const int*)p[k++];
  B.ch_country_len = (const int*)p[k++];
  B.ch_published = (const long*)p[k++];
  B.label = (const unsigned char*)p[k++];
  B.lang_pool = (const unsigned char*)p[k++];
  B.created_str = (const unsigned char*)p[k++];
  B.capture_str = (const unsigned char*)p[k++];
  B.n = (int)s[0];
  B.label_len = (int)s[1];
  B.created_len = (int)s[2];
  B.capture_len = (int)s[3];
  return B;
}

}  // namespace crawl

extern "C" {

int crawl_yt_ptr_count() { return 28; }

int crawl_yt_measure(void** ptrs, const long* scalars, void* line_len,
                     int grid, void* stream) {
  crawl::YtView B = crawl::yt_make_view(ptrs, scalars);
  hipLaunchKernelGGL(crawl::yt_measure_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, (int*)line_len);
  return (int)hipGetLastError();
}

int crawl_yt_write(void** ptrs, const long* scalars, const void* line_off,
                   void* out, int grid, void* stream) {
  crawl::YtView B = crawl::yt_make_view(ptrs, scalars);
  hipLaunchKernelGGL(crawl::yt_write_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, (const long*)line_off,
                     (unsigned char*)out);
  return (int)hipGetLastError();
}

}  // extern "C"
