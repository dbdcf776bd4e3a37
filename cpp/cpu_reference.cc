// CPU reference implementation of the per-post hot path (parse + link
// extraction + Go-JSON encode) over the packed batch format — the measured
// "reference logic on host CPU, rate limiters disabled" baseline that
// BASELINE.md calls for. Multithreaded (std::thread) so the comparison is
// against a full host socket, not one core.
//
// Semantics match crawler_amd/ops/golden.py (and therefore the reference's
// telegramhelper/tdutils.go); the harness tools/measure_cpu_baseline.py
// asserts byte-equality against the Python oracle before timing.
//
// Input: a directory of raw little-endian arrays dumped by the harness
// (see BatchDump in measure_cpu_baseline.py). Output: posts/sec JSON line.

#include <atomic>
#include <chrono>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <string>
#include <thread>
#include <vector>

namespace {

std::vector<char> read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary | std::ios::ate);
  if (!f) { fprintf(stderr, "missing %s\n", path.c_str()); exit(2); }
  size_t n = (size_t)f.tellg();
  f.seekg(0);
  std::vector<char> buf(n);
  f.read(buf.data(), n);
  return buf;
}

template <typename T>
struct Arr {
  std::vector<char> raw;
  const T* p = nullptr;
  size_t n = 0;
  void load(const std::string& path) {
    raw = read_file(path);
    p = (const T*)raw.data();
    n = raw.size() / sizeof(T);
  }
  const T& operator[](size_t i) const { return p[i]; }
};

struct Batch {
  Arr<int64_t> chat_id, msg_id, text_off;
  Arr<int32_t> date, content_type, views, forwards, media_album_id,
      channel_idx, flags, text_len, aux_off, aux_len, ent_off, ent_cnt,
      react_off, react_cnt, com_off, com_cnt, poster_off, poster_len;
  Arr<uint8_t> pool;
  Arr<int32_t> entities;
  Arr<int32_t> react_emoji, react_count;
  Arr<int32_t> com_text_off, com_text_len, com_handle_off, com_handle_len,
      com_views, com_replies, com_react_off, com_react_cnt;
  Arr<int64_t> ch_chat_id;
  Arr<int32_t> ch_member, ch_postcount, ch_totalviews, ch_user_off,
      ch_user_len, ch_title_off, ch_title_len;
  int n = 0;
};

// ---- Go JSON escaping (oracle: models/post.py go_json_escape) ----

inline void esc_append(std::string& o, const uint8_t* s, int n) {
  static const char* hexd = "0123456789abcdef";
  for (int p = 0; p < n; ++p) {
    uint8_t c = s[p];
    switch (c) {
      case '"': o += "\\\""; continue;
      case '\\': o += "\\\\"; continue;
      case '\n': o += "\\n"; continue;
      case '\r': o += "\\r"; continue;
      case '\t': o += "\\t"; continue;
      default: break;
    }
    if (c < 0x20 || c == '<' || c == '>' || c == '&') {
      o += "\\u00";
      o += hexd[c >> 4];
      o += hexd[c & 15];
      continue;
    }
    if (c == 0xE2 && p + 2 < n && s[p + 1] == 0x80 &&
        (s[p + 2] == 0xA8 || s[p + 2] == 0xA9)) {
      o += (s[p + 2] == 0xA8) ? "\\u2028" : "\\u2029";
      p += 2;
      continue;
    }
    o += (char)c;
  }
}

inline void rfc3339(std::string& o, int64_t secs) {
  int64_t days = secs / 86400, rem = secs % 86400;
  if (rem < 0) { rem += 86400; days -= 1; }
  int hh = (int)(rem / 3600), mm = (int)((rem % 3600) / 60),
      ss = (int)(rem % 60);
  int64_t z = days + 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  int64_t doe = z - era * 146097;
  int64_t yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t y = yoe + era * 400;
  int64_t doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  int64_t mp = (5 * doy + 2) / 153;
  int64_t d = doy - (153 * mp + 2) / 5 + 1;
  int64_t m = mp + (mp < 10 ? 3 : -9);
  y += (m <= 2);
  char buf[32];
  snprintf(buf, sizeof buf, "%04d-%02d-%02dT%02d:%02d:%02dZ", (int)y,
           (int)m, (int)d, hh, mm, ss);
  o += buf;
}

// ---- link extraction (oracle: ops/golden.py) ----

inline bool is_letter(uint8_t c) {
  return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z');
}
inline bool is_word(uint8_t c) {
  return is_letter(c) || (c >= '0' && c <= '9') || c == '_';
}

bool is_reserved(const std::string& n) {
  static const char* tbl[] = {"joinchat", "addlist", "addstickers",
                              "addtheme", "setlanguage", "share", "proxy",
                              "socks", "login", "confirm"};
  for (auto* t : tbl)
    if (n == t) return true;
  return false;
}

int extract_name_at(const uint8_t* s, int hi, int q, std::string& name) {
  if (q >= hi || !is_letter(s[q])) return 0;
  int len = 0;
  while (len < 32 && q + len < hi && is_word(s[q + len])) ++len;
  if (len < 5) return 0;
  name.assign((const char*)s + q, len);
  for (auto& c : name) c = (char)tolower((unsigned char)c);
  if (is_reserved(name)) return 0;
  return len;
}

int find_tme(const uint8_t* s, int from, int hi) {
  for (int p = from; p + 5 <= hi; ++p)
    if (s[p] == 't' && s[p + 1] == '.' && s[p + 2] == 'm' &&
        s[p + 3] == 'e' && s[p + 4] == '/')
      return p;
  return -1;
}

void utf16_to_bytes(const uint8_t* s, int n, int off16, int len16,
                    int* pstart, int* pend) {
  int i = 0, u16 = 0, start = -1;
  const int t1 = off16, t2 = off16 + len16;
  while (i < n) {
    if (u16 == t1 && start < 0) start = i;
    if (u16 == t2) { *pstart = start; *pend = i; return; }
    uint8_t b = s[i];
    int size, units;
    if (b < 0x80) { size = 1; units = 1; }
    else if (b < 0xE0) { size = 2; units = 1; }
    else if (b < 0xF0) { size = 3; units = 1; }
    else { size = 4; units = 2; }
    u16 += units;
    i += size;
  }
  if (start < 0) { *pstart = 0; *pend = 0; return; }
  *pstart = start;
  *pend = n;
}

struct Links {
  std::vector<std::string> names;
  void add(const std::string& n) {
    for (auto& e : names)
      if (e == n) return;
    if (names.size() < 8) names.push_back(n);
  }
};

void extract_links(const Batch& B, int i, Links& L) {
  int ct = B.content_type[i];
  if (ct > 6) return;
  const uint8_t* text = B.pool.p + B.text_off[i];
  const int tn = B.text_len[i];
  std::string name;
  const int e0 = B.ent_off[i], ec = B.ent_cnt[i];
  for (int e = e0; e < e0 + ec; ++e) {
    const int32_t* row = B.entities.p + (size_t)e * 5;
    if (row[0] == 1) {
      const uint8_t* url = B.pool.p + row[3];
      int p = find_tme(url, 0, row[4]);
      if (p >= 0 && extract_name_at(url, row[4], p + 5, name)) L.add(name);
    } else {
      int lo, hi;
      utf16_to_bytes(text, tn, row[1], row[2], &lo, &hi);
      if (!(lo < hi && hi <= tn)) continue;
      if (row[0] == 0) {
        for (int q = lo; q < hi; ++q) {
          if (is_letter(text[q]) && q + 5 <= hi && is_word(text[q + 1]) &&
              is_word(text[q + 2]) && is_word(text[q + 3]) &&
              is_word(text[q + 4])) {
            int len = 0;
            while (len < 32 && q + len < hi && is_word(text[q + len])) ++len;
            name.assign((const char*)text + q, len);
            for (auto& c : name) c = (char)tolower((unsigned char)c);
            L.add(name);
            break;
          }
        }
      } else {
        int p = find_tme(text, lo, hi);
        if (p >= 0 && extract_name_at(text, hi, p + 5, name)) L.add(name);
      }
    }
  }
  int cursor = 0, from = 0;
  while (true) {
    int p = find_tme(text, from, tn);
    if (p < 0) break;
    from = p + 1;
    if (p < cursor) continue;
    if (int len = extract_name_at(text, tn, p + 5, name)) {
      L.add(name);
      cursor = p + 5 + len;
      from = cursor;
    }
  }
}

// ---- the line emitter (mirror of parse_encode.hip emit_line) ----

struct Tables {
  std::vector<std::string> emoji, ctnames;
  std::string created, capture;
};

void emit_line(const Batch& B, const Tables& T, int i, std::string& o,
               const Links& L) {
  const int c = B.channel_idx[i];
  const uint8_t* user = B.pool.p + B.ch_user_off[c];
  const int user_n = B.ch_user_len[c];
  const uint8_t* title = B.pool.p + B.ch_title_off[c];
  const int title_n = B.ch_title_len[c];
  const long pub = B.msg_id[i] >> 20;
  const int ct = B.content_type[i];
  const int ncom = B.com_cnt[i];
  char num[32];

  auto I = [&](long long v) { o += std::to_string(v); };
  auto post_link = [&]() {
    if (!user_n) { o += "\"\""; return; }
    o += "\"https://t.me/";
    esc_append(o, user, user_n);
    o += '/';
    I(pub);
    if (B.media_album_id[i] != 0) o += "?single";
    o += '"';
  };
  (void)num;
  o += "{\"post_link\":";
  post_link();
  o += ",\"channel_id\":\"";
  I(B.chat_id[i]);
  o += "\",\"post_uid\":\"";
  I(pub);
  o += '-';
  esc_append(o, user, user_n);
  o += "\",\"url\":";
  post_link();
  o += ",\"published_at\":\"";
  rfc3339(o, B.date[i]);
  o += "\",\"created_at\":\"";
  o += T.created;
  o += "\",\"language_code\":\"\",\"engagement\":";
  I(B.views[i]);
  o += ",\"view_count\":";
  I(B.views[i]);
  o += ",\"like_count\":0,\"share_count\":";
  I(B.forwards[i]);
  o += ",\"comment_count\":";
  I(ncom);
  o += ",\"crawl_label\":\"\",\"list_ids\":null,\"channel_name\":\"";
  esc_append(o, title, title_n);
  o += "\",\"search_terms\":null,\"search_term_ids\":null,"
       "\"project_ids\":null,\"exercise_ids\":null,\"label_data\":null,"
       "\"labels_metadata\":null,\"project_labeled_post_ids\":null,"
       "\"labeler_ids\":null,\"all_labels\":null,\"label_ids\":null,"
       "\"is_ad\":false,\"transcript_text\":\"\",\"image_text\":\"\","
       "\"video_length\":null,\"is_verified\":null,"
       "\"channel_data\":{\"channel_id\":\"";
  I(B.chat_id[i]);
  o += "\",\"channel_name\":\"";
  esc_append(o, title, title_n);
  o += "\",\"channel_description\":\"\",\"channel_profile_image\":\"\","
       "\"channel_engagement_data\":{\"follower_count\":";
  I(B.ch_member[c]);
  o += ",\"following_count\":0,\"like_count\":0,\"post_count\":";
  I(B.ch_postcount[c]);
  o += ",\"views_count\":";
  I(B.ch_totalviews[c]);
  o += ",\"comment_count\":0,\"share_count\":0},"
       "\"channel_url_external\":\"https://t.me/c/";
  esc_append(o, user, user_n);
  o += "\",\"channel_url\":\"https://t.me/c/";
  esc_append(o, user, user_n);
  o += "\",\"country_code\":\"\",\"published_at\":"
       "\"0001-01-01T00:00:00Z\"},\"platform_name\":\"Telegram\","
       "\"shared_id\":null,\"quoted_id\":null,\"replied_id\":null,"
       "\"ai_label\":null,\"root_post_id\":null,"
       "\"engagement_steps_count\":0,\"ocr_data\":null,"
       "\"performance_scores\":{\"likes\":null,\"shares\":null,"
       "\"comments\":null,\"views\":0},\"has_embed_media\":null,"
       "\"description\":\"";
  {
    const uint8_t* d = nullptr;
    int dn = 0;
    if (ct == 0 || ct == 1 || ct == 2 || ct == 4 || ct == 14) {
      d = B.pool.p + B.text_off[i];
      dn = B.text_len[i];
    } else if (ct == 3 || ct == 9 || ct == 10 || ct == 11) {
      d = B.pool.p + B.aux_off[i];
      dn = B.aux_len[i];
    }
    if (dn) esc_append(o, d, dn);
  }
  o += "\",\"repost_channel_data\":null,\"post_type\":[\"";
  o += T.ctnames[ct];
  o += "\"],\"inner_link\":{},\"post_title\":null,\"media_data\":"
       "{\"document_name\":\"\"},\"is_reply\":null,\"ad_fields\":null,"
       "\"likes_count\":0,\"shares_count\":";
  I(B.forwards[i]);
  o += ",\"comments_count\":";
  I(ncom);
  o += ",\"views_count\":";
  I(B.views[i]);
  o += ",\"searchable_text\":\"\",\"all_text\":\"\","
       "\"contrast_agent_project_ids\":null,\"agent_ids\":null,"
       "\"segment_ids\":null,\"thumb_url\":\"\",\"media_url\":\"";
  if ((ct == 3 || ct == 8) && (B.flags[i] & 2)) {
    o += "AgAD";
    I(pub);
    o += 'v';
  }
  o += "\",\"comments\":[";
  {
    const int c0 = B.com_off[i];
    for (int k = 0; k < ncom; ++k) {
      if (k) o += ',';
      const int cc = c0 + k;
      o += "{\"text\":\"";
      esc_append(o, B.pool.p + B.com_text_off[cc], B.com_text_len[cc]);
      o += "\",\"reactions\":{";
      const int r0 = B.com_react_off[cc], rc = B.com_react_cnt[cc];
      for (int r = 0; r < rc; ++r) {
        if (r) o += ',';
        o += '"';
        o += T.emoji[B.react_emoji[r0 + r]];
        o += "\":";
        I(B.react_count[r0 + r]);
      }
      o += "},\"view_count\":";
      I(B.com_views[cc]);
      o += ",\"reply_count\":";
      I(B.com_replies[cc]);
      o += ",\"handle\":\"";
      esc_append(o, B.pool.p + B.com_handle_off[cc], B.com_handle_len[cc]);
      o += "\"}";
    }
  }
  o += "],\"reactions\":{";
  {
    const int r0 = B.react_off[i], rc = B.react_cnt[i];
    for (int r = 0; r < rc; ++r) {
      if (r) o += ',';
      o += '"';
      o += T.emoji[B.react_emoji[r0 + r]];
      o += "\":";
      I(B.react_count[r0 + r]);
    }
  }
  o += "},\"outlinks\":[";
  for (size_t k = 0; k < L.names.size(); ++k) {
    if (k) o += ',';
    o += '"';
    o += L.names[k];
    o += '"';
  }
  o += "],\"capture_time\":\"";
  o += T.capture;
  o += "\",\"handle\":\"";
  esc_append(o, B.pool.p + B.poster_off[i], B.poster_len[i]);
  o += "\"}\n";
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 5) {
    fprintf(stderr,
            "usage: %s <dump_dir> <threads> <reps> <out.jsonl|-> \n",
            argv[0]);
    return 2;
  }
  std::string dir = argv[1];
  int threads = atoi(argv[2]);
  int reps = atoi(argv[3]);
  std::string outpath = argv[4];

  Batch B;
  auto L = [&](auto& a, const char* name) { a.load(dir + "/" + name); };
  L(B.chat_id, "chat_id"); L(B.msg_id, "msg_id"); L(B.text_off, "text_off");
  L(B.date, "date"); L(B.content_type, "content_type"); L(B.views, "views");
  L(B.forwards, "forwards"); L(B.media_album_id, "media_album_id");
  L(B.channel_idx, "channel_idx"); L(B.flags, "flags");
  L(B.text_len, "text_len"); L(B.aux_off, "aux_off");
  L(B.aux_len, "aux_len"); L(B.ent_off, "ent_off");
  L(B.ent_cnt, "ent_cnt"); L(B.react_off, "react_off");
  L(B.react_cnt, "react_cnt"); L(B.com_off, "com_off");
  L(B.com_cnt, "com_cnt"); L(B.poster_off, "poster_off");
  L(B.poster_len, "poster_len"); L(B.pool, "pool");
  L(B.entities, "entities"); L(B.react_emoji, "react_emoji");
  L(B.react_count, "react_count"); L(B.com_text_off, "com_text_off");
  L(B.com_text_len, "com_text_len");
  L(B.com_handle_off, "com_handle_off");
  L(B.com_handle_len, "com_handle_len"); L(B.com_views, "com_views");
  L(B.com_replies, "com_replies"); L(B.com_react_off, "com_react_off");
  L(B.com_react_cnt, "com_react_cnt"); L(B.ch_chat_id, "ch_chat_id");
  L(B.ch_member, "ch_member"); L(B.ch_postcount, "ch_postcount");
  L(B.ch_totalviews, "ch_totalviews"); L(B.ch_user_off, "ch_user_off");
  L(B.ch_user_len, "ch_user_len"); L(B.ch_title_off, "ch_title_off");
  L(B.ch_title_len, "ch_title_len");
  B.n = (int)B.msg_id.n;

  Tables T;
  {
    auto meta = read_file(dir + "/tables.txt");
    std::string s(meta.begin(), meta.end());
    size_t pos = 0;
    auto next = [&]() {
      size_t e = s.find('\n', pos);
      std::string line = s.substr(pos, e - pos);
      pos = e + 1;
      return line;
    };
    int ne = atoi(next().c_str());
    for (int k = 0; k < ne; ++k) T.emoji.push_back(next());
    int nc = atoi(next().c_str());
    for (int k = 0; k < nc; ++k) T.ctnames.push_back(next());
    T.created = next();
    T.capture = next();
  }

  // correctness output (rep 0, single-threaded order)
  std::vector<std::string> outs((size_t)threads);
  auto run = [&](int rep) {
    std::vector<std::thread> ts;
    std::atomic<int> next_block{0};
    const int BLK = 4096;
    for (int t = 0; t < threads; ++t) {
      outs[t].clear();
      ts.emplace_back([&, t]() {
        std::string local;
        local.reserve(1 << 22);
        while (true) {
          int b = next_block.fetch_add(1);
          int lo = b * BLK, hi = std::min(B.n, lo + BLK);
          if (lo >= B.n) break;
          for (int i = lo; i < hi; ++i) {
            Links L2;
            extract_links(B, i, L2);
            emit_line(B, T, i, local, L2);
          }
        }
        outs[t] = std::move(local);
      });
    }
    for (auto& th : ts) th.join();
    (void)rep;
  };

  // warmup + write correctness dump (ordered single-thread pass)
  if (outpath != "-") {
    std::string all;
    for (int i = 0; i < B.n; ++i) {
      Links L2;
      extract_links(B, i, L2);
      emit_line(B, T, i, all, L2);
    }
    std::ofstream f(outpath, std::ios::binary);
    f.write(all.data(), (std::streamsize)all.size());
  }

  run(0);  // warmup
  auto t0 = std::chrono::steady_clock::now();
  for (int r = 0; r < reps; ++r) run(r);
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  size_t bytes = 0;
  for (auto& s : outs) bytes += s.size();
  double pps = (double)B.n * reps / secs;
  printf(
      "{\"posts_per_sec\": %.1f, \"threads\": %d, \"posts\": %d, "
      "\"reps\": %d, \"secs\": %.3f, \"bytes_per_rep\": %zu}\n",
      pps, threads, B.n, reps, secs, bytes);
  return 0;
}
