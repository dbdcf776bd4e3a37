// CDNA4 (gfx950) parse + JSONL-encode kernels — the per-post hot path.
//
// MI355X-native replacement for the reference's per-message hot loop:
//   - telegramhelper.ParseMessage content switch + field extraction
//     (tdutils.go:380-732)
//   - entity walk + UTF-16 offset conversion + t.me regex + username rules
//     (tdutils.go:55-78, 897-1002; username_filter.go:26-81)
//   - Go-compatible JSONL emission of model.Post (model/data.go:9-75,
//     storageproviders.go:275-298)
//
// Geometry: 256-thread blocks = 4 waves; ONE WAVE PER MESSAGE, grid-stride.
// Each message's JSON line is emitted wave-cooperatively (striped copies,
// wave-scanned escape expansion); scalar fields are formatted by lane 0
// with a wave-uniform cursor. Two passes share ONE templated emitter:
//   pass A (WRITE=false): exact line length + link extraction into scratch
//   pass B (WRITE=true):  byte emission at exclusive-scanned offsets
// so measure and write can never drift. The oracle is
// crawler_amd/ops/golden_batch.py (byte-identical output required).

#include "common.h"
#include "tg_lits.h"

namespace crawl {

#define MAX_LINKS 8
// per-wave LDS staging budget for user+title+poster+description
// (oversize fields fall back to their global pointers)
#define TG_STAGE_BYTES 3968

struct BatchView {
  // per-message
  const long* __restrict__ chat_id;
  const long* __restrict__ msg_id;
  const long* __restrict__ text_off;
  const int* __restrict__ date;
  const int* __restrict__ content_type;
  const int* __restrict__ views;
  const int* __restrict__ forwards;
  const int* __restrict__ media_album_id;
  const int* __restrict__ channel_idx;
  const int* __restrict__ flags;
  const int* __restrict__ text_len;
  const int* __restrict__ aux_off;
  const int* __restrict__ aux_len;
  const int* __restrict__ ent_off;
  const int* __restrict__ ent_cnt;
  const int* __restrict__ react_off;
  const int* __restrict__ react_cnt;
  const int* __restrict__ com_off;
  const int* __restrict__ com_cnt;
  const int* __restrict__ poster_off;
  const int* __restrict__ poster_len;
  // pools / tables
  const unsigned char* __restrict__ pool;
  const int* __restrict__ entities;  // [E,5] etype,off16,len16,url_off,url_len
  const int* __restrict__ react_emoji;
  const int* __restrict__ react_count;
  const int* __restrict__ com_text_off;
  const int* __restrict__ com_text_len;
  const int* __restrict__ com_handle_off;
  const int* __restrict__ com_handle_len;
  const int* __restrict__ com_views;
  const int* __restrict__ com_replies;
  const int* __restrict__ com_react_off;
  const int* __restrict__ com_react_cnt;
  // channel table
  const long* __restrict__ ch_chat_id;
  const int* __restrict__ ch_member;
  const int* __restrict__ ch_postcount;
  const int* __restrict__ ch_totalviews;
  const int* __restrict__ ch_user_off;
  const int* __restrict__ ch_user_len;
  const int* __restrict__ ch_title_off;
  const int* __restrict__ ch_title_len;
  // emoji vocabulary
  const unsigned char* __restrict__ emoji_pool;
  const int* __restrict__ emoji_off;
  const int* __restrict__ emoji_len;
  // timestamps (preformatted Go time strings)
  const unsigned char* __restrict__ created_str;
  int created_len;
  const unsigned char* __restrict__ capture_str;
  int capture_len;
  // config
  int n;
  int skip_media;
  long min_post_date;  // unix secs; <= filters nothing when INT64_MIN
  // content-type name table
  const unsigned char* __restrict__ ctname_pool;
  const int* __restrict__ ctname_off;
  const int* __restrict__ ctname_len;
};

struct LinkOut {
  unsigned char* __restrict__ name;     // [N, MAX_LINKS, 32]
  unsigned char* __restrict__ name_len; // [N, MAX_LINKS]
  unsigned char* __restrict__ src;      // [N, MAX_LINKS] 0=mention 1=text_url 2=url 3=plain
  int* __restrict__ cnt;                // [N]
  unsigned long long* __restrict__ hash;// [N, MAX_LINKS] fnv1a64 of name
};

// ---------- UTF-16 offset resolution (oracle: golden.utf16_offset_to_bytes)

DEV void utf16_to_bytes(const unsigned char* s, int n, int off16, int len16,
                        int* out_start, int* out_end) {
  int lane = lane_id();
  int run = 0;
  int start = -1, end = -1;
  const int t1 = off16, t2 = off16 + len16;
  for (int base = 0; base < n; base += WAVE) {
    int p = base + lane;
    int u = (p < n) ? u16_units_of_byte(s[p]) : 0;
    int pre = run + wave_prefix_excl(u);
    bool boundary = (p < n) && u > 0;
    unsigned long long m1 = __ballot(boundary && pre == t1);
    unsigned long long m2 = __ballot(boundary && pre == t2);
    if (start < 0 && m1) start = base + __ffsll(m1) - 1;
    if (end < 0 && m2) end = base + __ffsll(m2) - 1;
    run += wave_sum(u);
    if (end >= 0) break;
  }
  if (start < 0) { *out_start = 0; *out_end = 0; return; }
  *out_start = start;
  *out_end = (end >= 0) ? end : n;
}

// ---------- t.me link scanning (oracle: golden.CHANNEL_LINK_RE walk) ------

// Reserved t.me path segments >= 5 chars (shorter ones can't match the
// username group). tdutils.go:27-32.
DEV bool is_reserved_path(const unsigned char* name, int n) {
  const char* tbl[10] = {"joinchat", "addlist", "addstickers", "addtheme",
                         "setlanguage", "share", "proxy", "socks", "login",
                         "confirm"};
  const int lens[10] = {8, 7, 11, 8, 11, 5, 5, 5, 5, 7};
  for (int k = 0; k < 10; ++k) {
    if (lens[k] != n) continue;
    bool eq = true;
    for (int j = 0; j < n; ++j) eq &= (name[j] == (unsigned char)tbl[k][j]);
    if (eq) return true;
  }
  return false;
}

// Regex-group match after "t.me/" at byte q (exclusive bound hi).
// Returns the MATCH length (5..32, lowercased into name) or 0 when the
// regex does not match here; *reserved reports the reserved-path
// filter separately, because the two failures behave differently:
// no-regex-match means re.search keeps scanning for the NEXT t.me,
// while a reserved MATCH ends a .search() (and, in the plaintext
// FindAll walk, consumes its span). Runs uniformly on all lanes.
DEV int match_name_at(const unsigned char* s, int hi, int q,
                      unsigned char* name, bool* reserved) {
  *reserved = false;
  if (q >= hi || !is_ascii_letter(s[q])) return 0;
  int len = 0;
  while (len < 32 && q + len < hi && is_word_char(s[q + len])) ++len;
  if (len < 5) return 0;
  for (int j = 0; j < len; ++j) name[j] = to_lower(s[q + j]);
  *reserved = is_reserved_path(name, len);
  return len;
}

// First occurrence of "t.me/" at p in [from, hi-5]; wave-parallel.
DEV int find_tme(const unsigned char* s, int from, int hi, int lane) {
  for (int base = from; base + 5 <= hi; base += WAVE) {
    int p = base + lane;
    bool m = (p + 5 <= hi) && s[p] == 't' && s[p + 1] == '.' &&
             s[p + 2] == 'm' && s[p + 3] == 'e' && s[p + 4] == '/';
    unsigned long long bal = __ballot(m);
    if (bal) return base + __ffsll(bal) - 1;
  }
  return -1;
}

// usernameRegex.search: first letter in [lo,hi) followed by >=4 word chars.
DEV int find_username(const unsigned char* s, int lo, int hi, int lane,
                      unsigned char* name) {
  for (int base = lo; base < hi; base += WAVE) {
    int p = base + lane;
    bool m = false;
    if (p < hi && p + 5 <= hi && is_ascii_letter(s[p])) {
      m = is_word_char(s[p + 1]) && is_word_char(s[p + 2]) &&
          is_word_char(s[p + 3]) && is_word_char(s[p + 4]);
    }
    unsigned long long bal = __ballot(m);
    if (bal) {
      int q = base + __ffsll(bal) - 1;
      int len = 0;
      while (len < 32 && q + len < hi && is_word_char(s[q + len])) ++len;
      for (int j = 0; j < len; ++j) name[j] = to_lower(s[q + j]);
      return len;
    }
  }
  return 0;
}

struct LinkList {
  unsigned char* names;      // MAX_LINKS * 32 (global scratch)
  unsigned char* lens;
  unsigned char* srcs;
  unsigned long long* hashes;
  int cnt;

  DEV void add(const unsigned char* name, int len, int src, int lane) {
    // first-wins dedup by name (tdutils.go:903-907 addIfNew)
    for (int k = 0; k < cnt; ++k) {
      if (lens[k] != (unsigned char)len) continue;
      bool eq = true;
      for (int j = 0; j < len; ++j) eq &= (names[k * 32 + j] == name[j]);
      if (eq) return;
    }
    if (cnt >= MAX_LINKS) return;
    // lane 0 stores; same-address stores from all 64 lanes serialize
    // (64-way conflict in LDS, redundant traffic in global). Reads
    // broadcast, and every lane keeps `cnt` in step.
    if (lane == 0) {
      for (int j = 0; j < len; ++j) names[cnt * 32 + j] = name[j];
      lens[cnt] = (unsigned char)len;
      srcs[cnt] = (unsigned char)src;
      hashes[cnt] = fnv1a64(name, len);
    }
    ++cnt;
  }
};

// Full link extraction for one message; fills the LinkList.
// Source enum: 0=mention 1=text_url 2=url 3=plaintext (golden SOURCE_*).
DEV void extract_links(const BatchView& B, int i, LinkList& L, int lane) {
  int ct = B.content_type[i];
  if (ct > 6) return;  // not a FormattedText-bearing type (tdutils.go:953-975)
  const unsigned char* text = B.pool + B.text_off[i];
  const int tn = B.text_len[i];
  unsigned char name[32];

  // entity walk, in entity order (tdutils.go:909-941)
  const int e0 = B.ent_off[i], ec = B.ent_cnt[i];
  for (int e = e0; e < e0 + ec; ++e) {
    const int* row = B.entities + (long)e * 5;
    int etype = row[0];
    if (etype == 1) {  // text_url: .search() over the URL attribute
      const unsigned char* url = B.pool + row[3];
      int un = row[4];
      for (int from = 0;;) {
        int p = find_tme(url, from, un, lane);
        if (p < 0) break;
        bool rsv;
        int len = match_name_at(url, un, p + 5, name, &rsv);
        if (len) {  // first regex match decides the .search()
          if (!rsv) L.add(name, len, 1, lane);
          break;
        }
        from = p + 1;  // not a regex match here; search continues
      }
    } else {  // mention (0) / url (2): slice text at utf16 offsets
      int lo, hi;
      utf16_to_bytes(text, tn, row[1], row[2], &lo, &hi);
      if (!(lo < hi && hi <= tn)) continue;
      if (etype == 0) {
        int len = find_username(text, lo, hi, lane, name);
        if (len) L.add(name, len, 0, lane);
      } else {
        for (int from = lo;;) {
          int p = find_tme(text, from, hi, lane);
          if (p < 0) break;
          bool rsv;
          int len = match_name_at(text, hi, p + 5, name, &rsv);
          if (len) {
            if (!rsv) L.add(name, len, 2, lane);
            break;
          }
          from = p + 1;
        }
      }
    }
  }

  // plaintext scan with non-overlap cursor (FindAllStringSubmatch
  // semantics: every regex match — reserved or not — consumes its
  // span; only non-matches keep probing forward byte by byte)
  int cursor = 0;
  int from = 0;
  while (true) {
    int p = find_tme(text, from, tn, lane);
    if (p < 0) break;
    from = p + 1;  // next candidate search position
    if (p < cursor) continue;
    bool rsv;
    int len = match_name_at(text, tn, p + 5, name, &rsv);
    if (len) {
      if (!rsv) L.add(name, len, 3, lane);
      cursor = p + 5 + len;
      from = cursor;
    }
  }
}

// ---------- the templated line emitter ----------
// JsonEmit<W> (common.h) provides raw/esc/u64/i64/rfc3339 and the LIT
// macro; Emit adds the quoted-escape shorthand used below.

template <bool W>
struct Emit : JsonEmit<W> {
  DEV void qesc(const unsigned char* s, int n) {
    LIT(*this, "\"");
    this->esc(s, n);
    LIT(*this, "\"");
  }
};

// Block-LDS copies of the per-batch constant tables (staged kernel):
// emoji vocabulary, created/capture timestamp strings, content-type
// names. All tiny; staging them turns their per-line loads (issued
// AFTER line stores -> full store-FIFO drains) into lgkmcnt ds_reads.
struct StagedTabs {
  const unsigned char* emoji_pool;
  const int* emoji_off;
  const int* emoji_len;
  const unsigned char* created;
  const unsigned char* capture;
  const unsigned char* ctname_pool;
  const int* ctname_off;
  const int* ctname_len;
};

template <bool W, bool STAGED>
DEV int emit_line(const BatchView& B, int i, unsigned char* out,
                  LinkList& L, const unsigned char* lds_lits,
                  unsigned char* lds_stage, const StagedTabs* T) {
  const int lane = lane_id();
  Emit<W> e{};
  e.out = out;
  e.cur = 0;
  e.lds = lds_lits;
  // Load EVERY per-line scalar BEFORE the first store: vmcnt is a
  // single FIFO over loads AND stores on CDNA, so a load issued after
  // stores waits for those stores to retire (the write kernel spent
  // ~85 full-pipeline waits per line on lazily-loaded fields).
  const int c = B.channel_idx[i];
  const unsigned char* user = B.pool + B.ch_user_off[c];
  const int user_n = B.ch_user_len[c];
  const unsigned char* title = B.pool + B.ch_title_off[c];
  const int title_n = B.ch_title_len[c];
  const long pub_id = B.msg_id[i] >> 20;
  const int ct = B.content_type[i];
  const int ncom = B.com_cnt[i];
  const bool has_user = user_n > 0;
  const long chat_id_v = B.chat_id[i];
  const int media_album = B.media_album_id[i];
  const int date_v = B.date[i];
  const int views_v = B.views[i];
  const int forwards_v = B.forwards[i];
  const int ch_member_v = B.ch_member[c];
  const int ch_postcount_v = B.ch_postcount[c];
  const int ch_totalviews_v = B.ch_totalviews[c];
  const int flags_v = B.flags[i];
  const int com0_v = B.com_off[i];
  const int react0_v = B.react_off[i];
  const int reactc_v = B.react_cnt[i];
  const int poster_off_v = B.poster_off[i];
  const int poster_len_v = B.poster_len[i];
  const int text_off_v = B.text_off[i];
  const int text_len_v = B.text_len[i];
  const int aux_off_v = B.aux_off[i];
  const int aux_len_v = B.aux_len[i];
  const int ctname_off_v = B.ctname_off[ct];
  const int ctname_len_v = B.ctname_len[ct];
  // description source per content switch (tdutils.go:443-587)
  const unsigned char* desc_p;
  int desc_n;
  if (ct == 0 || ct == 1 || ct == 2 || ct == 4 || ct == 14) {
    desc_p = B.pool + text_off_v;
    desc_n = text_len_v;
  } else if (ct == 3 || ct == 9 || ct == 10 || ct == 11) {
    desc_p = B.pool + aux_off_v;
    desc_n = aux_len_v;
  } else {
    desc_p = nullptr;
    desc_n = 0;
  }
  const unsigned char* poster_p = B.pool + poster_off_v;
  // Stage the escape-scanned variable fields in LDS at line START: the
  // staging loads batch into ONE vmcnt wait before any store exists,
  // and every later esc() read is a ds_read (lgkmcnt) that never waits
  // on the line's store stream (the vmcnt FIFO orders loads AFTER
  // stores; see profiles/r02_valu_diet.md).
  // STAGED instantiations copy unconditionally (the HOST verified the
  // batch's max field lengths fit TG_STAGE_BYTES), so every staged
  // pointer is provably addrspace(3): the escape reads become ds_read
  // (lgkmcnt) instead of flat_load, which waits BOTH counters — the
  // mixed-provenance select of an earlier revision generated exactly
  // those flat loads and re-serialized the emitter.
  const int* re_src = B.react_emoji + react0_v;
  const int* rcnt_src = B.react_count + react0_v;
  const unsigned char* lnames_src = L.names;
  const unsigned char* llens_src = L.lens;
  if (STAGED) {
    int stage_o = 0;
    auto staged = [&](const unsigned char* ptr,
                      int n) -> const unsigned char* {
      unsigned char* dst = lds_stage + stage_o;
      stage_o += n;
      for (int j = lane; j < n; j += WAVE) dst[j] = ptr[j];
      return dst;
    };
    user = staged(user, user_n);
    title = staged(title, title_n);
    poster_p = staged(poster_p, poster_len_v);
    desc_p = staged(desc_p, desc_n);
    lnames_src = staged(L.names, L.cnt * 32);
    llens_src = staged(L.lens, L.cnt);
    // reaction id/count arrays (int32), 4B-aligned carve
    stage_o = (stage_o + 3) & ~3;
    int* s_re = (int*)(lds_stage + stage_o);
    stage_o += reactc_v * 4;
    int* s_rc = (int*)(lds_stage + stage_o);
    stage_o += reactc_v * 4;
    for (int j = lane; j < reactc_v; j += WAVE) {
      s_re[j] = B.react_emoji[react0_v + j];
      s_rc[j] = B.react_count[react0_v + j];
    }
    re_src = s_re;
    rcnt_src = s_rc;
  }
  const unsigned char* emoji_pool_s = STAGED ? T->emoji_pool : B.emoji_pool;
  const int* emoji_off_s = STAGED ? T->emoji_off : B.emoji_off;
  const int* emoji_len_s = STAGED ? T->emoji_len : B.emoji_len;
  const unsigned char* created_s = STAGED ? T->created : B.created_str;
  const unsigned char* capture_s = STAGED ? T->capture : B.capture_str;
  const unsigned char* ctpool_s = STAGED ? T->ctname_pool : B.ctname_pool;

  // post_link / url (tdutils.go:1005-1031; empty for private channels)
  auto post_link = [&]() {
    if (!has_user) { LIT(e, "\"\""); return; }
    TGLIT(e, TGL_0);
    e.esc(user, user_n);
    LIT(e, "/");
    e.i64(pub_id);
    if (media_album != 0) TGLIT(e, TGL_1);
    LIT(e, "\"");
  };

  TGLIT(e, TGL_2);
  post_link();
  TGLIT(e, TGL_3);
  e.i64(chat_id_v);
  TGLIT(e, TGL_4);
  e.i64(pub_id);
  LIT(e, "-");
  e.esc(user, user_n);
  TGLIT(e, TGL_5);
  post_link();
  TGLIT(e, TGL_6);
  e.rfc3339(date_v);
  TGLIT(e, TGL_7);
  e.raw(created_s, B.created_len);
  TGLIT(e, TGL_8);
  e.i64(views_v);
  TGLIT(e, TGL_9);
  e.i64(views_v);
  TGLIT(e, TGL_10);
  e.i64(forwards_v);
  TGLIT(e, TGL_11);
  e.i64(ncom);
  TGLIT(e, TGL_12);
  e.qesc(title, title_n);
  TGLIT(e, TGL_13);
  e.i64(chat_id_v);
  TGLIT(e, TGL_14);
  e.qesc(title, title_n);
  TGLIT(e, TGL_15);
  e.i64(ch_member_v);
  TGLIT(e, TGL_16);
  e.i64(ch_postcount_v);
  TGLIT(e, TGL_17);
  e.i64(ch_totalviews_v);
  TGLIT(e, TGL_18);
  e.esc(user, user_n);
  TGLIT(e, TGL_19);
  e.esc(user, user_n);
  TGLIT(e, TGL_20);
  e.qesc(desc_p, desc_n);
  TGLIT(e, TGL_21);
  e.raw(ctpool_s + (STAGED ? T->ctname_off[ct] : ctname_off_v),
        STAGED ? T->ctname_len[ct] : ctname_len_v);
  TGLIT(e, TGL_22);
  e.i64(forwards_v);
  TGLIT(e, TGL_23);
  e.i64(ncom);
  TGLIT(e, TGL_17);
  e.i64(views_v);
  TGLIT(e, TGL_24);
  // media (fetchAndUploadMedia skip rules, tdutils.go:233-239): GPU path
  // always runs skip_media (media-on is staged host-side).
  TGLIT(e, TGL_25);
  if ((ct == 3 || ct == 8) && (flags_v & 2)) {
    LIT(e, "AgAD");
    e.i64(pub_id);
    LIT(e, "v");
  }
  TGLIT(e, TGL_26);
  {
    const int c0 = com0_v;
    for (int k = 0; k < ncom; ++k) {
      if (k) LIT(e, ",");
      const int cc = c0 + k;
      TGLIT(e, TGL_27);
      e.qesc(B.pool + B.com_text_off[cc], B.com_text_len[cc]);
      TGLIT(e, TGL_28);
      const int r0 = B.com_react_off[cc], rc = B.com_react_cnt[cc];
      for (int r = 0; r < rc; ++r) {
        if (r) LIT(e, ",");
        int em = B.react_emoji[r0 + r];
        LIT(e, "\"");
        e.raw(emoji_pool_s + emoji_off_s[em], emoji_len_s[em]);
        LIT(e, "\":");
        e.i64(B.react_count[r0 + r]);
      }
      TGLIT(e, TGL_29);
      e.i64(B.com_views[cc]);
      TGLIT(e, TGL_30);
      e.i64(B.com_replies[cc]);
      TGLIT(e, TGL_31);
      e.qesc(B.pool + B.com_handle_off[cc], B.com_handle_len[cc]);
      LIT(e, "}");
    }
  }
  TGLIT(e, TGL_32);
  {
    const int rc = reactc_v;
    for (int r = 0; r < rc; ++r) {
      if (r) LIT(e, ",");
      int em = re_src[r];
      LIT(e, "\"");
      e.raw(emoji_pool_s + emoji_off_s[em], emoji_len_s[em]);
      LIT(e, "\":");
      e.i64(rcnt_src[r]);
    }
  }
  TGLIT(e, TGL_33);
  for (int k = 0; k < L.cnt; ++k) {
    if (k) LIT(e, ",");
    LIT(e, "\"");
    e.raw(lnames_src + k * 32, llens_src[k]);
    LIT(e, "\"");
  }
  TGLIT(e, TGL_34);
  e.raw(capture_s, B.capture_len);
  TGLIT(e, TGL_35);
  e.qesc(poster_p, poster_len_v);
  LIT(e, "}\n");
  return e.cur;
}

// ---------- kernels ----------

__global__ void __launch_bounds__(256, 6)
measure_extract_kernel(BatchView B, LinkOut LO, int* __restrict__ line_len) {
  __shared__ unsigned char s_lits[TG_POOL_BYTES];
  // LDS-resident LinkList per wave: the extractor's dedup scans and
  // name stores stay in LDS (single-lane writes, broadcast reads) and
  // spill to the global LinkOut arrays ONCE per message.
  __shared__ unsigned char s_ln[4][MAX_LINKS * 32];
  __shared__ unsigned char s_ll[4][MAX_LINKS];
  __shared__ unsigned char s_ls[4][MAX_LINKS];
  __shared__ unsigned long long s_lh[4][MAX_LINKS];
  for (int t = threadIdx.x; t < TG_POOL_BYTES; t += blockDim.x)
    s_lits[t] = (unsigned char)tg_lit_pool.v[t];
  __syncthreads();
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += waves_per_grid) {
    // min_post_date filter (tdutils.go:419-421)
    if ((long)B.date[i] < B.min_post_date) {
      if (lane == 0) { line_len[i] = 0; LO.cnt[i] = 0; }
      continue;
    }
    LinkList L{&s_ln[wave][0], &s_ll[wave][0], &s_ls[wave][0],
               &s_lh[wave][0], 0};
    extract_links(B, i, L, lane);
    int len = emit_line<false, false>(B, i, nullptr, L, s_lits,
                                     nullptr, nullptr);
    // spill the wave's links to the global LinkOut (lane-striped)
    const int nb = L.cnt * 32;
    unsigned char* gname = LO.name + (size_t)i * MAX_LINKS * 32;
    for (int j = lane; j < nb; j += WAVE) gname[j] = s_ln[wave][j];
    if (lane < L.cnt) {
      LO.name_len[(size_t)i * MAX_LINKS + lane] = s_ll[wave][lane];
      LO.src[(size_t)i * MAX_LINKS + lane] = s_ls[wave][lane];
      LO.hash[(size_t)i * MAX_LINKS + lane] = s_lh[wave][lane];
    }
    if (lane == 0) {
      line_len[i] = len;
      LO.cnt[i] = L.cnt;
    }
  }
}

__global__ void __launch_bounds__(256, 4)
write_kernel(BatchView B, LinkOut LO, const long* __restrict__ line_off,
             const int* __restrict__ line_len, unsigned char* __restrict__ out) {
  __shared__ unsigned char s_lits[TG_POOL_BYTES];
  for (int t = threadIdx.x; t < TG_POOL_BYTES; t += blockDim.x)
    s_lits[t] = (unsigned char)tg_lit_pool.v[t];
  __syncthreads();
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += waves_per_grid) {
    if (line_len[i] == 0) continue;
    LinkList L{LO.name + (size_t)i * MAX_LINKS * 32,
               LO.name_len + (size_t)i * MAX_LINKS,
               LO.src + (size_t)i * MAX_LINKS,
               LO.hash + (size_t)i * MAX_LINKS, LO.cnt[i]};
    emit_line<true, false>(B, i, out + line_off[i], L, s_lits, nullptr,
                           nullptr);
  }
}

// Staged write: the host launches this when the batch's max field
// lengths fit TG_STAGE_BYTES (flagship corpus always does); escape
// reads come from LDS so no in-line load waits on the store FIFO.
#define TG_EMOJI_POOL_CAP 512
#define TG_EMOJI_CAP 64
#define TG_TS_CAP 64
#define TG_CTNAME_POOL_CAP 512
#define TG_CTNAME_CAP 32

__global__ void __launch_bounds__(256, 4)
write_staged_kernel(BatchView B, LinkOut LO, const long* __restrict__ line_off,
                    const int* __restrict__ line_len,
                    unsigned char* __restrict__ out,
                    int n_emoji, int emoji_pool_bytes,
                    int n_ctnames, int ctname_pool_bytes) {
  __shared__ unsigned char s_lits[TG_POOL_BYTES];
  __shared__ unsigned char s_stage[4][TG_STAGE_BYTES];
  __shared__ unsigned char s_emoji_pool[TG_EMOJI_POOL_CAP];
  __shared__ int s_emoji_off[TG_EMOJI_CAP];
  __shared__ int s_emoji_len[TG_EMOJI_CAP];
  __shared__ unsigned char s_created[TG_TS_CAP];
  __shared__ unsigned char s_capture[TG_TS_CAP];
  __shared__ unsigned char s_ctpool[TG_CTNAME_POOL_CAP];
  __shared__ int s_ctoff[TG_CTNAME_CAP];
  __shared__ int s_ctlen[TG_CTNAME_CAP];
  for (int t = threadIdx.x; t < TG_POOL_BYTES; t += blockDim.x)
    s_lits[t] = (unsigned char)tg_lit_pool.v[t];
  for (int t = threadIdx.x; t < emoji_pool_bytes; t += blockDim.x)
    s_emoji_pool[t] = B.emoji_pool[t];
  for (int t = threadIdx.x; t < n_emoji; t += blockDim.x) {
    s_emoji_off[t] = B.emoji_off[t];
    s_emoji_len[t] = B.emoji_len[t];
  }
  for (int t = threadIdx.x; t < B.created_len; t += blockDim.x)
    s_created[t] = B.created_str[t];
  for (int t = threadIdx.x; t < B.capture_len; t += blockDim.x)
    s_capture[t] = B.capture_str[t];
  for (int t = threadIdx.x; t < ctname_pool_bytes; t += blockDim.x)
    s_ctpool[t] = B.ctname_pool[t];
  for (int t = threadIdx.x; t < n_ctnames; t += blockDim.x) {
    s_ctoff[t] = B.ctname_off[t];
    s_ctlen[t] = B.ctname_len[t];
  }
  __syncthreads();
  StagedTabs T{s_emoji_pool, s_emoji_off, s_emoji_len, s_created,
               s_capture, s_ctpool, s_ctoff, s_ctlen};
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += waves_per_grid) {
    if (line_len[i] == 0) continue;
    LinkList L{LO.name + (size_t)i * MAX_LINKS * 32,
               LO.name_len + (size_t)i * MAX_LINKS,
               LO.src + (size_t)i * MAX_LINKS,
               LO.hash + (size_t)i * MAX_LINKS, LO.cnt[i]};
    emit_line<true, true>(B, i, out + line_off[i], L, s_lits,
                          &s_stage[wave][0], &T);
  }
}

// LDS-staged write: emit each line into a per-wave LDS buffer (fast byte
// stores, no vmem latency on the hot emission path), then stream it to its
// exact global offset with the funnel-shift dword copy. Lines larger than
// the LDS budget (rare) fall back to direct global emission.

// 3 KB covers the typical ~2 KB line at FULL occupancy (4 waves x 3KB
// + literal pool ~= 14 KB/block, 8 blocks/CU); bigger lines (comment
// threads) take the direct-global fallback. The round-1 12 KB buffer
// capped occupancy at 3 blocks/CU and lost 2x.
#define LDS_LINE_BYTES 3072

DEV void copy_line(const unsigned char* src, unsigned char* dst, int n,
                   int lane);

__global__ void __launch_bounds__(256)
write_lds_kernel(BatchView B, LinkOut LO, const long* __restrict__ line_off,
                 const int* __restrict__ line_len, unsigned char* __restrict__ out) {
  __shared__ unsigned char lbuf[4][LDS_LINE_BYTES];
  __shared__ unsigned char s_lits[TG_POOL_BYTES];
  for (int t = threadIdx.x; t < TG_POOL_BYTES; t += blockDim.x)
    s_lits[t] = (unsigned char)tg_lit_pool.v[t];
  __syncthreads();
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += waves_per_grid) {
    const int len = line_len[i];
    if (len == 0) continue;
    LinkList L{LO.name + (size_t)i * MAX_LINKS * 32,
               LO.name_len + (size_t)i * MAX_LINKS,
               LO.src + (size_t)i * MAX_LINKS,
               LO.hash + (size_t)i * MAX_LINKS, LO.cnt[i]};
    if (len > LDS_LINE_BYTES) {
      emit_line<true, false>(B, i, out + line_off[i], L, s_lits,
                             nullptr, nullptr);
      continue;
    }
    emit_line<true, false>(B, i, &lbuf[wave][0], L, s_lits, nullptr,
                           nullptr);
    copy_line(&lbuf[wave][0], out + line_off[i], len, lane);
  }
}

// Single-pass variant: emit straight into per-message scratch slots (the
// host supplies a sound stride bound), then a vectorized compaction gathers
// the final contiguous JSONL. Replaces measure+write (saves the whole
// measuring pass; compaction is pure memcpy-rate).

__global__ void __launch_bounds__(256)
write_scratch_kernel(BatchView B, LinkOut LO, unsigned char* __restrict__ scratch,
                     long stride, int* __restrict__ line_len, int* __restrict__ overflow) {
  __shared__ unsigned char s_lits[TG_POOL_BYTES];
  for (int t = threadIdx.x; t < TG_POOL_BYTES; t += blockDim.x)
    s_lits[t] = (unsigned char)tg_lit_pool.v[t];
  __syncthreads();
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < B.n; i += waves_per_grid) {
    if ((long)B.date[i] < B.min_post_date) {
      if (lane == 0) { line_len[i] = 0; LO.cnt[i] = 0; }
      continue;
    }
    LinkList L{LO.name + (size_t)i * MAX_LINKS * 32,
               LO.name_len + (size_t)i * MAX_LINKS,
               LO.src + (size_t)i * MAX_LINKS,
               LO.hash + (size_t)i * MAX_LINKS, 0};
    extract_links(B, i, L, lane);
    int len = emit_line<true, false>(B, i, scratch + (size_t)i * stride, L,
                                     s_lits, nullptr, nullptr);
    if (lane == 0) {
      line_len[i] = len;
      LO.cnt[i] = L.cnt;
      if (len > stride) atomicMax(overflow, len);  // host asserts 0
    }
  }
}

// Wave-cooperative aligned-dst copy: head bytes to 4B alignment, then
// funnel-shifted dword stores (src may be misaligned), byte tail.
DEV void copy_line(const unsigned char* src, unsigned char* dst, int n,
                   int lane) {
  int h = (int)((4 - ((unsigned long)(size_t)dst & 3)) & 3);
  if (h > n) h = n;
  if (lane == 0)
    for (int j = 0; j < h; ++j) dst[j] = src[j];
  const unsigned char* s = src + h;
  unsigned char* d = dst + h;
  int rem = n - h;
  int nw = rem >> 2;
  int sh = (int)((unsigned long)(size_t)s & 3);
  const unsigned int* sw = (const unsigned int*)(s - sh);
  for (int k = lane; k < nw; k += WAVE) {
    unsigned int w0 = sw[k];
    unsigned int v;
    if (sh) {
      unsigned int w1 = sw[k + 1];
      v = (unsigned int)((((unsigned long long)w1 << 32) | w0) >> (8 * sh));
    } else {
      v = w0;
    }
    ((unsigned int*)d)[k] = v;
  }
  if (lane == 0)
    for (int j = h + (nw << 2); j < n; ++j) dst[j] = src[j];
}

__global__ void __launch_bounds__(256)
compact_kernel(const unsigned char* scratch, long stride,
               const long* line_off, const int* line_len,
               unsigned char* out, int n) {
  const int lane = lane_id();
  const int wave = wave_id();
  const int waves_per_grid = gridDim.x * 4;
  for (int i = blockIdx.x * 4 + wave; i < n; i += waves_per_grid) {
    int len = line_len[i];
    if (len == 0) continue;
    copy_line(scratch + (size_t)i * stride, out + line_off[i], len, lane);
  }
}

}  // namespace crawl

// ---- C ABI (ctypes-friendly; torch-header-free) ----
//
// Pointer/scalar array layouts MUST stay in sync with ops/gpu.py
// (_BATCH_PTR_ORDER / _SCALAR_ORDER). The Python side passes
// tensor.data_ptr() values in that order.

namespace crawl {

static BatchView make_view(void** p, const long* s) {
  BatchView B;
  int k = 0;
  B.chat_id = (const long*)p[k++];
  B.msg_id = (const long*)p[k++];
  B.text_off = (const long*)p[k++];
  B.date = (const int*)p[k++];
  B.content_type = (const int*)p[k++];
  B.views = (const int*)p[k++];
  B.forwards = (const int*)p[k++];
  B.media_album_id = (const int*)p[k++];
  B.channel_idx = (const int*)p[k++];
  B.flags = (const int*)p[k++];
  B.text_len = (const int*)p[k++];
  B.aux_off = (const int*)p[k++];
  B.aux_len = (const int*)p[k++];
  B.ent_off = (const int*)p[k++];
  B.ent_cnt = (const int*)p[k++];
  B.react_off = (const int*)p[k++];
  B.react_cnt = (const int*)p[k++];
  B.com_off = (const int*)p[k++];
  B.com_cnt = (const int*)p[k++];
  B.poster_off = (const int*)p[k++];
  B.poster_len = (const int*)p[k++];
  B.pool = (const unsigned char*)p[k++];
  B.entities = (const int*)p[k++];
  B.react_emoji = (const int*)p[k++];
  B.react_count = (const int*)p[k++];
  B.com_text_off = (const int*)p[k++];
  B.com_text_len = (const int*)p[k++];
  B.com_handle_off = (const int*)p[k++];
  B.com_handle_len = (const int*)p[k++];
  B.com_views = (const int*)p[k++];
  B.com_replies = (const int*)p[k++];
  B.com_react_off = (const int*)p[k++];
  B.com_react_cnt = (const int*)p[k++];
  B.ch_chat_id = (const long*)p[k++];
  B.ch_member = (const int*)p[k++];
  B.ch_postcount = (const int*)p[k++];
  B.ch_totalviews = (const int*)p[k++];
  B.ch_user_off = (const int*)p[k++];
  B.ch_user_len = (const int*)p[k++];
  B.ch_title_off = (const int*)p[k++];
  B.ch_title_len = (const int*)p[k++];
  B.emoji_pool = (const unsigned char*)p[k++];
  B.emoji_off = (const int*)p[k++];
  B.emoji_len = (const int*)p[k++];
  B.created_str = (const unsigned char*)p[k++];
  B.capture_str = (const unsigned char*)p[k++];
  B.ctname_pool = (const unsigned char*)p[k++];
  B.ctname_off = (const int*)p[k++];
  B.ctname_len = (const int*)p[k++];
  B.n = (int)s[0];
  B.skip_media = (int)s[1];
  B.min_post_date = s[2];
  B.created_len = (int)s[3];
  B.capture_len = (int)s[4];
  return B;
}

static LinkOut make_links(void** p) {
  LinkOut L;
  L.name = (unsigned char*)p[0];
  L.name_len = (unsigned char*)p[1];
  L.src = (unsigned char*)p[2];
  L.cnt = (int*)p[3];
  L.hash = (unsigned long long*)p[4];
  return L;
}

}  // namespace crawl

extern "C" {

int crawl_batch_ptr_count() { return 49; }

int crawl_measure_extract(void** batch_ptrs, const long* scalars,
                          void** link_ptrs, void* line_len, int grid,
                          void* stream) {
  crawl::BatchView B = crawl::make_view(batch_ptrs, scalars);
  crawl::LinkOut LO = crawl::make_links(link_ptrs);
  hipLaunchKernelGGL(crawl::measure_extract_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, LO, (int*)line_len);
  return (int)hipGetLastError();
}

int crawl_write(void** batch_ptrs, const long* scalars, void** link_ptrs,
                const void* line_off, const void* line_len, void* out,
                int grid, void* stream) {
  crawl::BatchView B = crawl::make_view(batch_ptrs, scalars);
  crawl::LinkOut LO = crawl::make_links(link_ptrs);
  hipLaunchKernelGGL(crawl::write_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, LO, (const long*)line_off,
                     (const int*)line_len, (unsigned char*)out);
  return (int)hipGetLastError();
}

int crawl_write_staged(void** batch_ptrs, const long* scalars,
                       void** link_ptrs, const void* line_off,
                       const void* line_len, void* out,
                       int n_emoji, int emoji_pool_bytes,
                       int n_ctnames, int ctname_pool_bytes,
                       int grid, void* stream) {
  crawl::BatchView B = crawl::make_view(batch_ptrs, scalars);
  crawl::LinkOut LO = crawl::make_links(link_ptrs);
  hipLaunchKernelGGL(crawl::write_staged_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, LO, (const long*)line_off,
                     (const int*)line_len, (unsigned char*)out,
                     n_emoji, emoji_pool_bytes, n_ctnames,
                     ctname_pool_bytes);
  return (int)hipGetLastError();
}

// max per-line staged bytes the staged writer supports (host checks the
// batch's max field lengths against this before choosing it)
int crawl_stage_budget() { return TG_STAGE_BYTES; }

int crawl_write_scratch(void** batch_ptrs, const long* scalars,
                        void** link_ptrs, void* scratch, long stride,
                        void* line_len, void* overflow, int grid,
                        void* stream) {
  crawl::BatchView B = crawl::make_view(batch_ptrs, scalars);
  crawl::LinkOut LO = crawl::make_links(link_ptrs);
  hipLaunchKernelGGL(crawl::write_scratch_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, LO, (unsigned char*)scratch,
                     stride, (int*)line_len, (int*)overflow);
  return (int)hipGetLastError();
}

int crawl_write_lds(void** batch_ptrs, const long* scalars, void** link_ptrs,
                    const void* line_off, const void* line_len, void* out,
                    int grid, void* stream) {
  crawl::BatchView B = crawl::make_view(batch_ptrs, scalars);
  crawl::LinkOut LO = crawl::make_links(link_ptrs);
  hipLaunchKernelGGL(crawl::write_lds_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, B, LO, (const long*)line_off,
                     (const int*)line_len, (unsigned char*)out);
  return (int)hipGetLastError();
}

int crawl_compact(const void* scratch, long stride, const void* line_off,
                  const void* line_len, void* out, int n, int grid,
                  void* stream) {
  hipLaunchKernelGGL(crawl::compact_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, (const unsigned char*)scratch,
                     stride, (const long*)line_off, (const int*)line_len,
                     (unsigned char*)out, n);
  return (int)hipGetLastError();
}

}  // extern "C"
