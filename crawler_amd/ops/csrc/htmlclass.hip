// Batched t.me HTML classification ("HTML title/meta parse
// (validator) -> batch substring-search kernel (64 KB docs)").
//
// One wave per document: locate <title>...</title>, strip it, apply the
// reference's title rules (telegramhelper/channelvalidator.go:132-153),
// and for "Contact @" pages scan <meta ...> tags for a robots/noindex
// marker. Oracle: crawler_amd/engine/htmlvalidator.py parse_channel_html.
//
// status: 0=valid 1=not_channel 2=invalid
// reason: 0="" 1=not_supergroup 2=not_found(noindex) 3=not_found
//         4=unrecognized

#include "common.h"

namespace crawl {

#define HTML_BODY_CAP (64 * 1024)  // channelvalidator.go:103

// Case-SENSITIVE wave-parallel substring find (patterns are <=20 bytes).
DEV int find_sub(const unsigned char* s, int lo, int hi, const char* pat,
                 int plen, int lane) {
  for (int base = lo; base + plen <= hi; base += WAVE) {
    int p = base + lane;
    bool m = (p + plen <= hi);
    if (m) {
      for (int j = 0; j < plen; ++j) m &= (s[p + j] == (unsigned char)pat[j]);
    }
    unsigned long long bal = __ballot(m);
    if (bal) return base + __ffsll(bal) - 1;
  }
  return -1;
}

DEV bool is_space(unsigned char c) {
  return c == ' ' || c == '\t' || c == '\n' || c == '\r';
}

DEV bool starts_with(const unsigned char* s, int lo, int hi,
                     const char* pat, int plen) {
  if (lo + plen > hi) return false;
  for (int j = 0; j < plen; ++j)
    if (s[lo + j] != (unsigned char)pat[j]) return false;
  return true;
}

__global__ void __launch_bounds__(256)
html_classify_kernel(const unsigned char* pool, const long* doc_off,
                     const int* doc_len, int n, int* status, int* reason) {
  const int lane = lane_id();
  const int wave = wave_id();
  for (int i = blockIdx.x * 4 + wave; i < n; i += gridDim.x * 4) {
    const unsigned char* s = pool + doc_off[i];
    int len = doc_len[i];
    if (len > HTML_BODY_CAP) len = HTML_BODY_CAP;

    // ---- locate the <title> tag content ----
    int tlo = 0, thi = 0;
    int tag = find_sub(s, 0, len, "<title", 6, lane);
    if (tag >= 0) {
      int gt = find_sub(s, tag, len, ">", 1, lane);
      if (gt >= 0) {
        int close = find_sub(s, gt + 1, len, "</title", 7, lane);
        if (close >= 0) {
          tlo = gt + 1;
          thi = close;
          while (tlo < thi && is_space(s[tlo])) ++tlo;
          while (thi > tlo && is_space(s[thi - 1])) --thi;
        }
      }
    }

    int st = 2, rs = 4;  // default invalid/unrecognized
    const bool view_at =
        starts_with(s, tlo, thi, "View @", 6) ||
        find_sub(s, tlo, thi, "Telegram: View @", 16, lane) >= 0;
    const bool contact_at =
        starts_with(s, tlo, thi, "Contact @", 9) ||
        find_sub(s, tlo, thi, "Telegram: Contact @", 19, lane) >= 0;
    if (view_at) {
      st = 0; rs = 0;
    } else if (contact_at) {
      // scan <meta ...> tags for name="robots" (or ') + noindex in-tag
      bool noindex = false;
      int from = 0;
      while (!noindex) {
        int m = find_sub(s, from, len, "<meta", 5, lane);
        if (m < 0) break;
        int gt = find_sub(s, m, len, ">", 1, lane);
        int end = gt < 0 ? len : gt;
        int rpos = find_sub(s, m, end, "name=\"robots\"", 13, lane);
        if (rpos < 0) rpos = find_sub(s, m, end, "name='robots'", 13, lane);
        if (rpos >= 0) {
          // oracle regex requires noindex AFTER the name attribute
          // within the same tag (htmlvalidator._NOINDEX_RE)
          if (find_sub(s, rpos + 13, end, "noindex", 7, lane) >= 0)
            noindex = true;
        }
        from = m + 5;
      }
      if (noindex) { st = 2; rs = 2; }       // not_found (noindex)
      else { st = 1; rs = 1; }               // not_channel/not_supergroup
    } else if (find_sub(s, tlo, thi, "Telegram Messenger", 18, lane) >= 0) {
      st = 2; rs = 3;                        // not_found
    }
    if (lane == 0) {
      status[i] = st;
      reason[i] = rs;
    }
  }
}

}  // namespace crawl

extern "C" {

int crawl_html_classify(const void* pool, const void* doc_off,
                        const void* doc_len, int n, void* status,
                        void* reason, int grid, void* stream) {
  hipLaunchKernelGGL(crawl::html_classify_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, (const unsigned char*)pool,
                     (const long*)doc_off, (const int*)doc_len, n,
                     (int*)status, (int*)reason);
  return (int)hipGetLastError();
}

}  // extern "C"
