// Device helpers for the CDNA4 (gfx950) crawl hot-path kernels.
//
// Wave-cooperative primitives over 64-lane wavefronts: striped memcpy,
// Go-compatible JSON string escaping, integer/date formatting, UTF-16
// offset resolution, and the t.me link scanner. Every routine here has a
// pure-Python oracle in crawler_amd/ops/golden.py (reference semantics
// cited there against telegramhelper/tdutils.go).
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64
#define DEV __device__ __forceinline__

// Non-temporal (L2-streaming) stores for the JSONL output: the line
// bytes are written once and never read on device, and 2.5 GB/batch of
// regular stores evicts the literal/template and text-pool lines the
// emitters re-read every line — the write kernel stalls ~one
// HBM-latency per emitter call without this.
typedef unsigned int __attribute__((aligned(1))) u32_unal;
#ifndef CRAWL_NT_STORES
#define CRAWL_NT_STORES 0
#endif
DEV void nt_store_u32(unsigned char* p, unsigned v) {
#if CRAWL_NT_STORES
  __builtin_nontemporal_store(v, (u32_unal*)p);
#else
  __builtin_memcpy(p, &v, 4);
#endif
}
DEV void nt_store_u8(unsigned char* p, unsigned char v) {
#if CRAWL_NT_STORES
  __builtin_nontemporal_store(v, p);
#else
  *p = v;
#endif
}

namespace crawl {

DEV int lane_id() { return threadIdx.x & (WAVE - 1); }
DEV int wave_id() { return threadIdx.x >> 6; }

// ---------- byte classification ----------

DEV bool is_ascii_letter(uint8_t c) {
  return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z');
}
DEV bool is_word_char(uint8_t c) {  // [a-zA-Z0-9_]
  return is_ascii_letter(c) || (c >= '0' && c <= '9') || c == '_';
}
DEV uint8_t to_lower(uint8_t c) {
  return (c >= 'A' && c <= 'Z') ? (c + 32) : c;
}

// UTF-16 code units contributed by the byte at p (0 for continuation bytes).
DEV int u16_units_of_byte(uint8_t b) {
  if (b < 0x80) return 1;          // ascii
  if (b < 0xC0) return 0;          // continuation
  if (b < 0xF0) return 1;          // 2- or 3-byte leader (BMP)
  return 2;                        // 4-byte leader -> surrogate pair
}

// ---------- Go encoding/json escape model ----------
// Output bytes for input byte s[p] (with lookahead for U+2028/U+2029).
// 1 = verbatim; 2 = two-char escape; 6 = \u00xx / \u202x; 0 = swallowed
// (continuation bytes of an escaped U+2028/29 sequence).
// Per-byte Go-JSON escape class: 1 verbatim, 2 two-char escape, 6
// \u00xx, 0xFF = positional (0xE2/0x80/0xA8/0xA9 around U+2028/29).
// Generated from the rules below; keep in sync with escape_write_at.
__device__ const uint8_t ESC_CLS[256] = {
    0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x02, 0x02, 0x06, 0x06, 0x02, 0x06, 0x06,
    0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06, 0x06,
    0x01, 0x01, 0x02, 0x01, 0x01, 0x01, 0x06, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x06, 0x01, 0x06, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x02, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0xFF, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0xFF, 0xFF, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0xFF, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
    0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01, 0x01,
};

DEV int escape_len_at(const uint8_t* s, int n, int p) {
  uint8_t c = s[p];
  uint8_t v = ESC_CLS[c];
  if (v != 0xFF) return v;
  if (c == 0xE2) {
    if (p + 2 < n && s[p + 1] == 0x80 &&
        (s[p + 2] == 0xA8 || s[p + 2] == 0xA9))
      return 6;
    return 1;
  }
  if (c == 0x80) {
    if (p >= 1 && s[p - 1] == 0xE2 && p + 1 < n &&
        (s[p + 1] == 0xA8 || s[p + 1] == 0xA9))
      return 0;
    return 1;
  }
  // c == 0xA8 || c == 0xA9
  if (p >= 2 && s[p - 2] == 0xE2 && s[p - 1] == 0x80) return 0;
  return 1;
}

DEV const char* HEXD() { return "0123456789abcdef"; }

// Write the escape expansion of s[p] to out; returns bytes written.
DEV int escape_write_at(const uint8_t* s, int n, int p, uint8_t* out) {
  uint8_t c = s[p];
  switch (c) {
    case '"':  out[0] = '\\'; out[1] = '"';  return 2;
    case '\\': out[0] = '\\'; out[1] = '\\'; return 2;
    case '\n': out[0] = '\\'; out[1] = 'n';  return 2;
    case '\r': out[0] = '\\'; out[1] = 'r';  return 2;
    case '\t': out[0] = '\\'; out[1] = 't';  return 2;
    default: break;
  }
  if (c < 0x20 || c == '<' || c == '>' || c == '&') {
    out[0] = '\\'; out[1] = 'u'; out[2] = '0'; out[3] = '0';
    out[4] = HEXD()[c >> 4]; out[5] = HEXD()[c & 15];
    return 6;
  }
  if (c == 0xE2 && p + 2 < n && s[p + 1] == 0x80 &&
      (s[p + 2] == 0xA8 || s[p + 2] == 0xA9)) {
    out[0] = '\\'; out[1] = 'u'; out[2] = '2'; out[3] = '0'; out[4] = '2';
    out[5] = (s[p + 2] == 0xA8) ? '8' : '9';
    return 6;
  }
  if (escape_len_at(s, n, p) == 0) return 0;
  out[0] = c;
  return 1;
}

// ---------- wave primitives ----------

DEV int wave_prefix_excl(int v) {
  // exclusive prefix sum across the 64-lane wave
  int lane = lane_id();
  for (int d = 1; d < WAVE; d <<= 1) {
    int up = __shfl_up(v, d);
    if (lane >= d) v += up;
  }
  return __shfl_up(v, 1) * (lane > 0 ? 1 : 0);
}

DEV int wave_sum(int v) {
  for (int d = WAVE / 2; d > 0; d >>= 1) v += __shfl_down(v, d);
  return __shfl(v, 0);
}

// ---------- cursor-based cooperative JSON emitter ----------
// One wave emits one JSON line; the cursor is wave-uniform (every helper
// leaves all 64 lanes with the same cursor). The W=false instantiation
// measures (stores compiled out; constant segments fold to `cur += K`).
// Shared by the Telegram (parse_encode.hip) and YouTube (yt_encode.hip)
// emitters; LIT(e, "...") supplies compile-time literal lengths.

DEV int u64_dec_len(uint64_t v) {
  // comparison ladder (v is wave-uniform at every call site: scalar
  // branches, no divide chain)
  if (v < 10ULL) return 1;
  if (v < 100ULL) return 2;
  if (v < 1000ULL) return 3;
  if (v < 10000ULL) return 4;
  if (v < 100000ULL) return 5;
  if (v < 1000000ULL) return 6;
  if (v < 10000000ULL) return 7;
  if (v < 100000000ULL) return 8;
  if (v < 1000000000ULL) return 9;
  if (v < 10000000000ULL) return 10;
  if (v < 100000000000ULL) return 11;
  if (v < 1000000000000ULL) return 12;
  if (v < 10000000000000ULL) return 13;
  if (v < 100000000000000ULL) return 14;
  if (v < 1000000000000000ULL) return 15;
  if (v < 10000000000000000ULL) return 16;
  if (v < 100000000000000000ULL) return 17;
  if (v < 1000000000000000000ULL) return 18;
  if (v < 10000000000000000000ULL) return 19;
  return 20;
}
DEV int i64_dec_len(int64_t v) {
  return (v < 0) ? 1 + u64_dec_len((uint64_t)(-v)) : u64_dec_len((uint64_t)v);
}

DEV void rfc3339_write(uint8_t* out, int cur, long long secs) {
  long long days = secs / 86400;
  long long rem = secs % 86400;
  if (rem < 0) { rem += 86400; days -= 1; }
  int hh = (int)(rem / 3600), mm = (int)((rem % 3600) / 60),
      ss = (int)(rem % 60);
  // civil_from_days (Howard Hinnant's algorithm)
  long long z = days + 719468;
  long long era = (z >= 0 ? z : z - 146096) / 146097;
  long long doe = z - era * 146097;
  long long yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  long long y = yoe + era * 400;
  long long doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  long long mp = (5 * doy + 2) / 153;
  long long d = doy - (153 * mp + 2) / 5 + 1;
  long long m = mp + (mp < 10 ? 3 : -9);
  y += (m <= 2);
  // Pack the 20 chars into 5 dwords in REGISTERS (a char buf[20] read
  // with a lane-indexed subscript spills to scratch — the write kernel
  // was latency-bound on private-segment traffic, not VALU).
  int yy = (int)y, mo = (int)m, dd = (int)d;
  auto pk = [](int a, int b, int c, int e) {
    return (unsigned)a | ((unsigned)b << 8) | ((unsigned)c << 16) |
           ((unsigned)e << 24);
  };
  unsigned w0 = pk('0' + (yy / 1000) % 10, '0' + (yy / 100) % 10,
                   '0' + (yy / 10) % 10, '0' + yy % 10);
  unsigned w1 = pk('-', '0' + mo / 10, '0' + mo % 10, '-');
  unsigned w2 = pk('0' + dd / 10, '0' + dd % 10, 'T', '0' + hh / 10);
  unsigned w3 = pk('0' + hh % 10, ':', '0' + mm / 10, '0' + mm % 10);
  unsigned w4 = pk(':', '0' + ss / 10, '0' + ss % 10, 'Z');
  int lane = lane_id();
  if (lane < 5) {
    unsigned v = w0;
    if (lane == 1) v = w1;
    else if (lane == 2) v = w2;
    else if (lane == 3) v = w3;
    else if (lane == 4) v = w4;
    nt_store_u32(out + cur + 4 * lane, v);
  }
}

// Tiny literals (<= 4 bytes: separators, quotes) become ONE immediate
// store instead of raw()'s striped-loop setup (~6 instrs + a global
// load per byte); longer literals keep the wave-cooperative copy. The
// all-immediate variant was measured SLOWER for large literals
// (single-lane serialization) — this is scoped to the tiny ones.
#define LIT(e, s)                                              \
  do {                                                         \
    if ((int)sizeof(s) - 1 <= 4)                               \
      (e).template lit_small<(int)sizeof(s) - 1>(s);           \
    else                                                       \
      (e).raw((const unsigned char*)(s), (int)sizeof(s) - 1);  \
  } while (0)

template <bool W>
struct JsonEmit {
  uint8_t* out;
  int cur;
  const uint8_t* lds;  // LDS literal pool base (tg_lits.h), may be null

  // Pooled constant fragment served from LDS: ds_read waits on
  // lgkmcnt, NOT the loads+stores vmcnt FIFO — so a literal after line
  // stores doesn't stall on those stores retiring (the write kernel's
  // dominant stall; see profiles/r02_valu_diet.md).
  DEV void lds_lit(int off, int n) {
    if (W) {
      const uint8_t* s = lds + off;
      int lane = lane_id();
      int nw = n >> 2;
      for (int k = lane; k < nw; k += WAVE) {
        unsigned w;
        __builtin_memcpy(&w, s + 4 * k, 4);
        nt_store_u32(out + cur + 4 * k, w);
      }
      for (int j = (nw << 2) + lane; j < n; j += WAVE)
        nt_store_u8(out + cur + j, s[j]);
    }
    cur += n;
  }

  template <int N, int M>
  DEV void lit_small(const char (&str)[M]) {
    if (W && lane_id() == 0) {
      if (N == 1) {
        out[cur] = (unsigned char)str[0];
      } else if (N == 2) {
        unsigned short v = (unsigned short)((unsigned char)str[0] |
                                            ((unsigned char)str[1] << 8));
        __builtin_memcpy(out + cur, &v, 2);
      } else {
        unsigned v = (unsigned)(unsigned char)str[0] |
                     ((unsigned)(unsigned char)str[1] << 8) |
                     ((N > 2 ? (unsigned)(unsigned char)str[2] : 0u) << 16) |
                     ((N > 3 ? (unsigned)(unsigned char)str[3] : 0u) << 24);
        if (N == 4)
          __builtin_memcpy(out + cur, &v, 4);
        else {
          __builtin_memcpy(out + cur, &v, 2);
          out[cur + 2] = (unsigned char)str[2];
        }
      }
    }
    cur += N;
  }

  // Striped copy, dword-granular: the emitters are memory-op bound on
  // the literal/template bytes (~1.3 KB of raw copies per 2 KB line) —
  // unaligned 4-byte loads/stores quarter the op count vs byte stripes.
  DEV void raw(const uint8_t* s, int n) {
    if (W) {
      int lane = lane_id();
      int nw = n >> 2;
      for (int k = lane; k < nw; k += WAVE) {
        unsigned w;
        __builtin_memcpy(&w, s + 4 * k, 4);
        nt_store_u32(out + cur + 4 * k, w);
      }
      for (int j = (nw << 2) + lane; j < n; j += WAVE)
        nt_store_u8(out + cur + j, s[j]);
    }
    cur += n;
  }

  // SWAR probe: does dword w contain any byte that Go-JSON escapes
  // (<0x20, ", \, <, >, &) or an 0xE2 (possible U+2028/29 leader)?
  // Classic haszero trick: (x-0x01010101) & ~x & 0x80808080 != 0 iff a
  // zero byte exists in x.
  static DEV bool swar_special(unsigned w) {
    unsigned hz;
    unsigned nonascii = w & 0x80808080u;
    // hasless(w, 0x20): exact for the "any byte" question (borrow
    // propagation can only add false positives, which just take the
    // exact path)
    unsigned lt20 = (w - 0x20202020u) & ~w & 0x80808080u;
    if (lt20) return true;
    hz = ((w ^ 0x22222222u) - 0x01010101u) & ~(w ^ 0x22222222u);   // "
    hz |= ((w ^ 0x5C5C5C5Cu) - 0x01010101u) & ~(w ^ 0x5C5C5C5Cu);  // \ .
    hz |= ((w ^ 0x3C3C3C3Cu) - 0x01010101u) & ~(w ^ 0x3C3C3C3Cu);  // <
    hz |= ((w ^ 0x3E3E3E3Eu) - 0x01010101u) & ~(w ^ 0x3E3E3E3Eu);  // >
    hz |= ((w ^ 0x26262626u) - 0x01010101u) & ~(w ^ 0x26262626u);  // &
    if (hz & 0x80808080u) return true;
    if (nonascii) {  // only 0xE2 needs the slow path among high bytes
      unsigned e2 = ((w ^ 0xE2E2E2E2u) - 0x01010101u) & ~(w ^ 0xE2E2E2E2u);
      if (e2 & 0x80808080u) return true;
    }
    return false;
  }

  // Escaped JSON string content (no quotes). BOTH passes scan clean
  // 256-byte blocks 4 bytes/lane with a SWAR special-byte probe; a
  // clean block is copied verbatim with dword stores on the write pass
  // (same ballot -> measure and write can never disagree on length).
  // Dirty or boundary-risk blocks fall back to the exact 64-byte
  // stripe path below.
  DEV void esc(const uint8_t* s, int n) {
    int lane = lane_id();
    int start = 0;
    while (start < n) {
      if (start + 4 * WAVE <= n) {
        // a U+2028/29 leader just before the block would swallow bytes
        // INSIDE it; route those rare blocks to the exact path
        bool risk = (start >= 1 && s[start - 1] == 0xE2) ||
                    (start >= 2 && s[start - 2] == 0xE2 &&
                     s[start - 1] == 0x80);
        if (!risk) {
          unsigned w;
          __builtin_memcpy(&w, s + start + 4 * lane, 4);
          if (__ballot(swar_special(w)) == 0) {
            if (W)
              nt_store_u32(out + cur + 4 * lane, w);
            cur += 4 * WAVE;   // every byte emits verbatim
            start += 4 * WAVE;
            continue;
          }
        }
      }
      esc_stripe(s, n, start, lane);
      start += WAVE;
    }
  }

  // One exact 64-byte stripe of the escape walk (the original path).
  DEV void esc_stripe(const uint8_t* s, int n, int start, int lane) {
    {
      int p = start + lane;
      int span = n - start;
      if (span > WAVE) span = WAVE;
      int el = (p < n) ? escape_len_at(s, n, p) : 0;
      unsigned long long dirty = __ballot(p < n && el != 1);
      if (dirty == 0) {
        if (W && p < n) nt_store_u8(out + cur + p - start, s[p]);
        cur += span;
        return;
      }
      if (W) {
        int off = wave_prefix_excl(el);
        if (p < n && el > 0) {
          // write the expansion straight to its global slot — a local
          // tmp[6] staging array spills to scratch (private-segment
          // latency dominated the write kernel)
          escape_write_at(s, n, p, out + cur + off);
        }
      }
      cur += wave_sum(el);
    }
  }

  // Wave-parallel decimal emission: lane j computes digit j-from-the-right
  // ((v / 10^j) % 10) so a 10-digit number costs ~1 division per LANE in
  // parallel instead of a serial divide chain on lane 0. 10^lane comes
  // from a constant table (the per-lane multiply loop cost ~lane VALU
  // ops per call).
  DEV void u64(unsigned long long v) {
    static const unsigned long long POW10[20] = {
        1ULL, 10ULL, 100ULL, 1000ULL, 10000ULL, 100000ULL, 1000000ULL,
        10000000ULL, 100000000ULL, 1000000000ULL, 10000000000ULL,
        100000000000ULL, 1000000000000ULL, 10000000000000ULL,
        100000000000000ULL, 1000000000000000ULL, 10000000000000000ULL,
        100000000000000000ULL, 1000000000000000000ULL,
        10000000000000000000ULL};
    int n = u64_dec_len(v);
    if (W) {
      int lane = lane_id();
      if (lane < n) {
        unsigned d;
        if (v <= 0xFFFFFFFFULL)
          d = ((unsigned)v / (unsigned)POW10[lane]) % 10u;
        else
          d = (unsigned)((v / POW10[lane]) % 10ULL);
        nt_store_u8(out + cur + n - 1 - lane, (uint8_t)('0' + d));
      }
    }
    cur += n;
  }
  DEV void i64(long long v) {
    if (v < 0) { LIT(*this, "-"); u64((unsigned long long)(-v)); }
    else u64((unsigned long long)v);
  }

  // RFC3339 UTC "YYYY-MM-DDTHH:MM:SSZ" from unix seconds (fixed 20 bytes).
  DEV void rfc3339(long long secs) {
    if (W) rfc3339_write(out, cur, secs);
    cur += 20;
  }
};

// ---------- hashing (FNV-1a 64, matches python oracle) ----------
DEV uint64_t fnv1a64(const uint8_t* s, int n) {
  uint64_t h = 1469598103934665603ULL;
  for (int i = 0; i < n; ++i) {
    h ^= s[i];
    h *= 1099511628211ULL;
  }
  return h;
}

}  // namespace crawl
