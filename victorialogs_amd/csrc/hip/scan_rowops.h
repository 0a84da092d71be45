// Per-row device operations of the block-scan engine: accessors, UTF-8,
// substring/phrase/prefix/sequence matchers, value formatters and parsers,
// regex fast paths + Glushkov NFA, and the per-row eval dispatchers.
//
// Included INSIDE `namespace vl` by scan_kernels.hip (device build) and by
// the host row-ops fuzz harness (host build, with __device__ & friends
// defined away) so the exact per-row code the GPU runs can be fuzzed at
// scale on the CPU against the oracle.  Keep this file free of wavefront
// intrinsics (ballots/shuffles/LDS) — those live in scan_kernels.hip.
#pragma once

// Non-ASCII token-rune ranges (tokenizer.go:142-148); see unicode_ranges.inc.
#include "../core/unicode_ranges.inc"

// Host local timezone offset (nsecs) for RFC3339 inputs without a timezone
// suffix — the reference uses the process-local zone of the current time
// (timeutil.GetLocalTimezoneOffsetNsecs).  Defined by scan_kernels.hip
// (device global, set via hipMemcpyToSymbol per device at stage build) and
// by the host harness/emu builds (plain global).
extern __device__ int64_t g_vl_local_tz_nsecs;

__device__ __forceinline__ bool d_is_token_char(uint8_t c) {
  // tokenizer.go:132-140: [a-zA-Z0-9_]
  return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') ||
         (c >= '0' && c <= '9') || c == '_';
}

__device__ bool d_is_token_rune(uint32_t r) {
  if (r < 0x80) return d_is_token_char(uint8_t(r));
  int lo = 0, hi = kTokenRuneRangesCount - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    if (r < kTokenRuneRanges[mid][0]) {
      hi = mid - 1;
    } else if (r > kTokenRuneRanges[mid][1]) {
      lo = mid + 1;
    } else {
      return true;
    }
  }
  return false;
}

// ---- memory accessors ----
// All string matching routines are templated over an accessor that serves
// 8-byte-aligned u64 loads and byte loads at tile/global byte offsets.

struct GlobalAcc {
  const uint8_t* base;  // 16-byte aligned (arena allocations)
  __device__ __forceinline__ uint64_t u64a(long off) const {  // off % 8 == 0
    return *(const uint64_t*)(base + off);
  }
  // off % 16 == 0: both halves of one 16-byte window (single dwordx4 load)
  __device__ __forceinline__ void u64a2(long off, uint64_t* x0,
                                        uint64_t* x1) const {
    const uint64_t* p = (const uint64_t*)(base + off);
    *x0 = p[0];
    *x1 = p[1];
  }
  __device__ __forceinline__ uint8_t u8(long off) const { return base[off]; }
};

// LDS tile with 16-byte-slot swizzle: slot u -> u ^ ((u>>4) & 15), bijective
// within every 4 KiB window; an aligned u64 never crosses its 16-byte slot.
struct TileAcc {
  const uint8_t* tile;
  __device__ __forceinline__ long swz(long off) const {
    long u = off >> 4;
    return ((u ^ ((u >> 4) & 15)) << 4) | (off & 15);
  }
  __device__ __forceinline__ uint64_t u64a(long off) const {
    return *(const uint64_t*)(tile + swz(off));
  }
  // off % 16 == 0: the whole window lives in ONE swizzled slot, so this is
  // a single ds_read_b128
  __device__ __forceinline__ void u64a2(long off, uint64_t* x0,
                                        uint64_t* x1) const {
    const uint64_t* p = (const uint64_t*)(tile + swz(off));
    *x0 = p[0];
    *x1 = p[1];
  }
  __device__ __forceinline__ uint8_t u8(long off) const { return tile[swz(off)]; }
};

// Byte-wise ASCII tolower view over another accessor.  Only correct for
// ASCII bytes; rows containing non-ASCII bytes are resolved on the host via
// the override bitmaps (DevLeafBlock.ovr_mask), so corruption of >=0x80
// bytes is harmless.  The SWAR form is carry-safe across bytes.
template <typename A>
struct LowerAcc {
  A a;
  __device__ __forceinline__ static uint64_t lower64(uint64_t x) {
    uint64_t low7 = x & 0x7F7F7F7F7F7F7F7FULL;
    uint64_t ge_a = low7 + 0x3F3F3F3F3F3F3F3FULL;   // high bit: byte >= 0x41
    uint64_t ge_z1 = low7 + 0x2525252525252525ULL;  // high bit: byte >= 0x5B
    uint64_t is_az = ge_a & ~ge_z1 & ~x & 0x8080808080808080ULL;
    return x | (is_az >> 2);
  }
  __device__ __forceinline__ uint64_t u64a(long off) const {
    return lower64(a.u64a(off));
  }
  __device__ __forceinline__ void u64a2(long off, uint64_t* x0,
                                        uint64_t* x1) const {
    a.u64a2(off, x0, x1);
    *x0 = lower64(*x0);
    *x1 = lower64(*x1);
  }
  __device__ __forceinline__ uint8_t u8(long off) const {
    uint8_t c = a.u8(off);
    return uint8_t(c - 'A') < 26 ? uint8_t(c + 0x20) : c;
  }
};

// Go utf8.DecodeRuneInString semantics (0xFFFD,1 on invalid), reading bytes
// [s, s+n) of the accessor at base offset `off`.
template <typename A>
__device__ uint32_t d_utf8_decode(const A& a, long off, long n, int* size) {
  *size = 1;
  if (n <= 0) return 0xFFFD;
  uint8_t c0 = a.u8(off);
  if (c0 < 0x80) return c0;
  int len;
  uint32_t r, lo;
  if ((c0 & 0xE0) == 0xC0) {
    len = 2; r = c0 & 0x1F; lo = 0x80;
  } else if ((c0 & 0xF0) == 0xE0) {
    len = 3; r = c0 & 0x0F; lo = 0x800;
  } else if ((c0 & 0xF8) == 0xF0) {
    len = 4; r = c0 & 0x07; lo = 0x10000;
  } else {
    return 0xFFFD;
  }
  if (len > n) return 0xFFFD;
  for (int i = 1; i < len; i++) {
    uint8_t c = a.u8(off + i);
    if ((c & 0xC0) != 0x80) return 0xFFFD;
    r = (r << 6) | (c & 0x3F);
  }
  if (r < lo || r > 0x10FFFF || (r >= 0xD800 && r <= 0xDFFF)) return 0xFFFD;
  *size = len;
  return r;
}

template <typename A>
__device__ uint32_t d_utf8_decode_last(const A& a, long off, long n, int* size) {
  *size = 1;
  if (n <= 0) return 0xFFFD;
  long start = n - 1;
  if (a.u8(off + start) < 0x80) return a.u8(off + start);
  long lim = n >= 4 ? n - 4 : 0;
  while (start > lim && (a.u8(off + start) & 0xC0) == 0x80) start--;
  int sz;
  uint32_t r = d_utf8_decode(a, off + start, n - start, &sz);
  if (start + sz != n) return 0xFFFD;
  *size = sz;
  return r;
}

// strings.Index over accessor bytes [s0, s0+sn): first occurrence of the
// operand (global memory, byte-addressable) or -1.  SWAR scan over ALIGNED
// u64 windows with a TWO-byte anchor (first byte at k AND second byte at
// k+1; the window's top byte keeps its candidate bit since its successor
// lives in the next window) — a first-byte-only anchor left ~6 candidate
// verifies per 256 B of random text, and the divergent verify loop was the
// dominant cost of non-matching scans (18.4 ms vs 5.1 ms per 100M-row pass
// measured on the phrase kernel).  Head/tail bytes are masked out.
__device__ __forceinline__ uint64_t d_swar_zero(uint64_t t) {
  return (t - 0x0101010101010101ULL) & ~t & 0x8080808080808080ULL;
}
// NOTE (measured dead end): replacing the byte-wise candidate verify with
// aligned u64-window compares regressed EVERY config (headline 11.7 -> 9.2
// G rows/s, non-matching 10.5 -> 18.4 ms) — the extra live values collapse
// the hot clone's codegen (same class as the round-1 descriptor-reload
// regression).  The byte loop below stays.

template <typename A>
__device__ long d_index_at(const A& a, long s0, long sn, const uint8_t* sub,
                           long subn) {
  if (subn == 0) return 0;
  if (subn > sn) return -1;
  const uint8_t c0 = sub[0];
  const uint64_t pat = 0x0101010101010101ULL * c0;
  const uint64_t pat1 = subn > 1 ? 0x0101010101010101ULL * sub[1] : 0;
  const long last = s0 + sn - subn;  // last valid start (absolute)
  // 32-byte iterations: two ds_read_b128 windows issued together (a
  // swizzled tile slot holds one 16-byte window), overlapping the LDS
  // latency that serialized the original one-8-byte-window-per-iteration
  // loop.  Scratch-buffer accessors are sized for reads up to 32 bytes
  // past the aligned start of the last valid position.
  for (long w = s0 & ~15L; w <= last; w += 32) {
    uint64_t x0, x1, x2, x3;
    a.u64a2(w, &x0, &x1);
    a.u64a2(w + 16, &x2, &x3);
    uint64_t hitA = d_swar_zero(x0 ^ pat);
    uint64_t hitB = d_swar_zero(x1 ^ pat);
    uint64_t hitC = d_swar_zero(x2 ^ pat);
    uint64_t hitD = d_swar_zero(x3 ^ pat);
    if (subn > 1 && (hitA | hitB | hitC | hitD)) {
      // two-byte anchor: require the second pattern byte at k+1 (the
      // zero-scan has no false negatives, so pruning is sound; the last
      // byte's successor lives in the next window and keeps its bit)
      const uint64_t h1A = d_swar_zero(x0 ^ pat1);
      const uint64_t h1B = d_swar_zero(x1 ^ pat1);
      const uint64_t h1C = d_swar_zero(x2 ^ pat1);
      const uint64_t h1D = d_swar_zero(x3 ^ pat1);
      hitA &= (h1A >> 8) | ((h1B & 0x80) << 56);
      hitB &= (h1B >> 8) | ((h1C & 0x80) << 56);
      hitC &= (h1C >> 8) | ((h1D & 0x80) << 56);
      hitD &= (h1D >> 8) | 0x8000000000000000ULL;
    }
    // mask hits before s0 (first iteration only) -- hit bit for byte k is
    // bit 8k+7; s0 - w is in [0, 15]
    if (w < s0) {
      const long off = s0 - w;
      if (off >= 8) {
        hitA = 0;
        hitB &= ~((uint64_t(1) << ((off - 8) * 8)) - 1);
      } else {
        hitA &= ~((uint64_t(1) << (off * 8)) - 1);
      }
    }
    while (hitA | hitB | hitC | hitD) {
      long k;
      if (hitA) {
        k = long(__builtin_ctzll(hitA) >> 3);
        hitA &= hitA - 1;
      } else if (hitB) {
        k = 8 + long(__builtin_ctzll(hitB) >> 3);
        hitB &= hitB - 1;
      } else if (hitC) {
        k = 16 + long(__builtin_ctzll(hitC) >> 3);
        hitC &= hitC - 1;
      } else {
        k = 24 + long(__builtin_ctzll(hitD) >> 3);
        hitD &= hitD - 1;
      }
      const long pos = w + k;
      if (pos > last) return -1;
      // verify from byte 0: the SWAR zero-scan's borrow cascade can flag a
      // byte equal to c0^1 right after a true candidate ("101" vs "11")
      bool eq = true;
      for (long i = 0; i < subn; i++) {
        if (a.u8(pos + i) != sub[i]) {
          eq = false;
          break;
        }
      }
      if (eq) return pos - s0;
    }
  }
  return -1;
}

// getPhrasePos (filter_phrase.go:220-270) over accessor bytes [s0, s0+sn);
// returns the match position or -1.
template <typename A>
__device__ long d_get_phrase_pos_at(const A& a, long s0, long sn, const uint8_t* ph,
                                    long phn, uint8_t flags) {
  if (phn == 0) return 0;
  if (phn > sn) return -1;
  long pos = 0;
  for (;;) {
    long n = d_index_at(a, s0 + pos, sn - pos, ph, phn);
    if (n < 0) return -1;
    pos += n;
    if ((flags & kPhraseStartsToken) && pos > 0) {
      uint32_t rb = a.u8(s0 + pos - 1);
      if (rb >= 0x80) {
        int sz;
        rb = d_utf8_decode_last(a, s0, pos, &sz);
      }
      if (rb == 0xFFFD || d_is_token_rune(rb)) {
        pos++;
        continue;
      }
    }
    if ((flags & kPhraseEndsToken) && pos + phn < sn) {
      uint32_t ra = a.u8(s0 + pos + phn);
      if (ra >= 0x80) {
        int sz;
        ra = d_utf8_decode(a, s0 + pos + phn, sn - pos - phn, &sz);
      }
      if (ra == 0xFFFD || d_is_token_rune(ra)) {
        pos++;
        continue;
      }
    }
    return pos;
  }
}

template <typename A>
__device__ __forceinline__ bool d_match_phrase_at(const A& a, long s0, long sn,
                                                  const uint8_t* ph, long phn,
                                                  uint8_t flags) {
  if (phn == 0) return sn == 0;  // filter_phrase.go:212-215
  return d_get_phrase_pos_at(a, s0, sn, ph, phn, flags) >= 0;
}

// matchPrefix (filter_prefix.go:318-352): empty prefix matches non-empty s;
// boundary check only at the start.
template <typename A>
__device__ bool d_match_prefix_at(const A& a, long s0, long sn, const uint8_t* pf,
                                  long pfn, uint8_t flags) {
  if (pfn == 0) return sn > 0;
  if (pfn > sn) return false;
  long off = 0;
  for (;;) {
    long n = d_index_at(a, s0 + off, sn - off, pf, pfn);
    if (n < 0) return false;
    off += n;
    if ((flags & kPhraseStartsToken) && off > 0) {
      uint32_t rb = a.u8(s0 + off - 1);
      if (rb >= 0x80) {
        int sz;
        rb = d_utf8_decode_last(a, s0, off, &sz);
      }
      if (rb == 0xFFFD || d_is_token_rune(rb)) {
        off++;
        continue;
      }
    }
    return true;
  }
}

template <typename A>
__device__ __forceinline__ bool d_has_prefix_bytes(const A& a, long s0, long sn,
                                                   const uint8_t* pf, long pfn) {
  if (pfn > sn) return false;
  for (long i = 0; i < pfn; i++) {
    if (a.u8(s0 + i) != pf[i]) return false;
  }
  return true;
}

// matchSequence (filter_sequence.go:260-269) over a serialized phrase list:
// blob = u16 n, then per phrase { u16 len, u8 flags, bytes }.
template <typename A>
__device__ bool d_match_sequence_at(const A& a, long s0, long sn,
                                    const uint8_t* blob) {
  uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t* p = blob + 2;
  for (uint16_t i = 0; i < n; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    uint8_t flags = p[2];
    p += 3;
    long pos = d_get_phrase_pos_at(a, s0, sn, p, len, flags);
    if (pos < 0) return false;
    s0 += pos + len;
    sn -= pos + len;
    p += len;
  }
  return true;
}

// ---- number/ip/timestamp formatting (device mirrors of values.cpp) ----

__device__ int d_format_u64(char* buf, uint64_t v) {
  char tmp[20];
  int n = 0;
  do {
    tmp[n++] = char('0' + v % 10);
    v /= 10;
  } while (v);
  for (int i = 0; i < n; i++) buf[i] = tmp[n - 1 - i];
  return n;
}

__device__ int d_format_i64(char* buf, int64_t v) {
  if (v < 0) {
    buf[0] = '-';
    uint64_t u = ~uint64_t(v) + 1;  // handles INT64_MIN
    return 1 + d_format_u64(buf + 1, u);
  }
  return d_format_u64(buf, uint64_t(v));
}

__device__ int d_format_ipv4(char* buf, uint32_t ip) {
  int n = d_format_u64(buf, (ip >> 24) & 255);
  buf[n++] = '.';
  n += d_format_u64(buf + n, (ip >> 16) & 255);
  buf[n++] = '.';
  n += d_format_u64(buf + n, (ip >> 8) & 255);
  buf[n++] = '.';
  n += d_format_u64(buf + n, ip & 255);
  return n;
}

__device__ void d_pad2(char* buf, int v) {
  buf[0] = char('0' + v / 10);
  buf[1] = char('0' + v % 10);
}

// time.Unix(0,nsecs).UTC() "2006-01-02T15:04:05.000Z" (values_encoder.go:1420-1424)
__device__ int d_format_iso8601(char* buf, int64_t nsecs) {
  int64_t secs = nsecs / 1000000000;
  int64_t rem = nsecs % 1000000000;
  if (rem < 0) {
    rem += 1000000000;
    secs--;
  }
  int64_t days = secs / 86400;
  int64_t sod = secs % 86400;
  if (sod < 0) {
    sod += 86400;
    days--;
  }
  int64_t z = days + 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  int64_t doe = z - era * 146097;
  int64_t yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t yy = yoe + era * 400;
  int64_t doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  int64_t mp = (5 * doy + 2) / 153;
  int d = int(doy - (153 * mp + 2) / 5 + 1);
  int m = int(mp + (mp < 10 ? 3 : -9));
  int64_t y = yy + (m <= 2);
  int msec = int(rem / 1000000);
  int n = 0;
  if (y >= 1000) {
    n = d_format_u64(buf, uint64_t(y));
  } else {
    buf[0] = '0';
    buf[1] = char('0' + (y / 100) % 10);
    buf[2] = char('0' + (y / 10) % 10);
    buf[3] = char('0' + y % 10);
    n = 4;
  }
  buf[n++] = '-';
  d_pad2(buf + n, m); n += 2;
  buf[n++] = '-';
  d_pad2(buf + n, d); n += 2;
  buf[n++] = 'T';
  d_pad2(buf + n, int(sod / 3600)); n += 2;
  buf[n++] = ':';
  d_pad2(buf + n, int(sod % 3600 / 60)); n += 2;
  buf[n++] = ':';
  d_pad2(buf + n, int(sod % 60)); n += 2;
  buf[n++] = '.';
  buf[n++] = char('0' + msec / 100);
  buf[n++] = char('0' + msec / 10 % 10);
  buf[n++] = char('0' + msec % 10);
  buf[n++] = 'Z';
  return n;
}

// Accessor over a tiny per-thread scratch buffer (formatted numbers).
struct BufAcc {
  const uint8_t* base;
  __device__ __forceinline__ uint64_t u64a(long off) const {
    uint64_t v;
    __builtin_memcpy(&v, base + off, 8);
    return v;
  }
  __device__ __forceinline__ void u64a2(long off, uint64_t* x0,
                                        uint64_t* x1) const {
    __builtin_memcpy(x0, base + off, 8);
    __builtin_memcpy(x1, base + off + 8, 8);
  }
  __device__ __forceinline__ uint8_t u8(long off) const { return base[off]; }
};


// ---- device number parsing (mirrors values.cpp; for range-on-string) ----

// Go math.Pow10 restated (values.cpp go_pow10)
__device__ static const double kPow10Tab[32] = {
    1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10, 1e11, 1e12,
    1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22, 1e23,
    1e24, 1e25, 1e26, 1e27, 1e28, 1e29, 1e30, 1e31};
__device__ static const double kPow10PosTab32[10] = {
    1e0, 1e32, 1e64, 1e96, 1e128, 1e160, 1e192, 1e224, 1e256, 1e288};

__device__ inline double d_go_pow10(int n) {
  if (n >= 0 && n <= 308) return kPow10PosTab32[unsigned(n) / 32] * kPow10Tab[unsigned(n) % 32];
  if (n <= 0 && n >= -323) {
    return 1.0 / (kPow10PosTab32[unsigned(-n) / 32] * kPow10Tab[unsigned(-n) % 32]);
  }
  return n > 0 ? __builtin_inf() : 0.0;
}

// tryParseUint64 (values_encoder.go:553-585)
template <typename A>
__device__ bool d_try_parse_uint64(const A& a, long s0, long sn, uint64_t* out) {
  if (sn == 0 || sn > 26) return false;
  if (sn > 1 && a.u8(s0) == '0') return false;
  uint64_t n = 0;
  for (long i = 0; i < sn; i++) {
    uint8_t ch = a.u8(s0 + i);
    if (ch == '_') continue;
    if (ch < '0' || ch > '9') return false;
    if (n > 1844674407370955161ULL) return false;  // UINT64_MAX/10
    n *= 10;
    uint64_t d = ch - '0';
    uint64_t n1 = n + d;
    if (n1 < n) return false;
    n = n1;
  }
  *out = n;
  return true;
}

// tryParseFloat64 non-exact (values_encoder.go:788-848, isExact=false)
template <typename A>
__device__ bool d_try_parse_float64(const A& a, long s0, long sn, double* out) {
  if (sn == 0 || sn > 27) return false;
  bool minus = a.u8(s0) == '-';
  if (minus) {
    s0++;
    sn--;
  }
  long ndot = -1;
  for (long i = 0; i < sn; i++) {
    if (a.u8(s0 + i) == '.') {
      ndot = i;
      break;
    }
  }
  if (ndot < 0) {
    uint64_t n;
    if (!d_try_parse_uint64(a, s0, sn, &n)) return false;
    double f = double(n);
    *out = minus ? -f : f;
    return true;
  }
  if (ndot == 0 || ndot == sn - 1) return false;
  uint64_t n_int;
  if (!d_try_parse_uint64(a, s0, ndot, &n_int)) return false;
  long f0 = s0 + ndot + 1, fn = sn - ndot - 1;
  long skip = 0;
  while (skip < fn - 1 && a.u8(f0 + skip) == '0') skip++;
  uint64_t n_frac;
  if (!d_try_parse_uint64(a, f0 + skip, fn - skip, &n_frac)) return false;
  int underscores = 0;
  for (long i = 0; i < fn; i++) {
    if (a.u8(f0 + i) == '_') underscores++;
  }
  double p10 = d_go_pow10(underscores - int(fn));
  double f = fma(double(n_frac), p10, double(n_int));
  *out = minus ? -f : f;
  return true;
}

// tryParseFloat64Prefix (values_encoder.go:762-773); advances *s0/*sn
template <typename A>
__device__ bool d_parse_float64_prefix(const A& a, long* s0, long* sn, double* f) {
  long i = 0;
  while (i < *sn) {
    uint8_t c = a.u8(*s0 + i);
    if ((c >= '0' && c <= '9') || c == '.' || c == '_') {
      i++;
    } else {
      break;
    }
  }
  if (i == 0) return false;
  if (!d_try_parse_float64(a, *s0, i, f)) return false;
  *s0 += i;
  *sn -= i;
  return true;
}

__device__ inline long long d_add_i64_no_overflow(long long n, double f) {
  long long x = (long long)(f);
  if (n < 0 || x < 0 || x > 0x7FFFFFFFFFFFFFFFLL - n) return 0x7FFFFFFFFFFFFFFFLL;
  return n + x;
}

template <typename A>
__device__ bool d_has_prefix(const A& a, long s0, long sn, const char* p, int np) {
  if (sn < np) return false;
  for (int i = 0; i < np; i++) {
    if (a.u8(s0 + i) != uint8_t(p[i])) return false;
  }
  return true;
}

// tryParseDuration (values_encoder.go:990-1061)
template <typename A>
__device__ bool d_try_parse_duration(const A& a, long s0, long sn, long long* out) {
  if (sn == 0) return false;
  bool minus = a.u8(s0) == '-';
  if (minus) {
    s0++;
    sn--;
  }
  long long nsecs = 0;
  while (sn > 0) {
    double f;
    if (!d_parse_float64_prefix(a, &s0, &sn, &f)) return false;
    if (sn == 0) return false;
    if (sn >= 3 && a.u8(s0) == 0xC2 && a.u8(s0 + 1) == 0xB5 && a.u8(s0 + 2) == 's') {
      nsecs = d_add_i64_no_overflow(nsecs, f * 1000);
      s0 += 3; sn -= 3;
      continue;
    }
    if (d_has_prefix(a, s0, sn, "ms", 2)) {
      nsecs = d_add_i64_no_overflow(nsecs, f * 1000000);
      s0 += 2; sn -= 2;
      continue;
    }
    if (d_has_prefix(a, s0, sn, "ns", 2)) {
      nsecs = d_add_i64_no_overflow(nsecs, f);
      s0 += 2; sn -= 2;
      continue;
    }
    uint8_t c = a.u8(s0);
    double mult;
    switch (c) {
      case 'y': mult = 365.0 * 24 * 3600 * 1e9; break;
      case 'w': mult = 7.0 * 24 * 3600 * 1e9; break;
      case 'd': mult = 24.0 * 3600 * 1e9; break;
      case 'h': mult = 3600e9; break;
      case 'm': mult = 60e9; break;
      case 's': mult = 1e9; break;
      default: return false;
    }
    nsecs = d_add_i64_no_overflow(nsecs, f * mult);
    s0 += 1; sn -= 1;
  }
  *out = minus ? -nsecs : nsecs;
  return true;
}

// tryParseBytes (values_encoder.go:855-966)
template <typename A>
__device__ bool d_try_parse_bytes(const A& a, long s0, long sn, long long* out) {
  if (sn == 0) return false;
  bool minus = a.u8(s0) == '-';
  if (minus) {
    s0++;
    sn--;
  }
  long long n = 0;
  while (sn > 0) {
    double f;
    if (!d_parse_float64_prefix(a, &s0, &sn, &f)) return false;
    if (sn == 0) {
      double ip = trunc(f);
      if (f != ip) return false;  // no suffix: integers only
      n = d_add_i64_no_overflow(n, f);
      continue;
    }
    bool matched = false;
    if (sn >= 3) {
      const char* s3[] = {"KiB", "MiB", "GiB", "TiB"};
      const double m3[] = {1024.0, 1048576.0, 1073741824.0, 1099511627776.0};
      for (int i = 0; i < 4 && !matched; i++) {
        if (d_has_prefix(a, s0, sn, s3[i], 3)) {
          n = d_add_i64_no_overflow(n, f * m3[i]);
          s0 += 3; sn -= 3;
          matched = true;
        }
      }
    }
    if (!matched && sn >= 2) {
      const char* s2[] = {"Ki", "Mi", "Gi", "Ti", "KB", "MB", "GB", "TB"};
      const double m2[] = {1024.0, 1048576.0, 1073741824.0, 1099511627776.0,
                           1e3, 1e6, 1e9, 1e12};
      for (int i = 0; i < 8 && !matched; i++) {
        if (d_has_prefix(a, s0, sn, s2[i], 2)) {
          n = d_add_i64_no_overflow(n, f * m2[i]);
          s0 += 2; sn -= 2;
          matched = true;
        }
      }
    }
    if (!matched) {
      const char* s1 = "BKMGT";
      const double m1[] = {1.0, 1e3, 1e6, 1e9, 1e12};
      for (int i = 0; i < 5 && !matched; i++) {
        if (a.u8(s0) == uint8_t(s1[i])) {
          n = d_add_i64_no_overflow(n, f * m1[i]);
          s0 += 1; sn -= 1;
          matched = true;
        }
      }
    }
    if (!matched) return false;
  }
  *out = minus ? -n : n;
  return true;
}

// parseMathNumber subset (pipe_math.go:1066-1080; same legs as the host
// parse_math_number in values.cpp -- float, duration, bytes; others NaN)
// tryParseDateUint64 subset used by RFC3339 parsing (values_encoder.go:552+):
// plain decimal digits, with the reference's 2-digit fast-path quirk of
// checking only the first char (mirrored in host try_parse_date_uint64)
template <typename A>
__device__ bool d_parse_date_u64(const A& a, long s0, long sn, uint64_t* out) {
  if (sn == 0 || sn > 18) return false;
  if (sn == 2) {
    uint8_t c0 = a.u8(s0);
    if (c0 < '0' || c0 > '9') return false;
    *out = 10 * uint64_t(c0 - '0') + uint64_t(uint8_t(a.u8(s0 + 1) - '0'));
    return true;
  }
  uint64_t v = 0;
  for (long i = 0; i < sn; i++) {
    uint8_t c = a.u8(s0 + i);
    if (c < '0' || c > '9') return false;
    v = v * 10 + (c - '0');
  }
  *out = v;
  return true;
}

__device__ inline int64_t d_days_from_civil(int64_t y, int m, int64_t d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  int64_t yoe = y - era * 400;
  int64_t doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  int64_t doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + doe - 719468;
}

// tryParseTimestampSecs (values_encoder.go:469-550); consumes the leading
// "YYYY-MM-DD[T ]hh:mm:ss", returns seconds + the consumed length
template <typename A>
__device__ bool d_parse_ts_secs(const A& a, long s0, long sn, int64_t* secs,
                                long* consumed) {
  if (sn < 19) return false;
  uint64_t n;
  if (a.u8(s0 + 4) != '-') return false;
  if (!d_parse_date_u64(a, s0, 4, &n) || n < 1677 || n > 2262) return false;
  int64_t year = int64_t(n);
  long i = 5;
  if (a.u8(s0 + i + 2) != '-') return false;
  if (!d_parse_date_u64(a, s0 + i, 2, &n)) return false;
  int64_t month = int64_t(n);
  i += 3;
  uint8_t delim = a.u8(s0 + i + 2);
  if (delim != 'T' && delim != ' ') return false;
  if (!d_parse_date_u64(a, s0 + i, 2, &n)) return false;
  int64_t day = int64_t(n);
  i += 3;
  if (a.u8(s0 + i + 2) != ':') return false;
  if (!d_parse_date_u64(a, s0 + i, 2, &n) || n > 60) return false;
  int64_t hour = int64_t(n);
  i += 3;
  if (a.u8(s0 + i + 2) != ':') return false;
  if (!d_parse_date_u64(a, s0 + i, 2, &n) || n > 60) return false;
  int64_t minute = int64_t(n);
  i += 3;
  if (!d_parse_date_u64(a, s0 + i, 2, &n) || n > 60) return false;
  int64_t sec = int64_t(n);
  i += 2;
  // Go time.Date normalization: month/day may overflow their ranges
  int64_t ym = (month - 1);
  int64_t yadd = ym >= 0 ? ym / 12 : -((-ym + 11) / 12);
  int64_t mo = ym - yadd * 12 + 1;
  int64_t days = d_days_from_civil(year + yadd, int(mo), day);
  int64_t sv = days * 86400 + hour * 3600 + minute * 60 + sec;
  // values_encoder.go:545-548: reject timestamps whose nsecs overflow int64
  if (sv < INT64_MIN / 1000000000 || sv >= INT64_MAX / 1000000000) {
    return false;
  }
  *secs = sv;
  *consumed = i;
  return true;
}

// TryParseTimestampRFC3339Nano (values_encoder.go:340-381); no-timezone
// inputs use a zero local offset (the runtime boxes are UTC)
template <typename A>
__device__ bool d_parse_rfc3339(const A& a, long s0, long sn, int64_t* out) {
  if (sn < 19) return false;
  int64_t secs;
  long used;
  if (!d_parse_ts_secs(a, s0, sn, &secs, &used)) return false;
  long i = s0 + used, n = sn - used;
  int64_t nsecs = secs * 1000000000;
  // parseTimezoneOffset (values_encoder.go:383-406)
  if (n > 0 && a.u8(i + n - 1) == 'Z') {
    n--;
  } else {
    long tz = -1;
    for (long k = n - 1; k >= 0; k--) {
      uint8_t c = a.u8(i + k);
      if (c == '+' || c == '-') {
        tz = k;
        break;
      }
    }
    if (tz >= 0) {
      long on = n - tz - 1;
      if (on != 5 || a.u8(i + tz + 3) != ':') return false;
      uint64_t hh, mm;
      if (!d_parse_date_u64(a, i + tz + 1, 2, &hh) || hh > 24) return false;
      if (!d_parse_date_u64(a, i + tz + 4, 2, &mm) || mm > 60) return false;
      int64_t off = int64_t(hh) * 3600000000000LL + int64_t(mm) * 60000000000LL;
      if (a.u8(i + tz) == '-') off = -off;
      nsecs -= off;
      n = tz;
    } else {
      // no timezone suffix: host local offset, set once per device at stage
      // build (GetLocalTimezoneOffsetNsecs semantics, timezone.go:9-19)
      nsecs -= g_vl_local_tz_nsecs;
    }
  }
  if (n == 0) {
    *out = nsecs;
    return true;
  }
  if (a.u8(i) == '.') {
    i++;
    n--;
  }
  if (n > 9) return false;
  uint64_t frac;
  if (!d_parse_date_u64(a, i, n, &frac)) return false;
  for (long k = n; k < 9; k++) frac *= 10;
  *out = nsecs + int64_t(frac);
  return true;
}

template <typename A>
struct AccReader {
  const A* a;
  long s0;
  __device__ uint8_t u8(long i) const { return a->u8(s0 + i); }
};

template <typename A>
__device__ double d_parse_math_number(const A& a, long s0, long sn) {
  double f;
  if (sn > 0 && d_try_parse_float64(a, s0, sn, &f)) return f;
  long long v;
  if (sn > 0 && d_try_parse_duration(a, s0, sn, &v)) return double(v);
  if (sn > 0 && d_try_parse_bytes(a, s0, sn, &v)) return double(v);
  AccReader<A> r{&a, s0};
  if (sn > 0 && vl_pf::pf_is_likely_number(r, sn)) {
    double d;
    if (vl_pf::go_parse_float(r, sn, &d)) return d;
    int64_t iv;
    if (vl_pf::go_parse_int0(r, sn, &iv)) return double(iv);
  }
  int64_t ts;
  if (d_parse_rfc3339(a, s0, sn, &ts)) return double(ts);
  uint32_t ip;
  if (d_try_parse_ipv4(a, s0, sn, &ip)) return double(ip);
  return __builtin_nan("");
}


// value in a sorted string set: blob = u32 n, u32 offs[n+1], bytes
template <typename A>
__device__ bool d_in_sorted_str(const uint8_t* blob, const A& a, long s0, long sn) {
  uint32_t n;
  __builtin_memcpy(&n, blob, 4);
  const uint8_t* offs = blob + 4;
  const uint8_t* data = blob + 4 + size_t(n + 1) * 4;
  auto off_at = [&](uint32_t i) {
    uint32_t o;
    __builtin_memcpy(&o, offs + size_t(i) * 4, 4);
    return o;
  };
  uint32_t lo = 0, hi = n;
  while (lo < hi) {
    uint32_t mid = (lo + hi) / 2;
    uint32_t mo = off_at(mid), ml = off_at(mid + 1) - mo;
    // lexicographic compare set[mid] vs row
    int c = 0;
    long k = 0;
    long lim = ml < uint32_t(sn) ? ml : uint32_t(sn);
    for (; k < lim; k++) {
      uint8_t cb = data[mo + k], rb = a.u8(s0 + k);
      if (cb != rb) {
        c = cb < rb ? -1 : 1;
        break;
      }
    }
    if (c == 0) c = long(ml) < sn ? -1 : (long(ml) > sn ? 1 : 0);
    if (c == 0) return true;
    if (c < 0) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return false;
}

// fixed-width value in a sorted packed binary set (bytewise order == value
// order for BE encodings)
__device__ bool d_in_sorted_bin(const uint8_t* vals, uint32_t n, uint8_t width,
                                const uint8_t* p) {
  uint32_t lo = 0, hi = n;
  while (lo < hi) {
    uint32_t mid = (lo + hi) / 2;
    const uint8_t* m = vals + size_t(mid) * width;
    int c = 0;
    for (int k = 0; k < width; k++) {
      if (m[k] != p[k]) {
        c = m[k] < p[k] ? -1 : 1;
        break;
      }
    }
    if (c == 0) return true;
    if (c < 0) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return false;
}

// matchAnyPhrase / matchAllPhrases over a serialized phrase list (the
// sequence blob layout: u16 n, { u16 len, u8 flags, bytes })
template <typename A>
__device__ bool d_match_any_phrase_at(const A& a, long s0, long sn,
                                      const uint8_t* blob) {
  uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t* p = blob + 2;
  for (uint16_t i = 0; i < n; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    uint8_t flags = p[2];
    p += 3;
    if (len == 0 ? sn == 0 : d_get_phrase_pos_at(a, s0, sn, p, len, flags) >= 0) {
      return true;
    }
    p += len;
  }
  return false;
}

template <typename A>
__device__ bool d_match_all_phrases_at(const A& a, long s0, long sn,
                                       const uint8_t* blob) {
  uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t* p = blob + 2;
  for (uint16_t i = 0; i < n; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    uint8_t flags = p[2];
    p += 3;
    if (len != 0 && d_get_phrase_pos_at(a, s0, sn, p, len, flags) < 0) {
      return false;  // empty phrases match everything (filter_contains_all.go:312-315)
    }
    p += len;
  }
  return true;
}

// matchStringRange: s >= min && s < max; blob = u32 minlen, u32 maxlen, bytes
template <typename A>
__device__ bool d_string_range_at(const A& a, long s0, long sn,
                                  const uint8_t* blob) {
  uint32_t mn_len, mx_len;
  __builtin_memcpy(&mn_len, blob, 4);
  __builtin_memcpy(&mx_len, blob + 4, 4);
  const uint8_t* mn = blob + 8;
  const uint8_t* mx = mn + mn_len;
  auto cmp = [&](const uint8_t* b, uint32_t bl) {
    long lim = long(bl) < sn ? long(bl) : sn;
    for (long k = 0; k < lim; k++) {
      uint8_t rb = a.u8(s0 + k);
      if (rb != b[k]) return rb < b[k] ? -1 : 1;
    }
    return sn < long(bl) ? -1 : (sn > long(bl) ? 1 : 0);
  };
  return cmp(mn, mn_len) >= 0 && cmp(mx, mx_len) < 0;
}

template <typename A>
__device__ uint64_t d_rune_count(const A& a, long s0, long sn) {
  uint64_t n = 0;
  for (long i = 0; i < sn; i++) {
    if ((a.u8(s0 + i) & 0xC0) != 0x80) n++;
  }
  return n;
}

// tryParseIPv4 over an accessor (values_encoder.go:675-730)
template <typename A>
__device__ bool d_try_parse_ipv4(const A& a, long s0, long sn, uint32_t* out) {
  if (sn < 7 || sn > 15) return false;
  int dots = 0;
  for (long i = 0; i < sn; i++) {
    if (a.u8(s0 + i) == '.') dots++;
  }
  if (dots != 3) return false;
  uint32_t ip = 0;
  for (int oct = 0; oct < 4; oct++) {
    long seg_end = s0 + sn;
    if (oct < 3) {
      long j = s0;
      while (j < s0 + sn && a.u8(j) != '.') j++;
      seg_end = j;
    }
    long len = seg_end - s0;
    if (len <= 0 || len > 3) return false;
    // tryParseDateUint64 two-digit fast path quirk: only first char checked
    uint32_t v = 0;
    if (len == 2) {
      uint8_t c0 = a.u8(s0);
      if (c0 < '0' || c0 > '9') return false;
      v = 10 * uint32_t(c0 - '0') + uint32_t(uint8_t(a.u8(s0 + 1) - '0'));
    } else {
      for (long k = 0; k < len; k++) {
        uint8_t c = a.u8(s0 + k);
        if (c < '0' || c > '9') return false;
        v = v * 10 + (c - '0');
      }
    }
    if (v > 255) return false;
    ip = ip << 8 | v;
    s0 = seg_end + 1;
    sn -= len + 1;
  }
  *out = ip;
  return true;
}

// ---- regex fast paths on serialized blob (regex.go:86-212) ----

struct DRegex {
  uint8_t flags;
  uint16_t prefix_len, substr_len, n_or;
  const uint8_t* prefix;
  const uint8_t* substr;
  const uint8_t* ors;  // sequence of {u16 len, bytes}
};

__device__ DRegex d_regex_load(const uint8_t* blob) {
  DRegex re;
  re.flags = blob[0];
  re.prefix_len = uint16_t(blob[1]) | uint16_t(blob[2]) << 8;
  re.substr_len = uint16_t(blob[3]) | uint16_t(blob[4]) << 8;
  re.n_or = uint16_t(blob[5]) | uint16_t(blob[6]) << 8;
  re.prefix = blob + 7;
  re.substr = re.prefix + re.prefix_len;
  re.ors = re.substr + re.substr_len;
  return re;
}

template <typename A>
__device__ bool d_regex_or_contains(const DRegex& re, const A& a, long s0, long sn) {
  const uint8_t* p = re.ors;
  for (int i = 0; i < re.n_or; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    p += 2;
    if (d_index_at(a, s0, sn, p, len) >= 0) return true;
    p += len;
  }
  return false;
}

template <typename A>
__device__ bool d_regex_or_hasprefix(const DRegex& re, const A& a, long s0, long sn) {
  const uint8_t* p = re.ors;
  for (int i = 0; i < re.n_or; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    p += 2;
    if (long(len) <= sn) {
      bool eq = true;
      for (int k = 0; k < len; k++) {
        if (a.u8(s0 + k) != p[k]) {
          eq = false;
          break;
        }
      }
      if (eq) return true;
    }
    p += len;
  }
  return false;
}

// Glushkov NFA executor (general regex class; see core/regex.cpp g_build).
// blob: u16 nstates, u8 flags (1 '^', 2 '$', 4 nullable, 8 wide), pad to 8,
// then first/last/follow[n]/table[256] masks — 8-byte masks, or 16-byte
// pairs when the wide flag is set (65..128 positions).
// assert-layout executor (flag 16): per-class {plain, \b, \B} masks; the
// ASCII word boundary between the previous and current byte selects which
// class's transitions are live (core/regex.cpp nfa_match_assert mirror;
// d_is_token_char IS ASCII \w)
template <typename A>
__device__ bool d_nfa_match_assert_at(const uint8_t* blob, const A& a, long s0,
                                      long sn) {
  const uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t anchors = blob[2];
  const uint8_t null_mask = blob[3];
  const bool a_start = anchors & 1, a_end = anchors & 2;
  uint64_t first[3], last[3];
  for (int c = 0; c < 3; c++) {
    __builtin_memcpy(&first[c], blob + 8 + 8 * c, 8);
    __builtin_memcpy(&last[c], blob + 32 + 8 * c, 8);
  }
  const uint8_t* follow = blob + 56;
  const uint8_t* table = follow + size_t(n) * 24;
  if (sn == 0) return (null_mask >> 0) & 1;  // "" has no boundary
  if (((null_mask >> 0) & 1) && !(a_start && a_end)) return true;
  uint64_t active = 0;
  bool prev_w = false;  // BOF behaves as a non-word char
  for (long i = 0; i < sn; i++) {
    const bool cur_w = d_is_token_char(a.u8(s0 + i));
    const bool bnd = prev_w != cur_w;
    if (!a_end && (!a_start || i == 0)) {
      if ((null_mask >> (bnd ? 1 : 2)) & 1) return true;
    }
    uint64_t targets = 0;
    if (!a_start || i == 0) {
      targets = first[0] | (bnd ? first[1] : first[2]);
    }
    uint64_t m = active;
    while (m) {
      const int x = __builtin_ctzll(m);
      m &= m - 1;
      const uint8_t* f = follow + size_t(x) * 24;
      uint64_t f0, fb;
      __builtin_memcpy(&f0, f, 8);
      __builtin_memcpy(&fb, f + (bnd ? 8 : 16), 8);
      targets |= f0 | fb;
    }
    uint64_t tb;
    __builtin_memcpy(&tb, table + size_t(a.u8(s0 + i)) * 8, 8);
    const uint64_t entered = targets & tb;
    if (entered && !a_end) {
      const bool next_w =
          i + 1 < sn ? d_is_token_char(a.u8(s0 + i + 1)) : false;
      const bool bnd2 = cur_w != next_w;
      if (entered & (last[0] | (bnd2 ? last[1] : last[2]))) return true;
    }
    active = entered;
    prev_w = cur_w;
  }
  const bool bnd_eof = d_is_token_char(a.u8(s0 + sn - 1));
  if ((active & (last[0] | (bnd_eof ? last[1] : last[2]))) && a_end) {
    return true;
  }
  if (!a_start && ((null_mask >> (bnd_eof ? 1 : 2)) & 1)) return true;
  return false;
}

template <typename A>
__device__ bool d_nfa_match_at(const uint8_t* blob, const A& a, long s0, long sn) {
  const uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t anchors = blob[2];
  if (anchors & 16) return d_nfa_match_assert_at(blob, a, s0, sn);
  const bool a_start = anchors & 1, a_end = anchors & 2;
  if (anchors & 8) {
    // wide: two-word position masks
    const uint8_t* p = blob + 8;
    uint64_t first0, first1, last0, last1;
    __builtin_memcpy(&first0, p, 8);
    __builtin_memcpy(&first1, p + 8, 8);
    __builtin_memcpy(&last0, p + 16, 8);
    __builtin_memcpy(&last1, p + 24, 8);
    const uint8_t* follow = p + 32;
    const uint8_t* table = follow + size_t(n) * 16;
    if (sn == 0) return (anchors & 4) != 0;
    uint64_t active0 = 0, active1 = 0;
    for (long i = 0; i < sn; i++) {
      uint64_t t0 = (a_start && i > 0) ? 0 : first0;
      uint64_t t1 = (a_start && i > 0) ? 0 : first1;
      uint64_t m = active0;
      while (m) {
        int x = __builtin_ctzll(m);
        m &= m - 1;
        uint64_t f0, f1;
        __builtin_memcpy(&f0, follow + size_t(x) * 16, 8);
        __builtin_memcpy(&f1, follow + size_t(x) * 16 + 8, 8);
        t0 |= f0;
        t1 |= f1;
      }
      m = active1;
      while (m) {
        int x = 64 + __builtin_ctzll(m);
        m &= m - 1;
        uint64_t f0, f1;
        __builtin_memcpy(&f0, follow + size_t(x) * 16, 8);
        __builtin_memcpy(&f1, follow + size_t(x) * 16 + 8, 8);
        t0 |= f0;
        t1 |= f1;
      }
      uint64_t tb0, tb1;
      __builtin_memcpy(&tb0, table + size_t(a.u8(s0 + i)) * 16, 8);
      __builtin_memcpy(&tb1, table + size_t(a.u8(s0 + i)) * 16 + 8, 8);
      const uint64_t e0 = t0 & tb0, e1 = t1 & tb1;
      if (!a_end && ((e0 & last0) | (e1 & last1))) return true;
      active0 = e0;
      active1 = e1;
    }
    return a_end && ((active0 & last0) | (active1 & last1)) != 0;
  }
  uint64_t first, last;
  __builtin_memcpy(&first, blob + 8, 8);
  __builtin_memcpy(&last, blob + 16, 8);
  const uint8_t* follow = blob + 24;
  const uint8_t* table = blob + 24 + size_t(n) * 8;
  if (sn == 0) return (anchors & 4) != 0;
  uint64_t active = 0;
  for (long i = 0; i < sn; i++) {
    uint64_t targets = (a_start && i > 0) ? 0 : first;
    uint64_t m = active;
    while (m) {
      int x = __builtin_ctzll(m);
      m &= m - 1;
      uint64_t f;
      __builtin_memcpy(&f, follow + size_t(x) * 8, 8);
      targets |= f;
    }
    uint64_t tb;
    __builtin_memcpy(&tb, table + size_t(a.u8(s0 + i)) * 8, 8);
    const uint64_t entered = targets & tb;
    if (!a_end && (entered & last)) return true;
    active = entered;
  }
  return a_end && (active & last) != 0;
}

// Regex.MatchString (regex.go:86-212) over accessor bytes [s0, s0+sn)
// for a single compiled branch (no alt-list marker).
template <typename A>
__device__ bool d_regex_match_single(const uint8_t* blob, const A& a, long s0,
                                     long sn) {
  if (blob[0] & kReAlways) return true;
  if (blob[0] & kReNfa) {
    // NFA blob sits after the (empty) prefix/substr/or-values header
    DRegex hdr = d_regex_load(blob);
    const uint8_t* p = hdr.ors;
    for (int i = 0; i < hdr.n_or; i++) {
      uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
      p += 2 + len;
    }
    return d_nfa_match_at(p, a, s0, sn);
  }
  DRegex re = d_regex_load(blob);
  if (re.flags & kReOnlyPrefix) {
    if (re.prefix_len == 0) return true;
    return d_index_at(a, s0, sn, re.prefix, re.prefix_len) >= 0;
  }
  if (re.prefix_len == 0) {
    // matchStringNoPrefix (regex.go:131-160)
    if (re.flags & kReDotStar) return true;
    if (re.flags & kReDotPlus) return sn > 0;
    if (re.flags & kReSubstrStar) {
      return d_index_at(a, s0, sn, re.substr, re.substr_len) >= 0;
    }
    if (re.flags & kReSubstrPlus) {
      long n = d_index_at(a, s0, sn, re.substr, re.substr_len);
      return n > 0 && n + re.substr_len < sn;
    }
    return d_regex_or_contains(re, a, s0, sn);
  }
  // matchStringWithPrefix (regex.go:162-212)
  long n = d_index_at(a, s0, sn, re.prefix, re.prefix_len);
  if (n < 0) return false;
  long next0 = s0 + n + 1, next_n = sn - n - 1;
  long t0 = s0 + n + re.prefix_len, tn = sn - n - re.prefix_len;

  if (re.flags & kReDotStar) return true;
  if (re.flags & kReDotPlus) return tn > 0;
  if (re.flags & kReSubstrStar) {
    return d_index_at(a, t0, tn, re.substr, re.substr_len) >= 0;
  }
  if (re.flags & kReSubstrPlus) {
    long k = d_index_at(a, t0, tn, re.substr, re.substr_len);
    return k > 0 && k + re.substr_len < tn;
  }
  for (;;) {
    if (d_regex_or_hasprefix(re, a, t0, tn)) return true;
    s0 = next0;
    sn = next_n;
    n = d_index_at(a, s0, sn, re.prefix, re.prefix_len);
    if (n < 0) return false;
    next0 = s0 + n + 1;
    next_n = sn - n - 1;
    t0 = s0 + n + re.prefix_len;
    tn = sn - n - re.prefix_len;
  }
}

// Entry point: a 0x00 marker byte means an alt-list blob (top-level
// alternation with per-branch anchors, e.g. "^01|04$"): u16 n_alts, then
// per branch {u32 len, sub-blob}; match = any branch matches.
template <typename A>
__device__ bool d_regex_match_at(const uint8_t* blob, const A& a, long s0,
                                 long sn) {
  if (blob[0] != 0) return d_regex_match_single(blob, a, s0, sn);
  const int n_alts = int(blob[1]) | int(blob[2]) << 8;
  const uint8_t* p = blob + 3;
  for (int i = 0; i < n_alts; i++) {
    uint32_t len = uint32_t(p[0]) | uint32_t(p[1]) << 8 |
                   uint32_t(p[2]) << 16 | uint32_t(p[3]) << 24;
    p += 4;
    if (d_regex_match_single(p, a, s0, sn)) return true;
    p += len;
  }
  return false;
}

// ---- per-row predicates ----

__device__ __forceinline__ uint64_t d_get_u64be(const uint8_t* p) {
  uint64_t v;
  __builtin_memcpy(&v, p, 8);
  return __builtin_bswap64(v);
}
__device__ __forceinline__ uint32_t d_get_u32be(const uint8_t* p) {
  uint32_t v;
  __builtin_memcpy(&v, p, 4);
  return __builtin_bswap32(v);
}
__device__ __forceinline__ uint16_t d_get_u16be(const uint8_t* p) {
  return uint16_t(p[0]) << 8 | p[1];
}

// Cold string kinds, kept OUT of the hot scan loop: inlining every matcher
// into the workgroup loop grew the loop body past the instruction cache and
// cost ~18% on the phrase workload (798 vs 677 us/launch, profiles/r01b).
template <typename A>
__device__ __noinline__ bool d_eval_string_row_cold(const DevLeafBlock& lb,
                                                    const A& a, long s0,
                                                    long sn) {
  switch (lb.kind) {
    case kScanRangeStr: {
      // matchRange (filter_range.go:369-372)
      double x = d_parse_math_number(a, s0, sn);
      double mn = __builtin_bit_cast(double, lb.vmin);
      double mx = __builtin_bit_cast(double, lb.vmax);
      return x >= mn && x <= mx;
    }
    case kScanPrefixStr:
      return d_match_prefix_at(a, s0, sn, lb.operand, lb.operand_len, lb.flags);
    case kScanExactPrefixStr:
      return d_has_prefix_bytes(a, s0, sn, lb.operand, lb.operand_len);
    case kScanSeqStr:
      return d_match_sequence_at(a, s0, sn, lb.operand);
    case kScanInStr:
      return d_in_sorted_str(lb.operand, a, s0, sn);
    case kScanAnyPhraseStr:
      return d_match_any_phrase_at(a, s0, sn, lb.operand);
    case kScanAllPhrasesStr:
      return d_match_all_phrases_at(a, s0, sn, lb.operand);
    case kScanStrRange:
      return d_string_range_at(a, s0, sn, lb.operand);
    case kScanIPv4RangeStr: {
      uint32_t ip;
      if (!d_try_parse_ipv4(a, s0, sn, &ip)) return false;
      return ip >= uint32_t(lb.vmin) && ip <= uint32_t(lb.vmax);
    }
    case kScanLenRangeStr: {
      uint64_t n = d_rune_count(a, s0, sn);
      return n >= lb.vmin && n <= lb.vmax;
    }
    case kScanAnyCasePhraseStr: {
      // matchAnyCasePhrase (filter_any_case_phrase.go:159-181); operand is
      // the lowercase phrase, flags from the lowercase phrase
      if (lb.operand_len == 0) return sn == 0;
      if (long(lb.operand_len) > sn) return false;
      LowerAcc<A> la{a};
      return d_get_phrase_pos_at(la, s0, sn, lb.operand, lb.operand_len,
                                 lb.flags & 15) >= 0;
    }
    case kScanAnyCasePrefixStr: {
      // matchAnyCasePrefix (filter_any_case_prefix.go:161-183)
      if (lb.operand_len == 0) return sn > 0;
      if (long(lb.operand_len) > sn) return false;
      LowerAcc<A> la{a};
      return d_match_prefix_at(la, s0, sn, lb.operand, lb.operand_len,
                               lb.flags & 15);
    }
    default:  // kScanRegexStr
      return d_regex_match_at(lb.operand, a, s0, sn);
  }
}

// String-kind predicate over an accessor (tile or global); the two kinds the
// steady-state workloads hammer stay inline, everything else is a call.
template <typename A>
__device__ __forceinline__ bool d_eval_string_row(const DevLeafBlock& lb,
                                                  const A& a, long s0,
                                                  long sn) {
  if (lb.kind == kScanPhraseStr) {
    return d_match_phrase_at(a, s0, sn, lb.operand, lb.operand_len, lb.flags);
  }
  if (lb.kind == kScanEqStr) {
    if (sn != long(lb.operand_len)) return false;
    for (long k = 0; k < sn; k++) {
      if (a.u8(s0 + k) != lb.operand[k]) return false;
    }
    return true;
  }
  return d_eval_string_row_cold(lb, a, s0, sn);
}


// ---- two-column filters (filter_eq_field.go, filter_le_field.go) ----

// leValuesString (filter_le_field.go:284-299) over two global byte spans
__device__ inline bool d_le_values_string(const uint8_t* ap, long an,
                                          const uint8_t* bp, long bn,
                                          bool exclude_equal) {
  GlobalAcc aa{ap}, ab{bp};
  double fa = d_parse_math_number(aa, 0, an);
  if (!(fa != fa)) {  // !isnan
    double fb = d_parse_math_number(ab, 0, bn);
    if (!(fb != fb)) return exclude_equal ? fa < fb : fa <= fb;
  }
  long lim = an < bn ? an : bn;
  int c = 0;
  for (long i = 0; i < lim; i++) {
    if (ap[i] != bp[i]) {
      c = ap[i] < bp[i] ? -1 : 1;
      break;
    }
  }
  if (c == 0) c = an < bn ? -1 : (an > bn ? 1 : 0);
  return exclude_equal ? c < 0 : c <= 0;
}

// One side of a two-column string-form compare.  Blob layout per side:
// u8 mode (0 raw string col, 1 const, 2 missing, 3 formatted fixed-width,
// 4 dict), u8 fmt<<4|width, u16 aux_len; aux bytes follow both descriptors
// (mode 1: the const value; mode 4: dict table {u8 n, u16 offs[n+1], bytes}).
struct DFieldSide {
  const uint8_t* p;
  long n;
  char buf[64];
};

__device__ inline void d_field_side_resolve(DFieldSide& out, uint8_t mode,
                                            uint8_t fmtw, const uint8_t* cval,
                                            uint16_t clen,
                                            const uint8_t* data,
                                            const uint32_t* offs,
                                            uint32_t row) {
  switch (mode) {
    case 0:  // raw string column
      out.p = data + offs[row];
      out.n = long(offs[row + 1]) - long(offs[row]);
      return;
    case 1:  // const value
      out.p = cval;
      out.n = clen;
      return;
    case 2:  // missing column => ""
      out.p = (const uint8_t*)out.buf;
      out.n = 0;
      return;
    case 4: {  // dict: 1-byte codes + table in the aux area
      const uint8_t cnt = cval[0];
      const uint8_t* offs16 = cval + 1;
      const uint8_t code = data[row];
      uint16_t o0 = uint16_t(offs16[code * 2]) |
                    uint16_t(offs16[code * 2 + 1]) << 8;
      uint16_t o1 = uint16_t(offs16[code * 2 + 2]) |
                    uint16_t(offs16[code * 2 + 3]) << 8;
      out.p = cval + 1 + 2 * (cnt + 1) + o0;
      out.n = long(o1) - long(o0);
      return;
    }
    default: {  // formatted fixed-width value
      const uint8_t fmt = fmtw >> 4, w = fmtw & 15;
      const uint8_t* q = data + size_t(row) * w;
      int n;
      switch (fmt) {
        case kFmtU64: {
          uint64_t v;
          switch (w) {
            case 1: v = q[0]; break;
            case 2: v = d_get_u16be(q); break;
            case 4: v = d_get_u32be(q); break;
            default: v = d_get_u64be(q); break;
          }
          n = d_format_u64(out.buf, v);
          break;
        }
        case kFmtI64: {
          uint64_t u = d_get_u64be(q);
          n = d_format_i64(out.buf, int64_t(u >> 1) ^ (int64_t(u << 63) >> 63));
          break;
        }
        case kFmtF64:
          n = vl_ryu::format_f64(out.buf, d_get_u64be(q));
          break;
        case kFmtIp:
          n = d_format_ipv4(out.buf, d_get_u32be(q));
          break;
        default:
          n = d_format_iso8601(out.buf, int64_t(d_get_u64be(q)));
          break;
      }
      out.p = (const uint8_t*)out.buf;
      out.n = n;
      return;
    }
  }
}

__device__ inline bool d_eval_field_pair(const DevLeafBlock& lb, uint32_t row,
                                         bool le) {
  const uint8_t* blob = lb.operand;
  const uint8_t amode = blob[0], afmtw = blob[1];
  const uint8_t bmode = blob[4], bfmtw = blob[5];
  uint16_t aclen = uint16_t(blob[2]) | uint16_t(blob[3]) << 8;
  uint16_t bclen = uint16_t(blob[6]) | uint16_t(blob[7]) << 8;
  const uint8_t* aconst = blob + 8;
  const uint8_t* bconst = aconst + aclen;
  DFieldSide sa, sb;
  d_field_side_resolve(sa, amode, afmtw, aconst, aclen, lb.data, lb.offsets,
                       row);
  d_field_side_resolve(sb, bmode, bfmtw, bconst, bclen,
                       (const uint8_t*)lb.hashes, (const uint32_t*)lb.bloom,
                       row);
  if (le) {
    return d_le_values_string(sa.p, sa.n, sb.p, sb.n, (lb.flags & 1) != 0);
  }
  if (sa.n != sb.n) return false;
  for (long i = 0; i < sa.n; i++) {
    if (sa.p[i] != sb.p[i]) return false;
  }
  return true;
}

// Cold fixed-width kinds (formatters, regex, parsers) behind a call so the
// hot scan loop stays small (see d_eval_string_row_cold).
__device__ __noinline__ bool d_eval_fixed_row_cold(const DevLeafBlock& lb,
                                                   uint32_t row) {
  switch (lb.kind) {
    case kScanRangeU: {
      const uint8_t* p = lb.data + size_t(row) * lb.width;
      uint64_t v;
      switch (lb.width) {
        case 1: v = p[0]; break;
        case 2: v = d_get_u16be(p); break;
        case 4: v = d_get_u32be(p); break;
        default: v = d_get_u64be(p); break;
      }
      return v >= lb.vmin && v <= lb.vmax;
    }
    case kScanRangeI: {
      uint64_t u = d_get_u64be(lb.data + size_t(row) * 8);
      // flags bit0: plain BE i64 (iso8601 nsecs); else zig-zag (int.go:79-84)
      int64_t v = (lb.flags & 1) ? int64_t(u)
                                 : int64_t(u >> 1) ^ (int64_t(u << 63) >> 63);
      return v >= int64_t(lb.vmin) && v <= int64_t(lb.vmax);
    }
    case kScanRangeF: {
      uint64_t u = d_get_u64be(lb.data + size_t(row) * 8);
      double v = __builtin_bit_cast(double, u);
      double mn = __builtin_bit_cast(double, lb.vmin);
      double mx = __builtin_bit_cast(double, lb.vmax);
      return v >= mn && v <= mx;
    }
    case kScanPhraseIp: {
      char buf[32];
      int n = d_format_ipv4(buf, d_get_u32be(lb.data + size_t(row) * 4));
      BufAcc a{(const uint8_t*)buf};
      return d_match_phrase_at(a, 0, n, lb.operand, lb.operand_len, lb.flags);
    }
    case kScanPhraseIso: {
      char buf[48];
      int n = d_format_iso8601(buf, int64_t(d_get_u64be(lb.data + size_t(row) * 8)));
      BufAcc a{(const uint8_t*)buf};
      return d_match_phrase_at(a, 0, n, lb.operand, lb.operand_len, lb.flags);
    }
    case kScanRegexU: {
      const uint8_t* p = lb.data + size_t(row) * lb.width;
      uint64_t v;
      switch (lb.width) {
        case 1: v = p[0]; break;
        case 2: v = d_get_u16be(p); break;
        case 4: v = d_get_u32be(p); break;
        default: v = d_get_u64be(p); break;
      }
      char buf[48];
      int n = d_format_u64(buf, v);
      BufAcc a{(const uint8_t*)buf};
      return d_regex_match_at(lb.operand, a, 0, n);
    }
    case kScanRegexI: {
      uint64_t u = d_get_u64be(lb.data + size_t(row) * 8);
      int64_t v = int64_t(u >> 1) ^ (int64_t(u << 63) >> 63);
      char buf[48];
      int n = d_format_i64(buf, v);
      BufAcc a{(const uint8_t*)buf};
      return d_regex_match_at(lb.operand, a, 0, n);
    }
    case kScanRegexIp: {
      char buf[32];
      int n = d_format_ipv4(buf, d_get_u32be(lb.data + size_t(row) * 4));
      BufAcc a{(const uint8_t*)buf};
      return d_regex_match_at(lb.operand, a, 0, n);
    }
    case kScanRegexIso: {
      char buf[48];
      int n = d_format_iso8601(buf, int64_t(d_get_u64be(lb.data + size_t(row) * 8)));
      BufAcc a{(const uint8_t*)buf};
      return d_regex_match_at(lb.operand, a, 0, n);
    }
    case kScanPhraseF64: {
      // matchFloat64ByPhrase slow path (filter_phrase.go:175-186): format the
      // stored float with Ryu (== Go strconv 'f' -1) and substring-match
      char buf[368];
      int n = vl_ryu::format_f64(buf, d_get_u64be(lb.data + size_t(row) * 8));
      BufAcc a{(const uint8_t*)buf};
      return d_match_phrase_at(a, 0, n, lb.operand, lb.operand_len, lb.flags);
    }
    case kScanRegexF64: {
      char buf[368];
      int n = vl_ryu::format_f64(buf, d_get_u64be(lb.data + size_t(row) * 8));
      BufAcc a{(const uint8_t*)buf};
      return d_regex_match_at(lb.operand, a, 0, n);
    }
    case kScanInBin:
      return d_in_sorted_bin(lb.operand, lb.operand_len / lb.width, lb.width,
                             lb.data + size_t(row) * lb.width);
    case kScanIPv4RangeBin: {
      const uint8_t* p = lb.data + size_t(row) * 4;
      uint32_t ip = uint32_t(p[0]) << 24 | uint32_t(p[1]) << 16 |
                    uint32_t(p[2]) << 8 | p[3];
      return ip >= uint32_t(lb.vmin) && ip <= uint32_t(lb.vmax);
    }
    case kScanDayRange: {
      long long off_tz;
      __builtin_memcpy(&off_tz, lb.operand, 8);
      int64_t off = (lb.ts[row] - off_tz) % (24LL * 3600 * 1000000000);
      return off >= int64_t(lb.vmin) && off <= int64_t(lb.vmax);
    }
    case kScanEqFieldBin: {
      const uint8_t* pa = lb.data + size_t(row) * lb.width;
      const uint8_t* pb = (const uint8_t*)lb.hashes + size_t(row) * lb.width;
      for (int i = 0; i < lb.width; i++) {
        if (pa[i] != pb[i]) return false;
      }
      return true;
    }
    case kScanEqFieldDict:
    case kScanLeFieldDict: {
      const uint8_t ca = lb.data[row];
      const uint8_t cb = ((const uint8_t*)lb.hashes)[row];
      return (lb.vmin >> (ca * 8 + cb)) & 1;
    }
    case kScanLeFieldI64: {
      uint64_t ua = d_get_u64be(lb.data + size_t(row) * 8);
      uint64_t ub = d_get_u64be((const uint8_t*)lb.hashes + size_t(row) * 8);
      int64_t va = int64_t(ua >> 1) ^ (int64_t(ua << 63) >> 63);
      int64_t vb = int64_t(ub >> 1) ^ (int64_t(ub << 63) >> 63);
      return (lb.flags & 1) ? va < vb : va <= vb;
    }
    case kScanLeFieldF64: {
      uint64_t ua = d_get_u64be(lb.data + size_t(row) * 8);
      uint64_t ub = d_get_u64be((const uint8_t*)lb.hashes + size_t(row) * 8);
      double va = __builtin_bit_cast(double, ua);
      double vb = __builtin_bit_cast(double, ub);
      return (lb.flags & 1) ? va < vb : va <= vb;
    }
    case kScanLeFieldBinStr: {
      return d_le_values_string(lb.data + size_t(row) * lb.width, lb.width,
                                (const uint8_t*)lb.hashes + size_t(row) * lb.width,
                                lb.width, (lb.flags & 1) != 0);
    }
    case kScanEqFieldStr:
      return d_eval_field_pair(lb, row, false);
    case kScanLeFieldStr:
      return d_eval_field_pair(lb, row, true);
    case kScanWeekRange: {
      long long off_tz;
      __builtin_memcpy(&off_tz, lb.operand, 8);
      const int64_t day = 24LL * 3600 * 1000000000;
      int64_t t = lb.ts[row] - off_tz;
      int64_t days = t / day;
      if (t % day < 0) days--;
      int64_t wd = (days + 4) % 7;
      if (wd < 0) wd += 7;
      return wd >= int64_t(lb.vmin) && wd <= int64_t(lb.vmax);
    }
    case kScanPrefixFmt:
    case kScanExactPrefixFmt:
    case kScanSeqFmt:
    case kScanAnyPhraseFmt:
    case kScanAllPhrasesFmt:
    case kScanStrRangeFmt:
    case kScanLenRangeFmt: {
      char buf[368];
      int n;
      switch (lb.flags >> 4) {
        case kFmtU64: {
          const uint8_t* p = lb.data + size_t(row) * lb.width;
          uint64_t v;
          switch (lb.width) {
            case 1: v = p[0]; break;
            case 2: v = d_get_u16be(p); break;
            case 4: v = d_get_u32be(p); break;
            default: v = d_get_u64be(p); break;
          }
          n = d_format_u64(buf, v);
          break;
        }
        case kFmtI64: {
          uint64_t u = d_get_u64be(lb.data + size_t(row) * 8);
          n = d_format_i64(buf, int64_t(u >> 1) ^ (int64_t(u << 63) >> 63));
          break;
        }
        case kFmtF64:
          n = vl_ryu::format_f64(buf, d_get_u64be(lb.data + size_t(row) * 8));
          break;
        case kFmtIp:
          n = d_format_ipv4(buf, d_get_u32be(lb.data + size_t(row) * 4));
          break;
        default:
          n = d_format_iso8601(buf, int64_t(d_get_u64be(lb.data + size_t(row) * 8)));
          break;
      }
      BufAcc a{(const uint8_t*)buf};
      switch (lb.kind) {
        case kScanPrefixFmt:
          return d_match_prefix_at(a, 0, n, lb.operand, lb.operand_len,
                                   lb.flags & 15);
        case kScanExactPrefixFmt:
          return d_has_prefix_bytes(a, 0, n, lb.operand, lb.operand_len);
        case kScanAnyPhraseFmt:
          return d_match_any_phrase_at(a, 0, n, lb.operand);
        case kScanAllPhrasesFmt:
          return d_match_all_phrases_at(a, 0, n, lb.operand);
        case kScanStrRangeFmt:
          return d_string_range_at(a, 0, n, lb.operand);
        case kScanLenRangeFmt: {
          uint64_t rc = d_rune_count(a, 0, n);
          return rc >= lb.vmin && rc <= lb.vmax;
        }
        default:
          return d_match_sequence_at(a, 0, n, lb.operand);
      }
    }
    default:
      return false;
  }
}

__device__ __forceinline__ bool d_is_string_kind(uint8_t kind) {
  constexpr uint64_t mask =
      (uint64_t(1) << kScanPhraseStr) | (uint64_t(1) << kScanEqStr) |
      (uint64_t(1) << kScanRegexStr) | (uint64_t(1) << kScanRangeStr) |
      (uint64_t(1) << kScanPrefixStr) | (uint64_t(1) << kScanExactPrefixStr) |
      (uint64_t(1) << kScanSeqStr) | (uint64_t(1) << kScanInStr) |
      (uint64_t(1) << kScanAnyPhraseStr) | (uint64_t(1) << kScanAllPhrasesStr) |
      (uint64_t(1) << kScanStrRange) | (uint64_t(1) << kScanIPv4RangeStr) |
      (uint64_t(1) << kScanLenRangeStr) | (uint64_t(1) << kScanAnyCasePhraseStr) |
      (uint64_t(1) << kScanAnyCasePrefixStr);
  return kind < 64 && ((mask >> kind) & 1);
}

// Fixed-width / dict / timestamp predicate (coalesced global reads); the
// steady-state kinds stay inline, the rest is a call.
__device__ __forceinline__ bool d_eval_fixed_row(const DevLeafBlock& lb,
                                                 uint32_t row) {
  switch (lb.kind) {
    case kScanEqBin: {
      const uint8_t* p = lb.data + size_t(row) * lb.width;
      switch (lb.width) {
        case 1: return p[0] == lb.operand[0];
        case 2: return p[0] == lb.operand[0] && p[1] == lb.operand[1];
        case 4: {
          uint32_t a, b;
          __builtin_memcpy(&a, p, 4);
          __builtin_memcpy(&b, lb.operand, 4);
          return a == b;
        }
        default: {
          uint64_t a, b;
          __builtin_memcpy(&a, p, 8);
          __builtin_memcpy(&b, lb.operand, 8);
          return a == b;
        }
      }
    }
    case kScanDict:
      return (lb.dict_mask >> lb.data[row]) & 1;
    case kScanTsRange: {
      int64_t v = lb.ts[row];
      return v >= int64_t(lb.vmin) && v <= int64_t(lb.vmax);
    }
    default:
      return d_eval_fixed_row_cold(lb, row);
  }
}

__device__ __forceinline__ int d_u64_declen(uint64_t v) {
  int n = 1;
  while (v >= 10) {
    v /= 10;
    n++;
  }
  return n;
}

// value byte length for one row (pass 1) -- must agree with d_gather_write
__device__ uint32_t d_gather_len(const DevGatherCol& gc, uint32_t row) {
  switch (gc.src) {
    case kGatherStr:
      return gc.offsets[row + 1] - gc.offsets[row];
    case kGatherConst:
      return gc.cval_len;
    case kGatherDict: {
      uint8_t code = gc.data[row];
      return gc.dict_offs[code + 1] - gc.dict_offs[code];
    }
    case kGatherFmtU: {
      const uint8_t* p = gc.data + size_t(row) * gc.width;
      uint64_t v;
      switch (gc.width) {
        case 1: v = p[0]; break;
        case 2: v = d_get_u16be(p); break;
        case 4: v = d_get_u32be(p); break;
        default: v = d_get_u64be(p); break;
      }
      return uint32_t(d_u64_declen(v));
    }
    case kGatherFmtI: {
      uint64_t u = d_get_u64be(gc.data + size_t(row) * 8);
      int64_t v = int64_t(u >> 1) ^ (int64_t(u << 63) >> 63);
      if (v < 0) return uint32_t(1 + d_u64_declen(~uint64_t(v) + 1));
      return uint32_t(d_u64_declen(uint64_t(v)));
    }
    case kGatherFmtF: {
      char buf[368];
      return uint32_t(vl_ryu::format_f64(buf, d_get_u64be(gc.data + size_t(row) * 8)));
    }
    case kGatherFmtIp: {
      uint32_t ip = d_get_u32be(gc.data + size_t(row) * 4);
      return uint32_t(d_u64_declen((ip >> 24) & 255) + d_u64_declen((ip >> 16) & 255) +
                      d_u64_declen((ip >> 8) & 255) + d_u64_declen(ip & 255) + 3);
    }
    case kGatherFmtIso:
      return 24;
    default:  // kGatherMissing
      return 0;
  }
}

__device__ uint32_t d_gather_write(const DevGatherCol& gc, uint32_t row,
                                   uint8_t* dst) {
  switch (gc.src) {
    case kGatherStr: {
      uint32_t off = gc.offsets[row];
      uint32_t len = gc.offsets[row + 1] - off;
      for (uint32_t i = 0; i < len; i++) dst[i] = gc.data[off + i];
      return len;
    }
    case kGatherConst: {
      for (uint32_t i = 0; i < gc.cval_len; i++) dst[i] = gc.cval[i];
      return gc.cval_len;
    }
    case kGatherDict: {
      uint8_t code = gc.data[row];
      uint32_t off = gc.dict_offs[code];
      uint32_t len = gc.dict_offs[code + 1] - off;
      for (uint32_t i = 0; i < len; i++) dst[i] = gc.dict_data[off + i];
      return len;
    }
    case kGatherFmtU: {
      const uint8_t* p = gc.data + size_t(row) * gc.width;
      uint64_t v;
      switch (gc.width) {
        case 1: v = p[0]; break;
        case 2: v = d_get_u16be(p); break;
        case 4: v = d_get_u32be(p); break;
        default: v = d_get_u64be(p); break;
      }
      return uint32_t(d_format_u64((char*)dst, v));
    }
    case kGatherFmtI: {
      uint64_t u = d_get_u64be(gc.data + size_t(row) * 8);
      return uint32_t(
          d_format_i64((char*)dst, int64_t(u >> 1) ^ (int64_t(u << 63) >> 63)));
    }
    case kGatherFmtF:
      return uint32_t(
          vl_ryu::format_f64((char*)dst, d_get_u64be(gc.data + size_t(row) * 8)));
    case kGatherFmtIp:
      return uint32_t(d_format_ipv4((char*)dst, d_get_u32be(gc.data + size_t(row) * 4)));
    case kGatherFmtIso:
      return uint32_t(d_format_iso8601(
          (char*)dst, int64_t(d_get_u64be(gc.data + size_t(row) * 8))));
    default:
      return 0;
  }
}
