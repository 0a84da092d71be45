// HIP/CDNA4 (gfx950) kernels for the VictoriaLogs block-scan hot path.
//
// Each workgroup (256 threads = 4 wavefronts) evaluates the compiled filter
// program over one chunk of up to kChunkRows rows of one block.  Leaf
// predicates are computed lane-per-row and assembled into bitmap words with
// 64-lane ballots (one wavefront ballot = one u64 bitmap word covering 64
// consecutive rows, LSB-first — exactly the bitmap.go:113-125 layout the
// drop-in boundary requires).  Bloom gates (bloomfilter.go:173-191) run
// cooperatively per chunk before any row is touched.
//
// This path is HBM-bandwidth-bound integer/byte work — no MFMA by design
// (BASELINE.json north star).
#include <hip/hip_runtime.h>

#include "scan_types.h"

namespace vl {

// Non-ASCII token-rune ranges (tokenizer.go:142-148); see unicode_ranges.inc.
#include "../core/unicode_ranges.inc"

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

// Go utf8.DecodeRuneInString semantics (0xFFFD,1 on invalid).
__device__ uint32_t d_utf8_decode(const char* p, long n, int* size) {
  *size = 1;
  if (n <= 0) return 0xFFFD;
  uint8_t c0 = uint8_t(p[0]);
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
    uint8_t c = uint8_t(p[i]);
    if ((c & 0xC0) != 0x80) return 0xFFFD;
    r = (r << 6) | (c & 0x3F);
  }
  if (r < lo || r > 0x10FFFF || (r >= 0xD800 && r <= 0xDFFF)) return 0xFFFD;
  *size = len;
  return r;
}

__device__ uint32_t d_utf8_decode_last(const char* p, long n, int* size) {
  *size = 1;
  if (n <= 0) return 0xFFFD;
  long start = n - 1;
  if (uint8_t(p[start]) < 0x80) return uint8_t(p[start]);
  long lim = n >= 4 ? n - 4 : 0;
  while (start > lim && (uint8_t(p[start]) & 0xC0) == 0x80) start--;
  int sz;
  uint32_t r = d_utf8_decode(p + start, n - start, &sz);
  if (start + sz != n) return 0xFFFD;
  *size = sz;
  return r;
}

// strings.Index: first occurrence of sub in s, -1 if absent.  SWAR first-byte
// candidate scan over 8-byte windows, then verify.
__device__ long d_index(const char* s, long sn, const char* sub, long subn) {
  if (subn == 0) return 0;
  if (subn > sn) return -1;
  const uint8_t c0 = uint8_t(sub[0]);
  const uint64_t pat = 0x0101010101010101ULL * c0;
  long last = sn - subn;
  long i = 0;
  while (i <= last) {
    // SWAR zero-byte trick over the next 8 bytes (bounded by last+1)
    long win = last + 1 - i;
    if (win >= 8) {
      uint64_t x;
      __builtin_memcpy(&x, s + i, 8);
      uint64_t t = x ^ pat;
      uint64_t hit = (t - 0x0101010101010101ULL) & ~t & 0x8080808080808080ULL;
      if (hit == 0) {
        i += 8;
        continue;
      }
      i += long(__builtin_ctzll(hit) >> 3);
      if (i > last) return -1;
    } else {
      if (uint8_t(s[i]) != c0) {
        i++;
        continue;
      }
    }
    if (uint8_t(s[i]) == c0) {
      bool eq = true;
      for (long k = 1; k < subn; k++) {
        if (s[i + k] != sub[k]) {
          eq = false;
          break;
        }
      }
      if (eq) return i;
    }
    i++;
  }
  return -1;
}

// getPhrasePos (filter_phrase.go:220-270); starts/ends-with-token flags are
// precomputed on the host from the phrase.
__device__ bool d_match_phrase(const char* s, long sn, const char* ph, long phn,
                               uint8_t flags) {
  if (phn == 0) return sn == 0;  // filter_phrase.go:212-215
  if (phn > sn) return false;
  long pos = 0;
  for (;;) {
    long n = d_index(s + pos, sn - pos, ph, phn);
    if (n < 0) return false;
    pos += n;
    if ((flags & kPhraseStartsToken) && pos > 0) {
      uint32_t rb = uint8_t(s[pos - 1]);
      if (rb >= 0x80) {
        int sz;
        rb = d_utf8_decode_last(s, pos, &sz);
      }
      if (rb == 0xFFFD || d_is_token_rune(rb)) {
        pos++;
        continue;
      }
    }
    if ((flags & kPhraseEndsToken) && pos + phn < sn) {
      uint32_t ra = uint8_t(s[pos + phn]);
      if (ra >= 0x80) {
        int sz;
        ra = d_utf8_decode(s + pos + phn, sn - pos - phn, &sz);
      }
      if (ra == 0xFFFD || d_is_token_rune(ra)) {
        pos++;
        continue;
      }
    }
    return true;
  }
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
    // careful with INT64_MIN
    uint64_t u = ~uint64_t(v) + 1;
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
  // civil_from_days
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
  // yyyy-mm-ddThh:mm:ss.mmmZ (year assumed 0..9999 for %04d)
  int n = 0;
  if (y >= 1000) {
    n = d_format_u64(buf, uint64_t(y));
  } else {
    buf[0] = '0'; buf[1] = '0'; buf[2] = '0'; buf[3] = char('0' + y % 10);
    if (y >= 10) d_pad2(buf + 2, int(y % 100));
    if (y >= 100) {
      buf[1] = char('0' + (y / 100) % 10);
    }
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

// ---- regex fast paths on serialized blob (regex.go:86-212) ----

struct DRegex {
  uint8_t flags;
  uint16_t prefix_len, substr_len, n_or;
  const char* prefix;
  const char* substr;
  const uint8_t* ors;  // sequence of {u16 len, bytes}
};

__device__ DRegex d_regex_load(const uint8_t* blob) {
  DRegex re;
  re.flags = blob[0];
  re.prefix_len = uint16_t(blob[1]) | uint16_t(blob[2]) << 8;
  re.substr_len = uint16_t(blob[3]) | uint16_t(blob[4]) << 8;
  re.n_or = uint16_t(blob[5]) | uint16_t(blob[6]) << 8;
  re.prefix = (const char*)blob + 7;
  re.substr = re.prefix + re.prefix_len;
  re.ors = (const uint8_t*)(re.substr + re.substr_len);
  return re;
}

__device__ bool d_regex_or_contains(const DRegex& re, const char* s, long sn) {
  const uint8_t* p = re.ors;
  for (int i = 0; i < re.n_or; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    p += 2;
    if (d_index(s, sn, (const char*)p, len) >= 0) return true;
    p += len;
  }
  return false;
}

__device__ bool d_regex_or_hasprefix(const DRegex& re, const char* s, long sn) {
  const uint8_t* p = re.ors;
  for (int i = 0; i < re.n_or; i++) {
    uint16_t len = uint16_t(p[0]) | uint16_t(p[1]) << 8;
    p += 2;
    if (long(len) <= sn) {
      bool eq = true;
      for (int k = 0; k < len; k++) {
        if (s[k] != ((const char*)p)[k]) {
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

__device__ bool d_regex_match(const uint8_t* blob, const char* s, long sn) {
  DRegex re = d_regex_load(blob);
  if (re.flags & kReOnlyPrefix) {
    if (re.prefix_len == 0) return true;
    return d_index(s, sn, re.prefix, re.prefix_len) >= 0;
  }
  if (re.prefix_len == 0) {
    // matchStringNoPrefix (regex.go:131-160)
    if (re.flags & kReDotStar) return true;
    if (re.flags & kReDotPlus) return sn > 0;
    if (re.flags & kReSubstrStar) return d_index(s, sn, re.substr, re.substr_len) >= 0;
    if (re.flags & kReSubstrPlus) {
      long n = d_index(s, sn, re.substr, re.substr_len);
      return n > 0 && n + re.substr_len < sn;
    }
    return d_regex_or_contains(re, s, sn);
  }
  // matchStringWithPrefix (regex.go:162-212)
  long n = d_index(s, sn, re.prefix, re.prefix_len);
  if (n < 0) return false;
  const char* snext = s + n + 1;
  long snext_n = sn - n - 1;
  const char* t = s + n + re.prefix_len;
  long tn = sn - n - re.prefix_len;

  if (re.flags & kReDotStar) return true;
  if (re.flags & kReDotPlus) return tn > 0;
  if (re.flags & kReSubstrStar) return d_index(t, tn, re.substr, re.substr_len) >= 0;
  if (re.flags & kReSubstrPlus) {
    long k = d_index(t, tn, re.substr, re.substr_len);
    return k > 0 && k + re.substr_len < tn;
  }
  for (;;) {
    if (d_regex_or_hasprefix(re, t, tn)) return true;
    s = snext;
    sn = snext_n;
    n = d_index(s, sn, re.prefix, re.prefix_len);
    if (n < 0) return false;
    snext = s + n + 1;
    snext_n = sn - n - 1;
    t = s + n + re.prefix_len;
    tn = sn - n - re.prefix_len;
  }
}

// ---- per-row leaf predicate ----

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

__device__ bool d_eval_row(const DevLeafBlock& lb, uint32_t row) {
  switch (lb.kind) {
    case kScanPhraseStr: {
      uint32_t off = lb.offsets[row];
      uint32_t len = lb.offsets[row + 1] - off;
      return d_match_phrase((const char*)lb.data + off, len, (const char*)lb.operand,
                            lb.operand_len, lb.flags);
    }
    case kScanEqStr: {
      uint32_t off = lb.offsets[row];
      uint32_t len = lb.offsets[row + 1] - off;
      if (len != lb.operand_len) return false;
      const char* s = (const char*)lb.data + off;
      for (uint32_t k = 0; k < len; k++) {
        if (s[k] != ((const char*)lb.operand)[k]) return false;
      }
      return true;
    }
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
    case kScanRegexStr: {
      uint32_t off = lb.offsets[row];
      uint32_t len = lb.offsets[row + 1] - off;
      return d_regex_match(lb.operand, (const char*)lb.data + off, len);
    }
    case kScanPhraseIp: {
      char buf[16];
      int n = d_format_ipv4(buf, d_get_u32be(lb.data + size_t(row) * 4));
      return d_match_phrase(buf, n, (const char*)lb.operand, lb.operand_len, lb.flags);
    }
    case kScanPhraseIso: {
      char buf[32];
      int n = d_format_iso8601(buf, int64_t(d_get_u64be(lb.data + size_t(row) * 8)));
      return d_match_phrase(buf, n, (const char*)lb.operand, lb.operand_len, lb.flags);
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
      char buf[20];
      int n = d_format_u64(buf, v);
      return d_regex_match(lb.operand, buf, n);
    }
    case kScanRegexI: {
      uint64_t u = d_get_u64be(lb.data + size_t(row) * 8);
      int64_t v = int64_t(u >> 1) ^ (int64_t(u << 63) >> 63);
      char buf[21];
      int n = d_format_i64(buf, v);
      return d_regex_match(lb.operand, buf, n);
    }
    case kScanRegexIp: {
      char buf[16];
      int n = d_format_ipv4(buf, d_get_u32be(lb.data + size_t(row) * 4));
      return d_regex_match(lb.operand, buf, n);
    }
    case kScanRegexIso: {
      char buf[32];
      int n = d_format_iso8601(buf, int64_t(d_get_u64be(lb.data + size_t(row) * 8)));
      return d_regex_match(lb.operand, buf, n);
    }
    default:
      return false;
  }
}

// ---- the program kernel ----

__global__ __launch_bounds__(256) void scan_program_kernel(
    const DevOp* __restrict__ ops, int nops, const DevLeafBlock* __restrict__ lbs,
    int nleaves, const DevBlock* __restrict__ blocks,
    const DevChunk* __restrict__ chunks, unsigned long long* __restrict__ hits) {
  __shared__ uint64_t stack[kMaxStackDepth][kChunkWords];
  __shared__ int bloom_ok;
  __shared__ unsigned long long wave_sums[4];

  const DevChunk ck = chunks[blockIdx.x];
  const DevBlock blk = blocks[ck.block];
  const uint32_t r0 = ck.chunk * kChunkRows;
  const uint32_t r1 = min(blk.rows, r0 + kChunkRows);
  const uint32_t nwords = (r1 - r0 + 63) / 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  int sp = 0;
  for (int i = 0; i < nops; i++) {
    const DevOp op = ops[i];
    __syncthreads();
    if (op.kind == kOpLeaf) {
      const DevLeafBlock& lb = lbs[size_t(ck.block) * nleaves + op.leaf];
      uint64_t* out = stack[sp];
      if (lb.mode == kModeNone || lb.mode == kModeAll) {
        uint64_t fill = lb.mode == kModeAll ? ~uint64_t(0) : 0;
        for (uint32_t w = tid; w < nwords; w += blockDim.x) out[w] = fill;
        sp++;
        continue;
      }
      // bloom gate (bloomfilter.go:173-191), cooperative over the workgroup
      if (lb.nhashes) {
        if (tid == 0) bloom_ok = 1;
        __syncthreads();
        if (lb.bloom_words == 0) {
          // empty bloom = containsAll true (bloomfilter.go:174-177)
        } else {
          const uint64_t max_bits = uint64_t(lb.bloom_words) * 64;
          bool miss = false;
          for (uint32_t k = tid; k < lb.nhashes; k += blockDim.x) {
            uint64_t idx = lb.hashes[k] % max_bits;
            if (((lb.bloom[idx >> 6] >> (idx & 63)) & 1) == 0) miss = true;
          }
          if (miss) bloom_ok = 0;
        }
        __syncthreads();
        if (!bloom_ok) {
          for (uint32_t w = tid; w < nwords; w += blockDim.x) out[w] = 0;
          sp++;
          continue;
        }
      }
      for (uint32_t w = wave; w < nwords; w += nwaves) {
        uint32_t row = r0 + w * 64 + lane;
        bool pred = row < r1 ? d_eval_row(lb, row) : false;
        uint64_t word = __ballot(pred);
        if (lane == 0) out[w] = word;
      }
      sp++;
    } else if (op.kind == kOpNot) {
      uint64_t* top = stack[sp - 1];
      for (uint32_t w = tid; w < nwords; w += blockDim.x) top[w] = ~top[w];
    } else {
      const int n = op.nargs;
      uint64_t* dst = stack[sp - n];
      for (uint32_t w = tid; w < nwords; w += blockDim.x) {
        uint64_t acc = dst[w];
        if (op.kind == kOpAnd) {
          for (int k = 1; k < n; k++) acc &= stack[sp - n + k][w];
        } else {
          for (int k = 1; k < n; k++) acc |= stack[sp - n + k][w];
        }
        dst[w] = acc;
      }
      sp -= n - 1;
    }
  }
  __syncthreads();

  // mask the tail word, write the result, count hits
  unsigned long long local = 0;
  for (uint32_t w = tid; w < nwords; w += blockDim.x) {
    uint64_t word = stack[0][w];
    uint32_t base = r0 + w * 64;
    uint32_t valid = min(64u, r1 - base);
    if (valid < 64) word &= (uint64_t(1) << valid) - 1;
    blk.bitmap_out[r0 / 64 + w] = word;
    local += __popcll(word);
  }
  // workgroup reduction
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off, 64);
  if (lane == 0) wave_sums[wave] = local;
  __syncthreads();
  if (tid == 0) {
    unsigned long long s = 0;
    for (int k = 0; k < nwaves; k++) s += wave_sums[k];
    atomicAdd(hits, s);
  }
}

// host-side launcher (called from vql_api.cpp)
extern "C" hipError_t vql_launch_scan(const DevOp* ops, int nops,
                                      const DevLeafBlock* lbs, int nleaves,
                                      const DevBlock* blocks, const DevChunk* chunks,
                                      uint32_t nchunks, unsigned long long* hits,
                                      hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  hipLaunchKernelGGL(scan_program_kernel, dim3(nchunks), dim3(256), 0, stream, ops,
                     nops, lbs, nleaves, blocks, chunks, hits);
  return hipGetLastError();
}

}  // namespace vl
