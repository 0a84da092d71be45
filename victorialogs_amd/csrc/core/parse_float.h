// Shared host/device Go strconv.ParseFloat / strconv.ParseInt(s, 0, 64)
// restatement, for the parseMathNumber legs of the block-scan path
// (pipe_math.go:1066-1080, block_result.go:2710-2752).
//
// Decimal -> double conversion uses an exact-interval method instead of
// glibc strtod so that the oracle and the HIP kernels produce identical
// bits by construction:
//   value = w * 10^q with w the first <=19 significant digits.  10^q is
//   bracketed by a 128-bit mantissa table entry M (truncated for q>=0,
//   rounded up for q<0), so the true value lies in [W*(M-1), W*(M+1)] x 2^k.
//   Both 192-bit endpoint products are computed EXACTLY and rounded to
//   double (round-to-nearest-even with exact sticky bits); when all
//   endpoint roundings agree (always, except half-ULP ties of inputs with
//   >19 significant digits or q outside the exact table range) that is the
//   correctly rounded answer.  Otherwise a deterministic pick (the lower
//   endpoint's rounding) is used -- a documented, parity-stable divergence
//   from Go's big-decimal slow path (DESIGN.md limitations).
#pragma once

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define VL_PF_HD __host__ __device__ __forceinline__
#define VL_PF_CONST __device__ static const
#else
#define VL_PF_HD static inline
#define VL_PF_CONST static const
#endif

namespace vl_pf {

#include "pow10_el.inc"

VL_PF_HD int pf_clz64(uint64_t x) {
#ifdef __HIP_DEVICE_COMPILE__
  return __clzll((long long)x);
#else
  return __builtin_clzll(x);
#endif
}

VL_PF_HD double pf_bits(uint64_t u) {
  union { uint64_t u; double d; } c;
  c.u = u;
  return c.d;
}

// Rounds the exact value (t2:t1:t0 as a 192-bit integer) * 2^k to double.
// t2 != 0.  Round-to-nearest-even; all sticky bits are exact.
VL_PF_HD double pf_round192(uint64_t t2, uint64_t t1, uint64_t t0, int k) {
  const int msb = 191 - pf_clz64(t2);        // bit index of the MSB in P
  const int e = msb + k;                      // value in [2^e, 2^(e+1))
  int biased = e + 1023;
  int keep = 53;                              // mantissa bits incl leading 1
  if (biased <= 0) {                          // subnormal: fewer kept bits
    keep = 53 + biased - 1;                   // biased<=0 => keep<=52
    if (keep <= 0) {
      // underflow; can still round up to the smallest subnormal when the
      // value is >= 2^-1075 exactly at the halfway point
      if (keep == 0) {
        // value in [2^-1075-ish): round bit is the MSB itself
        // halfway = 2^-1075; value >= halfway rounds to 0x1 (ties: even=0)
        uint64_t below = (t2 << (64 - (msb % 64) )); // crude sticky
        (void)below;
        // value < 2^-1074 and >= 2^-1075 rounds to minimum subnormal only
        // when strictly above halfway; at exactly halfway rounds to even 0.
        // Sticky: any bit below msb.
        uint64_t sticky = (t2 & ~(uint64_t(1) << (msb - 128))) | t1 | t0;
        return pf_bits(sticky ? 1 : 0);
      }
      return pf_bits(0);
    }
    biased = 0;
  }
  // extract `keep` bits starting at msb
  const int low = msb - keep + 1;             // index of lowest kept bit
  uint64_t mant;
  // gather bits [low, msb] of (t2,t1,t0)
  {
    const int word = low >> 6, off = low & 63;
    uint64_t w0 = word == 0 ? t0 : (word == 1 ? t1 : t2);
    uint64_t w1 = word == 0 ? t1 : (word == 1 ? t2 : 0);
    mant = off ? (w0 >> off) | (w1 << (64 - off)) : w0;
    if (keep < 64) mant &= (uint64_t(1) << keep) - 1;
  }
  // round bit and sticky
  bool round_bit = false, sticky = false;
  if (low > 0) {
    const int rb = low - 1;
    const int word = rb >> 6, off = rb & 63;
    uint64_t w = word == 0 ? t0 : (word == 1 ? t1 : t2);
    round_bit = (w >> off) & 1;
    // sticky: any bit below rb
    if (word == 2) {
      sticky = (off ? (t2 & ((uint64_t(1) << off) - 1)) : 0) | t1 | t0;
    } else if (word == 1) {
      sticky = (off ? (t1 & ((uint64_t(1) << off) - 1)) : 0) | t0;
    } else {
      sticky = off ? (t0 & ((uint64_t(1) << off) - 1)) != 0 : false;
    }
  }
  if (round_bit && (sticky || (mant & 1))) {
    mant++;
    if (biased == 0) {
      if (mant >= (uint64_t(1) << 52)) biased = 1;       // rounded into normal
      if (biased == 1) mant &= (uint64_t(1) << 52) - 1;
    } else if (mant >= (uint64_t(1) << 53)) {
      mant >>= 1;
      biased++;
    }
  }
  if (biased >= 2047) return pf_bits(0x7FF0000000000000ULL);
  if (biased == 0) return pf_bits(mant);                  // subnormal
  mant &= (uint64_t(1) << 52) - 1;                        // drop implicit 1
  return pf_bits((uint64_t(biased) << 52) | mant);
}

// w * (mhi:mlo) -> exact 192-bit product
VL_PF_HD void pf_mul192(uint64_t w, uint64_t mhi, uint64_t mlo, uint64_t* t2,
                        uint64_t* t1, uint64_t* t0) {
  unsigned __int128 a = (unsigned __int128)w * mhi;
  unsigned __int128 b = (unsigned __int128)w * mlo;
  uint64_t b_hi = (uint64_t)(b >> 64);
  *t0 = (uint64_t)b;
  uint64_t a_lo = (uint64_t)a;
  *t1 = a_lo + b_hi;
  *t2 = (uint64_t)(a >> 64) + (*t1 < a_lo ? 1 : 0);
}

// w (u64) * 10^q -> double.  Returns false when the endpoint roundings
// disagree (ambiguous; caller decided to use lower endpoint anyway).
VL_PF_HD bool pf_decimal_to_double(uint64_t w, int q, bool truncated,
                                   double* out) {
  if (w == 0) { *out = 0.0; return true; }
  if (q < VL_POW10_MIN_Q) { *out = 0.0; return true; }
  if (q > VL_POW10_MAX_Q) {
    *out = pf_bits(0x7FF0000000000000ULL);
    return true;
  }
  const int idx = q - (VL_POW10_MIN_Q);
  const int lz = pf_clz64(w);
  const uint64_t W = w << lz;
  const uint64_t hi = kPow10Hi[idx], lo = kPow10Lo[idx];
  const int k = int(kPow10E2[idx]) - 127 - lz - 63;  // W*M128 * 2^(e2-127-lz-63-... )
  // value = (w<<lz) * M * 2^(e2 - 127 - lz); the 192-bit product P=W*M gives
  // value = P * 2^(e2 - 127 - lz - 127)   [since M in [2^127,2^128)] -- NO:
  // M's scale: 10^q = M * 2^(e2-127).  So value = W * M * 2^(e2-127-lz).
  const int kk = int(kPow10E2[idx]) - 127 - lz;
  uint64_t t2, t1, t0;
  pf_mul192(W, hi, lo, &t2, &t1, &t0);
  const bool exact = kPow10Exact[idx] != 0 && !truncated;
  if (exact) {
    *out = pf_round192(t2, t1, t0, kk);
    return true;
  }
  if (!truncated && q >= -27 && q < 0) {
    // exact path for small negative q: 5^27 < 2^63, so value = w * 2^q / 5^-q
    // can be rounded from an exact 128-bit quotient + remainder-sticky.
    // (True half-ULP ties require 5^-q | w, impossible for -q > 27.)
    uint64_t d = 5;
    for (int i = 1; i < -q; i++) d *= 5;
    unsigned __int128 num = (unsigned __int128)W << 63;
    unsigned __int128 Q = num / d;
    uint64_t sticky = (num % d) != 0 ? 1 : 0;
    // value = (Q + rem/d) * 2^(q - 63 - lz)
    uint64_t q2 = (uint64_t)(Q >> 64), q1 = (uint64_t)Q, q0 = sticky;
    int ke = q - 63 - lz - 64;  // P = Q*2^64 + sticky => value = P * 2^ke
    if (q2 == 0) {
      q2 = q1;
      q1 = q0;
      q0 = 0;
      ke -= 64;
    }
    *out = pf_round192(q2, q1, q0, ke);
    return true;
  }
  // interval: M_true in (M-1, M+1); w_true in [w, w+1) when truncated
  // lower endpoint: W*(M-1)  [ = P - W ]
  uint64_t l2 = t2, l1 = t1, l0 = t0;
  {
    uint64_t o = l0;
    l0 -= W;
    if (l0 > o) { if (l1-- == 0) l2--; }
  }
  double dl = pf_round192(l2, l1, l0, kk);
  // upper endpoint: (W + (truncated ? (1<<lz) : 0)) * (M+1) <= P + W + M + ...
  // compute exactly: P_up = (w + t?1:0)<<lz * (M+1)
  uint64_t wu = w + (truncated ? 1 : 0);
  int lzu = pf_clz64(wu);
  uint64_t Wu = wu << lzu;
  int kku = int(kPow10E2[idx]) - 127 - lzu;
  uint64_t u2, u1, u0;
  pf_mul192(Wu, hi, lo, &u2, &u1, &u0);
  {
    uint64_t o = u0;
    u0 += Wu;
    if (u0 < o) { if (++u1 == 0) u2++; }
  }
  double du = pf_round192(u2, u1, u0, kku);
  (void)k;
  if (dl == du) { *out = dl; return true; }
  *out = dl;  // deterministic pick, documented divergence
  return false;
}



// ---- Go strconv syntax layer (template over a byte reader R: r.u8(i)) ----

template <typename R>
VL_PF_HD bool pf_underscore_ok(const R& r, long n) {
  // strconv's underscoreOK: underscores only between digits or between a
  // base prefix and a digit
  char saw = '^';  // ^ start, 0 digit, _ underscore, ! other
  long i = 0;
  if (i < n && (r.u8(i) == '+' || r.u8(i) == '-')) i++;
  bool hex = false;
  if (i + 1 < n && r.u8(i) == '0' &&
      (r.u8(i + 1) == 'x' || r.u8(i + 1) == 'X' || r.u8(i + 1) == 'o' ||
       r.u8(i + 1) == 'O' || r.u8(i + 1) == 'b' || r.u8(i + 1) == 'B')) {
    hex = r.u8(i + 1) == 'x' || r.u8(i + 1) == 'X';
    saw = '0';
    i += 2;
  }
  for (; i < n; i++) {
    uint8_t c = r.u8(i);
    bool digit = (c >= '0' && c <= '9') ||
                 (hex && ((c | 32) >= 'a' && (c | 32) <= 'f'));
    if (digit) {
      saw = '0';
      continue;
    }
    if (c == '_') {
      if (saw != '0') return false;
      saw = '_';
      continue;
    }
    if (saw == '_') return false;
    saw = '!';
  }
  return saw != '_';
}

VL_PF_HD bool pf_lower_eq(uint8_t c, char l) { return (c | 32) == uint8_t(l); }

// strconv.ParseFloat(s, 64) (atof.go syntax + special values); returns false
// on syntax error or out-of-range (Go returns non-nil err for range, and the
// parseMathNumber leg skips on any err).
template <typename R>
VL_PF_HD bool go_parse_float(const R& r, long n, double* out) {
  if (n == 0) return false;
  long i = 0;
  bool neg = false;
  if (r.u8(0) == '+' || r.u8(0) == '-') {
    neg = r.u8(0) == '-';
    i = 1;
  }
  // special values: inf, infinity, nan (case-insensitive)
  if (i < n && (pf_lower_eq(r.u8(i), 'i') || pf_lower_eq(r.u8(i), 'n'))) {
    if (n - i == 3 && pf_lower_eq(r.u8(i), 'n') && pf_lower_eq(r.u8(i + 1), 'a') &&
        pf_lower_eq(r.u8(i + 2), 'n')) {
      if (neg || i != 0) {
        // Go: "nan" only without sign
        if (i != 0) return false;
      }
      *out = pf_bits(0x7FF8000000000001ULL);
      return i == 0;
    }
    const char* inf = "infinity";
    long m = n - i;
    if (m != 3 && m != 8) return false;
    for (long k = 0; k < m; k++) {
      if (!pf_lower_eq(r.u8(i + k), inf[k])) return false;
    }
    *out = pf_bits(neg ? 0xFFF0000000000000ULL : 0x7FF0000000000000ULL);
    return true;
  }
  if (!pf_underscore_ok(r, n)) return false;
  // hex mantissa?
  bool hex = false;
  if (i + 1 < n && r.u8(i) == '0' && (r.u8(i + 1) | 32) == 'x') {
    hex = true;
    i += 2;
  }
  uint64_t mant = 0;
  int nd = 0;          // significant digits collected
  int dropped = 0;     // digits beyond capacity (integer part)
  bool truncated = false;
  bool any_digit = false, any_frac = false, seen_dot = false;
  int frac_digits = 0;
  const int max_nd = hex ? 16 : 19;
  for (; i < n; i++) {
    uint8_t c = r.u8(i);
    if (c == '_') continue;
    if (c == '.') {
      if (seen_dot) return false;
      seen_dot = true;
      continue;
    }
    int dv;
    if (c >= '0' && c <= '9') {
      dv = c - '0';
    } else if (hex && (c | 32) >= 'a' && (c | 32) <= 'f') {
      dv = (c | 32) - 'a' + 10;
    } else {
      break;
    }
    any_digit = true;
    if (seen_dot) any_frac = true;
    if (nd < max_nd && (mant != 0 || dv != 0)) {
      mant = mant * (hex ? 16 : 10) + uint64_t(dv);
      if (mant) nd++;
      if (seen_dot) frac_digits++;
    } else if (mant == 0 && dv == 0) {
      if (seen_dot) frac_digits++;  // leading zeros after the dot scale down
    } else {
      if (dv != 0) truncated = true;
      if (!seen_dot) dropped++;
    }
  }
  (void)any_frac;
  if (!any_digit) return false;
  int exp = 0;
  bool has_exp = false;
  if (i < n && ((!hex && (r.u8(i) | 32) == 'e') || (hex && (r.u8(i) | 32) == 'p'))) {
    long j = i + 1;
    bool eneg = false;
    if (j < n && (r.u8(j) == '+' || r.u8(j) == '-')) {
      eneg = r.u8(j) == '-';
      j++;
    }
    bool ed = false;
    long e = 0;
    for (; j < n; j++) {
      uint8_t c = r.u8(j);
      if (c == '_') continue;
      if (c < '0' || c > '9') break;
      ed = true;
      if (e < 100000) e = e * 10 + (c - '0');
    }
    if (!ed) return false;
    exp = int(eneg ? -e : e);
    has_exp = true;
    i = j;
  }
  if (i != n) return false;
  if (hex && !has_exp) return false;  // Go hex floats require the p exponent
  if (hex) {
    // value = mant * 16^(-frac_digits) * 2^exp, mant < 2^64 exact
    if (mant == 0) { *out = neg ? -0.0 : 0.0; return true; }
    int e2 = exp - 4 * frac_digits + 4 * dropped;
    // normalize into pf_round192 form: P = mant << 64
    int lz2 = pf_clz64(mant);
    uint64_t t2 = mant << lz2;
    double d = pf_round192(t2, truncated ? 1 : 0, 0, e2 - lz2 - 128);
    *out = neg ? -d : d;
    return (d != pf_bits(0x7FF0000000000000ULL));
  }
  int q = exp - frac_digits + dropped;
  if (mant == 0) { *out = neg ? -0.0 : 0.0; return true; }
  double d;
  pf_decimal_to_double(mant, q, truncated, &d);
  if (d == pf_bits(0x7FF0000000000000ULL)) return false;  // ErrRange
  if (d == 0.0 && (mant != 0)) {
    // underflowed to zero: Go reports ErrRange only when the value is too
    // small for a subnormal; 0 from nonzero mantissa means exactly that
    if (q < -400) return false;
    // tiny-but-nonzero handled in pf_decimal_to_double; reaching 0 here with
    // q >= -400 means true underflow as well
    return false;
  }
  *out = neg ? -d : d;
  return true;
}

// strconv.ParseInt(s, 0, 64): base from prefix (0x/0o/0b/leading-0 octal)
template <typename R>
VL_PF_HD bool go_parse_int0(const R& r, long n, int64_t* out) {
  if (n == 0) return false;
  long i = 0;
  bool neg = false;
  if (r.u8(0) == '+' || r.u8(0) == '-') {
    neg = r.u8(0) == '-';
    i = 1;
  }
  if (!pf_underscore_ok(r, n)) return false;
  int base = 10;
  if (i < n && r.u8(i) == '0') {
    if (i + 1 < n) {
      uint8_t c = r.u8(i + 1) | 32;
      if (c == 'x') { base = 16; i += 2; }
      else if (c == 'o') { base = 8; i += 2; }
      else if (c == 'b') { base = 2; i += 2; }
      else { base = 8; i += 1; }  // legacy octal, keep the 0 consumed
    }
  }
  uint64_t v = 0;
  bool any = false;
  const uint64_t lim = neg ? 0x8000000000000000ULL : 0x7FFFFFFFFFFFFFFFULL;
  for (; i < n; i++) {
    uint8_t c = r.u8(i);
    if (c == '_') continue;
    int dv;
    if (c >= '0' && c <= '9') dv = c - '0';
    else if ((c | 32) >= 'a' && (c | 32) <= 'z') dv = (c | 32) - 'a' + 10;
    else return false;
    if (dv >= base) return false;
    if (v > (lim - uint64_t(dv)) / uint64_t(base)) return false;  // overflow
    v = v * uint64_t(base) + uint64_t(dv);
    any = true;
  }
  // "0" parses with base switched to octal and zero digits left; Go accepts
  if (!any) {
    if (base == 8 && n - (neg ? 1 : 0) == 1) { *out = 0; return true; }
    return false;
  }
  *out = neg ? -int64_t(v) : int64_t(v);
  return true;
}

// isNumberPrefix (parser.go:3077-3097)
template <typename R>
VL_PF_HD bool pf_is_number_prefix(const R& r, long n) {
  if (n == 0) return false;
  long i = 0;
  if (r.u8(0) == '-' || r.u8(0) == '+') {
    i = 1;
    if (n == 1) return false;
  }
  if (n - i >= 3 && pf_lower_eq(r.u8(i), 'i') && pf_lower_eq(r.u8(i + 1), 'n') &&
      pf_lower_eq(r.u8(i + 2), 'f') && n - i == 3) {
    return true;
  }
  if (r.u8(i) == '.') {
    i++;
    if (i >= n) return false;
  }
  return r.u8(i) >= '0' && r.u8(i) <= '9';
}

// isLikelyNumber (block_result.go:2739-2752)
template <typename R>
VL_PF_HD bool pf_is_likely_number(const R& r, long n) {
  if (!pf_is_number_prefix(r, n)) return false;
  int dots = 0, dashes = 0;
  for (long i = 0; i < n; i++) {
    uint8_t c = r.u8(i);
    if (c == '.') dots++;
    if (c == ':') return false;
    if (c == '-') dashes++;
  }
  return dots <= 1 && dashes <= 2;
}

}  // namespace vl_pf
