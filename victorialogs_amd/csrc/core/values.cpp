#include "values.h"

#include "parse_float.h"

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <ctime>

namespace vl {

// ---- parsers ----

bool try_parse_uint64(strview s, uint64_t* out) {
  // values_encoder.go:553-585
  if (s.n == 0 || s.n > 26 /* len("18_446_744_073_709_551_615") */) return false;
  if (s.n > 1 && s.p[0] == '0') return false;
  uint64_t n = 0;
  for (size_t i = 0; i < s.n; i++) {
    char ch = s.p[i];
    if (ch == '_') continue;
    if (ch < '0' || ch > '9') return false;
    if (n > UINT64_MAX / 10) return false;
    n *= 10;
    uint64_t d = uint64_t(ch - '0');
    uint64_t n1 = n + d;
    if (n1 < n) return false;
    n = n1;
  }
  *out = n;
  return true;
}

// values_encoder.go:588-619
static bool try_parse_date_uint64(strview s, uint64_t* out) {
  if (s.n == 0 || s.n > 9) return false;
  if (s.n == 2) {
    // fast path: only the first char is validated (values_encoder.go:593-600)
    if (s.p[0] < '0' || s.p[0] > '9') return false;
    *out = 10 * uint64_t(s.p[0] - '0') + uint64_t(uint8_t(s.p[1] - '0'));
    return true;
  }
  uint64_t n = 0;
  for (size_t i = 0; i < s.n; i++) {
    char ch = s.p[i];
    if (ch < '0' || ch > '9') return false;
    n = n * 10 + uint64_t(ch - '0');
  }
  *out = n;
  return true;
}

bool try_parse_int64(strview s, int64_t* out) {
  // values_encoder.go:622-645
  if (s.n == 0) return false;
  bool minus = s.p[0] == '-';
  strview t = minus ? strview(s.p + 1, s.n - 1) : s;
  uint64_t n;
  if (!try_parse_uint64(t, &n)) return false;
  if (n >= (uint64_t(1) << 63)) {
    if (minus && n == (uint64_t(1) << 63)) {
      *out = INT64_MIN;
      return true;
    }
    return false;
  }
  int64_t ni = int64_t(n);
  *out = minus ? -ni : ni;
  return true;
}

// Go math.Pow10 restated (for bit-exact fractional scaling in
// tryParseFloat64Internal, values_encoder.go:842).
static double go_pow10(int n) {
  static const double tab[32] = {
      1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10, 1e11, 1e12,
      1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22, 1e23,
      1e24, 1e25, 1e26, 1e27, 1e28, 1e29, 1e30, 1e31};
  static const double postab32[10] = {1e0, 1e32, 1e64, 1e96, 1e128,
                                      1e160, 1e192, 1e224, 1e256, 1e288};
  if (n >= 0 && n <= 308) return postab32[unsigned(n) / 32] * tab[unsigned(n) % 32];
  if (n <= 0 && n >= -323) {
    return 1.0 / (postab32[unsigned(-n) / 32] * tab[unsigned(-n) % 32]);
  }
  if (n > 0) return HUGE_VAL;
  return 0;
}

static bool try_parse_float64_internal(strview s, bool exact, double* out) {
  // values_encoder.go:788-848
  if (s.n == 0 || s.n > 27 /* len("-18_446_744_073_709_551_615") */) return false;
  bool minus = s.p[0] == '-';
  if (minus) {
    s.p++;
    s.n--;
  }
  const char* dot = (const char*)memchr(s.p, '.', s.n);
  if (dot == nullptr) {
    uint64_t n;
    if (!try_parse_uint64(s, &n)) return false;
    if (exact && n >= (uint64_t(1) << 53)) return false;
    double f = double(n);
    *out = minus ? -f : f;
    return true;
  }
  size_t ndot = size_t(dot - s.p);
  if (ndot == 0 || ndot == s.n - 1) return false;
  strview s_int(s.p, ndot);
  strview s_frac(s.p + ndot + 1, s.n - ndot - 1);

  uint64_t n_int;
  if (!try_parse_uint64(s_int, &n_int)) return false;

  // skip leading zeros in sFrac keeping >=1 char (values_encoder.go:830-835)
  size_t skip = 0;
  while (skip < s_frac.n - 1 && s_frac.p[skip] == '0') skip++;
  uint64_t n_frac;
  if (!try_parse_uint64(strview(s_frac.p + skip, s_frac.n - skip), &n_frac)) return false;

  int underscores = 0;
  for (size_t i = 0; i < s_frac.n; i++) {
    if (s_frac.p[i] == '_') underscores++;
  }
  double p10 = go_pow10(underscores - int(s_frac.n));
  double f = std::fma(double(n_frac), p10, double(n_int));
  *out = minus ? -f : f;
  return true;
}

bool try_parse_float64_exact(strview s, double* out) {
  return try_parse_float64_internal(s, true, out);
}
bool try_parse_float64(strview s, double* out) {
  return try_parse_float64_internal(s, false, out);
}

bool try_parse_ipv4(strview s, uint32_t* out) {
  // values_encoder.go:675-730
  if (s.n < 7 || s.n > 15) return false;
  int dots = 0;
  for (size_t i = 0; i < s.n; i++) {
    if (s.p[i] == '.') dots++;
  }
  if (dots != 3) return false;

  uint8_t octets[4];
  for (int oct = 0; oct < 3; oct++) {
    const char* d = (const char*)memchr(s.p, '.', s.n);
    long n = d ? d - s.p : -1;
    if (n <= 0 || n > 3) return false;
    uint64_t v;
    if (!try_parse_date_uint64(strview(s.p, size_t(n)), &v) || v > 255) return false;
    octets[oct] = uint8_t(v);
    s.p += n + 1;
    s.n -= size_t(n) + 1;
  }
  uint64_t v;
  if (!try_parse_date_uint64(s, &v) || v > 255) return false;
  octets[3] = uint8_t(v);
  *out = get_u32be(octets);
  return true;
}

// days since 1970-01-01 for a (possibly out-of-range) civil date; Go
// time.Date normalizes overflowing fields (month/day may exceed their ranges).
static int64_t days_from_civil(int64_t y, int m /*1..12*/, int64_t d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  int64_t yoe = y - era * 400;
  int64_t doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  int64_t doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + doe - 719468;
}

// tryParseTimestampSecs (values_encoder.go:469-550)
static bool try_parse_timestamp_secs(strview s, int64_t* secs_out, strview* tail) {
  if (s.n < 19) return false;
  if (s.p[4] != '-') return false;
  uint64_t n;
  if (!try_parse_date_uint64(strview(s.p, 4), &n) || n < 1677 || n > 2262) return false;
  int64_t year = int64_t(n);
  s.p += 5; s.n -= 5;

  if (s.p[2] != '-') return false;
  if (!try_parse_date_uint64(strview(s.p, 2), &n)) return false;
  int64_t month = int64_t(n);
  s.p += 3; s.n -= 3;

  char delim = s.p[2];
  if (delim != 'T' && delim != ' ') return false;
  if (!try_parse_date_uint64(strview(s.p, 2), &n)) return false;
  int64_t day = int64_t(n);
  s.p += 3; s.n -= 3;

  if (s.p[2] != ':') return false;
  if (!try_parse_date_uint64(strview(s.p, 2), &n)) return false;
  int64_t hour = int64_t(n);
  s.p += 3; s.n -= 3;

  if (s.p[2] != ':') return false;
  if (!try_parse_date_uint64(strview(s.p, 2), &n)) return false;
  int64_t minute = int64_t(n);
  s.p += 3; s.n -= 3;

  if (!try_parse_date_uint64(strview(s.p, 2), &n)) return false;
  int64_t second = int64_t(n);
  s.p += 2; s.n -= 2;

  // time.Date(..., time.UTC).Unix() with field normalization
  int64_t m0 = month - 1;
  int64_t ny = year + (m0 >= 0 ? m0 / 12 : (m0 - 11) / 12);
  int64_t nm = m0 % 12;
  if (nm < 0) nm += 12;
  int64_t days = days_from_civil(ny, int(nm) + 1, day);
  int64_t secs = days * 86400 + hour * 3600 + minute * 60 + second;
  if (secs < INT64_MIN / 1000000000 || secs >= INT64_MAX / 1000000000) return false;
  *secs_out = secs;
  *tail = s;
  return true;
}

bool try_parse_timestamp_iso8601(strview s, int64_t* out) {
  // values_encoder.go:428-466: exactly "2006-01-02T15:04:05.000Z"
  if (s.n != 24) return false;
  int64_t secs;
  strview tail;
  if (!try_parse_timestamp_secs(s, &secs, &tail)) return false;
  s = tail;
  int64_t nsecs = secs * 1000000000;
  if (s.p[0] != '.') return false;
  s.p++; s.n--;
  if (s.p[3] != 'Z') return false;
  uint64_t msecs;
  if (!try_parse_date_uint64(strview(s.p, 3), &msecs)) return false;
  *out = nsecs + int64_t(msecs) * 1000000;
  return true;
}

// tryParseFloat64Prefix (values_encoder.go:762-773)
static bool try_parse_float64_prefix(strview s, double* f, strview* tail) {
  size_t i = 0;
  while (i < s.n && ((s.p[i] >= '0' && s.p[i] <= '9') || s.p[i] == '.' || s.p[i] == '_')) i++;
  if (i == 0) return false;
  if (!try_parse_float64(strview(s.p, i), f)) return false;
  *tail = strview(s.p + i, s.n - i);
  return true;
}

// addInt64NoOverflow (values_encoder.go:968-974)
static int64_t add_i64_no_overflow(int64_t n, double f) {
  int64_t x = int64_t(f);
  if (n < 0 || x < 0 || x > INT64_MAX - n) return INT64_MAX;
  return n + x;
}

static bool has_prefix(strview s, const char* p) {
  size_t n = strlen(p);
  return s.n >= n && memcmp(s.p, p, n) == 0;
}

bool try_parse_duration(strview s, int64_t* out) {
  // values_encoder.go:990-1061
  static constexpr int64_t kNsPerSecond = 1000000000;
  static constexpr int64_t kNsPerMinute = 60 * kNsPerSecond;
  static constexpr int64_t kNsPerHour = 3600 * kNsPerSecond;
  static constexpr int64_t kNsPerDay = 24 * kNsPerHour;
  static constexpr int64_t kNsPerWeek = 7 * kNsPerDay;
  static constexpr int64_t kNsPerYear = 365 * kNsPerDay;
  if (s.n == 0) return false;
  bool minus = s.p[0] == '-';
  if (minus) { s.p++; s.n--; }
  int64_t nsecs = 0;
  while (s.n > 0) {
    double f;
    strview tail;
    if (!try_parse_float64_prefix(s, &f, &tail)) return false;
    s = tail;
    if (s.n == 0) return false;
    if (s.n >= 3 && memcmp(s.p, "\xC2\xB5s", 3) == 0) {  // "µs"
      nsecs = add_i64_no_overflow(nsecs, f * 1000);
      s.p += 3; s.n -= 3;
      continue;
    }
    if (s.n >= 2 && has_prefix(s, "ms")) {
      nsecs = add_i64_no_overflow(nsecs, f * 1000000);
      s.p += 2; s.n -= 2;
      continue;
    }
    if (s.n >= 2 && has_prefix(s, "ns")) {
      nsecs = add_i64_no_overflow(nsecs, f);
      s.p += 2; s.n -= 2;
      continue;
    }
    char c = s.p[0];
    int64_t mult;
    switch (c) {
      case 'y': mult = kNsPerYear; break;
      case 'w': mult = kNsPerWeek; break;
      case 'd': mult = kNsPerDay; break;
      case 'h': mult = kNsPerHour; break;
      case 'm': mult = kNsPerMinute; break;
      case 's': mult = kNsPerSecond; break;
      default: return false;
    }
    nsecs = add_i64_no_overflow(nsecs, f * double(mult));
    s.p += 1; s.n -= 1;
  }
  *out = minus ? -nsecs : nsecs;
  return true;
}

bool try_parse_bytes(strview s, int64_t* out) {
  // values_encoder.go:855-966
  if (s.n == 0) return false;
  bool minus = s.p[0] == '-';
  if (minus) { s.p++; s.n--; }
  int64_t n = 0;
  while (s.n > 0) {
    double f;
    strview tail;
    if (!try_parse_float64_prefix(s, &f, &tail)) return false;
    if (tail.n == 0) {
      double ip;
      if (std::modf(f, &ip) != 0) return false;  // no suffix: integers only
    }
    s = tail;
    if (s.n == 0) {
      n = add_i64_no_overflow(n, f);
      continue;
    }
    struct Sfx { const char* s; double m; };
    static const Sfx sfx3[] = {{"KiB", 1 << 10}, {"MiB", 1 << 20}, {"GiB", 1 << 30}, {"TiB", double(1ULL << 40)}};
    static const Sfx sfx2[] = {{"Ki", 1 << 10}, {"Mi", 1 << 20}, {"Gi", 1 << 30}, {"Ti", double(1ULL << 40)},
                               {"KB", 1e3}, {"MB", 1e6}, {"GB", 1e9}, {"TB", 1e12}};
    static const Sfx sfx1[] = {{"B", 1}, {"K", 1e3}, {"M", 1e6}, {"G", 1e9}, {"T", 1e12}};
    bool matched = false;
    if (s.n >= 3) {
      for (const auto& x : sfx3) {
        if (has_prefix(s, x.s)) {
          n = add_i64_no_overflow(n, f * x.m);
          s.p += 3; s.n -= 3;
          matched = true;
          break;
        }
      }
    }
    if (!matched && s.n >= 2) {
      for (const auto& x : sfx2) {
        if (has_prefix(s, x.s)) {
          n = add_i64_no_overflow(n, f * x.m);
          s.p += 2; s.n -= 2;
          matched = true;
          break;
        }
      }
    }
    if (!matched) {
      for (const auto& x : sfx1) {
        if (has_prefix(s, x.s)) {
          n = add_i64_no_overflow(n, f * x.m);
          s.p += 1; s.n -= 1;
          matched = true;
          break;
        }
      }
    }
    if (!matched) {
      // The reference's loop would spin forever here; in practice an unknown
      // suffix never matches any case and the value is not a bytes quantity.
      return false;
    }
  }
  *out = minus ? -n : n;
  return true;
}

// GetLocalTimezoneOffsetNsecs (vendor/.../lib/timeutil/timezone.go:9-19):
// the reference samples time.Now().Zone() — the host local offset of the
// CURRENT time, not of the parsed timestamp — and caches it.  Same here.
int64_t local_tz_offset_nsecs() {
  static const int64_t cached = [] {
    tzset();
    time_t now = time(nullptr);
    struct tm tmv;
    if (localtime_r(&now, &tmv) == nullptr) return int64_t(0);
    return int64_t(tmv.tm_gmtoff) * int64_t(1000000000);
  }();
  return cached;
}

// parseTimezoneOffset + tryParseHHMM (values_encoder.go:383-426); inputs
// without a timezone suffix use the host local timezone offset, as the
// reference does
static bool parse_tz_offset(strview s, int64_t* off, strview* prefix) {
  if (s.n > 0 && s.p[s.n - 1] == 'Z') {
    *off = 0;
    *prefix = strview(s.p, s.n - 1);
    return true;
  }
  long n = -1;
  for (long i = long(s.n) - 1; i >= 0; i--) {
    if (s.p[i] == '+' || s.p[i] == '-') {
      n = i;
      break;
    }
  }
  if (n < 0) {
    *off = local_tz_offset_nsecs();
    *prefix = s;
    return true;
  }
  strview os(s.p + n + 1, s.n - size_t(n) - 1);
  if (os.n != 5 || os.p[2] != ':') return false;
  uint64_t hh, mm;
  if (!try_parse_date_uint64(strview(os.p, 2), &hh) || hh > 24) return false;
  if (!try_parse_date_uint64(strview(os.p + 3, 2), &mm) || mm > 60) return false;
  int64_t v = int64_t(hh) * 3600000000000LL + int64_t(mm) * 60000000000LL;
  *off = s.p[n] == '-' ? -v : v;
  *prefix = strview(s.p, size_t(n));
  return true;
}

bool try_parse_timestamp_rfc3339(strview s, int64_t* out) {
  // TryParseTimestampRFC3339Nano (values_encoder.go:340-381)
  if (s.n < 19) return false;
  int64_t secs;
  strview tail;
  if (!try_parse_timestamp_secs(s, &secs, &tail)) return false;
  s = tail;
  int64_t nsecs = secs * 1000000000;
  int64_t off;
  strview prefix;
  if (!parse_tz_offset(s, &off, &prefix)) return false;
  nsecs -= off;
  s = prefix;
  if (s.n == 0) {
    *out = nsecs;
    return true;
  }
  if (s.p[0] == '.') {
    s.p++;
    s.n--;
  }
  size_t digits = s.n;
  if (digits > 9) return false;
  uint64_t n64;
  if (!try_parse_date_uint64(s, &n64)) return false;
  for (size_t k = digits; k < 9; k++) n64 *= 10;
  *out = nsecs + int64_t(n64);
  return true;
}

bool le_values_string(strview a, strview b, bool exclude_equal) {
  // leValuesString (filter_le_field.go:284-299)
  double fa = parse_math_number(a);
  if (!std::isnan(fa)) {
    double fb = parse_math_number(b);
    if (!std::isnan(fb)) return exclude_equal ? fa < fb : fa <= fb;
  }
  int c = memcmp(a.p, b.p, a.n < b.n ? a.n : b.n);
  if (c == 0) c = a.n < b.n ? -1 : (a.n > b.n ? 1 : 0);
  return exclude_equal ? c < 0 : c <= 0;
}

struct SvReader {
  const char* p;
  uint8_t u8(long i) const { return uint8_t(p[i]); }
};

double parse_math_number(strview s) {
  // parseMathNumber (pipe_math.go:1066-1080) over tryParseNumber
  // (block_result.go:2710-2737)
  double f;
  if (s.n > 0 && try_parse_float64(s, &f)) return f;
  int64_t nsecs;
  if (s.n > 0 && try_parse_duration(s, &nsecs)) return double(nsecs);
  int64_t b;
  if (s.n > 0 && try_parse_bytes(s, &b)) return double(b);
  SvReader r{s.p};
  if (s.n > 0 && vl_pf::pf_is_likely_number(r, long(s.n))) {
    double d;
    if (vl_pf::go_parse_float(r, long(s.n), &d)) return d;
    int64_t iv;
    if (vl_pf::go_parse_int0(r, long(s.n), &iv)) return double(iv);
  }
  if (try_parse_timestamp_rfc3339(s, &nsecs)) return double(nsecs);
  uint32_t ip;
  if (try_parse_ipv4(s, &ip)) return double(ip);
  return NAN;
}

// ---- formatters ----

void format_uint64(std::string& dst, uint64_t n) {
  char buf[24];
  int len = snprintf(buf, sizeof(buf), "%llu", (unsigned long long)n);
  dst.append(buf, len);
}

void format_int64(std::string& dst, int64_t n) {
  char buf[24];
  int len = snprintf(buf, sizeof(buf), "%lld", (long long)n);
  dst.append(buf, len);
}

void format_float64(std::string& dst, double f) {
  // Go strconv.AppendFloat(dst, f, 'f', -1, 64): shortest digits that
  // round-trip, printed in fixed-point form.
  if (std::isnan(f)) { dst += "NaN"; return; }
  if (std::isinf(f)) { dst += f > 0 ? "+Inf" : "-Inf"; return; }
  if (std::signbit(f)) {
    dst += '-';
    f = -f;
  }
  if (f == 0) { dst += '0'; return; }
  // Find the minimal number of significant digits p (1..17) that round-trips,
  // via correctly-rounded %.*e + strtod (both correctly rounded in glibc).
  char buf[64];
  int p = 1;
  for (; p <= 17; p++) {
    snprintf(buf, sizeof(buf), "%.*e", p - 1, f);
    if (strtod(buf, nullptr) == f) break;
  }
  // buf = "d.ddddde±XX"; extract digits and exponent.
  char digits[32];
  int nd = 0;
  int exp10 = 0;
  for (char* q = buf; *q; q++) {
    if (*q >= '0' && *q <= '9' && nd < 31) {
      digits[nd++] = *q;
    } else if (*q == 'e') {
      exp10 = atoi(q + 1);
      break;
    }
  }
  // Strip trailing zeros from digits (keeps the shortest form).
  while (nd > 1 && digits[nd - 1] == '0') nd--;
  // Fixed-point rendering of 0.digits * 10^(exp10+1)
  if (exp10 >= nd - 1) {
    dst.append(digits, nd);
    dst.append(size_t(exp10 - (nd - 1)), '0');
  } else if (exp10 >= 0) {
    dst.append(digits, exp10 + 1);
    dst += '.';
    dst.append(digits + exp10 + 1, nd - exp10 - 1);
  } else {
    dst += "0.";
    dst.append(size_t(-exp10 - 1), '0');
    dst.append(digits, nd);
  }
}

static void format_uint8(std::string& dst, uint8_t n) {
  // marshalUint8String (values_encoder.go:1367-1386)
  format_uint64(dst, n);
}

void format_ipv4(std::string& dst, uint32_t ip) {
  // marshalIPv4String (values_encoder.go:1408-1417)
  format_uint8(dst, uint8_t(ip >> 24));
  dst += '.';
  format_uint8(dst, uint8_t(ip >> 16));
  dst += '.';
  format_uint8(dst, uint8_t(ip >> 8));
  dst += '.';
  format_uint8(dst, uint8_t(ip));
}

// civil date from days since 1970-01-01 (inverse of days_from_civil)
static void civil_from_days(int64_t z, int64_t* y, int* m, int* d) {
  z += 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  int64_t doe = z - era * 146097;
  int64_t yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t yy = yoe + era * 400;
  int64_t doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  int64_t mp = (5 * doy + 2) / 153;
  *d = int(doy - (153 * mp + 2) / 5 + 1);
  *m = int(mp + (mp < 10 ? 3 : -9));
  *y = yy + (*m <= 2);
}

void format_timestamp_iso8601(std::string& dst, int64_t nsecs) {
  // marshalTimestampISO8601String (values_encoder.go:1420-1424):
  // time.Unix(0, nsecs).UTC() with layout "2006-01-02T15:04:05.000Z"
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
  int64_t y;
  int m, d;
  civil_from_days(days, &y, &m, &d);
  int msec = int(rem / 1000000);
  char buf[48];
  int len = snprintf(buf, sizeof(buf), "%04lld-%02d-%02dT%02lld:%02lld:%02lld.%03dZ",
                     (long long)y, m, d, (long long)(sod / 3600),
                     (long long)(sod % 3600 / 60), (long long)(sod % 60), msec);
  dst.append(buf, len);
}

// ---- encoder ----

void encode_values(EncodedColumn& ec, const std::vector<std::string>& values) {
  // valuesEncoder.encode (values_encoder.go:109-154)
  ec = EncodedColumn();
  size_t n = values.size();
  if (n == 0) {
    ec.type = ValueType::String;
    return;
  }

  // Reserve buf up-front: strviews into it must stay stable.
  // Worst per-row encoded width is 8 bytes.
  ec.buf.reserve(n * 8);

  // 1. dict (values_encoder.go:1224-1241)
  {
    bool ok = true;
    std::vector<std::string> dict;
    std::vector<uint8_t> ids(n);
    for (size_t i = 0; i < n && ok; i++) {
      const std::string& v = values[i];
      // valuesDict.getOrAdd (values_encoder.go:1268-1287)
      if (v.size() > kMaxDictSizeBytes) {
        ok = false;
        break;
      }
      size_t j = 0;
      size_t dict_bytes = 0;
      for (; j < dict.size(); j++) {
        if (dict[j] == v) break;
        dict_bytes += dict[j].size();
      }
      if (j == dict.size()) {
        if (dict.size() >= kMaxDictLen || dict_bytes + v.size() > kMaxDictSizeBytes) {
          ok = false;
          break;
        }
        dict.push_back(v);
      }
      ids[i] = uint8_t(j);
    }
    if (ok) {
      ec.type = ValueType::Dict;
      ec.dict = std::move(dict);
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        ec.buf.push_back(ids[i]);
        ec.values.push_back(strview((const char*)ec.buf.data() + at, 1));
      }
      return;
    }
  }

  // 2. uint (values_encoder.go:1168-1222)
  {
    std::vector<uint64_t> a(n);
    bool ok = true;
    uint64_t mn = 0, mx = 0;
    for (size_t i = 0; i < n; i++) {
      if (!try_parse_uint64(strview(values[i]), &a[i])) {
        ok = false;
        break;
      }
      if (i == 0 || a[i] < mn) mn = a[i];
      if (i == 0 || a[i] > mx) mx = a[i];
    }
    if (ok) {
      int bits = 0;
      for (uint64_t v = mx; v; v >>= 1) bits++;
      int width = bits <= 8 ? 1 : bits <= 16 ? 2 : bits <= 32 ? 4 : 8;
      ec.type = width == 1 ? ValueType::Uint8
                : width == 2 ? ValueType::Uint16
                : width == 4 ? ValueType::Uint32 : ValueType::Uint64;
      ec.min_value = mn;
      ec.max_value = mx;
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        switch (width) {
          case 1: ec.buf.push_back(uint8_t(a[i])); break;
          case 2: put_u16be(ec.buf, uint16_t(a[i])); break;
          case 4: put_u32be(ec.buf, uint32_t(a[i])); break;
          default: put_u64be(ec.buf, a[i]); break;
        }
        ec.values.push_back(strview((const char*)ec.buf.data() + at, size_t(width)));
      }
      return;
    }
  }

  // 3. int64 (values_encoder.go:1141-1166); encoded zig-zag BE (int.go:69-74)
  {
    std::vector<int64_t> a(n);
    bool ok = true;
    int64_t mn = 0, mx = 0;
    for (size_t i = 0; i < n; i++) {
      if (!try_parse_int64(strview(values[i]), &a[i])) {
        ok = false;
        break;
      }
      if (i == 0 || a[i] < mn) mn = a[i];
      if (i == 0 || a[i] > mx) mx = a[i];
    }
    if (ok) {
      ec.type = ValueType::Int64;
      ec.min_value = uint64_t(mn);
      ec.max_value = uint64_t(mx);
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        put_i64be_zigzag(ec.buf, a[i]);
        ec.values.push_back(strview((const char*)ec.buf.data() + at, 8));
      }
      return;
    }
  }

  // 4. float64 (values_encoder.go:732-759); encoded Float64bits BE
  {
    std::vector<uint64_t> a(n);
    bool ok = true;
    double mn = 0, mx = 0;
    for (size_t i = 0; i < n; i++) {
      double f;
      if (!try_parse_float64_exact(strview(values[i]), &f)) {
        ok = false;
        break;
      }
      uint64_t u;
      memcpy(&u, &f, 8);
      a[i] = u;
      if (i == 0 || f < mn) mn = f;
      if (i == 0 || f > mx) mx = f;
    }
    if (ok) {
      ec.type = ValueType::Float64;
      uint64_t mnu, mxu;
      memcpy(&mnu, &mn, 8);
      memcpy(&mxu, &mx, 8);
      ec.min_value = mnu;
      ec.max_value = mxu;
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        put_u64be(ec.buf, a[i]);
        ec.values.push_back(strview((const char*)ec.buf.data() + at, 8));
      }
      return;
    }
  }

  // 5. ipv4 (values_encoder.go:647-672); encoded BE u32
  {
    std::vector<uint32_t> a(n);
    bool ok = true;
    uint32_t mn = 0, mx = 0;
    for (size_t i = 0; i < n; i++) {
      if (!try_parse_ipv4(strview(values[i]), &a[i])) {
        ok = false;
        break;
      }
      if (i == 0 || a[i] < mn) mn = a[i];
      if (i == 0 || a[i] > mx) mx = a[i];
    }
    if (ok) {
      ec.type = ValueType::IPv4;
      ec.min_value = mn;
      ec.max_value = mx;
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        put_u32be(ec.buf, a[i]);
        ec.values.push_back(strview((const char*)ec.buf.data() + at, 4));
      }
      return;
    }
  }

  // 6. iso8601 (values_encoder.go:308-333); encoded BE u64 of int64 nsecs
  {
    std::vector<int64_t> a(n);
    bool ok = true;
    int64_t mn = 0, mx = 0;
    for (size_t i = 0; i < n; i++) {
      if (!try_parse_timestamp_iso8601(strview(values[i]), &a[i])) {
        ok = false;
        break;
      }
      if (i == 0 || a[i] < mn) mn = a[i];
      if (i == 0 || a[i] > mx) mx = a[i];
    }
    if (ok) {
      ec.type = ValueType::TimestampISO8601;
      ec.min_value = uint64_t(mn);
      ec.max_value = uint64_t(mx);
      for (size_t i = 0; i < n; i++) {
        size_t at = ec.buf.size();
        put_u64be(ec.buf, uint64_t(a[i]));
        ec.values.push_back(strview((const char*)ec.buf.data() + at, 8));
      }
      return;
    }
  }

  // 7. fallback: string (values_encoder.go:151-153)
  ec.type = ValueType::String;
  for (size_t i = 0; i < n; i++) ec.values.push_back(strview(values[i]));
}

}  // namespace vl
