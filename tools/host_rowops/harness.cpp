// Host build of the EXACT per-row device code (scan_rowops.h, the same file
// scan_kernels.hip compiles for gfx950) so it can be differential-fuzzed at
// scale on the CPU against the host/oracle implementations.  Wavefront-level
// code (tile loops, ballots, kernels) is not included — its correctness is
// covered by the GPU parity suite; this harness covers the per-row matcher /
// parser / formatter logic where the subtle SWAR/NFA bugs live.
#include <cmath>
#include <cstdint>
#include <cstring>
#include <cstdio>

using std::fma;
using std::trunc;
using std::isnan;

#define __device__
#define __forceinline__ inline
#define __noinline__ __attribute__((noinline))

#include "../../victorialogs_amd/csrc/core/parse_float.h"
#include "../../victorialogs_amd/csrc/core/ryu.h"
#include "../../victorialogs_amd/csrc/hip/scan_types.h"

namespace vl {
int64_t g_vl_local_tz_nsecs = 0;
#include "../../victorialogs_amd/csrc/hip/scan_rowops.h"
}  // namespace vl

// keep the device-code mirror's local-tz behavior identical to the oracle's
__attribute__((constructor)) static void rowops_tz_init();

#include "../../victorialogs_amd/csrc/core/match.h"
#include "../../victorialogs_amd/csrc/core/op_serialize.h"
#include "../../victorialogs_amd/csrc/core/regex.h"
#include "../../victorialogs_amd/csrc/core/values.h"

using namespace vl;

__attribute__((constructor)) static void rowops_tz_init() {
  vl::g_vl_local_tz_nsecs = vl::local_tz_offset_nsecs();
}

extern "C" {

// device-code entry points over a plain byte buffer (BufAcc semantics)
long h_dev_phrase_pos(const char* s, long sn, const char* ph, long pn) {
  BufAcc a{(const uint8_t*)s};
  uint8_t flags = phrase_flags_of(std::string(ph, size_t(pn)));
  return d_get_phrase_pos_at(a, 0, sn, (const uint8_t*)ph, pn, flags);
}

int h_dev_match_prefix(const char* s, long sn, const char* pf, long pn) {
  BufAcc a{(const uint8_t*)s};
  uint8_t flags = phrase_flags_of(std::string(pf, size_t(pn)));
  return d_match_prefix_at(a, 0, sn, (const uint8_t*)pf, pn, flags) ? 1 : 0;
}

int h_dev_regex_match(const char* pattern, long pn, const char* s, long sn) {
  try {
    RegexProg re = regex_compile(std::string(pattern, size_t(pn)));
    bytes blob = serialize_regex(re);
    BufAcc a{(const uint8_t*)s};
    return d_regex_match_at(blob.data(), a, 0, sn) ? 1 : 0;
  } catch (...) {
    return -1;
  }
}

double h_dev_parse_math(const char* s, long sn) {
  BufAcc a{(const uint8_t*)s};
  return d_parse_math_number(a, 0, sn);
}

long h_dev_format(int what, long long v, char* out) {
  switch (what) {
    case 0: return d_format_u64(out, (uint64_t)v);
    case 1: return d_format_i64(out, v);
    case 2: return d_format_ipv4(out, (uint32_t)v);
    case 3: return d_format_iso8601(out, v);
    case 4: return vl_ryu::format_f64(out, (uint64_t)v);
    default: return -1;
  }
}

int h_dev_match_sequence(const char* s, long sn, const char* joined, long jn) {
  std::vector<std::string> phrases;
  const char* p = joined;
  const char* end = joined + jn;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
    if (!nl) nl = end;
    phrases.emplace_back(p, nl);
    p = nl + 1;
  }
  bytes blob = serialize_phrases(phrases);
  BufAcc a{(const uint8_t*)s};
  return d_match_sequence_at(a, 0, sn, blob.data()) ? 1 : 0;
}

int h_dev_any_case_phrase(const char* s, long sn, const char* lower, long ln) {
  // ASCII rows only (non-ASCII rows are host-resolved in staging)
  if (ln == 0) return sn == 0;
  if (ln > sn) return 0;
  BufAcc a{(const uint8_t*)s};
  LowerAcc<BufAcc> la{a};
  uint8_t flags = phrase_flags_of(std::string(lower, size_t(ln)));
  return d_get_phrase_pos_at(la, 0, sn, (const uint8_t*)lower, ln, flags) >= 0
             ? 1
             : 0;
}

int h_dev_le_values(const char* a, long an, const char* b, long bn, int excl) {
  return d_le_values_string((const uint8_t*)a, an, (const uint8_t*)b, bn,
                            excl != 0)
             ? 1
             : 0;
}

}  // extern "C"
