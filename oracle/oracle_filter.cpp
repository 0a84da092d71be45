// TEST INFRASTRUCTURE — see oracle_filter.h.  Each function cites the
// reference function it restates.
#include "oracle_filter.h"

#include <algorithm>
#include <cmath>
#include <cstring>

#include "../victorialogs_amd/csrc/core/bloom.h"
#include "../victorialogs_amd/csrc/core/match.h"
#include "../victorialogs_amd/csrc/core/unicode_case.h"
#include "../victorialogs_amd/csrc/core/values.h"

namespace vl {
namespace oracle {

const PartReader::BlockColumns& BlockCtx::columns() {
  if (!bc_loaded) {
    pr->read_block_columns(*bh, bc);
    bc_loaded = true;
  }
  return bc;
}

std::string BlockCtx::const_value(const std::string& name) {
  std::string v;
  if (!pr->get_const_column(columns(), name, &v)) return "";
  return v;
}

bool BlockCtx::column_header(const std::string& name, ColumnHeader* ch) {
  return pr->get_column_header(columns(), name, ch);
}

const StringsBlockDec& BlockCtx::values(const ColumnHeader& ch) {
  auto it = values_cache.find(ch.name);
  if (it == values_cache.end()) {
    StringsBlockDec dec;
    pr->read_values(ch, bh->rows_count, dec);
    it = values_cache.emplace(ch.name, std::move(dec)).first;
  }
  return it->second;
}

const std::vector<uint64_t>& BlockCtx::bloom(const ColumnHeader& ch) {
  auto it = bloom_cache.find(ch.name);
  if (it == bloom_cache.end()) {
    std::vector<uint64_t> words;
    pr->read_bloom(ch, words);
    it = bloom_cache.emplace(ch.name, std::move(words)).first;
  }
  return it->second;
}

const std::vector<int64_t>& BlockCtx::get_timestamps() {
  if (!ts_loaded) {
    pr->read_timestamps(*bh, timestamps);
    ts_loaded = true;
  }
  return timestamps;
}

// matchBloomFilterAllTokens (filter_phrase.go:302-308)
static bool match_bloom_all(BlockCtx& ctx, const ColumnHeader& ch,
                            const std::vector<uint64_t>& hashes) {
  if (hashes.empty()) return true;
  const auto& words = ctx.bloom(ch);
  return bloom_contains_all(words.data(), words.size(), hashes.data(), hashes.size());
}

// visitValues (filter_phrase.go:291-300)
template <typename F>
static void visit_values(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm, F&& f) {
  if (bm.is_zero()) return;
  const StringsBlockDec& vals = ctx.values(ch);
  bm.for_each_set_bit([&](uint64_t idx) { return f(vals.row(idx)); });
}

// matchBinaryValue (filter_exact.go:356-364)
static void match_binary_value(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                               const bytes& bin,
                               const std::vector<uint64_t>& hashes) {
  if (!match_bloom_all(ctx, ch, hashes)) {
    bm.reset_bits();
    return;
  }
  strview b((const char*)bin.data(), bin.size());
  visit_values(ctx, ch, bm, [&](strview v) { return v == b; });
}

// matchEncodedValuesDict (filter_phrase.go:272-289)
static void match_encoded_dict(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                               const std::vector<uint8_t>& encoded) {
  bool any = false;
  for (uint8_t c : encoded) any |= (c == 1);
  if (!any) {
    bm.reset_bits();
    return;
  }
  visit_values(ctx, ch, bm, [&](strview v) {
    if (v.n != 1) fail("unexpected dict value length");
    uint8_t idx = uint8_t(v.p[0]);
    if (idx >= encoded.size()) fail("too big dict index");
    return encoded[idx] == 1;
  });
}

// ---- exact-value matchers (filter_exact.go:237-364) ----

static void match_uint_by_exact(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                                strview phrase, const std::vector<uint64_t>& hashes,
                                int width) {
  uint64_t n;
  if (!try_parse_uint64(phrase, &n) || n < ch.min_value || n > ch.max_value) {
    bm.reset_bits();
    return;
  }
  bytes bin;
  switch (width) {
    case 1: bin.push_back(uint8_t(n)); break;
    case 2: put_u16be(bin, uint16_t(n)); break;
    case 4: put_u32be(bin, uint32_t(n)); break;
    default: put_u64be(bin, n); break;
  }
  match_binary_value(ctx, ch, bm, bin, hashes);
}

static void match_int64_by_exact(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                                 strview phrase, const std::vector<uint64_t>& hashes) {
  int64_t n;
  if (!try_parse_int64(phrase, &n) || n < int64_t(ch.min_value) ||
      n > int64_t(ch.max_value)) {
    bm.reset_bits();
    return;
  }
  bytes bin;
  put_i64be_zigzag(bin, n);
  match_binary_value(ctx, ch, bm, bin, hashes);
}

static void match_float64_by_exact(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                                   strview value, const std::vector<uint64_t>& hashes) {
  double f, mn, mx;
  uint64_t mnu = ch.min_value, mxu = ch.max_value;
  memcpy(&mn, &mnu, 8);
  memcpy(&mx, &mxu, 8);
  if (!try_parse_float64_exact(value, &f) || f < mn || f > mx) {
    bm.reset_bits();
    return;
  }
  bytes bin;
  uint64_t u;
  memcpy(&u, &f, 8);
  put_u64be(bin, u);
  match_binary_value(ctx, ch, bm, bin, hashes);
}

static void match_ipv4_by_exact(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                                strview value, const std::vector<uint64_t>& hashes) {
  uint32_t n;
  if (!try_parse_ipv4(value, &n) || uint64_t(n) < ch.min_value ||
      uint64_t(n) > ch.max_value) {
    bm.reset_bits();
    return;
  }
  bytes bin;
  put_u32be(bin, n);
  match_binary_value(ctx, ch, bm, bin, hashes);
}

static void match_iso8601_by_exact(BlockCtx& ctx, const ColumnHeader& ch, Bitmap& bm,
                                   strview value,
                                   const std::vector<uint64_t>& hashes) {
  int64_t n;
  if (!try_parse_timestamp_iso8601(value, &n) || n < int64_t(ch.min_value) ||
      n > int64_t(ch.max_value)) {
    bm.reset_bits();
    return;
  }
  bytes bin;
  put_u64be(bin, uint64_t(n));
  match_binary_value(ctx, ch, bm, bin, hashes);
}

// ---- string conversions (filter_phrase.go:321-346) ----

static std::string to_uint_string(strview v, int width) {
  uint64_t n = 0;
  switch (width) {
    case 1: n = uint8_t(v.p[0]); break;
    case 2: n = get_u16be((const uint8_t*)v.p); break;
    case 4: n = get_u32be((const uint8_t*)v.p); break;
    default: n = get_u64be((const uint8_t*)v.p); break;
  }
  std::string s;
  format_uint64(s, n);
  return s;
}

// decoded string form of a fixed-width encoded value (to*String helpers)
static std::string format_value(ValueType t, strview v) {
  std::string s;
  const uint8_t* p = (const uint8_t*)v.p;
  switch (t) {
    case ValueType::Uint8: format_uint64(s, p[0]); break;
    case ValueType::Uint16: format_uint64(s, get_u16be(p)); break;
    case ValueType::Uint32: format_uint64(s, get_u32be(p)); break;
    case ValueType::Uint64: format_uint64(s, get_u64be(p)); break;
    case ValueType::Int64: format_int64(s, get_i64be_zigzag(p)); break;
    case ValueType::Float64: {
      uint64_t u = get_u64be(p);
      double d;
      memcpy(&d, &u, 8);
      format_float64(s, d);
      break;
    }
    case ValueType::IPv4: format_ipv4(s, get_u32be(p)); break;
    case ValueType::TimestampISO8601:
      format_timestamp_iso8601(s, int64_t(get_u64be(p)));
      break;
    default:
      fail("format_value: unexpected type");
  }
  return s;
}

// ---- filterPhrase (filter_phrase.go:61-209) ----

static void apply_phrase(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  strview phrase(f.phrase);

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_phrase(strview(cv), phrase)) bm.reset_bits();
    return;
  }

  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (phrase.n > 0) bm.reset_bits();  // filter_phrase.go:76-83
    return;
  }

  switch (ch.type) {
    case ValueType::String: {
      // matchStringByPhrase (filter_phrase.go:201-209)
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) { return match_phrase(v, phrase); });
      return;
    }
    case ValueType::Dict: {
      // matchValuesDictByPhrase (filter_phrase.go:188-199)
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) {
        enc.push_back(match_phrase(strview(dv), phrase) ? 1 : 0);
      }
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
      match_uint_by_exact(ctx, ch, bm, phrase, f.token_hashes, 1);
      return;
    case ValueType::Uint16:
      match_uint_by_exact(ctx, ch, bm, phrase, f.token_hashes, 2);
      return;
    case ValueType::Uint32:
      match_uint_by_exact(ctx, ch, bm, phrase, f.token_hashes, 4);
      return;
    case ValueType::Uint64:
      match_uint_by_exact(ctx, ch, bm, phrase, f.token_hashes, 8);
      return;
    case ValueType::Int64:
      match_int64_by_exact(ctx, ch, bm, phrase, f.token_hashes);
      return;
    case ValueType::Float64: {
      // matchFloat64ByPhrase (filter_phrase.go:159-186)
      double ff;
      bool ok = try_parse_float64_exact(phrase, &ff);
      bool special = phrase == std::string(".") || phrase == std::string("+") ||
                     phrase == std::string("-");
      if (!ok && !special) {
        bm.reset_bits();
        return;
      }
      const char* dot = phrase.n ? (const char*)memchr(phrase.p, '.', phrase.n) : nullptr;
      long ndot = dot ? dot - phrase.p : -1;
      if (ndot > 0 && size_t(ndot) < phrase.n - 1) {
        match_float64_by_exact(ctx, ch, bm, phrase, f.token_hashes);
        return;
      }
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        if (v.n != 8) fail("unexpected float64 binary length");
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        std::string s;
        format_float64(s, d);
        return match_phrase(strview(s), phrase);
      });
      return;
    }
    case ValueType::IPv4: {
      // matchIPv4ByPhrase (filter_phrase.go:135-157)
      uint32_t ip;
      if (try_parse_ipv4(phrase, &ip)) {
        match_ipv4_by_exact(ctx, ch, bm, phrase, f.token_hashes);
        return;
      }
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        if (v.n != 4) fail("unexpected ipv4 binary length");
        std::string s;
        format_ipv4(s, get_u32be((const uint8_t*)v.p));
        return match_phrase(strview(s), phrase);
      });
      return;
    }
    case ValueType::TimestampISO8601: {
      // matchTimestampISO8601ByPhrase (filter_phrase.go:113-133)
      int64_t ts;
      if (try_parse_timestamp_iso8601(phrase, &ts)) {
        match_iso8601_by_exact(ctx, ch, bm, phrase, f.token_hashes);
        return;
      }
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        if (v.n != 8) fail("unexpected iso8601 binary length");
        std::string s;
        format_timestamp_iso8601(s, int64_t(get_u64be((const uint8_t*)v.p)));
        return match_phrase(strview(s), phrase);
      });
      return;
    }
    default:
      fail("unknown valueType in phrase filter");
  }
}

// delegate used by filterSequence's len==1 ipv4/iso cases: evaluate the
// phrase matcher with the sequence's single phrase + the sequence's tokens
static void apply_phrase_for(const FilterNode& seq, const std::string& phrase,
                             BlockCtx& ctx, Bitmap& bm) {
  FilterNode tmp;
  tmp.type = FilterNode::Phrase;
  tmp.field = seq.field;
  tmp.phrase = phrase;
  tmp.token_hashes = seq.token_hashes;
  apply_phrase(tmp, ctx, bm);
}

// ---- filterExact (filter_exact.go:178-235) ----

static void apply_exact(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  strview value(f.phrase);

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!(strview(cv) == value)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (value.n > 0) bm.reset_bits();
    return;
  }
  switch (ch.type) {
    case ValueType::String: {
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) { return v == value; });
      return;
    }
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) enc.push_back(strview(dv) == value ? 1 : 0);
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8: match_uint_by_exact(ctx, ch, bm, value, f.token_hashes, 1); return;
    case ValueType::Uint16: match_uint_by_exact(ctx, ch, bm, value, f.token_hashes, 2); return;
    case ValueType::Uint32: match_uint_by_exact(ctx, ch, bm, value, f.token_hashes, 4); return;
    case ValueType::Uint64: match_uint_by_exact(ctx, ch, bm, value, f.token_hashes, 8); return;
    case ValueType::Int64: match_int64_by_exact(ctx, ch, bm, value, f.token_hashes); return;
    case ValueType::Float64: match_float64_by_exact(ctx, ch, bm, value, f.token_hashes); return;
    case ValueType::IPv4: match_ipv4_by_exact(ctx, ch, bm, value, f.token_hashes); return;
    case ValueType::TimestampISO8601: match_iso8601_by_exact(ctx, ch, bm, value, f.token_hashes); return;
    default: fail("unknown valueType in exact filter");
  }
}

// ---- filterRegexp (filter_regexp.go:78-254) ----

static void apply_regexp(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!regex_match(f.re, strview(cv))) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!regex_match(f.re, strview("", 0))) bm.reset_bits();
    return;
  }

  if (ch.type == ValueType::Dict) {
    std::vector<uint8_t> enc;
    for (const auto& dv : ch.dict) enc.push_back(regex_match(f.re, strview(dv)) ? 1 : 0);
    match_encoded_dict(ctx, ch, bm, enc);
    return;
  }

  if (!match_bloom_all(ctx, ch, f.token_hashes)) {
    bm.reset_bits();
    return;
  }
  switch (ch.type) {
    case ValueType::String:
      visit_values(ctx, ch, bm, [&](strview v) { return regex_match(f.re, v); });
      return;
    case ValueType::Uint8:
    case ValueType::Uint16:
    case ValueType::Uint32:
    case ValueType::Uint64: {
      int width = ch.type == ValueType::Uint8 ? 1
                  : ch.type == ValueType::Uint16 ? 2
                  : ch.type == ValueType::Uint32 ? 4 : 8;
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string s = to_uint_string(v, width);
        return regex_match(f.re, strview(s));
      });
      return;
    }
    case ValueType::Int64:
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string s;
        format_int64(s, get_i64be_zigzag((const uint8_t*)v.p));
        return regex_match(f.re, strview(s));
      });
      return;
    case ValueType::Float64:
      visit_values(ctx, ch, bm, [&](strview v) {
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        std::string s;
        format_float64(s, d);
        return regex_match(f.re, strview(s));
      });
      return;
    case ValueType::IPv4:
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string s;
        format_ipv4(s, get_u32be((const uint8_t*)v.p));
        return regex_match(f.re, strview(s));
      });
      return;
    case ValueType::TimestampISO8601:
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string s;
        format_timestamp_iso8601(s, int64_t(get_u64be((const uint8_t*)v.p)));
        return regex_match(f.re, strview(s));
      });
      return;
    default:
      fail("unknown valueType in regexp filter");
  }
}

// ---- filterTime (filter_time.go:114-137) ----

static void apply_time(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  if (f.min_ts > f.max_ts) {
    bm.reset_bits();
    return;
  }
  const TimestampsHeader& th = ctx.bh->timestamps_header;
  if (f.min_ts > th.max_timestamp || f.max_ts < th.min_timestamp) {
    bm.reset_bits();
    return;
  }
  if (f.min_ts <= th.min_timestamp && f.max_ts >= th.max_timestamp) return;
  const auto& ts = ctx.get_timestamps();
  bm.for_each_set_bit(
      [&](uint64_t idx) { return ts[idx] >= f.min_ts && ts[idx] <= f.max_ts; });
}

// ---- filterRange (filter_range.go:180-372) ----

static uint64_t to_u64_clamp(double f) {
  if (f < 0) return 0;
  if (f > double(UINT64_MAX)) return UINT64_MAX;
  return uint64_t(f);
}
static int64_t to_i64_clamp(double f) {
  if (f < double(INT64_MIN)) return INT64_MIN;
  if (f >= double(INT64_MAX)) return INT64_MAX;
  return int64_t(f);
}
static uint32_t to_u32_clamp(double f) {
  if (f < 0) return 0;
  if (f > double(UINT32_MAX)) return UINT32_MAX;
  return uint32_t(f);
}

static void apply_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  double min_v = f.min_f, max_v = f.max_f;
  if (min_v > max_v) {
    bm.reset_bits();
    return;
  }
  std::string name = canonical_field(f.field);
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    double x = parse_math_number(strview(cv));
    if (!(x >= min_v && x <= max_v)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    bm.reset_bits();  // filter_range.go:199-204
    return;
  }

  switch (ch.type) {
    case ValueType::String:
      // matchStringByRange (filter_range.go:261-265)
      visit_values(ctx, ch, bm, [&](strview v) {
        double x = parse_math_number(v);
        return x >= min_v && x <= max_v;
      });
      return;
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) {
        double x = parse_math_number(strview(dv));
        enc.push_back(x >= min_v && x <= max_v ? 1 : 0);
      }
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
    case ValueType::Uint16:
    case ValueType::Uint32:
    case ValueType::Uint64: {
      // matchUintNByRange (filter_range.go:267-333): bounds via
      // ceil/floor + clamp (toUint64Range, :374-388)
      uint64_t mn = to_u64_clamp(std::ceil(min_v));
      uint64_t mx = to_u64_clamp(std::floor(max_v));
      if (max_v < 0 || mn > ch.max_value || mx < ch.min_value) {
        bm.reset_bits();
        return;
      }
      int width = ch.type == ValueType::Uint8 ? 1
                  : ch.type == ValueType::Uint16 ? 2
                  : ch.type == ValueType::Uint32 ? 4 : 8;
      visit_values(ctx, ch, bm, [&](strview v) {
        uint64_t n;
        switch (width) {
          case 1: n = uint8_t(v.p[0]); break;
          case 2: n = get_u16be((const uint8_t*)v.p); break;
          case 4: n = get_u32be((const uint8_t*)v.p); break;
          default: n = get_u64be((const uint8_t*)v.p); break;
        }
        return n >= mn && n <= mx;
      });
      return;
    }
    case ValueType::Int64: {
      // matchInt64ByRange (filter_range.go:335-350)
      int64_t mn = to_i64_clamp(std::ceil(min_v));
      int64_t mx = to_i64_clamp(std::floor(max_v));
      if (mn > int64_t(ch.max_value) || mx < int64_t(ch.min_value)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        int64_t n = get_i64be_zigzag((const uint8_t*)v.p);
        return n >= mn && n <= mx;
      });
      return;
    }
    case ValueType::Float64: {
      // matchFloat64ByRange (filter_range.go:233-246)
      double cmn, cmx;
      uint64_t mnu = ch.min_value, mxu = ch.max_value;
      memcpy(&cmn, &mnu, 8);
      memcpy(&cmx, &mxu, 8);
      if (min_v > cmx || max_v < cmn) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        if (v.n != 8) fail("unexpected float64 binary length");
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        return d >= min_v && d <= max_v;
      });
      return;
    }
    case ValueType::IPv4: {
      // filter_range.go:223-225 + matchIPv4ByRange
      uint32_t mn = to_u32_clamp(std::ceil(min_v));
      uint32_t mx = to_u32_clamp(std::floor(max_v));
      if (max_v < 0 || uint64_t(mn) > ch.max_value || uint64_t(mx) < ch.min_value) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        uint32_t n = get_u32be((const uint8_t*)v.p);
        return n >= mn && n <= mx;
      });
      return;
    }
    case ValueType::TimestampISO8601: {
      // matchTimestampISO8601ByRange (filter_range.go:352-367)
      int64_t mn = to_i64_clamp(std::ceil(min_v));
      int64_t mx = to_i64_clamp(std::floor(max_v));
      if (max_v < 0 || mn > int64_t(ch.max_value) || mx < int64_t(ch.min_value)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        int64_t n = int64_t(get_u64be((const uint8_t*)v.p));
        return n >= mn && n <= mx;
      });
      return;
    }
    default:
      fail("unknown valueType in range filter");
  }
}


// ---- filterPrefix (filter_prefix.go:58-316) ----

static void apply_prefix(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  strview prefix(f.phrase);

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_prefix(strview(cv), prefix)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    bm.reset_bits();  // filter_prefix.go:73-78
    return;
  }
  switch (ch.type) {
    case ValueType::String: {
      // matchStringByPrefix (filter_prefix.go:190-198)
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) { return match_prefix(v, prefix); });
      return;
    }
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) {
        enc.push_back(match_prefix(strview(dv), prefix) ? 1 : 0);
      }
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
    case ValueType::Uint16:
    case ValueType::Uint32:
    case ValueType::Uint64: {
      // matchUintNByPrefix (filter_prefix.go:200-285): no bloom gate
      if (prefix.n == 0) return;
      uint64_t n;
      if (!try_parse_uint64(prefix, &n) || n > ch.max_value) {
        bm.reset_bits();
        return;
      }
      int width = ch.type == ValueType::Uint8 ? 1
                  : ch.type == ValueType::Uint16 ? 2
                  : ch.type == ValueType::Uint32 ? 4 : 8;
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string str = to_uint_string(v, width);
        return match_prefix(strview(str), prefix);
      });
      return;
    }
    case ValueType::Int64: {
      // matchInt64ByPrefix (filter_prefix.go:287-310)
      if (prefix.n == 0) return;
      if (!(prefix == std::string("-"))) {
        int64_t n;
        if (!try_parse_int64(prefix, &n) || n < int64_t(ch.min_value) ||
            n > int64_t(ch.max_value)) {
          bm.reset_bits();
          return;
        }
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string str;
        format_int64(str, get_i64be_zigzag((const uint8_t*)v.p));
        return match_prefix(strview(str), prefix);
      });
      return;
    }
    case ValueType::Float64: {
      // matchFloat64ByPrefix (filter_prefix.go:148-176)
      if (prefix.n == 0) return;
      double ff;
      bool ok = try_parse_float64_exact(prefix, &ff);
      bool special = f.phrase == "." || f.phrase == "+" || f.phrase == "-" ||
                     (prefix.n > 0 && (prefix.p[0] == 'e' || prefix.p[0] == 'E'));
      if (!ok && !special) {
        bm.reset_bits();
        return;
      }
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        std::string str;
        format_float64(str, d);
        return match_prefix(strview(str), prefix);
      });
      return;
    }
    case ValueType::IPv4: {
      // matchIPv4ByPrefix (filter_prefix.go:128-147)
      if (prefix.n == 0) return;
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string str;
        format_ipv4(str, get_u32be((const uint8_t*)v.p));
        return match_prefix(strview(str), prefix);
      });
      return;
    }
    case ValueType::TimestampISO8601: {
      // matchTimestampISO8601ByPrefix (filter_prefix.go:107-126)
      if (prefix.n == 0) return;
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string str;
        format_timestamp_iso8601(str, int64_t(get_u64be((const uint8_t*)v.p)));
        return match_prefix(strview(str), prefix);
      });
      return;
    }
    default:
      fail("unknown valueType in prefix filter");
  }
}

// ---- filterExactPrefix (filter_exact_prefix.go:52-277) ----

static void apply_exact_prefix(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  strview prefix(f.phrase);

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_exact_prefix(strview(cv), prefix)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (prefix.n > 0) bm.reset_bits();  // matchExactPrefix("", prefix)
    return;
  }
  auto fmt_visit = [&](auto&& fmt) {
    visit_values(ctx, ch, bm, [&](strview v) {
      std::string str = fmt(v);
      return match_exact_prefix(strview(str), prefix);
    });
  };
  switch (ch.type) {
    case ValueType::String: {
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm,
                   [&](strview v) { return match_exact_prefix(v, prefix); });
      return;
    }
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) {
        enc.push_back(match_exact_prefix(strview(dv), prefix) ? 1 : 0);
      }
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
    case ValueType::Uint16:
    case ValueType::Uint32:
    case ValueType::Uint64: {
      // matchMinMaxExactPrefix (filter_exact_prefix.go:255-273)
      if (prefix.n == 0) return;
      if (!f.token_hashes.empty()) {
        bm.reset_bits();
        return;
      }
      uint64_t n;
      if (!try_parse_uint64(prefix, &n) || n > ch.max_value) {
        bm.reset_bits();
        return;
      }
      int width = ch.type == ValueType::Uint8 ? 1
                  : ch.type == ValueType::Uint16 ? 2
                  : ch.type == ValueType::Uint32 ? 4 : 8;
      fmt_visit([&](strview v) { return to_uint_string(v, width); });
      return;
    }
    case ValueType::Int64: {
      // matchInt64ByExactPrefix (filter_exact_prefix.go:226-253)
      if (prefix.n == 0) return;
      if (!f.token_hashes.empty()) {
        bm.reset_bits();
        return;
      }
      if (!(prefix == std::string("-"))) {
        int64_t n;
        if (!try_parse_int64(prefix, &n) || n > int64_t(ch.max_value) ||
            n < int64_t(ch.min_value)) {
          bm.reset_bits();
          return;
        }
      }
      fmt_visit([&](strview v) {
        std::string str;
        format_int64(str, get_i64be_zigzag((const uint8_t*)v.p));
        return str;
      });
      return;
    }
    case ValueType::Float64: {
      // matchFloat64ByExactPrefix (filter_exact_prefix.go:136-153)
      if (prefix.n == 0) return;
      if (f.token_hashes.size() > 2 * kBloomHashesCount ||
          !match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      fmt_visit([&](strview v) {
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        std::string str;
        format_float64(str, d);
        return str;
      });
      return;
    }
    case ValueType::IPv4: {
      // matchIPv4ByExactPrefix (filter_exact_prefix.go:119-134)
      if (prefix.n == 0) return;
      if (f.phrase < "0" || f.phrase > "9" ||
          f.token_hashes.size() > 3 * kBloomHashesCount ||
          !match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      fmt_visit([&](strview v) {
        std::string str;
        format_ipv4(str, get_u32be((const uint8_t*)v.p));
        return str;
      });
      return;
    }
    case ValueType::TimestampISO8601: {
      // matchTimestampISO8601ByExactPrefix (filter_exact_prefix.go:102-117)
      if (prefix.n == 0) return;
      if (f.phrase < "0" || f.phrase > "9" ||
          !match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      fmt_visit([&](strview v) {
        std::string str;
        format_timestamp_iso8601(str, int64_t(get_u64be((const uint8_t*)v.p)));
        return str;
      });
      return;
    }
    default:
      fail("unknown valueType in exact_prefix filter");
  }
}

// ---- filterSequence (filter_sequence.go:84-269) ----

static void apply_phrase_for(const FilterNode& seq, const std::string& phrase,
                             BlockCtx& ctx, Bitmap& bm);

static void apply_sequence(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  const auto& phrases = f.phrases;
  if (phrases.empty()) return;  // filter_sequence.go:88-90

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_sequence(strview(cv), phrases)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!match_sequence(strview("", 0), phrases)) bm.reset_bits();
    return;
  }
  switch (ch.type) {
    case ValueType::String: {
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm,
                   [&](strview v) { return match_sequence(v, phrases); });
      return;
    }
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) {
        enc.push_back(match_sequence(strview(dv), phrases) ? 1 : 0);
      }
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
    case ValueType::Uint16:
    case ValueType::Uint32:
    case ValueType::Uint64:
    case ValueType::Int64: {
      // matchUintNBySequence / matchInt64BySequence (filter_sequence.go:219-258)
      if (phrases.size() > 1) {
        bm.reset_bits();
        return;
      }
      if (ch.type == ValueType::Int64) {
        match_int64_by_exact(ctx, ch, bm, strview(phrases[0]), f.token_hashes);
      } else {
        int width = ch.type == ValueType::Uint8 ? 1
                    : ch.type == ValueType::Uint16 ? 2
                    : ch.type == ValueType::Uint32 ? 4 : 8;
        match_uint_by_exact(ctx, ch, bm, strview(phrases[0]), f.token_hashes, width);
      }
      return;
    }
    case ValueType::Float64: {
      // matchFloat64BySequence (filter_sequence.go:179-196): always slow path
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        uint64_t u = get_u64be((const uint8_t*)v.p);
        double d;
        memcpy(&d, &u, 8);
        std::string str;
        format_float64(str, d);
        return match_sequence(strview(str), phrases);
      });
      return;
    }
    case ValueType::IPv4:
    case ValueType::TimestampISO8601: {
      // len==1 delegates to the phrase matcher (filter_sequence.go:139-177)
      if (phrases.size() == 1) {
        apply_phrase_for(f, phrases[0], ctx, bm);
        return;
      }
      if (!match_bloom_all(ctx, ch, f.token_hashes)) {
        bm.reset_bits();
        return;
      }
      bool is_ip = ch.type == ValueType::IPv4;
      visit_values(ctx, ch, bm, [&](strview v) {
        std::string str;
        if (is_ip) {
          format_ipv4(str, get_u32be((const uint8_t*)v.p));
        } else {
          format_timestamp_iso8601(str, int64_t(get_u64be((const uint8_t*)v.p)));
        }
        return match_sequence(strview(str), phrases);
      });
      return;
    }
    default:
      fail("unknown valueType in sequence filter");
  }
}


// ---- filterIn (filter_in.go:120-234) ----

static bool in_set(const std::vector<std::string>& sorted_set, strview v) {
  // binary search over the sorted value set
  size_t lo = 0, hi = sorted_set.size();
  while (lo < hi) {
    size_t mid = (lo + hi) / 2;
    const std::string& m = sorted_set[mid];
    int c = memcmp(m.data(), v.p, std::min(m.size(), v.n));
    if (c == 0) c = m.size() < v.n ? -1 : (m.size() > v.n ? 1 : 0);
    if (c == 0) return true;
    if (c < 0) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return false;
}

static int bin_set_slot(ValueType t) {
  switch (t) {
    case ValueType::Uint8: return 0;
    case ValueType::Uint16: return 1;
    case ValueType::Uint32: return 2;
    case ValueType::Uint64: return 3;
    case ValueType::Int64: return 4;
    case ValueType::Float64: return 5;
    case ValueType::IPv4: return 6;
    case ValueType::TimestampISO8601: return 7;
    default: return -1;
  }
}

// matchBloomFilterAnyTokenSet (filter_in.go:202-218)
static bool match_bloom_any_token_set(const FilterNode& f, BlockCtx& ctx,
                                      const ColumnHeader& ch) {
  if (!match_bloom_all(ctx, ch, f.common_hashes)) return false;
  if (f.set_hashes.size() > 1000 ||
      f.set_hashes.size() > 10 * ctx.bh->rows_count) {
    return true;
  }
  const auto& words = ctx.bloom(ch);
  for (const auto& hs : f.set_hashes) {
    if (bloom_contains_all(words.data(), words.size(), hs.data(), hs.size())) {
      return true;
    }
  }
  return false;
}

static void apply_in(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  if (f.values.empty()) {
    bm.reset_bits();
    return;
  }
  auto has = [&](strview v) {
    for (const auto& s2 : f.values) {
      if (strview(s2) == v) return true;
    }
    return false;
  };
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!has(strview(cv))) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!has(strview("", 0))) bm.reset_bits();
    return;
  }
  if (ch.type == ValueType::Dict) {
    std::vector<uint8_t> enc;
    for (const auto& dv : ch.dict) enc.push_back(has(strview(dv)) ? 1 : 0);
    match_encoded_dict(ctx, ch, bm, enc);
    return;
  }
  // matchAnyValue (filter_in.go:187-200)
  const std::vector<std::string>* set;
  std::vector<std::string> str_sorted;
  if (ch.type == ValueType::String) {
    str_sorted = f.values;
    std::sort(str_sorted.begin(), str_sorted.end());
    str_sorted.erase(std::unique(str_sorted.begin(), str_sorted.end()),
                     str_sorted.end());
    set = &str_sorted;
  } else {
    set = &f.bin_sets[size_t(bin_set_slot(ch.type))];
  }
  if (set->empty()) {
    bm.reset_bits();
    return;
  }
  if (!match_bloom_any_token_set(f, ctx, ch)) {
    bm.reset_bits();
    return;
  }
  visit_values(ctx, ch, bm, [&](strview v) { return in_set(*set, v); });
}

// ---- filterContainsAny (filter_contains_any.go:105-296) ----

static bool match_any_phrase(strview v, const std::vector<std::string>& phrases) {
  for (const auto& ph : phrases) {
    if (match_phrase(v, strview(ph))) return true;
  }
  return false;
}

static void apply_contains_any(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  if (f.values.empty()) {
    bm.reset_bits();
    return;
  }
  for (const auto& v : f.values) {
    if (v.empty()) return;  // empty value matches everything (:110-113)
  }
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_any_phrase(strview(cv), f.values)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!match_any_phrase(strview("", 0), f.values)) bm.reset_bits();
    return;
  }
  if (ch.type == ValueType::Dict) {
    std::vector<uint8_t> enc;
    for (const auto& dv : ch.dict) {
      enc.push_back(match_any_phrase(strview(dv), f.values) ? 1 : 0);
    }
    match_encoded_dict(ctx, ch, bm, enc);
    return;
  }
  if (ch.type == ValueType::Uint8 || ch.type == ValueType::Uint16 ||
      ch.type == ValueType::Uint32 || ch.type == ValueType::Uint64) {
    // uint columns use the exact binary sets (filter_contains_any.go:141-152)
    const auto& set = f.bin_sets[size_t(bin_set_slot(ch.type))];
    if (set.empty()) {
      bm.reset_bits();
      return;
    }
    if (!match_bloom_any_token_set(f, ctx, ch)) {
      bm.reset_bits();
      return;
    }
    visit_values(ctx, ch, bm, [&](strview v) { return in_set(set, v); });
    return;
  }
  // common-token gate + per-phrase token-set survivor filter
  // (matchValuesAnyPhrase, filter_contains_any.go:179-198)
  if (!match_bloom_all(ctx, ch, f.common_hashes)) {
    bm.reset_bits();
    return;
  }
  std::vector<std::string> survivors;
  {
    const auto& words = ctx.bloom(ch);
    for (size_t i = 0; i < f.values.size(); i++) {
      if (bloom_contains_all(words.data(), words.size(), f.set_hashes[i].data(),
                             f.set_hashes[i].size())) {
        survivors.push_back(f.values[i]);
      }
    }
  }
  if (survivors.empty()) {
    bm.reset_bits();
    return;
  }
  visit_values(ctx, ch, bm, [&](strview v) {
    if (ch.type == ValueType::String) return match_any_phrase(v, survivors);
    std::string str = format_value(ch.type, v);
    return match_any_phrase(strview(str), survivors);
  });
}

// ---- filterContainsAll (filter_contains_all.go:123-321) ----

static bool match_all_phrases(strview v, const std::vector<std::string>& phrases) {
  for (const auto& ph : phrases) {
    if (ph.empty()) continue;  // empty phrase matches everything (:310-321)
    if (!match_phrase(v, strview(ph))) return false;
  }
  return true;
}

static void apply_contains_all(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  bool only_empty = f.values.size() == 1 && f.values[0].empty();
  if (f.values.empty() || only_empty) return;  // :124-126

  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_all_phrases(strview(cv), f.values)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!match_all_phrases(strview("", 0), f.values)) bm.reset_bits();
    return;
  }
  if (ch.type == ValueType::Dict) {
    std::vector<uint8_t> enc;
    for (const auto& dv : ch.dict) {
      enc.push_back(match_all_phrases(strview(dv), f.values) ? 1 : 0);
    }
    match_encoded_dict(ctx, ch, bm, enc);
    return;
  }
  if (ch.type == ValueType::Uint8 || ch.type == ValueType::Uint16 ||
      ch.type == ValueType::Uint32 || ch.type == ValueType::Uint64) {
    // matchAllValues (filter_contains_all.go:183-204)
    std::vector<std::string> distinct_nonempty;
    for (const auto& v : f.values) {
      if (!v.empty() &&
          std::find(distinct_nonempty.begin(), distinct_nonempty.end(), v) ==
              distinct_nonempty.end()) {
        distinct_nonempty.push_back(v);
      }
    }
    if (distinct_nonempty.empty()) return;
    const auto& set = f.bin_sets[size_t(bin_set_slot(ch.type))];
    if (distinct_nonempty.size() != 1 || set.size() != 1) {
      bm.reset_bits();
      return;
    }
    if (!match_bloom_all(ctx, ch, f.all_hashes)) {
      bm.reset_bits();
      return;
    }
    strview bin(set[0]);
    visit_values(ctx, ch, bm, [&](strview v) { return v == bin; });
    return;
  }
  if (!match_bloom_all(ctx, ch, f.all_hashes)) {
    bm.reset_bits();
    return;
  }
  visit_values(ctx, ch, bm, [&](strview v) {
    if (ch.type == ValueType::String) return match_all_phrases(v, f.values);
    std::string str = format_value(ch.type, v);
    return match_all_phrases(strview(str), f.values);
  });
}

// ---- filterStringRange (filter_string_range.go:47-230) ----

static bool match_string_range(strview s, const std::string& mn,
                               const std::string& mx) {
  // matchStringRange: s >= min && s < max (plain byte order)
  auto cmp = [](strview a, const std::string& b) {
    int c = memcmp(a.p, b.data(), std::min(a.n, b.size()));
    if (c != 0) return c;
    return a.n < b.size() ? -1 : (a.n > b.size() ? 1 : 0);
  };
  return cmp(s, mn) >= 0 && cmp(s, mx) < 0;
}

static void apply_string_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  if (f.min_s > f.max_s) {
    bm.reset_bits();
    return;
  }
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_string_range(strview(cv), f.min_s, f.max_s)) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!match_string_range(strview("", 0), f.min_s, f.max_s)) bm.reset_bits();
    return;
  }
  // per-type prunes (filter_string_range.go:100-225)
  switch (ch.type) {
    case ValueType::String:
    case ValueType::Dict:
      break;
    case ValueType::Int64:
      if ((f.min_s != "-" && f.min_s > "9") || (f.max_s != "-" && f.max_s < "0")) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Float64:
      if (f.min_s > "9" || f.max_s < "+") {
        bm.reset_bits();
        return;
      }
      break;
    default:
      if (f.min_s > "9" || f.max_s < "0") {
        bm.reset_bits();
        return;
      }
      break;
  }
  if (ch.type == ValueType::Dict) {
    std::vector<uint8_t> enc;
    for (const auto& dv : ch.dict) {
      enc.push_back(match_string_range(strview(dv), f.min_s, f.max_s) ? 1 : 0);
    }
    match_encoded_dict(ctx, ch, bm, enc);
    return;
  }
  visit_values(ctx, ch, bm, [&](strview v) {
    if (ch.type == ValueType::String) {
      return match_string_range(v, f.min_s, f.max_s);
    }
    std::string str = format_value(ch.type, v);
    return match_string_range(strview(str), f.min_s, f.max_s);
  });
}

// ---- filterIPv4Range (filter_ipv4_range.go:99-190) ----

static void apply_ipv4_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  uint32_t mn = uint32_t(f.min_u), mx = uint32_t(f.max_u);
  if (mn > mx) {
    bm.reset_bits();
    return;
  }
  auto match_str = [&](strview v) {
    uint32_t ip;
    if (!try_parse_ipv4(v, &ip)) return false;
    return ip >= mn && ip <= mx;
  };
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_str(strview(cv))) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    bm.reset_bits();
    return;
  }
  switch (ch.type) {
    case ValueType::String:
      visit_values(ctx, ch, bm, match_str);
      return;
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) enc.push_back(match_str(strview(dv)) ? 1 : 0);
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::IPv4: {
      // matchIPv4ByRange (filter_ipv4_range.go:166-181)
      if (ch.min_value > mx || ch.max_value < mn) {
        bm.reset_bits();
        return;
      }
      visit_values(ctx, ch, bm, [&](strview v) {
        uint32_t ip = get_u32be((const uint8_t*)v.p);
        return ip >= mn && ip <= mx;
      });
      return;
    }
    default:
      bm.reset_bits();
      return;
  }
}

// ---- filterLenRange (filter_len_range.go:126-348) ----

static uint64_t rune_count(strview s) {
  uint64_t n = 0;
  for (size_t i = 0; i < s.n; i++) {
    if ((uint8_t(s.p[i]) & 0xC0) != 0x80) n++;
  }
  return n;
}

static void apply_len_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  uint64_t mn = f.min_u, mx = f.max_u;
  if (mn > mx) {
    bm.reset_bits();
    return;
  }
  auto match_len = [&](strview v) {
    uint64_t n = rune_count(v);
    return n >= mn && n <= mx;
  };
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!match_len(strview(cv))) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    if (!match_len(strview("", 0))) bm.reset_bits();
    return;
  }
  // per-type prunes (filter_len_range.go:180-331)
  auto minmax_len_ok = [&]() {
    std::string s2;
    format_uint64(s2, ch.min_value);
    if (mx < s2.size()) return false;
    s2.clear();
    format_uint64(s2, ch.max_value);
    return mn <= s2.size();
  };
  switch (ch.type) {
    case ValueType::String:
      visit_values(ctx, ch, bm, match_len);
      return;
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) enc.push_back(match_len(strview(dv)) ? 1 : 0);
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    case ValueType::Uint8:
      if (mn > 3 || mx == 0 || !minmax_len_ok()) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Uint16:
      if (mn > 5 || mx == 0 || !minmax_len_ok()) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Uint32:
      if (mn > 10 || mx == 0 || !minmax_len_ok()) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Uint64:
      if (mn > 20 || mx == 0 || !minmax_len_ok()) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Int64:
      if (mn > 20 || mx == 0) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::Float64:
      if (mn > 24 || mx == 0) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::IPv4:
      if (mn > 15 || mx < 7) {
        bm.reset_bits();
        return;
      }
      break;
    case ValueType::TimestampISO8601:
      // matchTimestampISO8601ByLenRange (filter_len_range.go:180-185)
      if (mn > 24 || mx < 24) bm.reset_bits();
      return;
    default:
      fail("unknown valueType in len_range filter");
  }
  visit_values(ctx, ch, bm, [&](strview v) {
    std::string str = format_value(ch.type, v);
    return match_len(strview(str));
  });
}

// ---- filterDayRange / filterWeekRange (filter_day_range.go:126-139,
//      filter_week_range.go:128-141) ----

static constexpr int64_t kNsPerDay = 24LL * 3600 * 1000000000;

static int64_t floor_mod(int64_t a, int64_t m) {
  int64_t r = a % m;
  return r;  // Go's % (truncated); day offsets of pre-1970 stamps are negative
}

static void apply_day_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  int64_t start = int64_t(f.min_u), end = int64_t(f.max_u);
  if (start > end) {
    bm.reset_bits();
    return;
  }
  if (start == 0 && end == kNsPerDay - 1) return;
  const auto& ts = ctx.get_timestamps();
  bm.for_each_set_bit([&](uint64_t idx) {
    int64_t off = floor_mod(ts[idx] - f.tz_offset, kNsPerDay);
    return off >= start && off <= end;
  });
}

static void apply_week_range(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  int64_t start = int64_t(f.min_u), end = int64_t(f.max_u);
  if (start > end) {
    bm.reset_bits();
    return;
  }
  if (start == 0 /*Sunday*/ && end == 6 /*Saturday*/) return;
  const auto& ts = ctx.get_timestamps();
  bm.for_each_set_bit([&](uint64_t idx) {
    // Go time.Weekday: days since epoch + 4 (1970-01-01 = Thursday), floor
    int64_t t = ts[idx] - f.tz_offset;
    int64_t days = t / kNsPerDay;
    if (t % kNsPerDay < 0) days--;
    int64_t wd = (days + 4) % 7;
    if (wd < 0) wd += 7;
    return wd >= start && wd <= end;
  });
}

// ---- filterValueType (filter_value_type.go:44-67) ----

static const char* value_type_name(ValueType t) {
  switch (t) {
    case ValueType::String: return "string";
    case ValueType::Dict: return "dict";
    case ValueType::Uint8: return "uint8";
    case ValueType::Uint16: return "uint16";
    case ValueType::Uint32: return "uint32";
    case ValueType::Uint64: return "uint64";
    case ValueType::Int64: return "int64";
    case ValueType::Float64: return "float64";
    case ValueType::IPv4: return "ipv4";
    case ValueType::TimestampISO8601: return "iso8601";
    default: return "unknown";
  }
}

static void apply_value_type(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  std::string name = canonical_field(f.field);
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (f.min_s != "const") bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    bm.reset_bits();
    return;
  }
  if (f.min_s != value_type_name(ch.type)) bm.reset_bits();
}

// ---- filterStreamID (filter_stream_id.go:127-143) ----

static void apply_stream_id(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  if (f.stream_ids.empty()) {
    bm.reset_bits();
    return;
  }
  const StreamID& sid = ctx.bh->stream_id;
  uint64_t tp = uint64_t(sid.account_id) << 32 | sid.project_id;
  for (const auto& id : f.stream_ids) {
    if (id[0] == tp && id[1] == sid.id_hi && id[2] == sid.id_lo) return;
  }
  bm.reset_bits();
}

// ---- AND/OR bloom prefilters (filter_and.go:76-111, filter_or.go:80-115) ----

static bool and_match_bloom(const FilterNode& f, BlockCtx& ctx) {
  for (const auto& ft : f.by_field_tokens) {
    std::string name = canonical_field(ft.field);
    std::string cv = ctx.const_value(name);
    if (!cv.empty()) {
      if (match_string_by_all_tokens(strview(cv), ft.tokens)) continue;
      return false;
    }
    ColumnHeader ch;
    if (!ctx.column_header(name, &ch)) return false;
    if (ch.type == ValueType::Dict) {
      if (match_dict_values_by_all_tokens(ch.dict, ft.tokens)) continue;
      return false;
    }
    if (!match_bloom_all(ctx, ch, ft.hashes)) return false;
  }
  return true;
}

static bool or_match_bloom(const FilterNode& f, BlockCtx& ctx) {
  if (f.by_field_tokens.empty()) return true;
  for (const auto& ft : f.by_field_tokens) {
    std::string name = canonical_field(ft.field);
    std::string cv = ctx.const_value(name);
    if (!cv.empty()) {
      if (match_string_by_all_tokens(strview(cv), ft.tokens)) return true;
      continue;
    }
    ColumnHeader ch;
    if (!ctx.column_header(name, &ch)) continue;
    if (ch.type == ValueType::Dict) {
      if (match_dict_values_by_all_tokens(ch.dict, ft.tokens)) return true;
      continue;
    }
    if (match_bloom_all(ctx, ch, ft.hashes)) return true;
  }
  return false;
}


// ---- filterAnyCasePhrase / filterAnyCasePrefix
//      (filter_any_case_phrase.go:85-158, filter_any_case_prefix.go:90-160) ----

static void apply_phrase(const FilterNode& f, BlockCtx& ctx, Bitmap& bm);
static void apply_prefix(const FilterNode& f, BlockCtx& ctx, Bitmap& bm);

static void apply_any_case(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  const bool is_ph = f.type == FilterNode::AnyCasePhrase;
  std::string name = canonical_field(f.field);
  strview lower(f.min_s);
  auto host_match = [&](strview v) {
    return is_ph ? match_any_case_phrase(v, lower)
                 : match_any_case_prefix(v, lower);
  };
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    if (!host_match(strview(cv))) bm.reset_bits();
    return;
  }
  ColumnHeader ch;
  if (!ctx.column_header(name, &ch)) {
    // phrase matches missing columns only when empty; prefix never does
    if (!is_ph || lower.n > 0) bm.reset_bits();
    return;
  }
  switch (ch.type) {
    case ValueType::String:
      visit_values(ctx, ch, bm, host_match);
      return;
    case ValueType::Dict: {
      std::vector<uint8_t> enc;
      for (const auto& dv : ch.dict) enc.push_back(host_match(strview(dv)) ? 1 : 0);
      match_encoded_dict(ctx, ch, bm, enc);
      return;
    }
    default: {
      // numeric/ip/iso: same matchers as phrase/prefix with the lowercase
      // (iso: uppercase) pattern (filter_any_case_phrase.go:116-137)
      const bool iso = ch.type == ValueType::TimestampISO8601;
      FilterNode tmp;
      tmp.type = is_ph ? FilterNode::Phrase : FilterNode::Prefix;
      tmp.field = f.field;
      tmp.phrase = iso ? f.max_s : f.min_s;
      tmp.token_hashes = iso ? f.all_hashes : f.token_hashes;
      if (is_ph) {
        apply_phrase(tmp, ctx, bm);
      } else {
        apply_prefix(tmp, ctx, bm);
      }
      return;
    }
  }
}


// ---- filterEqField / filterLeField (filter_eq_field.go:122-220,
//      filter_le_field.go:155-282) ----

namespace {
struct FieldView {
  int kind = 0;  // 0 column, 1 const, 2 missing
  std::string cval;
  ColumnHeader ch;
  const StringsBlockDec* dec = nullptr;
};
}  // namespace

static FieldView make_field_view(BlockCtx& ctx, const std::string& name) {
  FieldView fv;
  std::string cv = ctx.const_value(name);
  if (!cv.empty()) {
    fv.kind = 1;
    fv.cval = cv;
    return fv;
  }
  if (!ctx.column_header(name, &fv.ch)) {
    fv.kind = 2;
    return fv;
  }
  fv.dec = &ctx.values(fv.ch);
  return fv;
}

// blockResult.getValues row semantics (block_result.go:306-478): const value
// for const columns, "" for missing, decoded string form for typed columns
static std::string fv_row_str(const FieldView& fv, uint64_t idx) {
  if (fv.kind == 1) return fv.cval;
  if (fv.kind == 2) return std::string();
  strview v = fv.dec->row(idx);
  if (fv.ch.type == ValueType::String) return std::string(v.p, v.n);
  if (fv.ch.type == ValueType::Dict) return fv.ch.dict[uint8_t(v.p[0])];
  return format_value(fv.ch.type, v);
}

static void apply_eq_field(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  const std::string name = canonical_field(f.field);
  const std::string other = canonical_field(f.min_s);
  if (name == other) return;
  FieldView a = make_field_view(ctx, name);
  FieldView b = make_field_view(ctx, other);
  if (a.kind == 1 && b.kind == 1) {
    if (a.cval != b.cval) bm.reset_bits();
    return;
  }
  if (a.kind == 2 && b.kind == 2) return;  // both missing: "" == ""
  const bool same_type = a.kind == 0 && b.kind == 0 && a.ch.type == b.ch.type;
  if (!same_type) {
    // applyFilterString (filter_eq_field.go:182-204)
    bm.for_each_set_bit([&](uint64_t idx) {
      return fv_row_str(a, idx) == fv_row_str(b, idx);
    });
    return;
  }
  if (a.ch.type == ValueType::String) {
    bm.for_each_set_bit([&](uint64_t idx) {
      return a.dec->row(idx) == b.dec->row(idx);
    });
    return;
  }
  if (a.ch.type == ValueType::Dict) {
    bm.for_each_set_bit([&](uint64_t idx) {
      return a.ch.dict[uint8_t(a.dec->row(idx).p[0])] ==
             b.ch.dict[uint8_t(b.dec->row(idx).p[0])];
    });
    return;
  }
  bm.for_each_set_bit([&](uint64_t idx) {
    return a.dec->row(idx) == b.dec->row(idx);  // encoded binary equality
  });
}

static void apply_le_field(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  const std::string name = canonical_field(f.field);
  const std::string other = canonical_field(f.min_s);
  const bool excl = f.min_u != 0;
  if (name == other) {
    if (excl) bm.reset_bits();
    return;
  }
  FieldView a = make_field_view(ctx, name);
  FieldView b = make_field_view(ctx, other);
  if (a.kind == 1 && b.kind == 1) {
    if (!le_values_string(strview(a.cval), strview(b.cval), excl)) {
      bm.reset_bits();
    }
    return;
  }
  if (a.kind == 2 && b.kind == 2) {
    if (excl) bm.reset_bits();
    return;
  }
  const bool same_type = a.kind == 0 && b.kind == 0 && a.ch.type == b.ch.type;
  if (!same_type || a.ch.type == ValueType::String) {
    bm.for_each_set_bit([&](uint64_t idx) {
      std::string sa = fv_row_str(a, idx);
      std::string sb = fv_row_str(b, idx);
      return le_values_string(strview(sa), strview(sb), excl);
    });
    return;
  }
  switch (a.ch.type) {
    case ValueType::Dict:
      bm.for_each_set_bit([&](uint64_t idx) {
        const std::string& va = a.ch.dict[uint8_t(a.dec->row(idx).p[0])];
        const std::string& vb = b.ch.dict[uint8_t(b.dec->row(idx).p[0])];
        return le_values_string(strview(va), strview(vb), excl);
      });
      return;
    case ValueType::Int64:
      bm.for_each_set_bit([&](uint64_t idx) {
        int64_t va = get_i64be_zigzag((const uint8_t*)a.dec->row(idx).p);
        int64_t vb = get_i64be_zigzag((const uint8_t*)b.dec->row(idx).p);
        return excl ? va < vb : va <= vb;
      });
      return;
    case ValueType::Float64:
      bm.for_each_set_bit([&](uint64_t idx) {
        uint64_t ua = get_u64be((const uint8_t*)a.dec->row(idx).p);
        uint64_t ub = get_u64be((const uint8_t*)b.dec->row(idx).p);
        double va, vb;
        memcpy(&va, &ua, 8);
        memcpy(&vb, &ub, 8);
        return excl ? va < vb : va <= vb;
      });
      return;
    default:
      // uint/ipv4/iso8601: leValuesString over the ENCODED bytes
      // (filter_le_field.go:257-263 quirk, mirrored bit-for-bit)
      bm.for_each_set_bit([&](uint64_t idx) {
        return le_values_string(a.dec->row(idx), b.dec->row(idx), excl);
      });
      return;
  }
}

void apply_filter(const FilterNode& f, BlockCtx& ctx, Bitmap& bm) {
  switch (f.type) {
    case FilterNode::Phrase:
      apply_phrase(f, ctx, bm);
      return;
    case FilterNode::Exact:
      apply_exact(f, ctx, bm);
      return;
    case FilterNode::Regexp:
      apply_regexp(f, ctx, bm);
      return;
    case FilterNode::Prefix:
      apply_prefix(f, ctx, bm);
      return;
    case FilterNode::ExactPrefix:
      apply_exact_prefix(f, ctx, bm);
      return;
    case FilterNode::Sequence:
      apply_sequence(f, ctx, bm);
      return;
    case FilterNode::In:
      apply_in(f, ctx, bm);
      return;
    case FilterNode::ContainsAny:
      apply_contains_any(f, ctx, bm);
      return;
    case FilterNode::ContainsAll:
      apply_contains_all(f, ctx, bm);
      return;
    case FilterNode::StringRange:
      apply_string_range(f, ctx, bm);
      return;
    case FilterNode::IPv4Range:
      apply_ipv4_range(f, ctx, bm);
      return;
    case FilterNode::LenRange:
      apply_len_range(f, ctx, bm);
      return;
    case FilterNode::DayRange:
      apply_day_range(f, ctx, bm);
      return;
    case FilterNode::WeekRange:
      apply_week_range(f, ctx, bm);
      return;
    case FilterNode::ValueTypeFilter:
      apply_value_type(f, ctx, bm);
      return;
    case FilterNode::StreamIdFilter:
      apply_stream_id(f, ctx, bm);
      return;
    case FilterNode::AnyCasePhrase:
    case FilterNode::AnyCasePrefix:
      apply_any_case(f, ctx, bm);
      return;
    case FilterNode::EqField:
      apply_eq_field(f, ctx, bm);
      return;
    case FilterNode::LeField:
      apply_le_field(f, ctx, bm);
      return;
    case FilterNode::Time:
      apply_time(f, ctx, bm);
      return;
    case FilterNode::Range:
      apply_range(f, ctx, bm);
      return;
    case FilterNode::Noop:
      return;
    case FilterNode::And: {
      // filter_and.go:58-74
      if (!and_match_bloom(f, ctx)) {
        bm.reset_bits();
        return;
      }
      for (const auto& c : f.children) {
        apply_filter(c, ctx, bm);
        if (bm.is_zero()) return;
      }
      return;
    }
    case FilterNode::Or: {
      // filter_or.go:55-78: andNot double-buffer
      if (!or_match_bloom(f, ctx)) {
        bm.reset_bits();
        return;
      }
      Bitmap bm_result, bm_tmp;
      bm_result.bits_len = bm.bits_len;
      bm_result.a = bm.a;
      for (const auto& c : f.children) {
        bm_tmp.bits_len = bm_result.bits_len;
        bm_tmp.a = bm_result.a;
        apply_filter(c, ctx, bm_tmp);
        bm_result.and_not(bm_tmp);
        if (bm_result.is_zero()) return;
      }
      bm.and_not(bm_result);
      return;
    }
    case FilterNode::Not: {
      // filter_not.go:38-46
      Bitmap bm_tmp;
      bm_tmp.bits_len = bm.bits_len;
      bm_tmp.a = bm.a;
      apply_filter(f.children[0], ctx, bm_tmp);
      bm.and_not(bm_tmp);
      return;
    }
  }
}

void search_block(const FilterNode& f, const PartReader& pr, const BlockHeader& bh,
                  Bitmap& bm) {
  BlockCtx ctx(&pr, &bh);
  bm.init_ones(bh.rows_count);
  apply_filter(f, ctx, bm);
}

}  // namespace oracle
}  // namespace vl
