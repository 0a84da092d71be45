// Filter tree restating the lib/logstorage filter interface (filter.go:8-20)
// for the hot-path filters: filterPhrase, filterExact, filterRegexp,
// filterAnd, filterOr, filterNot, filterTime, filterRange.
//
// Trees are built programmatically from JSON (the same shapes
// filter_test.go:34-59 builds in Go), compiled once per query: per-leaf
// tokens + bloom probe hashes (filter_phrase.go:52-55, bloomfilter.go:126-144)
// and the AND/OR cross-filter common tokens (filter_and.go:118-187,
// filter_or.go:122-193).
#pragma once

#include <array>
#include <memory>
#include <string>
#include <vector>

#include "regex.h"
#include "vl_base.h"

namespace vl {

struct FieldTokens {
  std::string field;  // canonical column name ("_msg" per getCanonicalColumnName)
  std::vector<std::string> tokens;
  std::vector<uint64_t> hashes;  // 6 probe hashes per token
};

struct FilterNode {
  enum Type {
    Phrase, Exact, Regexp, And, Or, Not, Time, Range, Noop,
    Prefix, ExactPrefix, Sequence,
    In, ContainsAny, ContainsAll, StringRange, IPv4Range, LenRange,
    DayRange, WeekRange, ValueTypeFilter, StreamIdFilter,
    AnyCasePhrase, AnyCasePrefix, EqField, LeField
  } type;

  std::string field;   // phrase/exact/regexp/range (as written in the query)
  std::string phrase;  // Phrase: phrase; Exact: value; Prefix/ExactPrefix: prefix
  std::vector<std::string> phrases;  // Sequence: non-empty phrases, in order
  RegexProg re;        // Regexp
  double min_f = 0, max_f = 0;    // Range
  int64_t min_ts = 0, max_ts = 0;  // Time
  std::vector<FilterNode> children;  // And/Or (n), Not (1)
  std::vector<std::string> values;   // In/ContainsAny/ContainsAll
  std::string min_s, max_s;          // StringRange; ValueTypeFilter: min_s=type
                                     // AnyCase*: min_s=lowercase, max_s=uppercase
                                     // EqField/LeField: min_s=otherFieldName
  uint64_t min_u = 0, max_u = 0;     // IPv4Range/LenRange; Day/WeekRange: start/end
                                     // LeField: min_u=excludeEqualValues
  int64_t tz_offset = 0;             // Day/WeekRange offset (nsecs)
  std::vector<std::array<uint64_t, 3>> stream_ids;  // {acct<<32|proj, hi, lo}

  // In/ContainsAny: inValues token structures (in_values.go:94-125)
  std::vector<uint64_t> common_hashes;             // probe hashes, common tokens
  std::vector<std::vector<uint64_t>> set_hashes;   // per-value probe hashes
  // ContainsAll: tokensHashesAll (in_values.go:94-102)
  // AnyCase*: uppercase token hashes (filter_any_case_phrase.go:53-62)
  std::vector<uint64_t> all_hashes;
  // In: per-type binary value sets, sorted (in_values.go:141-315); index by
  // width slot: [0]=u8 [1]=u16 [2]=u32 [3]=u64 [4]=i64 [5]=f64 [6]=ipv4 [7]=iso
  std::vector<std::vector<std::string>> bin_sets;

  // compiled:
  std::vector<std::string> tokens;
  std::vector<uint64_t> token_hashes;           // leaf bloom gate
  std::vector<FieldTokens> by_field_tokens;     // And/Or prefilter
};

// Parses the JSON filter tree and precomputes tokens/hashes.
// Throws vl::Error on malformed JSON or unsupported constructs.
FilterNode compile_filter(const std::string& json);

// getCanonicalFieldName (log_rows.go:508-513): "_msg" -> "" for column lookup.
inline std::string canonical_field(const std::string& name) {
  return name == "_msg" ? std::string() : name;
}

}  // namespace vl
