#include "filter.h"

#include <algorithm>
#include <cstring>
#include <cmath>
#include <cstdlib>
#include <map>

#include "bloom.h"
#include "json.h"
#include "match.h"
#include "unicode_case.h"
#include "values.h"
#include "tokenizer.h"

namespace vl {

namespace {


FilterNode build(const JValue& v);

std::vector<uint64_t> probe_hashes(const std::vector<std::string>& tokens) {
  std::vector<uint64_t> hashes;
  hashes.reserve(tokens.size() * kBloomHashesCount);
  for (const auto& t : tokens) append_token_hashes(hashes, strview(t));
  return hashes;
}

// getCommonTokensForAndFilters (filter_and.go:122-187).  `from_or` toggles the
// OR variant (filter_or.go:126-193): per-field token sets must cover EVERY
// child and are intersected.
std::vector<FieldTokens> common_tokens(const std::vector<FilterNode>& children,
                                       bool is_or) {
  struct Entry {
    std::vector<std::vector<std::string>> sets;  // OR: one set per child
    std::vector<std::string> merged;             // AND: concatenated
  };
  std::map<std::string, Entry> m;
  std::vector<std::string> field_order;

  auto merge = [&](const std::string& field, const std::vector<std::string>& tokens) {
    if (tokens.empty()) return;
    std::string f = field.empty() ? "_msg" : field;  // getCanonicalColumnName
    if (m.find(f) == m.end()) field_order.push_back(f);
    Entry& e = m[f];
    if (is_or) {
      e.sets.push_back(tokens);
    } else {
      e.merged.insert(e.merged.end(), tokens.begin(), tokens.end());
    }
  };

  for (const auto& c : children) {
    switch (c.type) {
      case FilterNode::Phrase:
      case FilterNode::Exact:
      case FilterNode::Regexp:
      case FilterNode::Prefix:
      case FilterNode::ExactPrefix:
      case FilterNode::Sequence:
        merge(c.field, c.tokens);
        break;
      case FilterNode::Or:
        if (!is_or) {
          for (const auto& bft : c.by_field_tokens) merge(bft.field, bft.tokens);
        } else {
          return {};  // OR of OR: not token-extractable (filter_or.go:167-170)
        }
        break;
      case FilterNode::And:
        if (is_or) {
          for (const auto& bft : c.by_field_tokens) merge(bft.field, bft.tokens);
        }
        // nested AND inside AND is not in the reference's switch; skip
        break;
      default:
        if (is_or) return {};  // filter_or.go:167-170 default case
        break;
    }
  }

  std::vector<FieldTokens> out;
  for (const auto& f : field_order) {
    Entry& e = m[f];
    std::vector<std::string> tokens;
    if (is_or) {
      // common tokens must be present in every OR child (filter_or.go:173-190)
      if (e.sets.size() != children.size()) continue;
      tokens = e.sets[0];
      for (size_t i = 1; i < e.sets.size() && !tokens.empty(); i++) {
        std::vector<std::string> kept;
        for (const auto& t : tokens) {
          if (std::find(e.sets[i].begin(), e.sets[i].end(), t) != e.sets[i].end()) {
            kept.push_back(t);
          }
        }
        tokens = std::move(kept);
      }
      if (tokens.empty()) continue;
    } else {
      // dedup preserving order (filter_and.go:166-186)
      std::vector<std::string> seen;
      for (const auto& t : e.merged) {
        if (std::find(seen.begin(), seen.end(), t) == seen.end()) seen.push_back(t);
      }
      tokens = std::move(seen);
    }
    FieldTokens ft;
    ft.field = f;
    ft.hashes = probe_hashes(tokens);
    ft.tokens = std::move(tokens);
    out.push_back(std::move(ft));
  }
  return out;
}

FilterNode build(const JValue& v) {
  if (v.kind != JValue::Obj) fail("filter json: node must be an object");
  const std::string& type = jget(v, "type").str;
  FilterNode n;
  if (type == "phrase") {
    n.type = FilterNode::Phrase;
    n.field = jget(v, "field").str;
    n.phrase = jget(v, "phrase").str;
    // filterPhrase.initTokens (filter_phrase.go:52-55)
    n.tokens = tokenize_strings({n.phrase});
    n.token_hashes = probe_hashes(n.tokens);
  } else if (type == "exact") {
    n.type = FilterNode::Exact;
    n.field = jget(v, "field").str;
    n.phrase = jget(v, "value").str;
    // filterExact.initTokens (filter_exact.go:44-47)
    n.tokens = tokenize_strings({n.phrase});
    n.token_hashes = probe_hashes(n.tokens);
  } else if (type == "regexp") {
    n.type = FilterNode::Regexp;
    n.field = jget(v, "field").str;
    n.re = regex_compile(jget(v, "re").str);
    // filterRegexp.initTokens (filter_regexp.go:44-51)
    std::vector<std::string> lits;
    for (const auto& lit : n.re.literals) lits.push_back(skip_first_last_token(lit));
    n.tokens = tokenize_strings(lits);
    n.token_hashes = probe_hashes(n.tokens);
  } else if (type == "prefix" || type == "exact_prefix") {
    n.type = type == "prefix" ? FilterNode::Prefix : FilterNode::ExactPrefix;
    n.field = jget(v, "field").str;
    n.phrase = jget(v, "prefix").str;
    // filterPrefix/filterExactPrefix initTokens: getTokensSkipLast
    // (filter_prefix.go:50-53, filter_exact_prefix.go:43-46)
    n.tokens = get_tokens_skip_last(n.phrase);
    n.token_hashes = probe_hashes(n.tokens);
  } else if (type == "sequence") {
    n.type = FilterNode::Sequence;
    n.field = jget(v, "field").str;
    for (const auto& pj : jget(v, "phrases").arr) {
      // getNonEmptyPhrases (filter_sequence.go:57-66)
      if (!pj.str.empty()) n.phrases.push_back(pj.str);
    }
    // filterSequence.initTokens (filter_sequence.go:51-55)
    n.tokens = tokenize_strings(n.phrases);
    n.token_hashes = probe_hashes(n.tokens);
  } else if (type == "and" || type == "or") {
    n.type = type == "and" ? FilterNode::And : FilterNode::Or;
    for (const auto& c : jget(v, "filters").arr) n.children.push_back(build(c));
    n.by_field_tokens = common_tokens(n.children, n.type == FilterNode::Or);
  } else if (type == "not") {
    n.type = FilterNode::Not;
    n.children.push_back(build(jget(v, "filter")));
  } else if (type == "time") {
    n.type = FilterNode::Time;
    n.min_ts = jget(v, "min").as_i64();
    n.max_ts = jget(v, "max").as_i64();
  } else if (type == "range") {
    n.type = FilterNode::Range;
    n.field = jget(v, "field").str;
    n.min_f = jget(v, "min").num;
    n.max_f = jget(v, "max").num;
  } else if (type == "in" || type == "contains_any" || type == "contains_all") {
    n.type = type == "in" ? FilterNode::In
             : type == "contains_any" ? FilterNode::ContainsAny
                                      : FilterNode::ContainsAll;
    n.field = jget(v, "field").str;
    for (const auto& vj : jget(v, "values").arr) n.values.push_back(vj.str);
    if (n.type == FilterNode::ContainsAll) {
      // getTokensHashesAll (in_values.go:94-102): tokenizeHashes over all
      // values, then appendHashesHashes
      std::vector<strview> vs;
      for (const auto& s2 : n.values) vs.push_back(strview(s2));
      for (uint64_t th : tokenize_hashes(vs)) {
        append_hash_hashes(n.all_hashes, th);
      }
    } else {
      // getTokensHashesAny (in_values.go:104-125)
      std::vector<std::string> common;
      std::vector<std::vector<std::string>> token_sets;
      get_common_tokens_and_sets(n.values, &common, &token_sets);
      n.common_hashes = probe_hashes(common);
      for (const auto& ts : token_sets) {
        n.set_hashes.push_back(probe_hashes(ts));
      }
    }
    if (n.type != FilterNode::Noop) {  // In/ContainsAny/ContainsAll
      // per-type binary value sets (in_values.go:141-315), sorted for the
      // device binary search; ContainsAny/ContainsAll use the uint slots
      // (filter_contains_any.go:141-152, filter_contains_all.go:183-204)
      n.bin_sets.resize(8);
      for (const auto& s2 : n.values) {
        strview sv(s2);
        uint64_t u;
        if (try_parse_uint64(sv, &u)) {
          bytes b;
          if (u < (1 << 8)) {
            b.push_back(uint8_t(u));
            n.bin_sets[0].emplace_back((const char*)b.data(), b.size());
          }
          b.clear();
          if (u < (1 << 16)) {
            put_u16be(b, uint16_t(u));
            n.bin_sets[1].emplace_back((const char*)b.data(), b.size());
          }
          b.clear();
          if (u < (uint64_t(1) << 32)) {
            put_u32be(b, uint32_t(u));
            n.bin_sets[2].emplace_back((const char*)b.data(), b.size());
          }
          b.clear();
          put_u64be(b, u);
          n.bin_sets[3].emplace_back((const char*)b.data(), b.size());
        }
        int64_t i64v;
        if (try_parse_int64(sv, &i64v)) {
          bytes b;
          put_i64be_zigzag(b, i64v);
          n.bin_sets[4].emplace_back((const char*)b.data(), b.size());
        }
        double f;
        if (try_parse_float64_exact(sv, &f)) {
          bytes b;
          uint64_t fu;
          memcpy(&fu, &f, 8);
          put_u64be(b, fu);
          n.bin_sets[5].emplace_back((const char*)b.data(), b.size());
        }
        uint32_t ip;
        if (try_parse_ipv4(sv, &ip)) {
          bytes b;
          put_u32be(b, ip);
          n.bin_sets[6].emplace_back((const char*)b.data(), b.size());
        }
        int64_t ts;
        if (try_parse_timestamp_iso8601(sv, &ts)) {
          bytes b;
          put_u64be(b, uint64_t(ts));
          n.bin_sets[7].emplace_back((const char*)b.data(), b.size());
        }
      }
      for (auto& bs2 : n.bin_sets) {
        std::sort(bs2.begin(), bs2.end());
        bs2.erase(std::unique(bs2.begin(), bs2.end()), bs2.end());
      }
    }
  } else if (type == "string_range") {
    n.type = FilterNode::StringRange;
    n.field = jget(v, "field").str;
    n.min_s = jget(v, "min").str;
    n.max_s = jget(v, "max").str;
  } else if (type == "ipv4_range") {
    n.type = FilterNode::IPv4Range;
    n.field = jget(v, "field").str;
    n.min_u = uint64_t(jget(v, "min").num);
    n.max_u = uint64_t(jget(v, "max").num);
  } else if (type == "len_range") {
    n.type = FilterNode::LenRange;
    n.field = jget(v, "field").str;
    n.min_u = uint64_t(jget(v, "min").num);
    n.max_u = uint64_t(jget(v, "max").num);
  } else if (type == "day_range" || type == "week_range") {
    n.type = type == "day_range" ? FilterNode::DayRange : FilterNode::WeekRange;
    n.min_u = uint64_t(jget(v, "start").as_i64());
    n.max_u = uint64_t(jget(v, "end").as_i64());
    auto it = v.obj.find("offset");
    if (it != v.obj.end()) n.tz_offset = it->second.as_i64();
  } else if (type == "value_type") {
    n.type = FilterNode::ValueTypeFilter;
    n.field = jget(v, "field").str;
    n.min_s = jget(v, "value_type").str;
  } else if (type == "stream_id") {
    n.type = FilterNode::StreamIdFilter;
    for (const auto& sj : jget(v, "ids").arr) {
      uint64_t acct = uint64_t(jget(sj, "account").num);
      uint64_t proj = uint64_t(jget(sj, "project").num);
      // hi/lo as decimal strings to keep full u64 precision through JSON
      uint64_t hi = strtoull(jget(sj, "hi").str.c_str(), nullptr, 10);
      uint64_t lo = strtoull(jget(sj, "lo").str.c_str(), nullptr, 10);
      n.stream_ids.push_back({acct << 32 | proj, hi, lo});
    }
  } else if (type == "any_case_phrase" || type == "any_case_prefix") {
    // filterAnyCasePhrase / filterAnyCasePrefix
    // (filter_any_case_phrase.go:19-62, filter_any_case_prefix.go:21-66)
    bool is_phrase = type == "any_case_phrase";
    n.type = is_phrase ? FilterNode::AnyCasePhrase : FilterNode::AnyCasePrefix;
    n.field = jget(v, "field").str;
    n.phrase = jget(v, is_phrase ? "phrase" : "prefix").str;
    n.min_s = to_lower_str(strview(n.phrase));
    n.max_s = to_upper_str(strview(n.phrase));
    std::vector<std::string> toks =
        is_phrase ? tokenize_strings({n.phrase}) : get_tokens_skip_last(n.phrase);
    n.tokens = toks;
    n.token_hashes = probe_hashes(toks);
    std::vector<std::string> toks_up;
    for (const auto& t : toks) toks_up.push_back(to_upper_str(strview(t)));
    n.all_hashes = probe_hashes(toks_up);
  } else if (type == "eq_field" || type == "le_field") {
    // filterEqField / filterLeField (filter_eq_field.go:15-21,
    // filter_le_field.go:16-24)
    n.type = type == "eq_field" ? FilterNode::EqField : FilterNode::LeField;
    n.field = jget(v, "field").str;
    n.min_s = jget(v, "other_field").str;
    auto it = v.obj.find("exclude_equal");
    if (it != v.obj.end() && it->second.kind == JValue::Bool) {
      n.min_u = it->second.b ? 1 : 0;
    }
  } else if (type == "noop") {
    n.type = FilterNode::Noop;
  } else {
    fail("filter json: unsupported filter type \"" + type + "\"");
  }
  return n;
}

}  // namespace

FilterNode compile_filter(const std::string& json) {
  return build(json_parse(json));
}

}  // namespace vl
