#include "filter.h"

#include <algorithm>
#include <cmath>
#include <cstdlib>
#include <map>

#include "bloom.h"
#include "json.h"
#include "match.h"
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
    n.min_ts = int64_t(jget(v, "min").num);
    n.max_ts = int64_t(jget(v, "max").num);
  } else if (type == "range") {
    n.type = FilterNode::Range;
    n.field = jget(v, "field").str;
    n.min_f = jget(v, "min").num;
    n.max_f = jget(v, "max").num;
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
