#include "filter.h"

#include <algorithm>
#include <cmath>
#include <cstdlib>
#include <map>

#include "bloom.h"
#include "match.h"
#include "tokenizer.h"

namespace vl {

// ---- minimal JSON parser (objects/arrays/strings/numbers/bools) ----
namespace {

struct JValue {
  enum Kind { Obj, Arr, Str, Num, Bool, Null } kind = Null;
  std::map<std::string, JValue> obj;
  std::vector<JValue> arr;
  std::string str;
  double num = 0;
  bool b = false;
};

struct JParser {
  const char* p;
  const char* end;
  [[noreturn]] void err(const char* msg) { fail(std::string("filter json: ") + msg); }
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) p++;
  }
  JValue parse() {
    ws();
    if (p >= end) err("unexpected end");
    switch (*p) {
      case '{': {
        p++;
        JValue v;
        v.kind = JValue::Obj;
        ws();
        if (p < end && *p == '}') {
          p++;
          return v;
        }
        for (;;) {
          ws();
          if (p >= end || *p != '"') err("expected key");
          std::string key = parse_string();
          ws();
          if (p >= end || *p != ':') err("expected :");
          p++;
          v.obj[key] = parse();
          ws();
          if (p < end && *p == ',') {
            p++;
            continue;
          }
          if (p < end && *p == '}') {
            p++;
            return v;
          }
          err("expected , or }");
        }
      }
      case '[': {
        p++;
        JValue v;
        v.kind = JValue::Arr;
        ws();
        if (p < end && *p == ']') {
          p++;
          return v;
        }
        for (;;) {
          v.arr.push_back(parse());
          ws();
          if (p < end && *p == ',') {
            p++;
            continue;
          }
          if (p < end && *p == ']') {
            p++;
            return v;
          }
          err("expected , or ]");
        }
      }
      case '"': {
        JValue v;
        v.kind = JValue::Str;
        v.str = parse_string();
        return v;
      }
      case 't':
        if (end - p >= 4 && memcmp(p, "true", 4) == 0) {
          p += 4;
          JValue v;
          v.kind = JValue::Bool;
          v.b = true;
          return v;
        }
        err("bad token");
      case 'f':
        if (end - p >= 5 && memcmp(p, "false", 5) == 0) {
          p += 5;
          JValue v;
          v.kind = JValue::Bool;
          return v;
        }
        err("bad token");
      case 'n':
        if (end - p >= 4 && memcmp(p, "null", 4) == 0) {
          p += 4;
          return JValue();
        }
        err("bad token");
      default: {
        char* endp;
        double d = strtod(p, &endp);
        if (endp == p) err("bad number");
        p = endp;
        JValue v;
        v.kind = JValue::Num;
        v.num = d;
        return v;
      }
    }
  }
  std::string parse_string() {
    p++;  // opening quote
    std::string s;
    while (p < end && *p != '"') {
      if (*p == '\\') {
        p++;
        if (p >= end) err("bad escape");
        switch (*p) {
          case 'n': s += '\n'; break;
          case 't': s += '\t'; break;
          case 'r': s += '\r'; break;
          case 'b': s += '\b'; break;
          case 'f': s += '\f'; break;
          case '"': s += '"'; break;
          case '\\': s += '\\'; break;
          case '/': s += '/'; break;
          case 'u': {
            if (end - p < 5) err("bad \\u");
            unsigned cp = 0;
            for (int i = 1; i <= 4; i++) {
              char c = p[i];
              cp <<= 4;
              if (c >= '0' && c <= '9') cp |= unsigned(c - '0');
              else if (c >= 'a' && c <= 'f') cp |= unsigned(c - 'a' + 10);
              else if (c >= 'A' && c <= 'F') cp |= unsigned(c - 'A' + 10);
              else err("bad \\u");
            }
            p += 4;
            // encode UTF-8 (surrogates unsupported)
            if (cp < 0x80) {
              s += char(cp);
            } else if (cp < 0x800) {
              s += char(0xC0 | (cp >> 6));
              s += char(0x80 | (cp & 0x3F));
            } else {
              s += char(0xE0 | (cp >> 12));
              s += char(0x80 | ((cp >> 6) & 0x3F));
              s += char(0x80 | (cp & 0x3F));
            }
            break;
          }
          default:
            err("bad escape");
        }
        p++;
      } else {
        s += *p++;
      }
    }
    if (p >= end) err("unterminated string");
    p++;
    return s;
  }
};

const JValue& jget(const JValue& o, const char* key) {
  auto it = o.obj.find(key);
  if (it == o.obj.end()) fail(std::string("filter json: missing field ") + key);
  return it->second;
}

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
  JParser jp{json.data(), json.data() + json.size()};
  JValue v = jp.parse();
  return build(v);
}

}  // namespace vl
