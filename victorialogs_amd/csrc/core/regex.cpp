#include "regex.h"

#include <cstring>

#include "match.h"
#include "tokenizer.h"

namespace vl {

namespace {

// Minimal RE2-subset AST.  Group nodes are preserved so GetLiterals sees the
// same literal boundaries as Go's parse tree (regex.go:101-124).
struct RNode {
  enum Kind { Lit, Dot, Concat, Alt, Star, Plus, Quest, Group, Empty } kind;
  std::string lit;
  std::vector<RNode> subs;
};

struct Parser {
  const std::string& s;
  size_t pos = 0;
  explicit Parser(const std::string& s_) : s(s_) {}

  [[noreturn]] void err(const std::string& msg) {
    fail("regex: " + msg + " in " + s);
  }

  bool eof() const { return pos >= s.size(); }
  char peek() const { return s[pos]; }

  RNode parse_alt() {
    std::vector<RNode> alts;
    alts.push_back(parse_concat());
    while (!eof() && peek() == '|') {
      pos++;
      alts.push_back(parse_concat());
    }
    if (alts.size() == 1) return std::move(alts[0]);
    RNode n;
    n.kind = RNode::Alt;
    n.subs = std::move(alts);
    return n;
  }

  RNode parse_concat() {
    std::vector<RNode> items;
    while (!eof() && peek() != '|' && peek() != ')') {
      RNode atom = parse_atom();
      // postfix quantifiers bind to the last atom
      while (!eof()) {
        char c = peek();
        if (c == '*' || c == '+' || c == '?') {
          pos++;
          RNode q;
          q.kind = c == '*' ? RNode::Star : c == '+' ? RNode::Plus : RNode::Quest;
          q.subs.push_back(std::move(atom));
          atom = std::move(q);
        } else if (c == '{') {
          err("{m,n} repetition is not supported");
        } else {
          break;
        }
      }
      items.push_back(std::move(atom));
    }
    if (items.empty()) {
      RNode n;
      n.kind = RNode::Empty;
      return n;
    }
    if (items.size() == 1) return std::move(items[0]);
    RNode n;
    n.kind = RNode::Concat;
    n.subs = std::move(items);
    return n;
  }

  RNode parse_atom() {
    char c = s[pos];
    switch (c) {
      case '(': {
        pos++;
        if (pos + 1 < s.size() && s[pos] == '?') {
          if (s[pos + 1] == ':') {
            pos += 2;  // non-capturing group
          } else {
            err("(?...) constructs other than (?:...) are not supported");
          }
        }
        RNode inner = parse_alt();
        if (eof() || peek() != ')') err("missing )");
        pos++;
        RNode g;
        g.kind = RNode::Group;
        g.subs.push_back(std::move(inner));
        return g;
      }
      case '[':
        return parse_char_class();
      case '.': {
        pos++;
        RNode n;
        n.kind = RNode::Dot;
        return n;
      }
      case '^':
      case '$':
        err("anchors are not supported (LogsQL regex filters are unanchored)");
      case '*':
      case '+':
      case '?':
        err("dangling quantifier");
      case '\\':
        return parse_escape();
      default: {
        pos++;
        RNode n;
        n.kind = RNode::Lit;
        n.lit.push_back(c);
        return n;
      }
    }
  }

  RNode parse_escape() {
    pos++;  // backslash
    if (eof()) err("trailing backslash");
    char c = s[pos++];
    RNode n;
    n.kind = RNode::Lit;
    switch (c) {
      case 'n': n.lit.push_back('\n'); return n;
      case 't': n.lit.push_back('\t'); return n;
      case 'r': n.lit.push_back('\r'); return n;
      case 'd': case 'D': case 'w': case 'W': case 's': case 'S':
      case 'b': case 'B': case 'p': case 'P':
        err(std::string("escape class \\") + c + " is not supported");
      default:
        if ((c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') ||
            (c >= '0' && c <= '9')) {
          err(std::string("escape \\") + c + " is not supported");
        }
        n.lit.push_back(c);  // escaped metacharacter
        return n;
    }
  }

  RNode parse_char_class() {
    pos++;  // '['
    if (!eof() && peek() == '^') err("negated char classes are not supported");
    std::vector<char> chars;
    bool first = true;
    while (!eof() && (peek() != ']' || first)) {
      char c = s[pos++];
      if (c == '\\') {
        if (eof()) err("trailing backslash in class");
        c = s[pos++];
        if (c == 'n') c = '\n';
        else if (c == 't') c = '\t';
        else if (c == 'r') c = '\r';
        else if ((c >= 'a' && c <= 'z' && c != 'n' && c != 't' && c != 'r') &&
                 strchr("dwsb", c)) {
          err("escape classes inside [...] are not supported");
        }
      }
      if (!eof() && peek() == '-' && pos + 1 < s.size() && s[pos + 1] != ']') {
        pos++;  // '-'
        char hi = s[pos++];
        if (hi == '\\') {
          if (eof()) err("trailing backslash in class");
          hi = s[pos++];
        }
        if (hi < c) err("invalid char class range");
        for (char x = c;; x++) {
          chars.push_back(x);
          if (x == hi) break;
          if (chars.size() > 100) err("char class too large for or-values");
        }
      } else {
        chars.push_back(c);
      }
      first = false;
      if (chars.size() > 100) err("char class too large for or-values");
    }
    if (eof()) err("missing ]");
    pos++;  // ']'
    // Expand as alternation of single chars (regexutil.go:93-107)
    RNode alt;
    alt.kind = RNode::Alt;
    for (char c : chars) {
      RNode l;
      l.kind = RNode::Lit;
      l.lit.push_back(c);
      alt.subs.push_back(std::move(l));
    }
    if (alt.subs.size() == 1) return std::move(alt.subs[0]);
    return alt;
  }
};

bool is_dot_star(const RNode& n) { return n.kind == RNode::Star && n.subs[0].kind == RNode::Dot; }
bool is_dot_plus(const RNode& n) { return n.kind == RNode::Plus && n.subs[0].kind == RNode::Dot; }

// Flatten groups/concats and merge adjacent literals (like sre.Simplify()'s
// effect on the shapes we support).
RNode simplify(const RNode& n) {
  switch (n.kind) {
    case RNode::Group:
      return simplify(n.subs[0]);
    case RNode::Concat: {
      std::vector<RNode> items;
      for (const auto& sub : n.subs) {
        RNode s = simplify(sub);
        if (s.kind == RNode::Empty) continue;
        if (s.kind == RNode::Concat) {
          for (auto& x : s.subs) items.push_back(std::move(x));
        } else {
          items.push_back(std::move(s));
        }
      }
      // merge adjacent literals
      std::vector<RNode> merged;
      for (auto& it : items) {
        if (it.kind == RNode::Lit && !merged.empty() &&
            merged.back().kind == RNode::Lit) {
          merged.back().lit += it.lit;
        } else {
          merged.push_back(std::move(it));
        }
      }
      if (merged.empty()) {
        RNode e;
        e.kind = RNode::Empty;
        return e;
      }
      if (merged.size() == 1) return std::move(merged[0]);
      RNode c;
      c.kind = RNode::Concat;
      c.subs = std::move(merged);
      return c;
    }
    case RNode::Alt: {
      RNode a;
      a.kind = RNode::Alt;
      for (const auto& sub : n.subs) a.subs.push_back(simplify(sub));
      return a;
    }
    case RNode::Star:
    case RNode::Plus:
    case RNode::Quest: {
      RNode q;
      q.kind = n.kind;
      q.subs.push_back(simplify(n.subs[0]));
      return q;
    }
    default:
      return n;
  }
}

// getOrValues (regexutil.go:67-139); returns empty vector if not or-able.
constexpr size_t kMaxOrValues = 100;
bool get_or_values(const RNode& n, std::vector<std::string>& out) {
  switch (n.kind) {
    case RNode::Lit:
      out.push_back(n.lit);
      return out.size() <= kMaxOrValues;
    case RNode::Empty:
      out.push_back("");
      return out.size() <= kMaxOrValues;
    case RNode::Group:
      return get_or_values(n.subs[0], out);
    case RNode::Alt: {
      for (const auto& sub : n.subs) {
        std::vector<std::string> vs;
        if (!get_or_values(sub, vs)) return false;
        if (vs.empty()) return false;
        for (auto& v : vs) out.push_back(std::move(v));
        if (out.size() > kMaxOrValues) return false;
      }
      return true;
    }
    case RNode::Concat: {
      if (n.subs.empty()) {
        out.push_back("");
        return true;
      }
      std::vector<std::string> prefixes;
      if (!get_or_values(n.subs[0], prefixes) || prefixes.empty()) return false;
      if (n.subs.size() == 1) {
        out = std::move(prefixes);
        return true;
      }
      RNode rest;
      rest.kind = RNode::Concat;
      rest.subs.assign(n.subs.begin() + 1, n.subs.end());
      std::vector<std::string> suffixes;
      if (!get_or_values(rest, suffixes) || suffixes.empty()) return false;
      if (prefixes.size() * suffixes.size() > kMaxOrValues) return false;
      for (const auto& p : prefixes) {
        for (const auto& q : suffixes) out.push_back(p + q);
      }
      return true;
    }
    default:
      return false;
  }
}

// GetLiterals (regex.go:101-124) on the RAW parse tree.
void collect_literals(const RNode& raw, std::vector<std::string>& out) {
  const RNode* n = &raw;
  while (n->kind == RNode::Group) n = &n->subs[0];
  if (n->kind == RNode::Lit) {
    out.push_back(n->lit);
    return;
  }
  if (n->kind != RNode::Concat) return;
  for (const auto& sub : n->subs) {
    const RNode* s = &sub;
    while (s->kind == RNode::Group) s = &s->subs[0];
    if (s->kind == RNode::Lit) out.push_back(s->lit);
  }
}

}  // namespace

RegexProg regex_compile(const std::string& expr) {
  Parser p(expr);
  RNode raw = p.parse_alt();
  if (!p.eof()) p.err("unexpected )");

  RegexProg re;
  re.expr = expr;

  // Bloom literals from the raw tree (filter_regexp.go:44-51 applies
  // skipFirstLastToken + tokenize later in filter compilation).
  collect_literals(raw, re.literals);

  RNode sre = simplify(raw);

  // Extract the literal prefix (simplifyRegex, regexutil.go:199-233).
  std::vector<RNode> items;
  if (sre.kind == RNode::Concat) {
    items = std::move(sre.subs);
  } else if (sre.kind != RNode::Empty) {
    items.push_back(std::move(sre));
  }
  if (!items.empty() && items[0].kind == RNode::Lit) {
    re.prefix = items[0].lit;
    items.erase(items.begin());
  }
  // SimplifyRegex (regexutil.go:157-185): drop leading .* when there is no
  // literal prefix; drop trailing .* always.
  if (re.prefix.empty()) {
    while (!items.empty() && is_dot_star(items[0])) items.erase(items.begin());
  }
  while (!items.empty() && is_dot_star(items.back())) items.pop_back();

  RNode suffix;
  if (items.empty()) {
    suffix.kind = RNode::Empty;
  } else if (items.size() == 1) {
    suffix = std::move(items[0]);
  } else {
    suffix.kind = RNode::Concat;
    suffix.subs = std::move(items);
  }

  // Classification (regex.go:49-82)
  std::vector<std::string> ors;
  bool or_ok = get_or_values(suffix, ors);
  re.is_only_prefix = or_ok && ors.size() == 1 && ors[0].empty();
  re.is_suffix_dot_star = is_dot_star(suffix);
  re.is_suffix_dot_plus = is_dot_plus(suffix);
  if (suffix.kind == RNode::Concat && suffix.subs.size() == 3 &&
      suffix.subs[1].kind == RNode::Lit) {
    if (is_dot_star(suffix.subs[0]) && is_dot_star(suffix.subs[2])) {
      re.substr_dot_star = suffix.subs[1].lit;
    }
    if (is_dot_plus(suffix.subs[0]) && is_dot_plus(suffix.subs[2])) {
      re.substr_dot_plus = suffix.subs[1].lit;
    }
  }
  // prefix + ".*lit" (e.g. "abc.*def"): Go's suffixRe here is "^(?:.*lit)",
  // whose unanchored-tail semantics in matchStringWithPrefix reduce to
  // strings.Contains(tail, lit) — identical to the substrDotStar branch
  // (regex.go:177-180), so classify it there.
  if (!re.prefix.empty() && suffix.kind == RNode::Concat &&
      suffix.subs.size() == 2 && is_dot_star(suffix.subs[0]) &&
      suffix.subs[1].kind == RNode::Lit) {
    re.substr_dot_star = suffix.subs[1].lit;
  }
  if (or_ok && !ors.empty()) {
    re.or_values = std::move(ors);
    re.has_or_values = true;
  }

  if (!re.is_only_prefix && !re.is_suffix_dot_star && !re.is_suffix_dot_plus &&
      re.substr_dot_star.empty() && re.substr_dot_plus.empty() && !re.has_or_values) {
    fail("regex: pattern \"" + expr +
         "\" falls outside the supported fast-path classes (literal / "
         "alternation / prefix.* / prefix.+ / .+substr.+ / or-values); the "
         "general NFA fallback is planned for round 2");
  }
  return re;
}

// strings.Contains
static bool contains(strview s, const std::string& sub) {
  if (sub.empty()) return true;
  if (sub.size() > s.n) return false;
  return memmem(s.p, s.n, sub.data(), sub.size()) != nullptr;
}
static long index_of(strview s, const std::string& sub) {
  if (sub.empty()) return 0;
  if (sub.size() > s.n) return -1;
  const char* f = (const char*)memmem(s.p, s.n, sub.data(), sub.size());
  return f ? long(f - s.p) : -1;
}

// matchStringNoPrefix (regex.go:131-160)
static bool match_no_prefix(const RegexProg& re, strview s) {
  if (re.is_suffix_dot_star) return true;
  if (re.is_suffix_dot_plus) return s.n > 0;
  if (!re.substr_dot_star.empty()) return contains(s, re.substr_dot_star);
  if (!re.substr_dot_plus.empty()) {
    long n = index_of(s, re.substr_dot_plus);
    return n > 0 && size_t(n) + re.substr_dot_plus.size() < s.n;
  }
  // or_values fast path (the general suffixRe fallback is rejected at compile)
  for (const auto& v : re.or_values) {
    if (contains(s, v)) return true;
  }
  return false;
}

// matchStringWithPrefix (regex.go:162-212)
static bool match_with_prefix(const RegexProg& re, strview s) {
  long n = index_of(s, re.prefix);
  if (n < 0) return false;
  strview snext(s.p + n + 1, s.n - size_t(n) - 1);
  strview t(s.p + n + re.prefix.size(), s.n - size_t(n) - re.prefix.size());

  if (re.is_suffix_dot_star) return true;
  if (re.is_suffix_dot_plus) return t.n > 0;
  if (!re.substr_dot_star.empty()) return contains(t, re.substr_dot_star);
  if (!re.substr_dot_plus.empty()) {
    long k = index_of(t, re.substr_dot_plus);
    return k > 0 && size_t(k) + re.substr_dot_plus.size() < t.n;
  }
  for (;;) {
    for (const auto& v : re.or_values) {
      // strings.HasPrefix(t, v)
      if (t.n >= v.size() && memcmp(t.p, v.data(), v.size()) == 0) return true;
    }
    s = snext;
    n = index_of(s, re.prefix);
    if (n < 0) return false;
    snext = strview(s.p + n + 1, s.n - size_t(n) - 1);
    t = strview(s.p + n + re.prefix.size(), s.n - size_t(n) - re.prefix.size());
  }
}

bool regex_match(const RegexProg& re, strview s) {
  // Regex.MatchString (regex.go:86-98)
  if (re.is_only_prefix) {
    if (re.prefix.empty()) return true;
    return contains(s, re.prefix);
  }
  if (re.prefix.empty()) return match_no_prefix(re, s);
  return match_with_prefix(re, s);
}

}  // namespace vl
