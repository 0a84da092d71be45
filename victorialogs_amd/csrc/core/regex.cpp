#include "regex.h"

#include <cstring>

#include "match.h"
#include "tokenizer.h"

namespace vl {

namespace {

// Minimal RE2-subset AST.  Group nodes are preserved so GetLiterals sees the
// same literal boundaries as Go's parse tree (regex.go:101-124).
struct RNode {
  enum Kind { Lit, Dot, Concat, Alt, Star, Plus, Quest, Group, Empty, Class } kind;
  std::string lit;
  std::vector<RNode> subs;
  // Class: 256-bit byte set (ASCII bits 0..127 used) + "all non-ASCII runes"
  uint8_t cls[32] = {0};
  bool cls_nonascii = false;
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
          // {m}, {m,}, {m,n}: expanded by cloning (Go regexp/syntax
          // semantics; a '{' that is not a valid repetition is a literal)
          size_t save = pos;
          pos++;
          long m = -1, nrep = -2;  // nrep -2: no comma; -1: open-ended
          long v = 0;
          bool any = false;
          while (!eof() && peek() >= '0' && peek() <= '9') {
            v = v * 10 + (peek() - '0');
            if (v > 100000) break;
            any = true;
            pos++;
          }
          if (any) m = v;
          if (!eof() && peek() == ',') {
            pos++;
            v = 0;
            bool any2 = false;
            while (!eof() && peek() >= '0' && peek() <= '9') {
              v = v * 10 + (peek() - '0');
              if (v > 100000) break;
              any2 = true;
              pos++;
            }
            nrep = any2 ? v : -1;
          }
          if (m < 0 || eof() || peek() != '}') {
            pos = save;  // literal '{'
            break;
          }
          pos++;
          if (nrep == -2) nrep = m;
          if (m > 1000 || nrep > 1000) err("invalid repeat count");
          if (nrep >= 0 && nrep < m) err("invalid repeat count");
          const long max_copies = nrep < 0 ? m + 1 : nrep;
          if (max_copies > 64) {
            err("repetition too large for the 64-position NFA");
          }
          // X{m,n} = X^m (X (X ... )?)? ; X{m,} = X^m X*
          RNode expanded;
          expanded.kind = RNode::Concat;
          for (long k = 0; k < m; k++) expanded.subs.push_back(atom);
          if (nrep < 0) {
            RNode st;
            st.kind = RNode::Star;
            st.subs.push_back(atom);
            expanded.subs.push_back(std::move(st));
          } else if (nrep > m) {
            RNode opt;
            opt.kind = RNode::Empty;
            for (long k = nrep; k > m; k--) {
              RNode q;
              q.kind = RNode::Quest;
              if (opt.kind == RNode::Empty) {
                q.subs.push_back(atom);
              } else {
                RNode cc;
                cc.kind = RNode::Concat;
                cc.subs.push_back(atom);
                cc.subs.push_back(std::move(opt));
                q.subs.push_back(std::move(cc));
              }
              opt = std::move(q);
            }
            expanded.subs.push_back(std::move(opt));
          }
          if (expanded.subs.size() == 1) {
            atom = std::move(expanded.subs[0]);
          } else if (expanded.subs.empty()) {
            expanded.kind = RNode::Empty;
            atom = std::move(expanded);
          } else {
            atom = std::move(expanded);
          }
          if (!eof() && (peek() == '*' || peek() == '+' || peek() == '?' ||
                         peek() == '{')) {
            err("nested repetition operator");
          }
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
        // consume a full UTF-8 rune so postfix quantifiers bind to the rune
        // (Go binds ? * + to the preceding rune, not byte)
        RNode n;
        n.kind = RNode::Lit;
        uint8_t c0 = uint8_t(c);
        int len = c0 < 0x80 ? 1 : (c0 & 0xE0) == 0xC0 ? 2
                  : (c0 & 0xF0) == 0xE0 ? 3 : (c0 & 0xF8) == 0xF0 ? 4 : 1;
        for (int i = 0; i < len && !eof(); i++) n.lit.push_back(s[pos++]);
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
      case 'd': case 'D': case 'w': case 'W': case 's': case 'S': {
        // Go regexp/syntax Perl classes: \d=[0-9], \w=[0-9A-Za-z_],
        // \s=[\t\n\f\r ]; upper-case = negation (includes non-ASCII runes)
        RNode cn;
        cn.kind = RNode::Class;
        auto set = [&](uint8_t ch) { cn.cls[ch >> 3] |= uint8_t(1) << (ch & 7); };
        char base = char(c | 0x20);
        if (base == 'd') {
          for (uint8_t ch = '0'; ch <= '9'; ch++) set(ch);
        } else if (base == 'w') {
          for (uint8_t ch = '0'; ch <= '9'; ch++) set(ch);
          for (uint8_t ch = 'a'; ch <= 'z'; ch++) set(ch);
          for (uint8_t ch = 'A'; ch <= 'Z'; ch++) set(ch);
          set('_');
        } else {
          set('\t'); set('\n'); set('\f'); set('\r'); set(' ');
        }
        if (c >= 'A' && c <= 'Z') {
          // negate within ASCII; all non-ASCII runes are included
          for (int i = 0; i < 16; i++) cn.cls[i] = uint8_t(~cn.cls[i]);
          for (int i = 16; i < 32; i++) cn.cls[i] = 0;
          cn.cls_nonascii = true;
        }
        return cn;
      }
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
    bool negated = false;
    if (!eof() && peek() == '^') {
      negated = true;
      pos++;
    }
    uint8_t bits[32] = {0};
    auto set = [&](uint8_t ch) { bits[ch >> 3] |= uint8_t(1) << (ch & 7); };
    bool first = true;
    while (!eof() && (peek() != ']' || first)) {
      uint8_t c = uint8_t(s[pos++]);
      if (c >= 0x80) err("non-ASCII characters in [...] are not supported");
      if (c == '\\') {
        if (eof()) err("trailing backslash in class");
        char e = s[pos++];
        if (e == 'n') c = '\n';
        else if (e == 't') c = '\t';
        else if (e == 'r') c = '\r';
        else if (e == 'f') c = '\f';
        else if (e == 'd' || e == 'w' || e == 's') {
          if (e == 'd') {
            for (uint8_t ch = '0'; ch <= '9'; ch++) set(ch);
          } else if (e == 'w') {
            for (uint8_t ch = '0'; ch <= '9'; ch++) set(ch);
            for (uint8_t ch = 'a'; ch <= 'z'; ch++) set(ch);
            for (uint8_t ch = 'A'; ch <= 'Z'; ch++) set(ch);
            set('_');
          } else {
            set('\t'); set('\n'); set('\f'); set('\r'); set(' ');
          }
          first = false;
          continue;
        } else if ((e >= 'a' && e <= 'z') || (e >= 'A' && e <= 'Z') ||
                   (e >= '0' && e <= '9')) {
          err(std::string("escape \\") + e + " inside [...] is not supported");
        } else {
          c = uint8_t(e);
        }
      }
      if (!eof() && peek() == '-' && pos + 1 < s.size() && s[pos + 1] != ']') {
        pos++;  // '-'
        uint8_t hi = uint8_t(s[pos++]);
        if (hi == '\\') {
          if (eof()) err("trailing backslash in class");
          hi = uint8_t(s[pos++]);
        }
        if (hi >= 0x80) err("non-ASCII characters in [...] are not supported");
        if (hi < c) err("invalid char class range");
        for (int x = c; x <= int(hi); x++) set(uint8_t(x));
      } else {
        set(c);
      }
      first = false;
    }
    if (eof()) err("missing ]");
    pos++;  // ']'

    if (negated) {
      RNode cn;
      cn.kind = RNode::Class;
      for (int i = 0; i < 16; i++) cn.cls[i] = uint8_t(~bits[i]);
      cn.cls_nonascii = true;  // Go: [^...] matches any rune outside the set
      return cn;
    }
    // expand small positive classes as alternation of single chars so the
    // or-values fast path classifies them like Go (regexutil.go:93-107)
    int count = 0;
    for (int b = 0; b < 128; b++) count += (bits[b >> 3] >> (b & 7)) & 1;
    if (count == 0) err("empty char class");
    if (count <= 100) {
      RNode alt;
      alt.kind = RNode::Alt;
      for (int b = 0; b < 128; b++) {
        if ((bits[b >> 3] >> (b & 7)) & 1) {
          RNode l;
          l.kind = RNode::Lit;
          l.lit.push_back(char(b));
          alt.subs.push_back(std::move(l));
        }
      }
      if (alt.subs.size() == 1) return std::move(alt.subs[0]);
      return alt;
    }
    RNode cn;
    cn.kind = RNode::Class;
    memcpy(cn.cls, bits, 32);
    return cn;
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


// ---- Glushkov position automaton for the general class ----
// Positions (atoms) are byte-consuming; first/last/follow sets are u64 masks
// (<= 64 positions).  Multi-byte runes (Dot, negated classes) use a relaxed
// UTF-8 shape: [00-7F] | [C0-DF][80-BF] | [E0-EF][80-BF]{2} | [F0-F4][80-BF]{3}
// -- identical to Go on valid UTF-8 input (DESIGN.md notes the invalid-UTF-8
// divergence).

struct GAtom {
  uint8_t set[32];
};

struct GBuild {
  std::vector<GAtom> atoms;
  std::vector<uint64_t> follow;
  [[noreturn]] void overflow(const std::string& expr) {
    fail("regex: NFA fallback for \"" + expr +
         "\" needs more than 64 positions; simplify the pattern");
  }
};

struct GInfo {
  bool nullable;
  uint64_t first, last;
};

static int g_add_atom(GBuild& b, const uint8_t* set, const std::string& expr) {
  if (b.atoms.size() >= 64) b.overflow(expr);
  GAtom a;
  memcpy(a.set, set, 32);
  b.atoms.push_back(a);
  b.follow.push_back(0);
  return int(b.atoms.size()) - 1;
}

static void g_range_set(uint8_t* set, int lo, int hi) {
  for (int c = lo; c <= hi; c++) set[c >> 3] |= uint8_t(1) << (c & 7);
}

static GInfo g_cat(GBuild& b, GInfo x, GInfo y) {
  // follow: last(x) -> first(y)
  uint64_t m = x.last;
  while (m) {
    int i = __builtin_ctzll(m);
    m &= m - 1;
    b.follow[i] |= y.first;
  }
  GInfo r;
  r.nullable = x.nullable && y.nullable;
  r.first = x.first | (x.nullable ? y.first : 0);
  r.last = y.last | (y.nullable ? x.last : 0);
  return r;
}

static GInfo g_alt(GInfo x, GInfo y) {
  return GInfo{x.nullable || y.nullable, x.first | y.first, x.last | y.last};
}

static void g_loop(GBuild& b, const GInfo& x) {
  uint64_t m = x.last;
  while (m) {
    int i = __builtin_ctzll(m);
    m &= m - 1;
    b.follow[i] |= x.first;
  }
}

// relaxed UTF-8 multi-byte rune
static GInfo g_multibyte(GBuild& b, const std::string& expr) {
  uint8_t cont[32] = {0}, l2[32] = {0}, l3[32] = {0}, l4[32] = {0};
  g_range_set(cont, 0x80, 0xBF);
  g_range_set(l2, 0xC0, 0xDF);
  g_range_set(l3, 0xE0, 0xEF);
  g_range_set(l4, 0xF0, 0xF4);
  auto seq = [&](const uint8_t* lead, int ncont) {
    int a = g_add_atom(b, lead, expr);
    GInfo r{false, uint64_t(1) << a, uint64_t(1) << a};
    for (int i = 0; i < ncont; i++) {
      int c = g_add_atom(b, cont, expr);
      r = g_cat(b, r, GInfo{false, uint64_t(1) << c, uint64_t(1) << c});
    }
    return r;
  };
  GInfo r = seq(l2, 1);
  r = g_alt(r, seq(l3, 2));
  r = g_alt(r, seq(l4, 3));
  return r;
}

// Re-compress alternations of single-byte literals/classes into ONE byte-set
// position before the Glushkov build: parse expands small positive classes
// into Alt-of-chars for the or-values classification, which would otherwise
// cost one NFA position per character ("[a-z]{3}" = 78 positions instead
// of 3).
static RNode compress_classes(const RNode& n) {
  RNode out;
  out.kind = n.kind;
  out.lit = n.lit;
  memcpy(out.cls, n.cls, sizeof(out.cls));
  out.cls_nonascii = n.cls_nonascii;
  out.subs.reserve(n.subs.size());
  for (const auto& sub : n.subs) out.subs.push_back(compress_classes(sub));
  if (out.kind != RNode::Alt) return out;
  uint8_t set[32] = {0};
  for (const auto& sub : out.subs) {
    if (sub.kind == RNode::Lit && sub.lit.size() == 1 &&
        uint8_t(sub.lit[0]) < 0x80) {
      uint8_t c = uint8_t(sub.lit[0]);
      set[c >> 3] |= uint8_t(1) << (c & 7);
    } else if (sub.kind == RNode::Class && !sub.cls_nonascii) {
      for (int i = 0; i < 32; i++) set[i] |= sub.cls[i];
    } else {
      return out;  // not a pure single-byte alternation
    }
  }
  RNode cn;
  cn.kind = RNode::Class;
  memcpy(cn.cls, set, 32);
  cn.cls_nonascii = false;
  return cn;
}

static GInfo g_build(GBuild& b, const RNode& n, const std::string& expr) {
  switch (n.kind) {
    case RNode::Empty:
      return GInfo{true, 0, 0};
    case RNode::Group:
      return g_build(b, n.subs[0], expr);
    case RNode::Lit: {
      GInfo r{true, 0, 0};
      for (unsigned char c : n.lit) {
        uint8_t set[32] = {0};
        set[c >> 3] = uint8_t(1) << (c & 7);
        int a = g_add_atom(b, set, expr);
        r = g_cat(b, r, GInfo{false, uint64_t(1) << a, uint64_t(1) << a});
      }
      return r;
    }
    case RNode::Dot: {
      uint8_t ascii[32] = {0};
      g_range_set(ascii, 0x00, 0x7F);  // (?s) DotNL: '.' matches any rune
      int a = g_add_atom(b, ascii, expr);
      GInfo r{false, uint64_t(1) << a, uint64_t(1) << a};
      return g_alt(r, g_multibyte(b, expr));
    }
    case RNode::Class: {
      int a = g_add_atom(b, n.cls, expr);
      GInfo r{false, uint64_t(1) << a, uint64_t(1) << a};
      if (n.cls_nonascii) r = g_alt(r, g_multibyte(b, expr));
      return r;
    }
    case RNode::Concat: {
      GInfo r{true, 0, 0};
      for (const auto& sub : n.subs) r = g_cat(b, r, g_build(b, sub, expr));
      return r;
    }
    case RNode::Alt: {
      GInfo r = g_build(b, n.subs[0], expr);
      for (size_t i = 1; i < n.subs.size(); i++) {
        r = g_alt(r, g_build(b, n.subs[i], expr));
      }
      return r;
    }
    case RNode::Star: {
      GInfo x = g_build(b, n.subs[0], expr);
      g_loop(b, x);
      return GInfo{true, x.first, x.last};
    }
    case RNode::Plus: {
      GInfo x = g_build(b, n.subs[0], expr);
      g_loop(b, x);
      return x;
    }
    case RNode::Quest: {
      GInfo x = g_build(b, n.subs[0], expr);
      return GInfo{true, x.first, x.last};
    }
  }
  fail("regex: unreachable node kind");
}

// blob: u16 nstates, u16 pad, u32 pad, u64 first, u64 last,
//       u64 follow[nstates], u64 byte_table[256]
static bytes g_serialize(const GBuild& b, const GInfo& root,
                         bool a_start = false, bool a_end = false,
                         bool nullable = false) {
  bytes out;
  uint16_t n = uint16_t(b.atoms.size());
  out.push_back(uint8_t(n));
  out.push_back(uint8_t(n >> 8));
  // byte 2: anchor flags (1 = ^ anchored, 2 = $ anchored, 4 = nullable root)
  out.push_back(uint8_t((a_start ? 1 : 0) | (a_end ? 2 : 0) |
                        (nullable ? 4 : 0)));
  out.resize(8, 0);
  auto put64 = [&](uint64_t v) {
    for (int i = 0; i < 8; i++) out.push_back(uint8_t(v >> (8 * i)));
  };
  put64(root.first);
  put64(root.last);
  for (uint16_t i = 0; i < n; i++) put64(b.follow[i]);
  for (int c = 0; c < 256; c++) {
    uint64_t m = 0;
    for (uint16_t i = 0; i < n; i++) {
      if ((b.atoms[i].set[c >> 3] >> (c & 7)) & 1) m |= uint64_t(1) << i;
    }
    put64(m);
  }
  return out;
}

}  // namespace

RegexProg regex_compile(const std::string& expr) {
  // Top-level anchors: a leading '^' / trailing unescaped '$' anchor the
  // whole pattern (Go regexp semantics without multiline).  Anchors inside
  // the pattern (alternation branches, groups) remain unsupported and error
  // in the parser.
  std::string body = expr;
  bool a_start = false, a_end = false;
  if (!body.empty() && body[0] == '^') {
    a_start = true;
    body.erase(body.begin());
  }
  if (!body.empty() && body.back() == '$') {
    size_t bs = 0;
    while (bs + 1 < body.size() && body[body.size() - 2 - bs] == '\\') bs++;
    if (bs % 2 == 0) {
      a_end = true;
      body.pop_back();
    }
  }
  Parser p(body);
  RNode raw = p.parse_alt();
  if (!p.eof()) p.err("unexpected )");
  if ((a_start || a_end) && raw.kind == RNode::Alt) {
    fail("regex: anchors with top-level alternation are not supported in " +
         expr);
  }

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

  if (a_start || a_end ||
      (!re.is_only_prefix && !re.is_suffix_dot_star && !re.is_suffix_dot_plus &&
       re.substr_dot_star.empty() && re.substr_dot_plus.empty() &&
       !re.has_or_values)) {
    // General class: Glushkov NFA over the WHOLE original pattern, matched
    // unanchored -- equivalent to Go's prefix-retry + anchored suffixRe loop
    // (regex.go:186-211) for pure regexes.  Anchored patterns always take
    // this path (the fast-path classes assume unanchored semantics).
    GBuild b;
    RNode packed = compress_classes(raw);
    GInfo root = g_build(b, packed, expr);
    re.has_nfa = true;
    // nullable root matches "": unanchored or half-anchored => every string
    // has an empty prefix/suffix match; both-anchored only matches ""
    re.always_true = root.nullable && !(a_start && a_end);
    re.is_only_prefix = false;
    re.is_suffix_dot_star = false;
    re.is_suffix_dot_plus = false;
    re.substr_dot_star.clear();
    re.substr_dot_plus.clear();
    re.has_or_values = false;
    re.or_values.clear();
    re.nfa_blob = g_serialize(b, root, a_start, a_end, root.nullable);
    re.prefix.clear();  // NFA matches the whole pattern; ignore the prefix
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

bool nfa_match(const uint8_t* blob, strview s) {
  uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t anchors = blob[2];
  const bool a_start = anchors & 1, a_end = anchors & 2;
  auto rd64 = [&](size_t off) {
    uint64_t v;
    memcpy(&v, blob + off, 8);
    return v;
  };
  const uint64_t first = rd64(8);
  const uint64_t last = rd64(16);
  const uint8_t* follow = blob + 24;
  const uint8_t* table = blob + 24 + size_t(n) * 8;
  if (s.n == 0) return (anchors & 4) != 0;  // nullable root matches ""
  uint64_t active = 0;
  for (size_t i = 0; i < s.n; i++) {
    // '^' anchored: new matches may start only at offset 0
    uint64_t targets = (a_start && i > 0) ? 0 : first;
    uint64_t m = active;
    while (m) {
      int x = __builtin_ctzll(m);
      m &= m - 1;
      uint64_t f;
      memcpy(&f, follow + size_t(x) * 8, 8);
      targets |= f;
    }
    uint64_t tb;
    memcpy(&tb, table + size_t(uint8_t(s.p[i])) * 8, 8);
    const uint64_t entered = targets & tb;
    if (!a_end && (entered & last)) return true;
    active = entered;
  }
  // '$' anchored: accept only with a final position active at string end
  return a_end && (active & last) != 0;
}

bool regex_match(const RegexProg& re, strview s) {
  // Regex.MatchString (regex.go:86-98)
  if (re.has_nfa) {
    if (re.always_true) return true;
    return nfa_match(re.nfa_blob.data(), s);
  }
  if (re.is_only_prefix) {
    if (re.prefix.empty()) return true;
    return contains(s, re.prefix);
  }
  if (re.prefix.empty()) return match_no_prefix(re, s);
  return match_with_prefix(re, s);
}

}  // namespace vl
