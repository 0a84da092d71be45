#include "regex.h"

#include <cstring>

#include "match.h"
#include "tokenizer.h"
#include "unicode_case.h"

namespace vl {

namespace {

// Minimal RE2-subset AST.  Group nodes are preserved so GetLiterals sees the
// same literal boundaries as Go's parse tree (regex.go:101-124).
struct RNode {
  enum Kind {
    Lit, Dot, Concat, Alt, Star, Plus, Quest, Group, Empty, Class, Assert
  } kind;
  std::string lit;
  std::vector<RNode> subs;
  // Class: 256-bit byte set (ASCII bits 0..127 used) + "all non-ASCII runes"
  uint8_t cls[32] = {0};
  bool cls_nonascii = false;
  // Assert: 1 = \b (word boundary), 2 = \B (non-boundary); zero-width,
  // checked between the previous and next byte (RE2 ASCII \w semantics)
  uint8_t assert_kind = 0;
};

// Case-fold orbits for (?i): closure over the simple lower/upper mappings
// plus the Unicode CaseFolding special orbits Go's unicode.SimpleFold walks
// (unicode.CaseOrbit — K/k/KELVIN, S/s/LONG S, the Greek symbol letters,
// ANGSTROM, MICRO, iota subscript, S-with-dot).  Simple-pair letters and
// ASCII are exact; exotic multi-member orbits outside this table fall back
// to the lower/upper closure.
static const uint32_t kFoldOrbits[][4] = {
    {0x004B, 0x006B, 0x212A, 0},       // K k KELVIN SIGN
    {0x0053, 0x0073, 0x017F, 0},       // S s LONG S
    {0x00B5, 0x039C, 0x03BC, 0},       // MICRO, CAP MU, mu
    {0x00C5, 0x00E5, 0x212B, 0},       // Å å ANGSTROM
    {0x0345, 0x0399, 0x03B9, 0x1FBE},  // iota subscript orbit
    {0x0392, 0x03B2, 0x03D0, 0},       // BETA beta beta-symbol
    {0x0395, 0x03B5, 0x03F5, 0},       // EPSILON epsilon lunate
    {0x0398, 0x03B8, 0x03D1, 0x03F4},  // THETA orbit
    {0x039A, 0x03BA, 0x03F0, 0},       // KAPPA kappa kappa-symbol
    {0x03A0, 0x03C0, 0x03D6, 0},       // PI pi omega-pi
    {0x03A1, 0x03C1, 0x03F1, 0},       // RHO rho rho-symbol
    {0x03A6, 0x03C6, 0x03D5, 0},       // PHI phi phi-symbol
    {0x03A9, 0x03C9, 0x2126, 0},       // OMEGA omega OHM
    {0x1E60, 0x1E61, 0x1E9B, 0},       // S-dot orbit
    {0x03A3, 0x03C2, 0x03C3, 0},       // SIGMA final-sigma sigma
    {0x01C4, 0x01C5, 0x01C6, 0},       // DŽ Dž dž (title-case digraphs)
    {0x01C7, 0x01C8, 0x01C9, 0},       // LJ Lj lj
    {0x01CA, 0x01CB, 0x01CC, 0},       // NJ Nj nj
    {0x01F1, 0x01F2, 0x01F3, 0},       // DZ Dz dz
};

static void case_orbit(uint32_t r, std::vector<uint32_t>& out) {
  out.clear();
  out.push_back(r);
  // closure over simple lower/upper (catches sigma/dz-digraph style orbits)
  for (size_t i = 0; i < out.size() && out.size() < 8; i++) {
    uint32_t lo = to_lower_rune(out[i]), up = to_upper_rune(out[i]);
    for (uint32_t x : {lo, up}) {
      bool seen = false;
      for (uint32_t y : out) seen |= (y == x);
      if (!seen) out.push_back(x);
    }
  }
  for (const auto& orbit : kFoldOrbits) {
    bool hit = false;
    for (uint32_t m : orbit) hit |= (m != 0 && m == r);
    if (!hit) {
      for (uint32_t y : out) {
        for (uint32_t m : orbit) hit |= (m != 0 && m == y);
      }
    }
    if (hit) {
      for (uint32_t m : orbit) {
        if (m == 0) continue;
        bool seen = false;
        for (uint32_t y : out) seen |= (y == m);
        if (!seen) out.push_back(m);
      }
    }
  }
}

static void utf8_encode(uint32_t r, std::string& out) {
  if (r < 0x80) {
    out.push_back(char(r));
  } else if (r < 0x800) {
    out.push_back(char(0xC0 | (r >> 6)));
    out.push_back(char(0x80 | (r & 0x3F)));
  } else if (r < 0x10000) {
    out.push_back(char(0xE0 | (r >> 12)));
    out.push_back(char(0x80 | ((r >> 6) & 0x3F)));
    out.push_back(char(0x80 | (r & 0x3F)));
  } else {
    out.push_back(char(0xF0 | (r >> 18)));
    out.push_back(char(0x80 | ((r >> 12) & 0x3F)));
    out.push_back(char(0x80 | ((r >> 6) & 0x3F)));
    out.push_back(char(0x80 | (r & 0x3F)));
  }
}

struct Parser {
  const std::string& s;
  size_t pos = 0;
  bool icase = false;  // leading (?i): fold literals/classes at parse time
  explicit Parser(const std::string& s_, bool icase_ = false)
      : s(s_), icase(icase_) {}

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
      // postfix quantifiers bind to the last atom; at most one quantifier
      // per atom (Go rejects "a**" as nested repetition), with an optional
      // trailing '?' lazy marker — lazy and greedy accept the same language,
      // and MatchString only asks for existence, so the marker is a no-op
      bool quantified = false;
      while (!eof()) {
        char c = peek();
        if (c == '*' || c == '+' || c == '?') {
          if (quantified) err("nested repetition operator");
          pos++;
          if (!eof() && peek() == '?') pos++;  // lazy variant
          RNode q;
          q.kind = c == '*' ? RNode::Star : c == '+' ? RNode::Plus : RNode::Quest;
          q.subs.push_back(std::move(atom));
          atom = std::move(q);
          quantified = true;
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
          if (quantified) err("nested repetition operator");
          if (!eof() && peek() == '?') pos++;  // lazy {m,n}?: same language
          if (nrep == -2) nrep = m;
          if (m > 1000 || nrep > 1000) err("invalid repeat count");
          if (nrep >= 0 && nrep < m) err("invalid repeat count");
          const long max_copies = nrep < 0 ? m + 1 : nrep;
          if (max_copies > 128) {
            err("repetition too large for the 128-position NFA");
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
          quantified = true;
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
        if (icase) return fold_rune_lit(n.lit);
        return n;
      }
    }
  }

  // (?i): a literal rune becomes the alternation of its case-fold orbit
  RNode fold_rune_lit(const std::string& lit) {
    int sz = 0;
    uint32_t r = uint8_t(lit[0]) < 0x80
                     ? uint32_t(uint8_t(lit[0]))
                     : utf8_decode(lit.data(), lit.size(), &sz);
    std::vector<uint32_t> orbit;
    case_orbit(r, orbit);
    if (orbit.size() == 1) {
      RNode n;
      n.kind = RNode::Lit;
      n.lit = lit;
      return n;
    }
    RNode alt;
    alt.kind = RNode::Alt;
    for (uint32_t m : orbit) {
      RNode l;
      l.kind = RNode::Lit;
      utf8_encode(m, l.lit);
      alt.subs.push_back(std::move(l));
    }
    return alt;
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
      case 'b': case 'B': {
        RNode an;
        an.kind = RNode::Assert;
        an.assert_kind = c == 'b' ? 1 : 2;
        return an;
      }
      case 'p': case 'P':
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

    // (?i): fold the ASCII set before negation (Go folds class members);
    // k/K and s/S drag in their non-ASCII orbit members (KELVIN, LONG S)
    std::vector<uint32_t> extra;
    if (icase) {
      uint8_t folded[32];
      memcpy(folded, bits, 32);
      for (int b = 0; b < 128; b++) {
        if (!((bits[b >> 3] >> (b & 7)) & 1)) continue;
        std::vector<uint32_t> orbit;
        case_orbit(uint32_t(b), orbit);
        for (uint32_t m : orbit) {
          if (m < 0x80) {
            folded[m >> 3] |= uint8_t(1) << (m & 7);
          } else {
            extra.push_back(m);
          }
        }
      }
      memcpy(bits, folded, 32);
      if (negated && !extra.empty()) {
        err("(?i) with a negated class containing k or s is not supported "
            "(non-ASCII fold-orbit members)");
      }
    }

    if (negated) {
      RNode cn;
      cn.kind = RNode::Class;
      for (int i = 0; i < 16; i++) cn.cls[i] = uint8_t(~bits[i]);
      cn.cls_nonascii = true;  // Go: [^...] matches any rune outside the set
      return cn;
    }
    // expand small positive classes as alternation of single chars so the
    // or-values fast path classifies them like Go (regexutil.go:93-107)
    RNode result;
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
      result = alt.subs.size() == 1 ? std::move(alt.subs[0]) : std::move(alt);
    } else {
      result.kind = RNode::Class;
      memcpy(result.cls, bits, 32);
    }
    if (!extra.empty()) {
      RNode alt;
      alt.kind = RNode::Alt;
      alt.subs.push_back(std::move(result));
      for (uint32_t m : extra) {
        RNode l;
        l.kind = RNode::Lit;
        utf8_encode(m, l.lit);
        alt.subs.push_back(std::move(l));
      }
      return alt;
    }
    return result;
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
// (<= 128 positions; two-u64 masks, narrow blob when <= 64).  Multi-byte runes (Dot, negated classes) use a relaxed
// UTF-8 shape: [00-7F] | [C0-DF][80-BF] | [E0-EF][80-BF]{2} | [F0-F4][80-BF]{3}
// -- identical to Go on valid UTF-8 input (DESIGN.md notes the invalid-UTF-8
// divergence).

struct GAtom {
  uint8_t set[32];
};

// position mask: up to 128 byte-level positions (two u64 words); patterns
// with <= 64 positions serialize to the narrow single-word blob
struct PMask {
  uint64_t w0 = 0, w1 = 0;
  void set(int i) { (i < 64 ? w0 : w1) |= uint64_t(1) << (i & 63); }
  void operator|=(const PMask& o) {
    w0 |= o.w0;
    w1 |= o.w1;
  }
};

template <typename F>
static void pm_foreach(const PMask& m, F f) {
  uint64_t v = m.w0;
  while (v) {
    f(__builtin_ctzll(v));
    v &= v - 1;
  }
  v = m.w1;
  while (v) {
    f(64 + __builtin_ctzll(v));
    v &= v - 1;
  }
}

// Edge/entry assertion classes: transitions (and fragment entries/exits)
// are classified by the zero-width assertion that must hold between the
// previous and next byte at that point.  The matcher evaluates the word
// boundary once per step (it has both bytes), so assertions cost one mask
// select.  plain composes as identity; b∘nb is infeasible (dead).
constexpr int kAcPlain = 0, kAcB = 1, kAcNB = 2, kAcDead = 3;

static int ac_compose(int a, int b) {
  if (a == kAcDead || b == kAcDead) return kAcDead;
  if (a == kAcPlain) return b;
  if (b == kAcPlain) return a;
  return a == b ? a : kAcDead;
}

struct CMask {
  PMask c[3];  // per assertion class
  void operator|=(const CMask& o) {
    for (int k = 0; k < 3; k++) c[k] |= o.c[k];
  }
  bool any_assert() const {
    return (c[1].w0 | c[1].w1 | c[2].w0 | c[2].w1) != 0;
  }
};

struct GBuild {
  std::vector<GAtom> atoms;
  std::vector<CMask> follow;
  bool has_assert = false;
  [[noreturn]] void overflow(const std::string& expr) {
    fail("regex: NFA fallback for \"" + expr +
         "\" needs more than 128 positions; simplify the pattern");
  }
};

struct GInfo {
  uint8_t null_mask = 0;  // bit c: fragment matches "" under class c
  CMask first, last;
};

static int g_add_atom(GBuild& b, const uint8_t* set, const std::string& expr) {
  if (b.atoms.size() >= 128) b.overflow(expr);
  GAtom a;
  memcpy(a.set, set, 32);
  b.atoms.push_back(a);
  b.follow.push_back(CMask{});
  return int(b.atoms.size()) - 1;
}

static void g_range_set(uint8_t* set, int lo, int hi) {
  for (int c = lo; c <= hi; c++) set[c >> 3] |= uint8_t(1) << (c & 7);
}

static GInfo g_cat(GBuild& b, GInfo x, GInfo y) {
  // follow: last(x) -> first(y), classes composed across the junction
  for (int cx = 0; cx < 3; cx++) {
    for (int cy = 0; cy < 3; cy++) {
      const int c = ac_compose(cx, cy);
      if (c == kAcDead) continue;
      pm_foreach(x.last.c[cx],
                 [&](int i) { b.follow[i].c[c] |= y.first.c[cy]; });
    }
  }
  GInfo r;
  r.first = x.first;
  r.last = y.last;
  for (int nx = 0; nx < 3; nx++) {
    if (!((x.null_mask >> nx) & 1)) continue;
    for (int cy = 0; cy < 3; cy++) {
      const int c = ac_compose(nx, cy);
      if (c != kAcDead) r.first.c[c] |= y.first.c[cy];
    }
  }
  for (int ny = 0; ny < 3; ny++) {
    if (!((y.null_mask >> ny) & 1)) continue;
    for (int cx = 0; cx < 3; cx++) {
      const int c = ac_compose(cx, ny);
      if (c != kAcDead) r.last.c[c] |= x.last.c[cx];
    }
  }
  r.null_mask = 0;
  for (int nx = 0; nx < 3; nx++) {
    if (!((x.null_mask >> nx) & 1)) continue;
    for (int ny = 0; ny < 3; ny++) {
      if (!((y.null_mask >> ny) & 1)) continue;
      const int c = ac_compose(nx, ny);
      if (c != kAcDead) r.null_mask |= uint8_t(1) << c;
    }
  }
  return r;
}

static GInfo g_alt(GInfo x, GInfo y) {
  GInfo r;
  r.null_mask = x.null_mask | y.null_mask;
  r.first = x.first;
  r.first |= y.first;
  r.last = x.last;
  r.last |= y.last;
  return r;
}

static void g_loop(GBuild& b, const GInfo& x) {
  for (int cx = 0; cx < 3; cx++) {
    for (int cy = 0; cy < 3; cy++) {
      const int c = ac_compose(cx, cy);
      if (c == kAcDead) continue;
      pm_foreach(x.last.c[cx],
                 [&](int i) { b.follow[i].c[c] |= x.first.c[cy]; });
    }
  }
}


static GInfo g_pos(int a) {
  GInfo r;
  r.first.c[kAcPlain].set(a);
  r.last.c[kAcPlain].set(a);
  return r;
}

static GInfo g_empty(bool nullable) {
  GInfo r;
  if (nullable) r.null_mask = uint8_t(1) << kAcPlain;
  return r;
}

static GInfo g_assert(GBuild& b, uint8_t kind) {
  // \b / \B: zero-width; matches "" only where the class condition holds
  b.has_assert = true;
  GInfo r;
  r.null_mask = uint8_t(1) << (kind == 1 ? kAcB : kAcNB);
  return r;
}

// relaxed UTF-8 multi-byte rune
static GInfo g_multibyte(GBuild& b, const std::string& expr) {
  uint8_t cont[32] = {0}, l2[32] = {0}, l3[32] = {0}, l4[32] = {0};
  g_range_set(cont, 0x80, 0xBF);
  g_range_set(l2, 0xC0, 0xDF);
  g_range_set(l3, 0xE0, 0xEF);
  g_range_set(l4, 0xF0, 0xF4);
  auto seq = [&](const uint8_t* lead, int ncont) {
    GInfo r = g_pos(g_add_atom(b, lead, expr));
    for (int i = 0; i < ncont; i++) {
      r = g_cat(b, r, g_pos(g_add_atom(b, cont, expr)));
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
  out.assert_kind = n.assert_kind;
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
      return g_empty(true);
    case RNode::Group:
      return g_build(b, n.subs[0], expr);
    case RNode::Lit: {
      GInfo r = g_empty(true);
      for (unsigned char c : n.lit) {
        uint8_t set[32] = {0};
        set[c >> 3] = uint8_t(1) << (c & 7);
        r = g_cat(b, r, g_pos(g_add_atom(b, set, expr)));
      }
      return r;
    }
    case RNode::Dot: {
      uint8_t ascii[32] = {0};
      g_range_set(ascii, 0x00, 0x7F);  // (?s) DotNL: '.' matches any rune
      GInfo r = g_pos(g_add_atom(b, ascii, expr));
      return g_alt(r, g_multibyte(b, expr));
    }
    case RNode::Class: {
      GInfo r = g_pos(g_add_atom(b, n.cls, expr));
      if (n.cls_nonascii) r = g_alt(r, g_multibyte(b, expr));
      return r;
    }
    case RNode::Concat: {
      GInfo r = g_empty(true);
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
      x.null_mask |= uint8_t(1) << kAcPlain;
      return x;
    }
    case RNode::Plus: {
      GInfo x = g_build(b, n.subs[0], expr);
      g_loop(b, x);
      return x;
    }
    case RNode::Quest: {
      GInfo x = g_build(b, n.subs[0], expr);
      x.null_mask |= uint8_t(1) << kAcPlain;
      return x;
    }
    case RNode::Assert:
      return g_assert(b, n.assert_kind);
  }
  fail("regex: unreachable node kind");
}

// blob layouts (selected by flags byte 2):
//   plain/wide:  u16 n, u8 flags, pad to 8, first, last, follow[n],
//                table[256] — 8-byte masks, or 16-byte pairs when flag 8
//                (wide, 65..128 positions) is set.
//   assertions (flag 16, <= 64 positions): u16 n, u8 flags, u8 null_mask,
//                pad to 8, u64 first[3], u64 last[3], u64 follow[n][3],
//                u64 table[256] — masks per assertion class
//                {plain, \b, \B}; null_mask bits say which classes admit
//                an empty match.
static bytes g_serialize(const GBuild& b, const GInfo& root,
                         const std::string& expr, bool a_start = false,
                         bool a_end = false, uint8_t null_mask = 0) {
  bytes out;
  const uint16_t n = uint16_t(b.atoms.size());
  const bool wide = n > 64;  // two-word position masks (65..128 positions)
  const bool has_assert = b.has_assert;
  if (has_assert && wide) {
    fail("regex: \\b/\\B with more than 64 NFA positions is not supported"
         " in " + expr);
  }
  out.push_back(uint8_t(n));
  out.push_back(uint8_t(n >> 8));
  out.push_back(uint8_t((a_start ? 1 : 0) | (a_end ? 2 : 0) |
                        (((null_mask >> kAcPlain) & 1) ? 4 : 0) |
                        (wide ? 8 : 0) | (has_assert ? 16 : 0)));
  out.push_back(has_assert ? null_mask : 0);
  out.resize(8, 0);
  auto put64 = [&](uint64_t v) {
    for (int i = 0; i < 8; i++) out.push_back(uint8_t(v >> (8 * i)));
  };
  auto put_mask = [&](const PMask& m) {
    put64(m.w0);
    if (wide) put64(m.w1);
  };
  if (has_assert) {
    for (int c = 0; c < 3; c++) put64(root.first.c[c].w0);
    for (int c = 0; c < 3; c++) put64(root.last.c[c].w0);
    for (uint16_t i = 0; i < n; i++) {
      for (int c = 0; c < 3; c++) put64(b.follow[i].c[c].w0);
    }
  } else {
    // no assertions anywhere: every mask lives in class 0
    put_mask(root.first.c[0]);
    put_mask(root.last.c[0]);
    for (uint16_t i = 0; i < n; i++) put_mask(b.follow[i].c[0]);
  }
  for (int c = 0; c < 256; c++) {
    PMask m;
    for (uint16_t i = 0; i < n; i++) {
      if ((b.atoms[i].set[c >> 3] >> (c & 7)) & 1) m.set(i);
    }
    put_mask(m);
  }
  return out;
}

}  // namespace

// Splits expr at top-level '|' (outside groups, classes and escapes).
// Returns a single element when there is no top-level alternation.
static std::vector<std::string> split_top_level_alt(const std::string& s) {
  std::vector<std::string> parts;
  size_t start = 0;
  int depth = 0;
  bool in_class = false, class_first = false;
  for (size_t i = 0; i < s.size(); i++) {
    char c = s[i];
    if (c == '\\') {
      i++;  // skip escaped char
      continue;
    }
    if (in_class) {
      if (c == ']' && !class_first) in_class = false;
      class_first = false;
      continue;
    }
    switch (c) {
      case '[':
        in_class = true;
        class_first = true;
        if (i + 1 < s.size() && s[i + 1] == '^') {
          i++;  // '^' right after '[' keeps the first-']'-is-literal rule
        }
        break;
      case '(':
        depth++;
        break;
      case ')':
        depth--;
        break;
      case '|':
        if (depth == 0) {
          parts.push_back(s.substr(start, i - start));
          start = i + 1;
        }
        break;
      default:
        break;
    }
  }
  parts.push_back(s.substr(start));
  return parts;
}

static bool strip_edge_anchors(std::string& body, bool* a_start, bool* a_end) {
  *a_start = *a_end = false;
  if (!body.empty() && body[0] == '^') {
    *a_start = true;
    body.erase(body.begin());
  }
  if (!body.empty() && body.back() == '$') {
    size_t bs = 0;
    while (bs + 1 < body.size() && body[body.size() - 2 - bs] == '\\') bs++;
    if (bs % 2 == 0) {
      *a_end = true;
      body.pop_back();
    }
  }
  return *a_start || *a_end;
}

static RegexProg compile_single(const std::string& expr,
                                const std::string& body_in, bool icase) {
  // Edge anchors: a leading '^' / trailing unescaped '$' anchor this
  // pattern (Go regexp semantics without multiline).  Anchors elsewhere
  // (inside groups, mid-pattern) remain unsupported and error in the
  // parser; anchored top-level alternatives are handled by the alt-list
  // split in regex_compile.
  std::string body = body_in;
  bool a_start = false, a_end = false;
  strip_edge_anchors(body, &a_start, &a_end);
  Parser p(body, icase);
  RNode raw = p.parse_alt();
  if (!p.eof()) p.err("unexpected )");
  if ((a_start || a_end) && raw.kind == RNode::Alt) {
    // only reachable for anchors around a group-free alternation inside a
    // single alt-list branch, e.g. "^(a|b)" is fine (Group) but "^a|b"
    // was already split; keep the loud error as a backstop
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
    re.always_true =
        ((root.null_mask >> kAcPlain) & 1) != 0 && !(a_start && a_end);
    re.is_only_prefix = false;
    re.is_suffix_dot_star = false;
    re.is_suffix_dot_plus = false;
    re.substr_dot_star.clear();
    re.substr_dot_plus.clear();
    re.has_or_values = false;
    re.or_values.clear();
    re.nfa_blob =
        g_serialize(b, root, expr, a_start, a_end, root.null_mask);
    re.prefix.clear();  // NFA matches the whole pattern; ignore the prefix
  }
  return re;
}

RegexProg regex_compile(const std::string& expr) {
  std::string body = expr;
  // a leading (?i) applies to the whole pattern (Go flag semantics); other
  // (?...) constructs still error in the parser
  bool icase = false;
  if (body.size() >= 4 && body.compare(0, 4, "(?i)") == 0) {
    icase = true;
    body.erase(0, 4);
  }

  // Top-level alternation whose branches carry their own anchors compiles
  // as an any-of list of independently compiled branches ("^01|04$" ==
  // starts-with-01 OR ends-with-04, Go per-branch anchoring).
  std::vector<std::string> parts = split_top_level_alt(body);
  if (parts.size() > 1) {
    bool any_anchor = false;
    for (const std::string& p : parts) {
      std::string tmp = p;
      bool s = false, e = false;
      if (strip_edge_anchors(tmp, &s, &e)) any_anchor = true;
    }
    if (any_anchor) {
      RegexProg re;
      re.expr = expr;
      re.is_alt_list = true;
      // no bloom literals: Go GetLiterals returns nil for OpAlternate
      // (regexutil.go:141-149 via regex.go:101-124)
      for (const std::string& p : parts) {
        re.alts.push_back(compile_single(expr, p, icase));
      }
      return re;
    }
  }
  return compile_single(expr, body, icase);
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

static bool ascii_word(uint8_t c) {
  // RE2 \b is over ASCII \w = [0-9A-Za-z_]
  return c == '_' || (c >= '0' && c <= '9') || (c >= 'a' && c <= 'z') ||
         (c >= 'A' && c <= 'Z');
}

// assert-layout matcher: per-class first/last/follow masks; the word
// boundary between the previous and current byte selects which class's
// transitions are live at each step (see g_serialize layout comment)
static bool nfa_match_assert(const uint8_t* blob, strview s) {
  const uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t anchors = blob[2];
  const uint8_t null_mask = blob[3];
  const bool a_start = anchors & 1, a_end = anchors & 2;
  auto rd64 = [](const uint8_t* p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
  };
  uint64_t first[3], last[3];
  for (int c = 0; c < 3; c++) {
    first[c] = rd64(blob + 8 + 8 * c);
    last[c] = rd64(blob + 32 + 8 * c);
  }
  const uint8_t* follow = blob + 56;
  const uint8_t* table = follow + size_t(n) * 24;
  if (s.n == 0) return (null_mask >> kAcPlain) & 1;  // "" has no boundary
  if (((null_mask >> kAcPlain) & 1) && !(a_start && a_end)) return true;
  uint64_t active = 0;
  bool prev_w = false;  // BOF behaves as a non-word char
  for (size_t i = 0; i < s.n; i++) {
    const bool cur_w = ascii_word(uint8_t(s.p[i]));
    const bool bnd = prev_w != cur_w;
    // empty match under an assertion class at offset i
    if (!a_end && (!a_start || i == 0)) {
      if ((null_mask >> (bnd ? kAcB : kAcNB)) & 1) return true;
    }
    uint64_t targets = 0;
    if (!a_start || i == 0) {
      targets = first[0] | (bnd ? first[1] : first[2]);
    }
    uint64_t m = active;
    while (m) {
      const int x = __builtin_ctzll(m);
      m &= m - 1;
      const uint8_t* f = follow + size_t(x) * 24;
      targets |= rd64(f) | (bnd ? rd64(f + 8) : rd64(f + 16));
    }
    const uint64_t entered = targets & rd64(table + size_t(uint8_t(s.p[i])) * 8);
    if (entered && !a_end) {
      const bool next_w = i + 1 < s.n ? ascii_word(uint8_t(s.p[i + 1])) : false;
      const bool bnd2 = cur_w != next_w;
      if (entered & (last[0] | (bnd2 ? last[1] : last[2]))) return true;
    }
    active = entered;
    prev_w = cur_w;
  }
  const bool bnd_eof = ascii_word(uint8_t(s.p[s.n - 1]));  // vs EOF non-word
  if (active & (last[0] | (bnd_eof ? last[1] : last[2]))) {
    if (a_end) return true;  // !a_end acceptances were taken in the loop
  }
  // empty match at offset s.n (allowed unless '^' pins the match to 0)
  if (!a_start && ((null_mask >> (bnd_eof ? kAcB : kAcNB)) & 1)) return true;
  return false;
}

bool nfa_match(const uint8_t* blob, strview s) {
  uint16_t n = uint16_t(blob[0]) | uint16_t(blob[1]) << 8;
  const uint8_t anchors = blob[2];
  if (anchors & 16) return nfa_match_assert(blob, s);
  const bool a_start = anchors & 1, a_end = anchors & 2;
  const bool wide = (anchors & 8) != 0;  // 65..128 positions: 16-byte masks
  const size_t msz = wide ? 16 : 8;
  auto rd = [&](const uint8_t* p, uint64_t* w0, uint64_t* w1) {
    memcpy(w0, p, 8);
    if (wide) {
      memcpy(w1, p + 8, 8);
    } else {
      *w1 = 0;
    }
  };
  uint64_t first0, first1, last0, last1;
  rd(blob + 8, &first0, &first1);
  rd(blob + 8 + msz, &last0, &last1);
  const uint8_t* follow = blob + 8 + 2 * msz;
  const uint8_t* table = follow + size_t(n) * msz;
  if (s.n == 0) return (anchors & 4) != 0;  // nullable root matches ""
  uint64_t active0 = 0, active1 = 0;
  for (size_t i = 0; i < s.n; i++) {
    // '^' anchored: new matches may start only at offset 0
    uint64_t t0 = (a_start && i > 0) ? 0 : first0;
    uint64_t t1 = (a_start && i > 0) ? 0 : first1;
    auto accum = [&](uint64_t m, int base) {
      while (m) {
        int x = base + __builtin_ctzll(m);
        m &= m - 1;
        uint64_t f0, f1;
        rd(follow + size_t(x) * msz, &f0, &f1);
        t0 |= f0;
        t1 |= f1;
      }
    };
    accum(active0, 0);
    accum(active1, 64);
    uint64_t tb0, tb1;
    rd(table + size_t(uint8_t(s.p[i])) * msz, &tb0, &tb1);
    const uint64_t e0 = t0 & tb0, e1 = t1 & tb1;
    if (!a_end && ((e0 & last0) | (e1 & last1))) return true;
    active0 = e0;
    active1 = e1;
  }
  // '$' anchored: accept only with a final position active at string end
  return a_end && ((active0 & last0) | (active1 & last1)) != 0;
}

bool regex_match(const RegexProg& re, strview s) {
  // Regex.MatchString (regex.go:86-98)
  if (re.is_alt_list) {
    for (const RegexProg& alt : re.alts) {
      if (regex_match(alt, s)) return true;
    }
    return false;
  }
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
