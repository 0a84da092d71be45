// Regex front-end restating vendor/.../lib/regexutil/regex.go (the Regex
// wrapper with its fast-path classification) over a self-contained parser for
// the RE2 subset the fast paths need.
//
// Supported syntax: literals (with \ escapes of metacharacters), '.',
// alternation '|', groups '(...)' (capturing or '(?:...)'), char classes
// '[...]' (expanded into or-values when small), postfix '*', '+', '?' and
// leading '^' / trailing '$' anchors (ignored the same way Go's
// SimplifyRegex drops them when simplifying unanchored matches is handled by
// MatchString semantics).  Patterns outside this subset (e.g. '\d', '{m,n}',
// flags, backreferences) are rejected at compile time with a clear error —
// the general-fallback NFA is scheduled for round 2.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "vl_base.h"

namespace vl {

// Classification result mirroring regexutil.Regex (regex.go:17-46).
struct RegexProg {
  std::string expr;
  std::string prefix;
  bool is_only_prefix = false;
  bool is_suffix_dot_star = false;
  bool is_suffix_dot_plus = false;
  std::string substr_dot_star;
  std::string substr_dot_plus;
  std::vector<std::string> or_values;
  bool has_or_values = false;  // distinguishes empty-list from ["" ] etc.
  // Literals for bloom tokens (regex.go:101-124 GetLiterals).
  std::vector<std::string> literals;

  // General-class fallback: Glushkov position automaton over bytes
  // (replaces Go's suffixRe slow path, regex.go:148-151,188-191 — an
  // unanchored whole-pattern existence match is equivalent to Go's
  // prefix-retry + anchored-suffix loop for pure regexes).
  bool has_nfa = false;
  bool always_true = false;  // pattern matches the empty string
  bytes nfa_blob;

  // Top-level alternation whose alternatives carry their own '^'/'$'
  // anchors (e.g. "^01|04$"): each alternative compiles independently and
  // the match is any-of.  Mirrors Go's per-branch anchoring semantics; no
  // bloom literals (Go GetLiterals returns nil for OpAlternate).
  bool is_alt_list = false;
  std::vector<RegexProg> alts;
};

// NFA blob layout: u16 nstates, u16 pad, u32 pad, u64 first_mask,
// u64 last_mask, u64 follow[nstates], u64 byte_table[256].
bool nfa_match(const uint8_t* blob, strview s);

// Compiles expr; patterns outside the fast-path classes compile to the NFA
// fallback.  Supported beyond the fast paths: {m,n} repetition, lazy
// quantifiers (same accepted language for existence matching), top-level
// and per-alternative '^'/'$' anchors, and a leading (?i) (simple case
// closure + the Unicode CaseFolding special orbits).  Word-boundary assertions
// \b/\B are supported (assertion-classed NFA edges).  Still rejected
// with a clear error: \p{...}, mid-pattern anchors, (?...) flags other
// than a leading (?i), >128 NFA positions, and \b with >64 positions.
RegexProg regex_compile(const std::string& expr);

// Regex.MatchString (regex.go:86-98,131-212).
bool regex_match(const RegexProg& re, strview s);

}  // namespace vl
