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
};

// Compiles expr; throws vl::Error with a message naming the unsupported
// construct when the pattern falls outside the fast-path classes.
RegexProg regex_compile(const std::string& expr);

// Regex.MatchString (regex.go:86-98,131-212).
bool regex_match(const RegexProg& re, strview s);

}  // namespace vl
