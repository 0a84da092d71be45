#include "unicode_case.h"

#include "match.h"
#include "tokenizer.h"

namespace vl {

#include "unicode_case.inc"

static uint32_t map_lookup(const uint32_t (*tab)[2], size_t n, uint32_t r) {
  size_t lo = 0, hi = n;
  while (lo < hi) {
    size_t mid = (lo + hi) / 2;
    if (tab[mid][0] == r) return tab[mid][1];
    if (tab[mid][0] < r) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  return r;
}

uint32_t to_lower_rune(uint32_t r) {
  if (r < 0x80) return r >= 'A' && r <= 'Z' ? r + 0x20 : r;
  return map_lookup(kLowerMap, sizeof(kLowerMap) / sizeof(kLowerMap[0]), r);
}

uint32_t to_upper_rune(uint32_t r) {
  if (r < 0x80) return r >= 'a' && r <= 'z' ? r - 0x20 : r;
  return map_lookup(kUpperMap, sizeof(kUpperMap) / sizeof(kUpperMap[0]), r);
}

// utf8.AppendRune
static void utf8_append(std::string& out, uint32_t r) {
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

static std::string map_case(strview s, bool lower) {
  std::string out;
  out.reserve(s.n);
  size_t i = 0;
  while (i < s.n) {
    int sz;
    uint32_t r = utf8_decode(s.p + i, s.n - i, &sz);
    utf8_append(out, lower ? to_lower_rune(r) : to_upper_rune(r));
    i += size_t(sz);
  }
  return out;
}

std::string to_lower_str(strview s) { return map_case(s, true); }
std::string to_upper_str(strview s) { return map_case(s, false); }

bool is_ascii_lowercase(strview s) {
  for (size_t i = 0; i < s.n; i++) {
    uint8_t c = uint8_t(s.p[i]);
    if (c >= 0x80 || (c >= 'A' && c <= 'Z')) return false;
  }
  return true;
}

bool match_any_case_phrase(strview s, strview phrase_lowercase) {
  if (phrase_lowercase.n == 0) return s.n == 0;
  if (phrase_lowercase.n > s.n) return false;
  if (is_ascii_lowercase(s)) return match_phrase(s, phrase_lowercase);
  std::string low = to_lower_str(s);
  return match_phrase(strview(low), phrase_lowercase);
}

bool match_any_case_prefix(strview s, strview prefix_lowercase) {
  if (prefix_lowercase.n == 0) return s.n > 0;
  if (prefix_lowercase.n > s.n) return false;
  if (is_ascii_lowercase(s)) return match_prefix(s, prefix_lowercase);
  std::string low = to_lower_str(s);
  return match_prefix(strview(low), prefix_lowercase);
}

}  // namespace vl
