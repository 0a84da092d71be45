#include "match.h"

#include <cstring>

#include "tokenizer.h"

namespace vl {

// strings.Index
static long str_index(strview s, strview sub) {
  if (sub.n == 0) return 0;
  if (sub.n > s.n) return -1;
  const char* found = (const char*)memmem(s.p, s.n, sub.p, sub.n);
  return found ? long(found - s.p) : -1;
}

long get_phrase_pos(strview s, strview phrase) {
  // getPhrasePos (filter_phrase.go:220-270)
  if (phrase.n == 0) return 0;
  if (phrase.n > s.n) return -1;

  int sz;
  uint32_t r = uint8_t(phrase.p[0]);
  if (r >= 0x80) r = utf8_decode(phrase.p, phrase.n, &sz);
  bool starts_with_token = is_token_rune(r);

  r = uint8_t(phrase.p[phrase.n - 1]);
  if (r >= 0x80) r = utf8_decode_last(phrase.p, phrase.n, &sz);
  bool ends_with_token = is_token_rune(r);

  long pos = 0;
  for (;;) {
    long n = str_index(strview(s.p + pos, s.n - pos), phrase);
    if (n < 0) return -1;
    pos += n;
    if (starts_with_token && pos > 0) {
      uint32_t rb = uint8_t(s.p[pos - 1]);
      if (rb >= 0x80) rb = utf8_decode_last(s.p, size_t(pos), &sz);
      // Go: if r == utf8.RuneError || isTokenRune(r) { pos++; continue }
      if (rb == 0xFFFD || is_token_rune(rb)) {
        pos++;
        continue;
      }
    }
    if (ends_with_token && size_t(pos) + phrase.n < s.n) {
      uint32_t ra = uint8_t(s.p[pos + phrase.n]);
      if (ra >= 0x80) {
        ra = utf8_decode(s.p + pos + phrase.n, s.n - size_t(pos) - phrase.n, &sz);
      }
      if (ra == 0xFFFD || is_token_rune(ra)) {
        pos++;
        continue;
      }
    }
    return pos;
  }
}

bool match_phrase(strview s, strview phrase) {
  if (phrase.n == 0) return s.n == 0;  // filter_phrase.go:212-215
  return get_phrase_pos(s, phrase) >= 0;
}

std::string skip_first_last_token(const std::string& s) {
  // filter_regexp.go:53-69
  const char* p = s.data();
  size_t n = s.size();
  for (;;) {
    if (n == 0) break;
    int sz;
    uint32_t r = utf8_decode(p, n, &sz);
    if (!is_token_rune(r)) break;
    p += sz;
    n -= sz;
  }
  for (;;) {
    if (n == 0) break;
    int sz;
    uint32_t r = utf8_decode_last(p, n, &sz);
    if (!is_token_rune(r)) break;
    n -= sz;
  }
  return std::string(p, n);
}

bool match_prefix(strview s, strview prefix) {
  // filter_prefix.go:318-352
  if (prefix.n == 0) return s.n > 0;
  if (prefix.n > s.n) return false;
  int sz;
  uint32_t r = uint8_t(prefix.p[0]);
  if (r >= 0x80) r = utf8_decode(prefix.p, prefix.n, &sz);
  bool starts_with_token = is_token_rune(r);
  long offset = 0;
  for (;;) {
    const char* found =
        (const char*)memmem(s.p + offset, s.n - size_t(offset), prefix.p, prefix.n);
    if (!found) return false;
    offset = found - s.p;
    if (starts_with_token && offset > 0) {
      uint32_t rb = uint8_t(s.p[offset - 1]);
      if (rb >= 0x80) rb = utf8_decode_last(s.p, size_t(offset), &sz);
      if (rb == 0xFFFD || is_token_rune(rb)) {
        offset++;
        continue;
      }
    }
    return true;
  }
}

bool match_exact_prefix(strview s, strview prefix) {
  return s.n >= prefix.n && memcmp(s.p, prefix.p, prefix.n) == 0;
}

bool match_sequence(strview s, const std::vector<std::string>& phrases) {
  // filter_sequence.go:260-269
  for (const auto& phrase : phrases) {
    long n = get_phrase_pos(s, strview(phrase));
    if (n < 0) return false;
    s.p += n + phrase.size();
    s.n -= size_t(n) + phrase.size();
  }
  return true;
}

std::vector<std::string> get_tokens_skip_last(const std::string& str) {
  // filter_prefix.go:354-363
  const char* p = str.data();
  size_t n = str.size();
  for (;;) {
    if (n == 0) break;
    int sz;
    uint32_t r = utf8_decode_last(p, n, &sz);
    if (!is_token_rune(r)) break;
    n -= sz;
  }
  return tokenize_strings({std::string(p, n)});
}

bool match_string_by_all_tokens(strview v, const std::vector<std::string>& tokens) {
  for (const auto& t : tokens) {
    if (!match_phrase(v, strview(t))) return false;
  }
  return true;
}

bool match_dict_values_by_all_tokens(const std::vector<std::string>& dict_values,
                                     const std::vector<std::string>& tokens) {
  // filter_and.go:198-208: "v1,v2,...," joined with trailing commas
  std::string joined;
  for (const auto& v : dict_values) {
    joined += v;
    joined += ',';
  }
  return match_string_by_all_tokens(strview(joined), tokens);
}

}  // namespace vl
