#include "tokenizer.h"

#include <algorithm>

#include <unordered_set>

#include "xxhash64.h"
#include "unicode_ranges.inc"

namespace vl {

static const uint8_t kTokenCharTable[256] = {
    // [a-zA-Z0-9_] per tokenizer.go:132-140
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,  // 0x00
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,  // 0x10
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,  // 0x20
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0,  // 0x30 '0'-'9'
    0, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,  // 0x40 'A'-
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 1,  // 0x50 -'Z', '_'
    0, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,  // 0x60 'a'-
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0,  // 0x70 -'z'
};

bool is_token_char(uint8_t c) { return c < 0x80 && kTokenCharTable[c] != 0; }

bool is_token_rune(uint32_t r) {
  if (r < 0x80) return kTokenCharTable[r] != 0;
  // Binary search the generated letter/digit ranges (tokenizer.go:142-148).
  int lo = 0, hi = kTokenRuneRangesCount - 1;
  while (lo <= hi) {
    int mid = (lo + hi) / 2;
    if (r < kTokenRuneRanges[mid][0]) {
      hi = mid - 1;
    } else if (r > kTokenRuneRanges[mid][1]) {
      lo = mid + 1;
    } else {
      return true;
    }
  }
  return false;
}

bool is_ascii(strview s) {
  for (size_t i = 0; i < s.n; i++) {
    if (uint8_t(s.p[i]) >= 0x80) return false;
  }
  return true;
}

// Go utf8.DecodeRuneInString: returns (0xFFFD, 1) on invalid encoding.
uint32_t utf8_decode(const char* p, size_t n, int* size) {
  *size = 1;
  if (n == 0) return 0xFFFD;
  uint8_t c0 = uint8_t(p[0]);
  if (c0 < 0x80) return c0;
  int len;
  uint32_t r, lo_bound;
  if ((c0 & 0xE0) == 0xC0) {
    len = 2; r = c0 & 0x1F; lo_bound = 0x80;
  } else if ((c0 & 0xF0) == 0xE0) {
    len = 3; r = c0 & 0x0F; lo_bound = 0x800;
  } else if ((c0 & 0xF8) == 0xF0) {
    len = 4; r = c0 & 0x07; lo_bound = 0x10000;
  } else {
    return 0xFFFD;
  }
  if (size_t(len) > n) return 0xFFFD;
  for (int i = 1; i < len; i++) {
    uint8_t c = uint8_t(p[i]);
    if ((c & 0xC0) != 0x80) return 0xFFFD;
    r = (r << 6) | (c & 0x3F);
  }
  if (r < lo_bound || r > 0x10FFFF || (r >= 0xD800 && r <= 0xDFFF)) return 0xFFFD;
  *size = len;
  return r;
}

uint32_t utf8_decode_last(const char* p, size_t n, int* size) {
  *size = 1;
  if (n == 0) return 0xFFFD;
  size_t start = n - 1;
  if (uint8_t(p[start]) < 0x80) return uint8_t(p[start]);
  // Walk back over at most 3 continuation bytes to the rune start
  // (Go utf8.DecodeLastRuneInString).
  size_t lim = n >= 4 ? n - 4 : 0;
  while (start > lim && (uint8_t(p[start]) & 0xC0) == 0x80) start--;
  int sz;
  uint32_t r = utf8_decode(p + start, n - start, &sz);
  if (start + size_t(sz) != n) return 0xFFFD;  // trailing garbage
  *size = sz;
  return r;
}

// Shared tokenize loop: calls f(token) for each token in s, in order.
// ASCII fast path restates tokenizer.go:34-80; unicode path :82-117.
template <typename F>
static void for_each_token(strview s, F&& f) {
  if (is_ascii(s)) {
    size_t i = 0;
    while (i < s.n) {
      size_t start = s.n;
      while (i < s.n) {
        if (!is_token_char(uint8_t(s.p[i]))) { i++; continue; }
        start = i; i++; break;
      }
      size_t end = s.n;
      while (i < s.n) {
        if (is_token_char(uint8_t(s.p[i]))) { i++; continue; }
        end = i; i++; break;
      }
      if (end <= start) break;
      f(strview(s.p + start, end - start));
    }
    return;
  }
  // Unicode slow path
  const char* p = s.p;
  size_t n = s.n;
  while (n > 0) {
    // Search for the next token start
    size_t off = n;
    for (size_t i = 0; i < n;) {
      int sz;
      uint32_t r = utf8_decode(p + i, n - i, &sz);
      if (is_token_rune(r)) { off = i; break; }
      i += sz;
    }
    p += off; n -= off;
    // Search for the token end
    size_t end = n;
    for (size_t i = 0; i < n;) {
      int sz;
      uint32_t r = utf8_decode(p + i, n - i, &sz);
      if (!is_token_rune(r)) { end = i; break; }
      i += sz;
    }
    if (end == 0) break;
    f(strview(p, end));
    p += end; n -= end;
  }
}

std::vector<std::string> tokenize_strings(const std::vector<std::string>& a) {
  std::vector<std::string> dst;
  std::unordered_set<std::string> seen;
  for (size_t i = 0; i < a.size(); i++) {
    if (i > 0 && a[i] == a[i - 1]) continue;  // tokenizer.go:15-18
    for_each_token(strview(a[i]), [&](strview tok) {
      std::string t = tok.str();
      if (seen.insert(t).second) dst.push_back(std::move(t));
    });
  }
  return dst;
}

std::vector<uint64_t> tokenize_hashes(const std::vector<strview>& a) {
  std::vector<uint64_t> dst;
  std::unordered_set<uint64_t> seen;  // dedup by hash, like hash_tokenizer.go:145-166
  for (size_t i = 0; i < a.size(); i++) {
    if (i > 0 && a[i] == a[i - 1]) continue;  // hash_tokenizer.go:18-21
    for_each_token(a[i], [&](strview tok) {
      uint64_t h = xxhash64(tok.p, tok.n);
      if (seen.insert(h).second) dst.push_back(h);
    });
  }
  return dst;
}


void get_common_tokens_and_sets(const std::vector<std::string>& values,
                                std::vector<std::string>* common,
                                std::vector<std::vector<std::string>>* sets) {
  // getCommonTokensAndTokenSets (in_values.go:104-139)
  sets->clear();
  common->clear();
  for (const auto& v : values) {
    sets->push_back(tokenize_strings({v}));
  }
  if (!sets->empty()) {
    *common = (*sets)[0];
    for (size_t i = 1; i < sets->size() && !common->empty(); i++) {
      std::vector<std::string> kept;
      for (const auto& t : *common) {
        if (std::find((*sets)[i].begin(), (*sets)[i].end(), t) !=
            (*sets)[i].end()) {
          kept.push_back(t);
        }
      }
      *common = std::move(kept);
    }
  }
  if (!common->empty()) {
    for (auto& ts : *sets) {
      std::vector<std::string> kept;
      for (auto& t : ts) {
        if (std::find(common->begin(), common->end(), t) == common->end()) {
          kept.push_back(std::move(t));
        }
      }
      ts = std::move(kept);
    }
  }
}

}  // namespace vl
