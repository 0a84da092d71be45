#include "op_serialize.h"

#include "../hip/scan_types.h"
#include "tokenizer.h"

namespace vl {

bytes serialize_regex(const RegexProg& re) {
  bytes b;
  if (re.is_alt_list) {
    // alt-list blob: marker 0x00 (a compiled non-alt regex always sets at
    // least one flag bit), u16 n_alts, then per branch {u32 len, sub-blob}
    b.push_back(0);
    b.push_back(uint8_t(re.alts.size()));
    b.push_back(uint8_t(re.alts.size() >> 8));
    for (const RegexProg& alt : re.alts) {
      bytes sb = serialize_regex(alt);
      uint32_t len = uint32_t(sb.size());
      for (int i = 0; i < 4; i++) b.push_back(uint8_t(len >> (8 * i)));
      b.insert(b.end(), sb.begin(), sb.end());
    }
    return b;
  }
  uint8_t flags = 0;
  if (re.is_only_prefix) flags |= kReOnlyPrefix;
  if (re.is_suffix_dot_star) flags |= kReDotStar;
  if (re.is_suffix_dot_plus) flags |= kReDotPlus;
  if (!re.substr_dot_star.empty()) flags |= kReSubstrStar;
  if (!re.substr_dot_plus.empty()) flags |= kReSubstrPlus;
  if (re.has_or_values) flags |= kReHasOr;
  if (re.has_nfa) flags |= kReNfa;
  if (re.always_true) flags |= kReAlways;
  const std::string& substr =
      !re.substr_dot_star.empty() ? re.substr_dot_star : re.substr_dot_plus;
  b.push_back(flags);
  auto put16 = [&](size_t v) {
    b.push_back(uint8_t(v));
    b.push_back(uint8_t(v >> 8));
  };
  put16(re.prefix.size());
  put16(substr.size());
  put16(re.or_values.size());
  b.insert(b.end(), re.prefix.begin(), re.prefix.end());
  b.insert(b.end(), substr.begin(), substr.end());
  for (const auto& v : re.or_values) {
    put16(v.size());
    b.insert(b.end(), v.begin(), v.end());
  }
  if (re.has_nfa) b.insert(b.end(), re.nfa_blob.begin(), re.nfa_blob.end());
  return b;
}

uint8_t phrase_flags_of(const std::string& phrase) {
  // getPhrasePos boundary-rune precomputation (filter_phrase.go:228-238)
  if (phrase.empty()) return 0;
  uint8_t flags = 0;
  int sz;
  uint32_t r = uint8_t(phrase[0]);
  if (r >= 0x80) r = utf8_decode(phrase.data(), phrase.size(), &sz);
  if (is_token_rune(r)) flags |= kPhraseStartsToken;
  r = uint8_t(phrase[phrase.size() - 1]);
  if (r >= 0x80) r = utf8_decode_last(phrase.data(), phrase.size(), &sz);
  if (is_token_rune(r)) flags |= kPhraseEndsToken;
  return flags;
}

bytes serialize_phrases(const std::vector<std::string>& phrases) {
  // blob = u16 n, then per phrase { u16 len, u8 flags, bytes }
  bytes b;
  b.push_back(uint8_t(phrases.size()));
  b.push_back(uint8_t(phrases.size() >> 8));
  for (const auto& ph : phrases) {
    b.push_back(uint8_t(ph.size()));
    b.push_back(uint8_t(ph.size() >> 8));
    b.push_back(phrase_flags_of(ph));
    b.insert(b.end(), ph.begin(), ph.end());
  }
  return b;
}

bytes serialize_str_set(const std::vector<std::string>& sorted_set) {
  bytes b;
  auto put32 = [&](uint32_t v) {
    b.push_back(uint8_t(v));
    b.push_back(uint8_t(v >> 8));
    b.push_back(uint8_t(v >> 16));
    b.push_back(uint8_t(v >> 24));
  };
  put32(uint32_t(sorted_set.size()));
  uint32_t off = 0;
  for (const auto& v : sorted_set) {
    put32(off);
    off += uint32_t(v.size());
  }
  put32(off);
  for (const auto& v : sorted_set) b.insert(b.end(), v.begin(), v.end());
  return b;
}

}  // namespace vl
