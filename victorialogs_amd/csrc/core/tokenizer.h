// Tokenizer restating lib/logstorage/tokenizer.go and hash_tokenizer.go.
//
// Token chars: [a-zA-Z0-9_] for ASCII (tokenizer.go:132-140); for non-ASCII
// runes, unicode letters/digits (tokenizer.go:142-148) via unicode_ranges.inc.
#pragma once

#include <cstdint>
#include <string>
#include <unordered_set>
#include <vector>

#include "vl_base.h"

namespace vl {

bool is_token_char(uint8_t c);      // tokenizer.go:128-140
bool is_token_rune(uint32_t r);     // tokenizer.go:142-148
bool is_ascii(strview s);           // tokenizer.go:119-126

// Decodes one UTF-8 rune at s[i..); returns rune and size.  Invalid encodings
// return 0xFFFD with size 1 (Go utf8.DecodeRuneInString semantics).
uint32_t utf8_decode(const char* p, size_t n, int* size);
// Decodes the rune ending at s[n) (Go utf8.DecodeLastRuneInString semantics).
uint32_t utf8_decode_last(const char* p, size_t n, int* size);

// tokenizeStrings (tokenizer.go:12-24): extracts unique word tokens in order.
// dedup state spans all input strings; adjacent equal strings are skipped.
std::vector<std::string> tokenize_strings(const std::vector<std::string>& a);

// tokenizeHashes (hash_tokenizer.go:15-27): per-string tokenize + global dedup
// by xxhash64; emits the hash of each first-seen token in order.
// (The 1024-bucket structure of the reference is an implementation detail; the
// observable output — first-occurrence-ordered unique token hashes — is what
// matters and is what this function produces.  Hash collisions dedup the same
// way as the reference since dedup is by hash value.)
std::vector<uint64_t> tokenize_hashes(const std::vector<strview>& a);

// getCommonTokensAndTokenSets (in_values.go:104-139): per-value token sets
// with the tokens common to every value factored out.
void get_common_tokens_and_sets(const std::vector<std::string>& values,
                                std::vector<std::string>* common,
                                std::vector<std::vector<std::string>>* sets);

}  // namespace vl
