#include "bloom.h"

#include "xxhash64.h"

namespace vl {

void append_hash_hashes(std::vector<uint64_t>& dst, uint64_t token_hash) {
  // bloomfilter.go:159-169: buf holds the native-LE image of an incrementing
  // u64; each iteration hashes buf then increments.
  uint64_t buf = token_hash;
  for (int i = 0; i < kBloomHashesCount; i++) {
    uint64_t h = xxhash64(&buf, 8);  // LE image on x86/gfx950 hosts
    buf++;
    dst.push_back(h);
  }
}

void append_token_hashes(std::vector<uint64_t>& dst, strview token) {
  // bloomfilter.go:133-143
  append_hash_hashes(dst, xxhash64(token.p, token.n));
}

void bloom_init(std::vector<uint64_t>& bits, const std::vector<uint64_t>& probe_hashes) {
  // initBloomFilter (bloomfilter.go:109-121)
  uint64_t max_bits = uint64_t(bits.size()) * 64;
  for (uint64_t h : probe_hashes) {
    uint64_t idx = h % max_bits;
    bits[idx / 64] |= uint64_t(1) << (idx % 64);
  }
}

static bytes bloom_marshal_words(const std::vector<uint64_t>& words) {
  bytes dst;
  dst.reserve(words.size() * 8);
  for (uint64_t w : words) put_u64be(dst, w);  // bloomfilter.go:49-55
  return dst;
}

bytes bloom_marshal_hashes(const std::vector<uint64_t>& token_hashes) {
  // mustInitHashes (bloomfilter.go:83-89): 16 bits per item, rounded to words.
  uint64_t bits_count = uint64_t(token_hashes.size()) * kBloomBitsPerItem;
  size_t words_count = size_t((bits_count + 63) / 64);
  std::vector<uint64_t> words(words_count, 0);
  if (words_count > 0) {
    std::vector<uint64_t> probes;
    probes.reserve(token_hashes.size() * kBloomHashesCount);
    for (uint64_t th : token_hashes) append_hash_hashes(probes, th);
    bloom_init(words, probes);
  }
  return bloom_marshal_words(words);
}

bytes bloom_marshal_tokens(const std::vector<std::string>& tokens) {
  // mustInitTokens (bloomfilter.go:74-80)
  uint64_t bits_count = uint64_t(tokens.size()) * kBloomBitsPerItem;
  size_t words_count = size_t((bits_count + 63) / 64);
  std::vector<uint64_t> words(words_count, 0);
  if (words_count > 0) {
    std::vector<uint64_t> probes;
    for (const auto& t : tokens) append_token_hashes(probes, strview(t));
    bloom_init(words, probes);
  }
  return bloom_marshal_words(words);
}

bool bloom_unmarshal(std::vector<uint64_t>& bits, const uint8_t* src, size_t n) {
  if (n % 8 != 0) return false;  // bloomfilter.go:59-61
  bits.resize(n / 8);
  for (size_t i = 0; i < bits.size(); i++) bits[i] = get_u64be(src + i * 8);
  return true;
}

bool bloom_contains_all(const uint64_t* bits, size_t nwords,
                        const uint64_t* probe_hashes, size_t nhashes) {
  // containsAll (bloomfilter.go:173-191)
  if (nwords == 0) return true;
  uint64_t max_bits = uint64_t(nwords) * 64;
  for (size_t k = 0; k < nhashes; k++) {
    uint64_t idx = probe_hashes[k] % max_bits;
    if ((bits[idx / 64] & (uint64_t(1) << (idx % 64))) == 0) return false;
  }
  return true;
}

}  // namespace vl
