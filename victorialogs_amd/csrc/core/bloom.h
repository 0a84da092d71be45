// Bloom filter restating lib/logstorage/bloomfilter.go.
//
// 6 hashes per token, 16 bits per token (bloomfilter.go:16-19).  The hash
// chain for one token hashes the 8-byte NATIVE little-endian image of an
// incrementing u64 seeded with xxhash64(token) (bloomfilter.go:126-144 — the
// unsafe-pointer store is LE on x86, do not "fix" it to BE).
// On-disk words are big-endian u64 (bloomfilter.go:49-55).
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "vl_base.h"

namespace vl {

constexpr int kBloomHashesCount = 6;   // bloomfilter.go:16
constexpr int kBloomBitsPerItem = 16;  // bloomfilter.go:19

// appendTokensHashes (bloomfilter.go:126-144): 6 probe hashes per token.
void append_token_hashes(std::vector<uint64_t>& dst, strview token);
// appendHashesHashes (bloomfilter.go:152-170): 6 probe hashes per tokenize_hashes output.
void append_hash_hashes(std::vector<uint64_t>& dst, uint64_t token_hash);

// initBloomFilter (bloomfilter.go:109-121) over probe hashes.
void bloom_init(std::vector<uint64_t>& bits, const std::vector<uint64_t>& probe_hashes);

// bloomFilterMarshalHashes (bloomfilter.go:31-37): build from tokenizeHashes
// output and marshal as BE u64 words.
bytes bloom_marshal_hashes(const std::vector<uint64_t>& token_hashes);
// bloomFilterMarshalTokens (bloomfilter.go:22-28).
bytes bloom_marshal_tokens(const std::vector<std::string>& tokens);

// unmarshal (bloomfilter.go:58-71): BE words -> host u64 vector.
// Returns false if size is not a multiple of 8.
bool bloom_unmarshal(std::vector<uint64_t>& bits, const uint8_t* src, size_t n);

// containsAll (bloomfilter.go:173-191) over probe hashes.
bool bloom_contains_all(const uint64_t* bits, size_t nwords,
                        const uint64_t* probe_hashes, size_t nhashes);

}  // namespace vl
