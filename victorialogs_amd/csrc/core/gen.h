// Synthetic data generator producing vlogsgenerator-shaped rows
// (spec: app/vlogsgenerator/main.go:234-297) written directly
// into reference-format parts.  Seeded mt19937_64 so runs are reproducible
// (the reference uses unseeded math/rand; shapes match, bytes differ — parity
// is oracle-vs-GPU on OUR parts).
#pragma once

#include <cstdint>
#include <string>

namespace vl {

struct GenConfig {
  uint64_t rows = 1000000;
  uint64_t streams = 1;       // blocks are per-stream, emitted in streamID order
  uint64_t rows_per_block = 8192;
  size_t msg_len = 256;       // _msg padded to this length (north-star: 256 B)
  uint64_t seed = 1;
  int64_t start_ts = 1700000000000000000LL;  // 2023-11-14T22:13:20Z, nsecs
  int64_t ts_step = 1000000;                 // 1ms between rows
  int num_dict_fields = 2;
  int num_var_fields = 1;
  int num_const_fields = 3;
  bool extra_typed_fields = true;  // u8/u16/u32/u64/i64/float/ip/iso8601 x1
};

// Writes one part under dir.  Returns total _msg bytes written (for
// GB-scanned accounting).
uint64_t generate_part(const std::string& dir, const GenConfig& cfg);

}  // namespace vl
