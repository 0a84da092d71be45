// TEST INFRASTRUCTURE — C ABI over the CPU oracle (see oracle_filter.h).
// Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
// load this library.
#include <atomic>
#include <cstring>
#include <thread>

#include "oracle_filter.h"
#include "../victorialogs_amd/csrc/core/gen.h"
#include "../victorialogs_amd/csrc/core/bloom.h"
#include "../victorialogs_amd/csrc/core/json.h"
#include "../victorialogs_amd/csrc/core/match.h"
#include "../victorialogs_amd/csrc/core/part_writer.h"
#include "../victorialogs_amd/csrc/core/tokenizer.h"
#include "../victorialogs_amd/csrc/core/codec.h"
#include "../victorialogs_amd/csrc/core/regex.h"
#include "../victorialogs_amd/csrc/core/unicode_case.h"
#include "../victorialogs_amd/csrc/core/values.h"
#include "../victorialogs_amd/csrc/core/xxhash64.h"

using namespace vl;
using namespace vl::oracle;

namespace {
thread_local std::string g_err;

struct OrcPart {
  PartReader pr;
  std::vector<BlockHeader> bhs;
  explicit OrcPart(const std::string& dir) : pr(dir) {
    bhs = pr.read_all_block_headers();
  }
};

int set_err(const std::exception& e) {
  g_err = e.what();
  return -1;
}
}  // namespace

extern "C" {

const char* orc_errstr() { return g_err.c_str(); }

void* orc_open_part(const char* dir) {
  try {
    return new OrcPart(dir);
  } catch (const std::exception& e) {
    g_err = e.what();
    return nullptr;
  }
}

void orc_close_part(void* p) { delete (OrcPart*)p; }

long orc_block_count(void* p) { return long(((OrcPart*)p)->bhs.size()); }
long orc_block_rows(void* p, long i) {
  return long(((OrcPart*)p)->bhs[size_t(i)].rows_count);
}
long orc_part_rows(void* p) { return long(((OrcPart*)p)->pr.header().rows_count); }

void* orc_compile_filter(const char* json) {
  try {
    return new FilterNode(compile_filter(json));
  } catch (const std::exception& e) {
    g_err = e.what();
    return nullptr;
  }
}

void orc_free_filter(void* f) { delete (FilterNode*)f; }

// Scans blocks [block_lo, block_hi) with `threads` worker threads.  Bitmap
// words for block i are written at out_words[word_off[i]] where word_off is
// the running sum of (rows+63)/64 over scanned blocks, in block order.
// Returns total matched rows, or -1 on error.
long long orc_scan_blocks(void* part, void* filter, long block_lo, long block_hi,
                          unsigned long long* out_words, long long out_words_cap,
                          int threads) {
  try {
    OrcPart* p = (OrcPart*)part;
    FilterNode* f = (FilterNode*)filter;
    if (block_hi < 0 || size_t(block_hi) > p->bhs.size()) block_hi = long(p->bhs.size());
    if (block_lo < 0) block_lo = 0;

    std::vector<uint64_t> word_off(size_t(block_hi - block_lo) + 1, 0);
    for (long i = block_lo; i < block_hi; i++) {
      word_off[size_t(i - block_lo) + 1] =
          word_off[size_t(i - block_lo)] + (p->bhs[size_t(i)].rows_count + 63) / 64;
    }
    if (out_words && int64_t(word_off.back()) > out_words_cap) {
      g_err = "orc_scan_blocks: output buffer too small";
      return -1;
    }

    std::atomic<long> next{block_lo};
    std::atomic<long long> hits{0};
    if (threads < 1) threads = 1;
    std::vector<std::thread> pool;
    auto worker = [&]() {
      long long local = 0;
      for (;;) {
        long i = next.fetch_add(1);
        if (i >= block_hi) break;
        Bitmap bm;
        search_block(*f, p->pr, p->bhs[size_t(i)], bm);
        local += (long long)bm.ones_count();
        if (out_words) {
          memcpy(out_words + word_off[size_t(i - block_lo)], bm.a.data(),
                 bm.a.size() * 8);
        }
      }
      hits += local;
    };
    for (int t = 1; t < threads; t++) pool.emplace_back(worker);
    worker();
    for (auto& t : pool) t.join();
    return hits.load();
  } catch (const std::exception& e) {
    return set_err(e);
  }
}

// Generates a synthetic part (vlogsgenerator shapes; gen.h).  Returns total
// _msg bytes or -1.
long long orc_generate_part(const char* dir, unsigned long long rows,
                            unsigned long long streams,
                            unsigned long long rows_per_block,
                            unsigned long long msg_len, unsigned long long seed) {
  try {
    GenConfig cfg;
    cfg.rows = rows;
    cfg.streams = streams;
    cfg.rows_per_block = rows_per_block;
    cfg.msg_len = size_t(msg_len);
    cfg.seed = seed;
    // lean parts for the 1B-row configs: only the columns the or8/dict_time
    // workloads scan (_msg, dict_*, host, worker_id, run_id) — halves the
    // on-disk fixture size and generation time on the GPU box
    if (getenv("VQL_GEN_LEAN") != nullptr) {
      cfg.extra_typed_fields = false;
      cfg.num_const_fields = 0;
      cfg.num_var_fields = 0;
    }
    return (long long)generate_part(dir, cfg);
  } catch (const std::exception& e) {
    return set_err(e);
  }
}


// ---- test-support exports (golden-vector checks from pytest) ----

unsigned long long orc_xxhash64(const uint8_t* p, long n) { return xxhash64(p, size_t(n)); }

// bloomFilterMarshalTokens over newline-separated tokens; returns marshaled
// size, writes up to cap bytes.
long orc_bloom_marshal_tokens(const char* tokens_nl, uint8_t* out, long cap) {
  std::vector<std::string> tokens;
  std::string s(tokens_nl);
  size_t pos = 0;
  while (pos <= s.size() && !s.empty()) {
    size_t nl = s.find('\n', pos);
    if (nl == std::string::npos) {
      tokens.push_back(s.substr(pos));
      break;
    }
    tokens.push_back(s.substr(pos, nl - pos));
    pos = nl + 1;
  }
  bytes b = bloom_marshal_tokens(tokens);
  long n = long(b.size()) < cap ? long(b.size()) : cap;
  memcpy(out, b.data(), size_t(n));
  return long(b.size());
}

int orc_match_phrase(const char* s, long sn, const char* ph, long pn) {
  return match_phrase(strview(s, size_t(sn)), strview(ph, size_t(pn))) ? 1 : 0;
}

// tokenizeStrings of one string; returns newline-joined tokens.
long orc_tokenize(const char* s, long sn, char* out, long cap) {
  auto tokens = tokenize_strings({std::string(s, size_t(sn))});
  std::string joined;
  for (size_t i = 0; i < tokens.size(); i++) {
    if (i) joined += '\n';
    joined += tokens[i];
  }
  long n = long(joined.size()) < cap ? long(joined.size()) : cap;
  memcpy(out, joined.data(), size_t(n));
  return long(joined.size());
}

long orc_format_float64(double f, char* out, long cap) {
  std::string s;
  format_float64(s, f);
  long n = long(s.size()) < cap ? long(s.size()) : cap;
  memcpy(out, s.data(), size_t(n));
  return long(s.size());
}

long orc_format_iso8601(long long nsecs, char* out, long cap) {
  std::string s;
  format_timestamp_iso8601(s, nsecs);
  long n = long(s.size()) < cap ? long(s.size()) : cap;
  memcpy(out, s.data(), size_t(n));
  return long(s.size());
}

int orc_parse_iso8601(const char* s, long sn, long long* out) {
  int64_t v;
  if (!try_parse_timestamp_iso8601(strview(s, size_t(sn)), &v)) return 0;
  *out = v;
  return 1;
}

// valuesEncoder.encode probe (values_encoder.go:109-154): encodes the
// newline-joined values and returns "<type> <min> <max>" + a decode
// round-trip verdict, for the TestValuesEncoder table port.
long orc_encode_values(const char* joined, long jn, char* out, long cap) {
  try {
    std::vector<std::string> values;
    const char* p = joined;
    const char* end = joined + jn;
    while (p < end) {
      const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
      if (!nl) nl = end;
      values.emplace_back(p, nl);
      p = nl + 1;
    }
    if (jn == 0) values.clear();
    EncodedColumn ec;
    encode_values(ec, values);
    // decode round trip through format_value-equivalents
    bool rt = true;
    if (ec.type == ValueType::String) {
      for (size_t i = 0; i < values.size(); i++) {
        if (!(ec.values[i] == strview(values[i]))) rt = false;
      }
    } else if (ec.type == ValueType::Dict) {
      for (size_t i = 0; i < values.size(); i++) {
        uint8_t id = uint8_t(ec.values[i].p[0]);
        if (id >= ec.dict.size() || ec.dict[id] != values[i]) rt = false;
      }
    } else {
      for (size_t i = 0; i < values.size(); i++) {
        std::string str;
        const uint8_t* q = (const uint8_t*)ec.values[i].p;
        switch (ec.type) {
          case ValueType::Uint8: format_uint64(str, q[0]); break;
          case ValueType::Uint16: format_uint64(str, get_u16be(q)); break;
          case ValueType::Uint32: format_uint64(str, get_u32be(q)); break;
          case ValueType::Uint64: format_uint64(str, get_u64be(q)); break;
          case ValueType::Int64: format_int64(str, get_i64be_zigzag(q)); break;
          case ValueType::Float64: {
            uint64_t u = get_u64be(q);
            double d;
            memcpy(&d, &u, 8);
            format_float64(str, d);
            break;
          }
          case ValueType::IPv4: format_ipv4(str, get_u32be(q)); break;
          case ValueType::TimestampISO8601:
            format_timestamp_iso8601(str, int64_t(get_u64be(q)));
            break;
          default: rt = false;
        }
        if (str != values[i]) rt = false;
      }
    }
    char buf[96];
    int n = snprintf(buf, sizeof(buf), "%d %llu %llu %d", int(ec.type),
                     (unsigned long long)ec.min_value,
                     (unsigned long long)ec.max_value, rt ? 1 : 0);
    long m = n < cap ? n : cap;
    memcpy(out, buf, size_t(m));
    return n;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

// getCommonTokensAndTokenSets probe (in_values_test.go:9-37)
long orc_common_tokens(const char* joined, long jn, char* out, long cap) {
  std::vector<std::string> values;
  const char* p = joined;
  const char* end = joined + jn;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
    if (!nl) nl = end;
    values.emplace_back(p, nl);
    p = nl + 1;
  }
  if (jn == 0) values.clear();
  std::vector<std::string> common;
  std::vector<std::vector<std::string>> sets;
  get_common_tokens_and_sets(values, &common, &sets);
  std::string outs;
  for (size_t i = 0; i < common.size(); i++) {
    if (i) outs += ' ';
    outs += common[i];
  }
  for (const auto& ts : sets) {
    outs += '|';
    for (size_t i = 0; i < ts.size(); i++) {
      if (i) outs += ' ';
      outs += ts[i];
    }
  }
  long n = long(outs.size()) < cap ? long(outs.size()) : cap;
  memcpy(out, outs.data(), size_t(n));
  return long(outs.size());
}

long orc_le_values(const char* a, long an, const char* b, long bn, int excl) {
  return le_values_string(strview(a, size_t(an)), strview(b, size_t(bn)),
                          excl != 0)
             ? 1
             : 0;
}

// host matcher probes for the device row-ops differential fuzz
long orc_match_prefix(const char* s, long sn, const char* pf, long pn) {
  return match_prefix(strview(s, size_t(sn)), strview(pf, size_t(pn))) ? 1 : 0;
}

long orc_match_sequence(const char* s, long sn, const char* joined, long jn) {
  std::vector<std::string> phrases;
  const char* p = joined;
  const char* end = joined + jn;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
    if (!nl) nl = end;
    phrases.emplace_back(p, nl);
    p = nl + 1;
  }
  return match_sequence(strview(s, size_t(sn)), phrases) ? 1 : 0;
}

long orc_any_case_phrase(const char* s, long sn, const char* lower, long ln) {
  return match_any_case_phrase(strview(s, size_t(sn)),
                               strview(lower, size_t(ln)))
             ? 1
             : 0;
}

// Regex probe for differential fuzzing against an independent engine:
// compiles `pat` (fast paths + Glushkov NFA) and matches `s` unanchored.
// Returns 1 match, 0 no match, -1 compile-reject.
long orc_regex_match(const char* pat, long pn, const char* s, long sn) {
  try {
    RegexProg re = regex_compile(std::string(pat, size_t(pn)));
    return regex_match(re, strview(s, size_t(sn))) ? 1 : 0;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

// Strings/uint64/int64 block codec round-trip probes
// (encoding_test.go:17-104,106-148 fixtures; compressed byte lengths are
// implementation-specific and deliberately not compared — SURVEY.md §8c)
long orc_strings_block_roundtrip(const char* joined, long jn) {
  try {
    std::vector<std::string> values;
    const char* p = joined;
    const char* end = joined + jn;
    while (p < end) {
      const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
      if (!nl) nl = end;
      values.emplace_back(p, nl);
      p = nl + 1;
    }
    if (jn == 0) values.clear();
    std::vector<strview> vs;
    for (const auto& v : values) vs.emplace_back(v);
    bytes data;
    marshal_strings_block(data, vs);
    StringsBlockDec dec;
    unmarshal_strings_block(dec, data.data(), data.size(), values.size());
    for (size_t i = 0; i < values.size(); i++) {
      if (!(dec.row(i) == strview(values[i]))) return 0;
    }
    return 1;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

long orc_uint64_block_roundtrip(const unsigned long long* a, long n) {
  try {
    bytes data;
    marshal_uint64_block(data, (const uint64_t*)a, size_t(n));
    std::vector<uint64_t> back;
    size_t used = unmarshal_uint64_block(back, data.data(), data.size(),
                                         uint64_t(n));
    if (used != data.size()) return 0;
    for (long i = 0; i < n; i++) {
      if (back[size_t(i)] != a[i]) return 0;
    }
    return 1;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

long orc_int64_array_roundtrip(const long long* a, long n) {
  try {
    bytes data;
    int64_t first;
    MarshalType mt = marshal_int64_array(data, (const int64_t*)a, size_t(n),
                                         &first);
    std::vector<int64_t> back;
    unmarshal_int64_array(back, data.data(), data.size(), mt, first,
                          uint64_t(n));
    for (long i = 0; i < n; i++) {
      if (back[size_t(i)] != a[i]) return 0;
    }
    return 1;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

// tokenizeHashes (hash_tokenizer.go:68-166) over newline-separated values:
// returns space-separated hex hashes (TestTokenizeHashes port)
long orc_tokenize_hashes(const char* joined, long jn, char* out, long cap) {
  std::vector<std::string> values;
  const char* p = joined;
  const char* end = joined + jn;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
    if (!nl) nl = end;
    values.emplace_back(p, nl);
    p = nl + 1;
  }
  if (jn == 0) values.clear();
  std::vector<strview> vs;
  for (const auto& v : values) vs.emplace_back(v);
  auto hashes = tokenize_hashes(vs);
  std::string outs;
  char buf[24];
  for (size_t i = 0; i < hashes.size(); i++) {
    snprintf(buf, sizeof(buf), "%s%016llX", i ? " " : "",
             (unsigned long long)hashes[i]);
    outs += buf;
  }
  long n = long(outs.size()) < cap ? long(outs.size()) : cap;
  memcpy(out, outs.data(), size_t(n));
  return long(outs.size());
}

// bloomFilter.containsAll over a marshaled bloom + one query token
// (bloomfilter_test.go equivalence / false-positive tables)
long orc_bloom_contains(const unsigned char* bloom, long bn, const char* token,
                        long tn) {
  std::vector<uint64_t> words;
  if (!bloom_unmarshal(words, bloom, size_t(bn))) return -1;
  std::vector<uint64_t> hashes;
  append_token_hashes(hashes, strview(token, size_t(tn)));
  return bloom_contains_all(words.data(), words.size(), hashes.data(),
                            hashes.size())
             ? 1
             : 0;
}

// tokenizeStrings over multiple values (newline-separated), preserving the
// cross-value dedup order — for the TestTokenizeStrings table port.
long orc_tokenize_multi(const char* joined, long jn, char* out, long cap) {
  std::vector<std::string> values;
  const char* p = joined;
  const char* end = joined + jn;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', size_t(end - p));
    if (!nl) nl = end;
    values.emplace_back(p, nl);
    p = nl + 1;
  }
  if (jn == 0) values.clear();
  auto tokens = tokenize_strings(values);
  std::string outs;
  for (size_t i = 0; i < tokens.size(); i++) {
    if (i) outs += '\n';
    outs += tokens[i];
  }
  long n = long(outs.size()) < cap ? long(outs.size()) : cap;
  memcpy(out, outs.data(), size_t(n));
  return long(outs.size());
}

// Unicode simple case mapping probes (strings.ToLower/ToUpper restatement)
long orc_to_lower(const char* s, long sn, char* out, long cap) {
  std::string r = to_lower_str(strview(s, size_t(sn)));
  long n = long(r.size()) < cap ? long(r.size()) : cap;
  memcpy(out, r.data(), size_t(n));
  return long(r.size());
}

long orc_to_upper(const char* s, long sn, char* out, long cap) {
  std::string r = to_upper_str(strview(s, size_t(sn)));
  long n = long(r.size()) < cap ? long(r.size()) : cap;
  memcpy(out, r.data(), size_t(n));
  return long(r.size());
}

// tryParseDuration / tryParseBytes / tryParseIPv4 probes for golden tests
long long orc_parse_duration(const char* s, long sn, long long* out) {
  int64_t v;
  if (!try_parse_duration(strview(s, size_t(sn)), &v)) return -1;
  *out = v;
  return 0;
}

long long orc_parse_bytes(const char* s, long sn, long long* out) {
  int64_t v;
  if (!try_parse_bytes(strview(s, size_t(sn)), &v)) return -1;
  *out = v;
  return 0;
}

long long orc_parse_ipv4(const char* s, long sn, unsigned int* out) {
  uint32_t v;
  if (!try_parse_ipv4(strview(s, size_t(sn)), &v)) return -1;
  *out = v;
  return 0;
}

// TryParseTimestampRFC3339Nano / parseMathNumber probes for golden tests
long long orc_parse_rfc3339(const char* s, long sn, long long* out) {
  int64_t v;
  if (!try_parse_timestamp_rfc3339(strview(s, size_t(sn)), &v)) return -1;
  *out = v;
  return 0;
}

double orc_parse_math_number(const char* s, long sn) {
  return parse_math_number(strview(s, size_t(sn)));
}

// CPU reference for the GPU ingest-side bloom build (test infrastructure):
// exactly the write path's tokenizeHashes + bloomFilterMarshalHashes
// (block.go:160-168 via part_writer.cpp:162).
long long orc_bloom_build(const unsigned char* data, long long nbytes,
                          const unsigned int* offsets, long long rows,
                          unsigned char* out, long long cap) {
  try {
    (void)nbytes;
    std::vector<strview> vs;
    vs.reserve(size_t(rows));
    for (long long i = 0; i < rows; i++) {
      vs.emplace_back((const char*)data + offsets[i],
                      size_t(offsets[i + 1] - offsets[i]));
    }
    bytes bf = bloom_marshal_hashes(tokenize_hashes(vs));
    long long n = (long long)bf.size();
    if (n <= cap) memcpy(out, bf.data(), bf.size());
    return n;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

// Writes a custom part from a JSON fixture (filter_test.go fixture style):
// {"blocks":[{"stream":0,"timestamps":[...],
//             "columns":[{"name":"x","values":["a","b",...]}]}]}
// Returns 0 on success, -1 on error.
int orc_write_custom_part(const char* dir, const char* spec_json) {
  try {
    JValue v = json_parse(spec_json);
    PartWriter w(dir, 1);
    for (const auto& bj : jget(v, "blocks").arr) {
      StreamID sid;
      auto it = bj.obj.find("stream");
      uint64_t snum = it != bj.obj.end() ? uint64_t(it->second.num) : 0;
      sid.id_hi = snum;
      sid.id_lo = 1;
      std::vector<int64_t> ts;
      for (const auto& t : jget(bj, "timestamps").arr) ts.push_back(t.as_i64());
      std::vector<InputColumn> cols;
      for (const auto& cj : jget(bj, "columns").arr) {
        InputColumn c;
        c.name = jget(cj, "name").str;
        if (c.name == "_msg") c.name = "";  // canonical (log_rows.go:508-513)
        for (const auto& vv : jget(cj, "values").arr) c.values.push_back(vv.str);
        if (c.values.size() != ts.size()) fail("column length != timestamps length");
        cols.push_back(std::move(c));
      }
      w.add_block(sid, ts, cols);
    }
    w.finish();
    return 0;
  } catch (const std::exception& e) {
    g_err = e.what();
    return -1;
  }
}

}  // extern "C"
