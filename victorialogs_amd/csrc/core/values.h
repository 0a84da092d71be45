// Value-type encoding restating lib/logstorage/values_encoder.go: the
// valueType enum (:22-60), the encode priority dict -> uint -> int -> float ->
// ipv4 -> iso8601 -> string (:109-154), the per-type parsers and the
// string formatters used when matching numeric columns by phrase/regexp.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "vl_base.h"

namespace vl {

enum class ValueType : uint8_t {
  Unknown = 0,
  String = 1,
  Dict = 2,
  Uint8 = 3,
  Uint16 = 4,
  Uint32 = 5,
  Uint64 = 6,
  Float64 = 7,
  IPv4 = 8,
  TimestampISO8601 = 9,
  Int64 = 10,
};

constexpr int kMaxDictLen = 8;          // consts.go:70
constexpr int kMaxDictSizeBytes = 256;  // consts.go:65

// ---- parsers (values_encoder.go) ----
bool try_parse_uint64(strview s, uint64_t* out);              // :553-585
bool try_parse_int64(strview s, int64_t* out);                // :622-645
bool try_parse_float64_exact(strview s, double* out);         // :784-848
bool try_parse_float64(strview s, double* out);               // :779-781 (non-exact)
bool try_parse_ipv4(strview s, uint32_t* out);                // :675-730
bool try_parse_timestamp_iso8601(strview s, int64_t* out);    // :428-466
bool try_parse_duration(strview s, int64_t* out);             // :990-1061
// TryParseTimestampRFC3339Nano (values_encoder.go:340-381); no-timezone
// inputs use the host local timezone offset, matching the reference's
// GetLocalTimezoneOffsetNsecs (sampled from the current time, cached)
bool try_parse_timestamp_rfc3339(strview s, int64_t* out);

// Host local timezone offset of the current time, in nanoseconds (cached;
// vendor/.../lib/timeutil/timezone.go:9-19 semantics)
int64_t local_tz_offset_nsecs();
// leValuesString (filter_le_field.go:284-299): numeric compare when both
// sides parse via parseMathNumber, else bytewise
bool le_values_string(strview a, strview b, bool exclude_equal);
bool try_parse_bytes(strview s, int64_t* out);                // :855-966

// parseMathNumber (pipe_math.go:1066-1080 / block_result.go:2710-2752), all
// legs: float64 -> duration -> bytes -> isLikelyNumber(ParseFloat/ParseInt
// base 0) -> RFC3339Nano -> ipv4; NaN otherwise.
double parse_math_number(strview s);

// ---- formatters (values_encoder.go:1367-1424) ----
void format_uint64(std::string& dst, uint64_t n);
void format_int64(std::string& dst, int64_t n);
// Go strconv.AppendFloat(dst, f, 'f', -1, 64): shortest round-trip decimal
// digits, fixed-point formatting.
void format_float64(std::string& dst, double f);
void format_ipv4(std::string& dst, uint32_t ip);
// time.Unix(0,nsecs).UTC().Format("2006-01-02T15:04:05.000Z")
void format_timestamp_iso8601(std::string& dst, int64_t nsecs);

// ---- encoder ----
// Result of valuesEncoder.encode (values_encoder.go:109-154): the chosen type,
// encoded per-row values (concatenated fixed-width, or 1-byte dict ids, or the
// original strings for valueTypeString), min/max, and the dict.
struct EncodedColumn {
  ValueType type = ValueType::String;
  uint64_t min_value = 0;
  uint64_t max_value = 0;
  std::vector<std::string> dict;   // valid for type==Dict
  // encoded values as strviews into buf (or into the caller's values for String)
  std::vector<strview> values;
  bytes buf;
};

// values must stay alive while ec.values is used.
void encode_values(EncodedColumn& ec, const std::vector<std::string>& values);

}  // namespace vl
