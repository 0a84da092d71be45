// Go unicode.ToLower/ToUpper simple case mappings + the host restatement of
// stringsutil.AppendLowercase (vendor/.../stringsutil/stringsutil.go:26-51)
// used by the any-case filters (filter_any_case_phrase.go:159-181,
// filter_any_case_prefix.go:161-183).
#pragma once

#include <string>

#include "vl_base.h"

namespace vl {

uint32_t to_lower_rune(uint32_t r);
uint32_t to_upper_rune(uint32_t r);

// strings.ToLower / strings.ToUpper (rune-wise simple mapping; invalid UTF-8
// bytes decode to U+FFFD like Go's range loop).
std::string to_lower_str(strview s);
std::string to_upper_str(strview s);

// isASCIILowercase (filter_any_case_phrase.go:149-157)
bool is_ascii_lowercase(strview s);

// matchAnyCasePhrase / matchAnyCasePrefix (filter_any_case_phrase.go:159-181,
// filter_any_case_prefix.go:161-183); the phrase/prefix argument must already
// be lowercase.
bool match_any_case_phrase(strview s, strview phrase_lowercase);
bool match_any_case_prefix(strview s, strview prefix_lowercase);

}  // namespace vl
