// Serialization of per-leaf operands into the device blobs the scan kernels
// consume (regex programs, phrase lists, sorted string sets) — shared by the
// staging layer (vql_api.cpp) and the host row-ops fuzz harness so both
// build byte-identical blobs.
#pragma once

#include <string>
#include <vector>

#include "regex.h"
#include "vl_base.h"

namespace vl {

bytes serialize_regex(const RegexProg& re);
// blob = u16 n, then per phrase { u16 len, u8 flags, bytes }
bytes serialize_phrases(const std::vector<std::string>& phrases);
// getPhrasePos boundary-rune flags (filter_phrase.go:228-238)
uint8_t phrase_flags_of(const std::string& phrase);
// sorted string set blob: u32 n, u32 offs[n+1], bytes (kScanInStr layout)
bytes serialize_str_set(const std::vector<std::string>& sorted_set);

}  // namespace vl
