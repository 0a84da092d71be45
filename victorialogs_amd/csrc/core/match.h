// Host-side match semantics restating lib/logstorage/filter_phrase.go:211-300
// and the token helpers used by the AND/OR bloom prefilters
// (filter_and.go:189-208).  The HIP kernels implement the same semantics
// independently on-device; the CPU oracle checks them row-for-row.
#pragma once

#include <string>
#include <vector>

#include "vl_base.h"

namespace vl {

// matchPhrase (filter_phrase.go:211-218): empty phrase matches only "".
bool match_phrase(strview s, strview phrase);
// getPhrasePos (filter_phrase.go:220-270): first occurrence with non-token
// boundary runes; -1 if none.
long get_phrase_pos(strview s, strview phrase);

// skipFirstLastToken (filter_regexp.go:53-69)
std::string skip_first_last_token(const std::string& s);

// matchPrefix (filter_prefix.go:318-352): empty prefix matches non-empty s.
bool match_prefix(strview s, strview prefix);
// matchExactPrefix = strings.HasPrefix (filter_exact_prefix.go:275-277)
bool match_exact_prefix(strview s, strview prefix);
// matchSequence (filter_sequence.go:260-269)
bool match_sequence(strview s, const std::vector<std::string>& phrases);
// getTokensSkipLast (filter_prefix.go:354-363)
std::vector<std::string> get_tokens_skip_last(const std::string& s);

// matchStringByAllTokens (filter_and.go:189-196)
bool match_string_by_all_tokens(strview v, const std::vector<std::string>& tokens);
// matchDictValuesByAllTokens (filter_and.go:198-208): match against
// comma-joined dict values.
bool match_dict_values_by_all_tokens(const std::vector<std::string>& dict_values,
                                     const std::vector<std::string>& tokens);

}  // namespace vl
