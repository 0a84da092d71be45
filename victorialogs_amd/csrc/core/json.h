// Minimal JSON parser used for programmatic filter trees and test fixtures.
#pragma once

#include <map>
#include <string>
#include <vector>

#include "vl_base.h"

namespace vl {

struct JValue {
  enum Kind { Obj, Arr, Str, Num, Bool, Null } kind = Null;
  std::map<std::string, JValue> obj;
  std::vector<JValue> arr;
  std::string str;
  double num = 0;
  // exact integer view: nanosecond timestamps exceed double's 53-bit
  // mantissa, so integral literals in int64 range keep their exact value
  long long ival = 0;
  bool is_int = false;
  bool b = false;

  long long as_i64() const { return is_int ? ival : (long long)num; }
};

JValue json_parse(const std::string& s);
const JValue& jget(const JValue& o, const char* key);

}  // namespace vl
