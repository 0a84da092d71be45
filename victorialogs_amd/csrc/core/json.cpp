#include "json.h"

#include <cerrno>
#include <cstdlib>

#include <cstdlib>
#include <cstring>

namespace vl {
namespace {

struct JParser {
  const char* p;
  const char* end;
  [[noreturn]] void err(const char* msg) { fail(std::string("filter json: ") + msg); }
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) p++;
  }
  JValue parse() {
    ws();
    if (p >= end) err("unexpected end");
    switch (*p) {
      case '{': {
        p++;
        JValue v;
        v.kind = JValue::Obj;
        ws();
        if (p < end && *p == '}') {
          p++;
          return v;
        }
        for (;;) {
          ws();
          if (p >= end || *p != '"') err("expected key");
          std::string key = parse_string();
          ws();
          if (p >= end || *p != ':') err("expected :");
          p++;
          v.obj[key] = parse();
          ws();
          if (p < end && *p == ',') {
            p++;
            continue;
          }
          if (p < end && *p == '}') {
            p++;
            return v;
          }
          err("expected , or }");
        }
      }
      case '[': {
        p++;
        JValue v;
        v.kind = JValue::Arr;
        ws();
        if (p < end && *p == ']') {
          p++;
          return v;
        }
        for (;;) {
          v.arr.push_back(parse());
          ws();
          if (p < end && *p == ',') {
            p++;
            continue;
          }
          if (p < end && *p == ']') {
            p++;
            return v;
          }
          err("expected , or ]");
        }
      }
      case '"': {
        JValue v;
        v.kind = JValue::Str;
        v.str = parse_string();
        return v;
      }
      case 't':
        if (end - p >= 4 && memcmp(p, "true", 4) == 0) {
          p += 4;
          JValue v;
          v.kind = JValue::Bool;
          v.b = true;
          return v;
        }
        err("bad token");
      case 'f':
        if (end - p >= 5 && memcmp(p, "false", 5) == 0) {
          p += 5;
          JValue v;
          v.kind = JValue::Bool;
          return v;
        }
        err("bad token");
      case 'n':
        if (end - p >= 4 && memcmp(p, "null", 4) == 0) {
          p += 4;
          return JValue();
        }
        err("bad token");
      default: {
        char* endp;
        double d = strtod(p, &endp);
        if (endp == p) err("bad number");
        JValue v;
        v.kind = JValue::Num;
        v.num = d;
        // integral literal (no '.', 'e'): keep the exact int64 too
        bool integral = true;
        for (const char* q = p; q < endp; q++) {
          if (*q == '.' || *q == 'e' || *q == 'E') integral = false;
        }
        if (integral) {
          errno = 0;
          char* iend;
          long long iv = strtoll(p, &iend, 10);
          if (iend == endp && errno != ERANGE) {
            v.ival = iv;
            v.is_int = true;
          }
        }
        p = endp;
        return v;
      }
    }
  }
  std::string parse_string() {
    p++;  // opening quote
    std::string s;
    while (p < end && *p != '"') {
      if (*p == '\\') {
        p++;
        if (p >= end) err("bad escape");
        switch (*p) {
          case 'n': s += '\n'; break;
          case 't': s += '\t'; break;
          case 'r': s += '\r'; break;
          case 'b': s += '\b'; break;
          case 'f': s += '\f'; break;
          case '"': s += '"'; break;
          case '\\': s += '\\'; break;
          case '/': s += '/'; break;
          case 'u': {
            if (end - p < 5) err("bad \\u");
            unsigned cp = 0;
            for (int i = 1; i <= 4; i++) {
              char c = p[i];
              cp <<= 4;
              if (c >= '0' && c <= '9') cp |= unsigned(c - '0');
              else if (c >= 'a' && c <= 'f') cp |= unsigned(c - 'a' + 10);
              else if (c >= 'A' && c <= 'F') cp |= unsigned(c - 'A' + 10);
              else err("bad \\u");
            }
            p += 4;
            // encode UTF-8 (surrogates unsupported)
            if (cp < 0x80) {
              s += char(cp);
            } else if (cp < 0x800) {
              s += char(0xC0 | (cp >> 6));
              s += char(0x80 | (cp & 0x3F));
            } else {
              s += char(0xE0 | (cp >> 12));
              s += char(0x80 | ((cp >> 6) & 0x3F));
              s += char(0x80 | (cp & 0x3F));
            }
            break;
          }
          default:
            err("bad escape");
        }
        p++;
      } else {
        s += *p++;
      }
    }
    if (p >= end) err("unterminated string");
    p++;
    return s;
  }
};

}  // namespace

JValue json_parse(const std::string& s) {
  JParser jp{s.data(), s.data() + s.size()};
  return jp.parse();
}

const JValue& jget(const JValue& o, const char* key) {
  auto it = o.obj.find(key);
  if (it == o.obj.end()) fail(std::string("json: missing field ") + key);
  return it->second;
}

}  // namespace vl
