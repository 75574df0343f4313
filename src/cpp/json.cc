#include "client_amd/json.h"

#include <cmath>
#include <cstdio>
#include <cstring>

namespace client_amd {

namespace {

void AppendEscaped(std::string& out, const std::string& s) {
  out.push_back('"');
  for (unsigned char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      case '\b': out += "\\b"; break;
      case '\f': out += "\\f"; break;
      default:
        if (c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof(buf), "\\u%04x", c);
          out += buf;
        } else {
          out.push_back((char)c);
        }
    }
  }
  out.push_back('"');
}

struct Parser {
  const char* p;
  const char* end;

  [[noreturn]] void Fail(const char* msg) {
    throw std::runtime_error(std::string("JSON parse error: ") + msg);
  }

  void SkipWs() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }

  char Peek() {
    if (p >= end) Fail("unexpected end");
    return *p;
  }

  void Expect(char c) {
    if (p >= end || *p != c) Fail("unexpected character");
    ++p;
  }

  Json ParseValue() {
    SkipWs();
    char c = Peek();
    switch (c) {
      case '{': return ParseObject();
      case '[': return ParseArray();
      case '"': return Json(ParseString());
      case 't':
        if (end - p >= 4 && memcmp(p, "true", 4) == 0) { p += 4; return Json(true); }
        Fail("bad literal");
      case 'f':
        if (end - p >= 5 && memcmp(p, "false", 5) == 0) { p += 5; return Json(false); }
        Fail("bad literal");
      case 'n':
        if (end - p >= 4 && memcmp(p, "null", 4) == 0) { p += 4; return Json(); }
        Fail("bad literal");
      default: return ParseNumber();
    }
  }

  std::string ParseString() {
    Expect('"');
    std::string out;
    while (true) {
      if (p >= end) Fail("unterminated string");
      char c = *p++;
      if (c == '"') break;
      if (c == '\\') {
        if (p >= end) Fail("bad escape");
        char e = *p++;
        switch (e) {
          case '"': out.push_back('"'); break;
          case '\\': out.push_back('\\'); break;
          case '/': out.push_back('/'); break;
          case 'n': out.push_back('\n'); break;
          case 'r': out.push_back('\r'); break;
          case 't': out.push_back('\t'); break;
          case 'b': out.push_back('\b'); break;
          case 'f': out.push_back('\f'); break;
          case 'u': {
            if (end - p < 4) Fail("bad \\u escape");
            unsigned cp = 0;
            for (int i = 0; i < 4; ++i) {
              char h = *p++;
              cp <<= 4;
              if (h >= '0' && h <= '9') cp |= h - '0';
              else if (h >= 'a' && h <= 'f') cp |= h - 'a' + 10;
              else if (h >= 'A' && h <= 'F') cp |= h - 'A' + 10;
              else Fail("bad hex digit");
            }
            // surrogate pair
            if (cp >= 0xD800 && cp <= 0xDBFF && end - p >= 6 && p[0] == '\\' &&
                p[1] == 'u') {
              unsigned lo = 0;
              const char* q = p + 2;
              for (int i = 0; i < 4; ++i) {
                char h = *q++;
                lo <<= 4;
                if (h >= '0' && h <= '9') lo |= h - '0';
                else if (h >= 'a' && h <= 'f') lo |= h - 'a' + 10;
                else if (h >= 'A' && h <= 'F') lo |= h - 'A' + 10;
                else Fail("bad hex digit");
              }
              if (lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                p = q;
              }
            }
            // utf-8 encode
            if (cp < 0x80) {
              out.push_back((char)cp);
            } else if (cp < 0x800) {
              out.push_back((char)(0xC0 | (cp >> 6)));
              out.push_back((char)(0x80 | (cp & 0x3F)));
            } else if (cp < 0x10000) {
              out.push_back((char)(0xE0 | (cp >> 12)));
              out.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
              out.push_back((char)(0x80 | (cp & 0x3F)));
            } else {
              out.push_back((char)(0xF0 | (cp >> 18)));
              out.push_back((char)(0x80 | ((cp >> 12) & 0x3F)));
              out.push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
              out.push_back((char)(0x80 | (cp & 0x3F)));
            }
            break;
          }
          default: Fail("bad escape");
        }
      } else {
        out.push_back(c);
      }
    }
    return out;
  }

  Json ParseNumber() {
    const char* start = p;
    bool is_double = false;
    if (p < end && (*p == '-' || *p == '+')) ++p;
    while (p < end && ((*p >= '0' && *p <= '9') || *p == '.' || *p == 'e' ||
                       *p == 'E' || *p == '-' || *p == '+')) {
      if (*p == '.' || *p == 'e' || *p == 'E') is_double = true;
      ++p;
    }
    std::string num(start, p - start);
    if (num.empty()) Fail("bad number");
    if (is_double) return Json(strtod(num.c_str(), nullptr));
    return Json((int64_t)strtoll(num.c_str(), nullptr, 10));
  }

  Json ParseArray() {
    Expect('[');
    JsonArray arr;
    SkipWs();
    if (Peek() == ']') { ++p; return Json(std::move(arr)); }
    while (true) {
      arr.push_back(ParseValue());
      SkipWs();
      char c = Peek();
      if (c == ',') { ++p; continue; }
      if (c == ']') { ++p; break; }
      Fail("expected , or ]");
    }
    return Json(std::move(arr));
  }

  Json ParseObject() {
    Expect('{');
    JsonObject obj;
    SkipWs();
    if (Peek() == '}') { ++p; return Json(std::move(obj)); }
    while (true) {
      SkipWs();
      std::string key = ParseString();
      SkipWs();
      Expect(':');
      obj[key] = ParseValue();
      SkipWs();
      char c = Peek();
      if (c == ',') { ++p; continue; }
      if (c == '}') { ++p; break; }
      Fail("expected , or }");
    }
    return Json(std::move(obj));
  }
};

}  // namespace

std::string Json::Dump() const {
  std::string out;
  switch (type_) {
    case Type::Null: out = "null"; break;
    case Type::Bool: out = bool_ ? "true" : "false"; break;
    case Type::Int: out = std::to_string(int_); break;
    case Type::Double: {
      char buf[32];
      snprintf(buf, sizeof(buf), "%.17g", dbl_);
      out = buf;
      break;
    }
    case Type::String: AppendEscaped(out, str_); break;
    case Type::Array: {
      out.push_back('[');
      bool first = true;
      for (const auto& v : arr_) {
        if (!first) out.push_back(',');
        first = false;
        out += v.Dump();
      }
      out.push_back(']');
      break;
    }
    case Type::Object: {
      out.push_back('{');
      bool first = true;
      for (const auto& kv : obj_) {
        if (!first) out.push_back(',');
        first = false;
        AppendEscaped(out, kv.first);
        out.push_back(':');
        out += kv.second.Dump();
      }
      out.push_back('}');
      break;
    }
  }
  return out;
}

Json Json::Parse(const char* begin, size_t len) {
  Parser parser{begin, begin + len};
  Json v = parser.ParseValue();
  parser.SkipWs();
  return v;
}

Json Json::Parse(const std::string& text) {
  return Parse(text.data(), text.size());
}

}  // namespace client_amd
