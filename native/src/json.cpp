#include "hipstore/json.h"

#include <cctype>
#include <cmath>
#include <cstdio>
#include <cstring>

namespace hipstore {

namespace {

struct Parser {
  const char* p;
  const char* end;
  bool incomplete = false;  // ran out of bytes mid-value

  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  }

  bool eof() { return p >= end; }

  [[noreturn]] void fail(const char* msg) {
    throw JsonError(std::string("json parse error: ") + msg);
  }

  // Each parse_* returns false if the input ended mid-value (incomplete).
  bool parse_value(Json* out) {
    skip_ws();
    if (eof()) { incomplete = true; return false; }
    switch (*p) {
      case '{': return parse_object(out);
      case '[': return parse_array(out);
      case '"': {
        std::string s;
        if (!parse_string(&s)) return false;
        *out = Json(std::move(s));
        return true;
      }
      case 't': return parse_lit("true", Json(true), out);
      case 'f': return parse_lit("false", Json(false), out);
      case 'n': return parse_lit("null", Json(nullptr), out);
      default: return parse_number(out);
    }
  }

  bool parse_lit(const char* lit, Json value, Json* out) {
    size_t n = strlen(lit);
    if (static_cast<size_t>(end - p) < n) {
      if (strncmp(p, lit, end - p) == 0) { incomplete = true; return false; }
      fail("bad literal");
    }
    if (strncmp(p, lit, n) != 0) fail("bad literal");
    p += n;
    *out = std::move(value);
    return true;
  }

  bool parse_number(Json* out) {
    const char* start = p;
    bool is_double = false;
    if (p < end && (*p == '-' || *p == '+')) ++p;
    while (p < end) {
      char c = *p;
      if (isdigit(static_cast<unsigned char>(c))) { ++p; continue; }
      if (c == '.' || c == 'e' || c == 'E' || c == '-' || c == '+') {
        if (c == '.' || c == 'e' || c == 'E') is_double = true;
        ++p;
        continue;
      }
      break;
    }
    if (p == start) fail("bad number");
    // A number at the very end of the buffer may be truncated.
    if (p == end) { incomplete = true; return false; }
    std::string text(start, p);
    if (is_double) {
      *out = Json(strtod(text.c_str(), nullptr));
    } else {
      *out = Json(static_cast<int64_t>(strtoll(text.c_str(), nullptr, 10)));
    }
    return true;
  }

  bool parse_string(std::string* out) {
    ++p;  // opening quote
    std::string s;
    while (true) {
      if (eof()) { incomplete = true; return false; }
      char c = *p++;
      if (c == '"') break;
      if (c == '\\') {
        if (eof()) { incomplete = true; return false; }
        char esc = *p++;
        switch (esc) {
          case '"': s += '"'; break;
          case '\\': s += '\\'; break;
          case '/': s += '/'; break;
          case 'b': s += '\b'; break;
          case 'f': s += '\f'; break;
          case 'n': s += '\n'; break;
          case 'r': s += '\r'; break;
          case 't': s += '\t'; break;
          case 'u': {
            if (end - p < 4) { incomplete = true; return false; }
            unsigned code = 0;
            for (int i = 0; i < 4; ++i) {
              char h = *p++;
              code <<= 4;
              if (h >= '0' && h <= '9') code |= h - '0';
              else if (h >= 'a' && h <= 'f') code |= h - 'a' + 10;
              else if (h >= 'A' && h <= 'F') code |= h - 'A' + 10;
              else fail("bad \\u escape");
            }
            // Surrogate pair: \uD800-\uDBFF must be followed by
            // \uDC00-\uDFFF; combine to the astral code point.
            // (Astral characters — emoji in a volume name — arrive
            // exactly this way from json.dumps; decoding the halves
            // separately would emit invalid UTF-8 that poisons the
            // response stream for stricter peers.)
            uint32_t cp = code;
            if (code >= 0xD800 && code <= 0xDBFF) {
              if (end - p < 6 || p[0] != '\\' || p[1] != 'u') {
                if (end - p < 6) { incomplete = true; return false; }
                fail("lone high surrogate");
              }
              uint32_t low = 0;
              for (int k = 2; k < 6; ++k) {
                low <<= 4;
                char h = p[k];
                if (h >= '0' && h <= '9') low |= h - '0';
                else if (h >= 'a' && h <= 'f') low |= h - 'a' + 10;
                else if (h >= 'A' && h <= 'F') low |= h - 'A' + 10;
                else fail("bad \\u escape");
              }
              if (low < 0xDC00 || low > 0xDFFF) fail("bad surrogate pair");
              p += 6;
              cp = 0x10000 + ((code - 0xD800) << 10) + (low - 0xDC00);
            } else if (code >= 0xDC00 && code <= 0xDFFF) {
              fail("lone low surrogate");
            }
            if (cp < 0x80) {
              s += static_cast<char>(cp);
            } else if (cp < 0x800) {
              s += static_cast<char>(0xC0 | (cp >> 6));
              s += static_cast<char>(0x80 | (cp & 0x3F));
            } else if (cp < 0x10000) {
              s += static_cast<char>(0xE0 | (cp >> 12));
              s += static_cast<char>(0x80 | ((cp >> 6) & 0x3F));
              s += static_cast<char>(0x80 | (cp & 0x3F));
            } else {
              s += static_cast<char>(0xF0 | (cp >> 18));
              s += static_cast<char>(0x80 | ((cp >> 12) & 0x3F));
              s += static_cast<char>(0x80 | ((cp >> 6) & 0x3F));
              s += static_cast<char>(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: fail("bad escape");
        }
      } else {
        s += c;
      }
    }
    *out = std::move(s);
    return true;
  }

  bool parse_array(Json* out) {
    ++p;  // '['
    JsonArray arr;
    skip_ws();
    if (eof()) { incomplete = true; return false; }
    if (*p == ']') { ++p; *out = Json(std::move(arr)); return true; }
    while (true) {
      Json v;
      if (!parse_value(&v)) return false;
      arr.push_back(std::move(v));
      skip_ws();
      if (eof()) { incomplete = true; return false; }
      if (*p == ',') { ++p; continue; }
      if (*p == ']') { ++p; break; }
      fail("expected ',' or ']'");
    }
    *out = Json(std::move(arr));
    return true;
  }

  bool parse_object(Json* out) {
    ++p;  // '{'
    JsonObject obj;
    skip_ws();
    if (eof()) { incomplete = true; return false; }
    if (*p == '}') { ++p; *out = Json(std::move(obj)); return true; }
    while (true) {
      skip_ws();
      if (eof()) { incomplete = true; return false; }
      if (*p != '"') fail("expected object key");
      std::string key;
      if (!parse_string(&key)) return false;
      skip_ws();
      if (eof()) { incomplete = true; return false; }
      if (*p != ':') fail("expected ':'");
      ++p;
      Json v;
      if (!parse_value(&v)) return false;
      obj[key] = std::move(v);
      skip_ws();
      if (eof()) { incomplete = true; return false; }
      if (*p == ',') { ++p; continue; }
      if (*p == '}') { ++p; break; }
      fail("expected ',' or '}'");
    }
    *out = Json(std::move(obj));
    return true;
  }
};

void dump_string(const std::string& s, std::string* out) {
  out->push_back('"');
  for (char c : s) {
    switch (c) {
      case '"': *out += "\\\""; break;
      case '\\': *out += "\\\\"; break;
      case '\b': *out += "\\b"; break;
      case '\f': *out += "\\f"; break;
      case '\n': *out += "\\n"; break;
      case '\r': *out += "\\r"; break;
      case '\t': *out += "\\t"; break;
      default:
        if (static_cast<unsigned char>(c) < 0x20) {
          char buf[8];
          snprintf(buf, sizeof(buf), "\\u%04x", c);
          *out += buf;
        } else {
          out->push_back(c);
        }
    }
  }
  out->push_back('"');
}

void dump_value(const Json& v, std::string* out) {
  switch (v.type()) {
    case Json::Type::Null: *out += "null"; break;
    case Json::Type::Bool: *out += v.as_bool() ? "true" : "false"; break;
    case Json::Type::Int: {
      char buf[32];
      snprintf(buf, sizeof(buf), "%lld", static_cast<long long>(v.as_int()));
      *out += buf;
      break;
    }
    case Json::Type::Double: {
      char buf[40];
      double d = v.as_double();
      if (std::isfinite(d)) snprintf(buf, sizeof(buf), "%.17g", d);
      else snprintf(buf, sizeof(buf), "null");  // JSON has no inf/nan
      *out += buf;
      break;
    }
    case Json::Type::String: dump_string(v.as_string(), out); break;
    case Json::Type::Array: {
      out->push_back('[');
      const auto& arr = v.as_array();
      for (size_t i = 0; i < arr.size(); ++i) {
        if (i) out->push_back(',');
        dump_value(arr[i], out);
      }
      out->push_back(']');
      break;
    }
    case Json::Type::Object: {
      out->push_back('{');
      bool first = true;
      for (const auto& [key, value] : v.as_object()) {
        if (!first) out->push_back(',');
        first = false;
        dump_string(key, out);
        out->push_back(':');
        dump_value(value, out);
      }
      out->push_back('}');
      break;
    }
  }
}

}  // namespace

std::string Json::dump() const {
  std::string out;
  dump_value(*this, &out);
  return out;
}

bool Json::parse_some(const char* begin, const char* end, Json* out,
                      size_t* consumed) {
  Parser parser{begin, end};
  Json value;
  if (!parser.parse_value(&value)) {
    if (parser.incomplete) return false;
    throw JsonError("json parse failed");
  }
  parser.skip_ws();
  *out = std::move(value);
  *consumed = parser.p - begin;
  return true;
}

Json Json::parse(const std::string& text) {
  Json value;
  size_t consumed = 0;
  if (!parse_some(text.data(), text.data() + text.size(), &value, &consumed)) {
    throw JsonError("incomplete json");
  }
  return value;
}

}  // namespace hipstore
