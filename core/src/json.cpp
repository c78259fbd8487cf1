#include "modelx/json.hpp"

#include <cmath>
#include <cstdio>
#include <cstring>

namespace modelx {
namespace json {

Value& Object::operator[](const std::string& k) {
  for (auto& kv : items_)
    if (kv.first == k) return kv.second;
  items_.emplace_back(k, Value());
  return items_.back().second;
}

const Value* Object::find(const std::string& k) const {
  for (auto& kv : items_)
    if (kv.first == k) return &kv.second;
  return nullptr;
}

Value& Value::set(const std::string& k, Value v) {
  if (kind_ != Kind::Object) {
    kind_ = Kind::Object;
    o_ = std::make_shared<Object>();
  } else if (!o_) {
    o_ = std::make_shared<Object>();
  }
  Value& slot = (*o_)[k];
  slot = std::move(v);
  return slot;
}

static void escape_to(const std::string& s, std::string& out) {
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
          snprintf(buf, sizeof buf, "\\u%04x", c);
          out += buf;
        } else {
          out.push_back(static_cast<char>(c));
        }
    }
  }
  out.push_back('"');
}

void Value::serialize(std::string& out) const {
  switch (kind_) {
    case Kind::Null: out += "null"; break;
    case Kind::Bool: out += b_ ? "true" : "false"; break;
    case Kind::Int: {
      char buf[24];
      snprintf(buf, sizeof buf, "%lld", static_cast<long long>(i_));
      out += buf;
      break;
    }
    case Kind::Double: {
      if (std::isfinite(d_)) {
        char buf[32];
        snprintf(buf, sizeof buf, "%.17g", d_);
        out += buf;
      } else {
        out += "null";
      }
      break;
    }
    case Kind::String: escape_to(s_, out); break;
    case Kind::Array: {
      out.push_back('[');
      bool first = true;
      if (a_)
        for (const auto& v : *a_) {
          if (!first) out.push_back(',');
          first = false;
          v.serialize(out);
        }
      out.push_back(']');
      break;
    }
    case Kind::Object: {
      out.push_back('{');
      bool first = true;
      if (o_)
        for (const auto& kv : *o_) {
          if (!first) out.push_back(',');
          first = false;
          escape_to(kv.first, out);
          out.push_back(':');
          kv.second.serialize(out);
        }
      out.push_back('}');
      break;
    }
  }
}

namespace {

struct Parser {
  const char* p;
  const char* end;

  [[noreturn]] void fail(const char* msg) {
    throw std::runtime_error(std::string("json: ") + msg);
  }
  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  }
  char peek() {
    if (p >= end) fail("unexpected end");
    return *p;
  }
  void expect(char c) {
    if (p >= end || *p != c) fail("unexpected character");
    ++p;
  }

  Value parse_value() {
    skip_ws();
    switch (peek()) {
      case '{': return parse_object();
      case '[': return parse_array();
      case '"': return Value(parse_string());
      case 't':
        if (end - p >= 4 && memcmp(p, "true", 4) == 0) {
          p += 4;
          return Value(true);
        }
        fail("bad literal");
      case 'f':
        if (end - p >= 5 && memcmp(p, "false", 5) == 0) {
          p += 5;
          return Value(false);
        }
        fail("bad literal");
      case 'n':
        if (end - p >= 4 && memcmp(p, "null", 4) == 0) {
          p += 4;
          return Value(nullptr);
        }
        fail("bad literal");
      default: return parse_number();
    }
  }

  Value parse_object() {
    expect('{');
    Object obj;
    skip_ws();
    if (peek() == '}') {
      ++p;
      return Value(std::move(obj));
    }
    while (true) {
      skip_ws();
      std::string key = parse_string();
      skip_ws();
      expect(':');
      obj[key] = parse_value();
      skip_ws();
      char c = peek();
      if (c == ',') {
        ++p;
        continue;
      }
      if (c == '}') {
        ++p;
        break;
      }
      fail("expected , or }");
    }
    return Value(std::move(obj));
  }

  Value parse_array() {
    expect('[');
    Array arr;
    skip_ws();
    if (peek() == ']') {
      ++p;
      return Value(std::move(arr));
    }
    while (true) {
      arr.push_back(parse_value());
      skip_ws();
      char c = peek();
      if (c == ',') {
        ++p;
        continue;
      }
      if (c == ']') {
        ++p;
        break;
      }
      fail("expected , or ]");
    }
    return Value(std::move(arr));
  }

  std::string parse_string() {
    expect('"');
    std::string out;
    while (true) {
      if (p >= end) fail("unterminated string");
      char c = *p++;
      if (c == '"') break;
      if (c == '\\') {
        if (p >= end) fail("bad escape");
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
            if (end - p < 4) fail("bad \\u escape");
            unsigned cp = 0;
            for (int i = 0; i < 4; i++) {
              char h = *p++;
              cp <<= 4;
              if (h >= '0' && h <= '9') cp |= h - '0';
              else if (h >= 'a' && h <= 'f') cp |= h - 'a' + 10;
              else if (h >= 'A' && h <= 'F') cp |= h - 'A' + 10;
              else fail("bad hex digit");
            }
            // surrogate pair
            if (cp >= 0xD800 && cp <= 0xDBFF && end - p >= 6 && p[0] == '\\' && p[1] == 'u') {
              p += 2;
              unsigned lo = 0;
              for (int i = 0; i < 4; i++) {
                char h = *p++;
                lo <<= 4;
                if (h >= '0' && h <= '9') lo |= h - '0';
                else if (h >= 'a' && h <= 'f') lo |= h - 'a' + 10;
                else if (h >= 'A' && h <= 'F') lo |= h - 'A' + 10;
                else fail("bad hex digit");
              }
              cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
            }
            // UTF-8 encode
            if (cp < 0x80) {
              out.push_back(static_cast<char>(cp));
            } else if (cp < 0x800) {
              out.push_back(static_cast<char>(0xC0 | (cp >> 6)));
              out.push_back(static_cast<char>(0x80 | (cp & 0x3F)));
            } else if (cp < 0x10000) {
              out.push_back(static_cast<char>(0xE0 | (cp >> 12)));
              out.push_back(static_cast<char>(0x80 | ((cp >> 6) & 0x3F)));
              out.push_back(static_cast<char>(0x80 | (cp & 0x3F)));
            } else {
              out.push_back(static_cast<char>(0xF0 | (cp >> 18)));
              out.push_back(static_cast<char>(0x80 | ((cp >> 12) & 0x3F)));
              out.push_back(static_cast<char>(0x80 | ((cp >> 6) & 0x3F)));
              out.push_back(static_cast<char>(0x80 | (cp & 0x3F)));
            }
            break;
          }
          default: fail("bad escape");
        }
      } else {
        out.push_back(c);
      }
    }
    return out;
  }

  Value parse_number() {
    const char* start = p;
    if (p < end && *p == '-') ++p;
    while (p < end && *p >= '0' && *p <= '9') ++p;
    bool is_double = false;
    if (p < end && *p == '.') {
      is_double = true;
      ++p;
      while (p < end && *p >= '0' && *p <= '9') ++p;
    }
    if (p < end && (*p == 'e' || *p == 'E')) {
      is_double = true;
      ++p;
      if (p < end && (*p == '+' || *p == '-')) ++p;
      while (p < end && *p >= '0' && *p <= '9') ++p;
    }
    if (p == start) fail("bad number");
    std::string tok(start, p);
    if (is_double) return Value(strtod(tok.c_str(), nullptr));
    errno = 0;
    long long v = strtoll(tok.c_str(), nullptr, 10);
    if (errno == ERANGE) return Value(strtod(tok.c_str(), nullptr));
    return Value(static_cast<int64_t>(v));
  }
};

}  // namespace

Value parse(const char* data, size_t len) {
  Parser ps{data, data + len};
  Value v = ps.parse_value();
  ps.skip_ws();
  if (ps.p != ps.end) throw std::runtime_error("json: trailing data");
  return v;
}

Value parse(const std::string& text) { return parse(text.data(), text.size()); }

}  // namespace json
}  // namespace modelx
