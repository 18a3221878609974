// Minimal JSON value + parser/serializer for the hypha_amd control plane.
// The reference speaks CBOR (crates/messages, ciborium); this framework
// preserves the message SCHEMAS (field names, variants) while framing
// payloads as length-prefixed JSON — simpler to debug, same semantics.
#pragma once

#include <cstdint>
#include <cstring>
#include <map>
#include <memory>
#include <sstream>
#include <stdexcept>
#include <string>
#include <variant>
#include <vector>

namespace hypha {

class Json;
using JsonArray = std::vector<Json>;
using JsonObject = std::map<std::string, Json>;

class Json {
 public:
  using Value = std::variant<std::nullptr_t, bool, double, std::string, JsonArray, JsonObject>;

  Json() : v_(nullptr) {}
  Json(std::nullptr_t) : v_(nullptr) {}
  Json(bool b) : v_(b) {}
  Json(int i) : v_(double(i)) {}
  Json(int64_t i) : v_(double(i)) {}
  Json(uint64_t i) : v_(double(i)) {}
  Json(double d) : v_(d) {}
  Json(const char* s) : v_(std::string(s)) {}
  Json(std::string s) : v_(std::move(s)) {}
  Json(JsonArray a) : v_(std::move(a)) {}
  Json(JsonObject o) : v_(std::move(o)) {}

  bool is_null() const { return std::holds_alternative<std::nullptr_t>(v_); }
  bool is_bool() const { return std::holds_alternative<bool>(v_); }
  bool is_number() const { return std::holds_alternative<double>(v_); }
  bool is_string() const { return std::holds_alternative<std::string>(v_); }
  bool is_array() const { return std::holds_alternative<JsonArray>(v_); }
  bool is_object() const { return std::holds_alternative<JsonObject>(v_); }

  bool as_bool() const { return std::get<bool>(v_); }
  double as_double() const { return std::get<double>(v_); }
  int64_t as_int() const { return (int64_t)std::get<double>(v_); }
  const std::string& as_string() const { return std::get<std::string>(v_); }
  const JsonArray& as_array() const { return std::get<JsonArray>(v_); }
  JsonArray& as_array() { return std::get<JsonArray>(v_); }
  const JsonObject& as_object() const { return std::get<JsonObject>(v_); }
  JsonObject& as_object() { return std::get<JsonObject>(v_); }

  // object access
  const Json& at(const std::string& k) const {
    auto& o = as_object();
    auto it = o.find(k);
    if (it == o.end()) throw std::runtime_error("json: missing key " + k);
    return it->second;
  }
  bool has(const std::string& k) const {
    return is_object() && as_object().count(k) > 0;
  }
  Json& operator[](const std::string& k) {
    if (is_null()) v_ = JsonObject{};
    return std::get<JsonObject>(v_)[k];
  }
  Json get_or(const std::string& k, Json def) const {
    if (has(k)) return at(k);
    return def;
  }

  std::string dump() const {
    std::ostringstream os;
    write(os);
    return os.str();
  }

  static Json parse(const std::string& s) {
    size_t pos = 0;
    Json j = parse_value(s, pos);
    skip_ws(s, pos);
    if (pos != s.size()) throw std::runtime_error("json: trailing data");
    return j;
  }

 private:
  Value v_;

  void write(std::ostringstream& os) const {
    if (is_null()) {
      os << "null";
    } else if (is_bool()) {
      os << (as_bool() ? "true" : "false");
    } else if (is_number()) {
      double d = as_double();
      if (d == (int64_t)d && d >= -9.0e15 && d <= 9.0e15) {
        os << (int64_t)d;
      } else {
        char buf[32];
        snprintf(buf, sizeof buf, "%.17g", d);
        os << buf;
      }
    } else if (is_string()) {
      write_string(os, as_string());
    } else if (is_array()) {
      os << '[';
      bool first = true;
      for (auto& e : as_array()) {
        if (!first) os << ',';
        first = false;
        e.write(os);
      }
      os << ']';
    } else {
      os << '{';
      bool first = true;
      for (auto& [k, val] : as_object()) {
        if (!first) os << ',';
        first = false;
        write_string(os, k);
        os << ':';
        val.write(os);
      }
      os << '}';
    }
  }

  static void write_string(std::ostringstream& os, const std::string& s) {
    os << '"';
    for (unsigned char c : s) {
      switch (c) {
        case '"': os << "\\\""; break;
        case '\\': os << "\\\\"; break;
        case '\n': os << "\\n"; break;
        case '\r': os << "\\r"; break;
        case '\t': os << "\\t"; break;
        default:
          if (c < 0x20) {
            char buf[8];
            snprintf(buf, sizeof buf, "\\u%04x", c);
            os << buf;
          } else {
            os << c;
          }
      }
    }
    os << '"';
  }

  static void skip_ws(const std::string& s, size_t& pos) {
    while (pos < s.size() && (s[pos] == ' ' || s[pos] == '\t' || s[pos] == '\n' || s[pos] == '\r'))
      ++pos;
  }

  static Json parse_value(const std::string& s, size_t& pos) {
    skip_ws(s, pos);
    if (pos >= s.size()) throw std::runtime_error("json: eof");
    char c = s[pos];
    if (c == '{') return parse_object(s, pos);
    if (c == '[') return parse_array(s, pos);
    if (c == '"') return Json(parse_string(s, pos));
    if (c == 't') {
      expect(s, pos, "true");
      return Json(true);
    }
    if (c == 'f') {
      expect(s, pos, "false");
      return Json(false);
    }
    if (c == 'n') {
      expect(s, pos, "null");
      return Json(nullptr);
    }
    return parse_number(s, pos);
  }

  static void expect(const std::string& s, size_t& pos, const char* lit) {
    size_t n = strlen(lit);
    if (s.compare(pos, n, lit) != 0) throw std::runtime_error("json: bad literal");
    pos += n;
  }

  static Json parse_number(const std::string& s, size_t& pos) {
    size_t start = pos;
    if (pos < s.size() && (s[pos] == '-' || s[pos] == '+')) ++pos;
    while (pos < s.size() &&
           (isdigit((unsigned char)s[pos]) || s[pos] == '.' || s[pos] == 'e' || s[pos] == 'E' ||
            s[pos] == '-' || s[pos] == '+'))
      ++pos;
    if (pos == start) throw std::runtime_error("json: bad number");
    return Json(std::stod(s.substr(start, pos - start)));
  }

  static std::string parse_string(const std::string& s, size_t& pos) {
    if (s[pos] != '"') throw std::runtime_error("json: expected string");
    ++pos;
    std::string out;
    while (pos < s.size() && s[pos] != '"') {
      char c = s[pos++];
      if (c == '\\') {
        if (pos >= s.size()) throw std::runtime_error("json: bad escape");
        char e = s[pos++];
        switch (e) {
          case '"': out += '"'; break;
          case '\\': out += '\\'; break;
          case '/': out += '/'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (pos + 4 > s.size()) throw std::runtime_error("json: bad \\u");
            unsigned code = std::stoul(s.substr(pos, 4), nullptr, 16);
            pos += 4;
            // UTF-16 surrogate pair -> full code point
            if (code >= 0xD800 && code <= 0xDBFF) {
              if (pos + 6 <= s.size() && s[pos] == '\\' && s[pos + 1] == 'u') {
                unsigned low = std::stoul(s.substr(pos + 2, 4), nullptr, 16);
                if (low >= 0xDC00 && low <= 0xDFFF) {
                  code = 0x10000 + ((code - 0xD800) << 10) + (low - 0xDC00);
                  pos += 6;
                } else {
                  code = 0xFFFD;  // lone high surrogate
                }
              } else {
                code = 0xFFFD;
              }
            } else if (code >= 0xDC00 && code <= 0xDFFF) {
              code = 0xFFFD;  // lone low surrogate
            }
            if (code < 0x80) {
              out += (char)code;
            } else if (code < 0x800) {
              out += (char)(0xC0 | (code >> 6));
              out += (char)(0x80 | (code & 0x3F));
            } else if (code < 0x10000) {
              out += (char)(0xE0 | (code >> 12));
              out += (char)(0x80 | ((code >> 6) & 0x3F));
              out += (char)(0x80 | (code & 0x3F));
            } else {
              out += (char)(0xF0 | (code >> 18));
              out += (char)(0x80 | ((code >> 12) & 0x3F));
              out += (char)(0x80 | ((code >> 6) & 0x3F));
              out += (char)(0x80 | (code & 0x3F));
            }
            break;
          }
          default: throw std::runtime_error("json: bad escape");
        }
      } else {
        out += c;
      }
    }
    if (pos >= s.size()) throw std::runtime_error("json: unterminated string");
    ++pos;
    return out;
  }

  static Json parse_array(const std::string& s, size_t& pos) {
    ++pos;  // [
    JsonArray arr;
    skip_ws(s, pos);
    if (pos < s.size() && s[pos] == ']') {
      ++pos;
      return Json(std::move(arr));
    }
    while (true) {
      arr.push_back(parse_value(s, pos));
      skip_ws(s, pos);
      if (pos >= s.size()) throw std::runtime_error("json: unterminated array");
      if (s[pos] == ',') {
        ++pos;
      } else if (s[pos] == ']') {
        ++pos;
        break;
      } else {
        throw std::runtime_error("json: bad array");
      }
    }
    return Json(std::move(arr));
  }

  static Json parse_object(const std::string& s, size_t& pos) {
    ++pos;  // {
    JsonObject obj;
    skip_ws(s, pos);
    if (pos < s.size() && s[pos] == '}') {
      ++pos;
      return Json(std::move(obj));
    }
    while (true) {
      skip_ws(s, pos);
      std::string key = parse_string(s, pos);
      skip_ws(s, pos);
      if (pos >= s.size() || s[pos] != ':') throw std::runtime_error("json: expected :");
      ++pos;
      obj[key] = parse_value(s, pos);
      skip_ws(s, pos);
      if (pos >= s.size()) throw std::runtime_error("json: unterminated object");
      if (s[pos] == ',') {
        ++pos;
      } else if (s[pos] == '}') {
        ++pos;
        break;
      } else {
        throw std::runtime_error("json: bad object");
      }
    }
    return Json(std::move(obj));
  }
};

}  // namespace hypha
