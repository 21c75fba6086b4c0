// json.hpp — minimal ordered JSON value for registrar payloads and configs.
//
// The ZooKeeper payloads this daemon writes must be shape-compatible with the
// reference registrar's JSON.stringify output (reference: lib/register.js:140-159,
// 45-75): absent/undefined keys are simply not emitted, and object key order is
// insertion order (deterministic output). This is a from-scratch ~500 LoC JSON
// implementation — no third-party deps.
#pragma once

#include <cassert>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <memory>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace registrar {

class Json;
using JsonArray = std::vector<Json>;
using JsonMember = std::pair<std::string, Json>;

class Json {
 public:
  enum class Type { Null, Bool, Int, Double, String, Array, Object };

  Json() : type_(Type::Null) {}
  Json(std::nullptr_t) : type_(Type::Null) {}
  Json(bool b) : type_(Type::Bool), bool_(b) {}
  Json(int v) : type_(Type::Int), int_(v) {}
  Json(int64_t v) : type_(Type::Int), int_(v) {}
  Json(uint64_t v) : type_(Type::Int), int_(static_cast<int64_t>(v)) {}
  Json(double v) : type_(Type::Double), dbl_(v) {}
  Json(const char* s) : type_(Type::String), str_(s) {}
  Json(std::string s) : type_(Type::String), str_(std::move(s)) {}

  static Json array() {
    Json j;
    j.type_ = Type::Array;
    return j;
  }
  static Json object() {
    Json j;
    j.type_ = Type::Object;
    return j;
  }

  Type type() const { return type_; }
  bool is_null() const { return type_ == Type::Null; }
  bool is_bool() const { return type_ == Type::Bool; }
  bool is_number() const { return type_ == Type::Int || type_ == Type::Double; }
  bool is_int() const { return type_ == Type::Int; }
  bool is_string() const { return type_ == Type::String; }
  bool is_array() const { return type_ == Type::Array; }
  bool is_object() const { return type_ == Type::Object; }

  bool as_bool() const {
    check(Type::Bool);
    return bool_;
  }
  int64_t as_int() const {
    if (type_ == Type::Double) return static_cast<int64_t>(dbl_);
    check(Type::Int);
    return int_;
  }
  double as_double() const {
    if (type_ == Type::Int) return static_cast<double>(int_);
    check(Type::Double);
    return dbl_;
  }
  const std::string& as_string() const {
    check(Type::String);
    return str_;
  }
  const JsonArray& items() const {
    check(Type::Array);
    return arr_;
  }
  JsonArray& items() {
    check(Type::Array);
    return arr_;
  }
  const std::vector<JsonMember>& members() const {
    check(Type::Object);
    return obj_;
  }

  size_t size() const {
    if (type_ == Type::Array) return arr_.size();
    if (type_ == Type::Object) return obj_.size();
    throw std::runtime_error("json: size() on non-container");
  }

  void push_back(Json v) {
    check(Type::Array);
    arr_.push_back(std::move(v));
  }

  // Object access. set() inserts or overwrites, preserving first-insertion
  // position. Setting is the only mutation; absent keys are never emitted
  // (the "dropped undefined key" contract).
  void set(const std::string& key, Json v) {
    check(Type::Object);
    for (auto& m : obj_) {
      if (m.first == key) {
        m.second = std::move(v);
        return;
      }
    }
    obj_.emplace_back(key, std::move(v));
  }

  bool contains(const std::string& key) const {
    if (type_ != Type::Object) return false;
    for (const auto& m : obj_)
      if (m.first == key) return true;
    return false;
  }

  const Json* find(const std::string& key) const {
    if (type_ != Type::Object) return nullptr;
    for (const auto& m : obj_)
      if (m.first == key) return &m.second;
    return nullptr;
  }

  // get(key) — null Json if absent.
  const Json& get(const std::string& key) const {
    static const Json kNull;
    const Json* p = find(key);
    return p ? *p : kNull;
  }

  // Typed getters with defaults, for config reading.
  int64_t get_int(const std::string& key, int64_t dflt) const {
    const Json* p = find(key);
    return (p && p->is_number()) ? p->as_int() : dflt;
  }
  bool get_bool(const std::string& key, bool dflt) const {
    const Json* p = find(key);
    return (p && p->is_bool()) ? p->as_bool() : dflt;
  }
  std::string get_string(const std::string& key, const std::string& dflt) const {
    const Json* p = find(key);
    return (p && p->is_string()) ? p->as_string() : dflt;
  }

  bool operator==(const Json& o) const {
    if (type_ != o.type_) {
      // int/double cross-compare
      if (is_number() && o.is_number()) return as_double() == o.as_double();
      return false;
    }
    switch (type_) {
      case Type::Null:
        return true;
      case Type::Bool:
        return bool_ == o.bool_;
      case Type::Int:
        return int_ == o.int_;
      case Type::Double:
        return dbl_ == o.dbl_;
      case Type::String:
        return str_ == o.str_;
      case Type::Array:
        return arr_ == o.arr_;
      case Type::Object: {
        // order-insensitive deep equality (tests use deepEqual semantics,
        // reference: test/register.test.js:122-130)
        if (obj_.size() != o.obj_.size()) return false;
        for (const auto& m : obj_) {
          const Json* p = o.find(m.first);
          if (!p || !(*p == m.second)) return false;
        }
        return true;
      }
    }
    return false;
  }
  bool operator!=(const Json& o) const { return !(*this == o); }

  std::string dump() const {
    std::string out;
    out.reserve(64);
    dump_to(out);
    return out;
  }

  static Json parse(const std::string& text) {
    Parser p(text);
    Json v = p.parse_value();
    p.skip_ws();
    if (!p.eof()) throw std::runtime_error("json: trailing characters at offset " + std::to_string(p.pos()));
    return v;
  }

 private:
  void check(Type t) const {
    if (type_ != t) throw std::runtime_error("json: wrong type access");
  }

  void dump_to(std::string& out) const {
    switch (type_) {
      case Type::Null:
        out += "null";
        break;
      case Type::Bool:
        out += bool_ ? "true" : "false";
        break;
      case Type::Int:
        out += std::to_string(int_);
        break;
      case Type::Double: {
        if (std::isfinite(dbl_)) {
          if (dbl_ == static_cast<double>(static_cast<int64_t>(dbl_)) && std::fabs(dbl_) < 1e15) {
            out += std::to_string(static_cast<int64_t>(dbl_));
          } else {
            char buf[32];
            snprintf(buf, sizeof(buf), "%.17g", dbl_);
            out += buf;
          }
        } else {
          out += "null";  // JSON has no Inf/NaN; match JSON.stringify
        }
        break;
      }
      case Type::String:
        dump_string(out, str_);
        break;
      case Type::Array: {
        out += '[';
        bool first = true;
        for (const auto& v : arr_) {
          if (!first) out += ',';
          first = false;
          v.dump_to(out);
        }
        out += ']';
        break;
      }
      case Type::Object: {
        out += '{';
        bool first = true;
        for (const auto& m : obj_) {
          if (!first) out += ',';
          first = false;
          dump_string(out, m.first);
          out += ':';
          m.second.dump_to(out);
        }
        out += '}';
        break;
      }
    }
  }

  static void dump_string(std::string& out, const std::string& s) {
    out += '"';
    for (unsigned char c : s) {
      switch (c) {
        case '"':
          out += "\\\"";
          break;
        case '\\':
          out += "\\\\";
          break;
        case '\b':
          out += "\\b";
          break;
        case '\f':
          out += "\\f";
          break;
        case '\n':
          out += "\\n";
          break;
        case '\r':
          out += "\\r";
          break;
        case '\t':
          out += "\\t";
          break;
        default:
          if (c < 0x20) {
            char buf[8];
            snprintf(buf, sizeof(buf), "\\u%04x", c);
            out += buf;
          } else {
            out += static_cast<char>(c);
          }
      }
    }
    out += '"';
  }

  class Parser {
   public:
    explicit Parser(const std::string& text) : s_(text) {}
    size_t pos() const { return i_; }
    bool eof() const { return i_ >= s_.size(); }

    void skip_ws() {
      while (i_ < s_.size()) {
        char c = s_[i_];
        if (c == ' ' || c == '\t' || c == '\n' || c == '\r')
          i_++;
        else
          break;
      }
    }

    Json parse_value() {
      skip_ws();
      if (eof()) fail("unexpected end of input");
      // recursive-descent depth cap: a hostile deeply-nested document must
      // produce a parse error, not a stack overflow (SIGSEGV)
      if (depth_ >= 256) fail("nesting too deep (max 256)");
      char c = s_[i_];
      switch (c) {
        case '{':
          return parse_object();
        case '[':
          return parse_array();
        case '"':
          return Json(parse_string());
        case 't':
          expect("true");
          return Json(true);
        case 'f':
          expect("false");
          return Json(false);
        case 'n':
          expect("null");
          return Json(nullptr);
        default:
          if (c == '-' || (c >= '0' && c <= '9')) return parse_number();
          fail(std::string("unexpected character '") + c + "'");
      }
      return Json();
    }

   private:
    Json parse_number();

    [[noreturn]] void fail(const std::string& msg) {
      throw std::runtime_error("json parse error at offset " + std::to_string(i_) + ": " + msg);
    }

    void expect(const char* lit) {
      size_t n = strlen(lit);
      if (s_.compare(i_, n, lit) != 0) fail(std::string("expected '") + lit + "'");
      i_ += n;
    }

    Json parse_object() {
      i_++;  // '{'
      depth_++;
      Json obj = Json::object();
      skip_ws();
      if (!eof() && s_[i_] == '}') {
        i_++;
        depth_--;
        return obj;
      }
      while (true) {
        skip_ws();
        if (eof() || s_[i_] != '"') fail("expected object key");
        std::string key = parse_string();
        skip_ws();
        if (eof() || s_[i_] != ':') fail("expected ':'");
        i_++;
        obj.set(key, parse_value());
        skip_ws();
        if (eof()) fail("unterminated object");
        if (s_[i_] == ',') {
          i_++;
          continue;
        }
        if (s_[i_] == '}') {
          i_++;
          depth_--;
          return obj;
        }
        fail("expected ',' or '}'");
      }
    }

    Json parse_array() {
      i_++;  // '['
      depth_++;
      Json arr = Json::array();
      skip_ws();
      if (!eof() && s_[i_] == ']') {
        i_++;
        depth_--;
        return arr;
      }
      while (true) {
        arr.push_back(parse_value());
        skip_ws();
        if (eof()) fail("unterminated array");
        if (s_[i_] == ',') {
          i_++;
          continue;
        }
        if (s_[i_] == ']') {
          i_++;
          depth_--;
          return arr;
        }
        fail("expected ',' or ']'");
      }
    }

    std::string parse_string() {
      i_++;  // '"'
      std::string out;
      while (true) {
        if (eof()) fail("unterminated string");
        char c = s_[i_++];
        if (c == '"') return out;
        if (c == '\\') {
          if (eof()) fail("unterminated escape");
          char e = s_[i_++];
          switch (e) {
            case '"':
              out += '"';
              break;
            case '\\':
              out += '\\';
              break;
            case '/':
              out += '/';
              break;
            case 'b':
              out += '\b';
              break;
            case 'f':
              out += '\f';
              break;
            case 'n':
              out += '\n';
              break;
            case 'r':
              out += '\r';
              break;
            case 't':
              out += '\t';
              break;
            case 'u': {
              unsigned cp = parse_hex4();
              if (cp >= 0xD800 && cp <= 0xDBFF) {
                // surrogate pair
                if (i_ + 1 < s_.size() && s_[i_] == '\\' && s_[i_ + 1] == 'u') {
                  i_ += 2;
                  unsigned lo = parse_hex4();
                  if (lo >= 0xDC00 && lo <= 0xDFFF) {
                    cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                  } else {
                    fail("invalid low surrogate");
                  }
                } else {
                  fail("lone high surrogate");
                }
              }
              append_utf8(out, cp);
              break;
            }
            default:
              fail("bad escape");
          }
        } else {
          out += c;
        }
      }
    }

    unsigned parse_hex4() {
      if (i_ + 4 > s_.size()) fail("short \\u escape");
      unsigned v = 0;
      for (int k = 0; k < 4; k++) {
        char c = s_[i_++];
        v <<= 4;
        if (c >= '0' && c <= '9')
          v |= static_cast<unsigned>(c - '0');
        else if (c >= 'a' && c <= 'f')
          v |= static_cast<unsigned>(c - 'a' + 10);
        else if (c >= 'A' && c <= 'F')
          v |= static_cast<unsigned>(c - 'A' + 10);
        else
          fail("bad hex digit");
      }
      return v;
    }

    static void append_utf8(std::string& out, unsigned cp) {
      if (cp < 0x80) {
        out += static_cast<char>(cp);
      } else if (cp < 0x800) {
        out += static_cast<char>(0xC0 | (cp >> 6));
        out += static_cast<char>(0x80 | (cp & 0x3F));
      } else if (cp < 0x10000) {
        out += static_cast<char>(0xE0 | (cp >> 12));
        out += static_cast<char>(0x80 | ((cp >> 6) & 0x3F));
        out += static_cast<char>(0x80 | (cp & 0x3F));
      } else {
        out += static_cast<char>(0xF0 | (cp >> 18));
        out += static_cast<char>(0x80 | ((cp >> 12) & 0x3F));
        out += static_cast<char>(0x80 | ((cp >> 6) & 0x3F));
        out += static_cast<char>(0x80 | (cp & 0x3F));
      }
    }

    const std::string& s_;
    size_t i_ = 0;
    int depth_ = 0;  // current object/array nesting (capped in parse_value)
  };

  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double dbl_ = 0.0;
  std::string str_;
  JsonArray arr_;
  std::vector<JsonMember> obj_;

  friend class Parser;
};

inline Json Json::Parser::parse_number() {
  size_t start = i_;
  if (s_[i_] == '-') i_++;
  bool is_double = false;
  while (i_ < s_.size()) {
    char c = s_[i_];
    if (c >= '0' && c <= '9') {
      i_++;
    } else if (c == '.' || c == 'e' || c == 'E' || c == '+' || c == '-') {
      is_double = true;
      i_++;
    } else {
      break;
    }
  }
  std::string tok = s_.substr(start, i_ - start);
  try {
    if (!is_double) {
      return Json(static_cast<int64_t>(std::stoll(tok)));
    }
    return Json(std::stod(tok));
  } catch (const std::exception&) {
    fail("bad number '" + tok + "'");
  }
}

}  // namespace registrar
