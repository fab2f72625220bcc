// Minimal JSON for the clawker-amd native runtime (ckrt/ckd).
// We control both ends of every document (spec files written by the Python
// engine, control frames to/from the Python CP client); arbitrary binary
// payloads travel base64-encoded, so this implementation only needs correct
// RFC8259 structure + string escapes. No external deps (no network in the
// build environment to vendor a JSON library).
#pragma once

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace mj {

class Value;
using Object = std::map<std::string, Value>;
using Array = std::vector<Value>;

class Value {
 public:
  enum class Type { Null, Bool, Int, Double, Str, Arr, Obj };

  Value() : type_(Type::Null) {}
  Value(std::nullptr_t) : type_(Type::Null) {}
  Value(bool b) : type_(Type::Bool), b_(b) {}
  Value(int i) : type_(Type::Int), i_(i) {}
  Value(int64_t i) : type_(Type::Int), i_(i) {}
  Value(uint64_t i) : type_(Type::Int), i_(static_cast<int64_t>(i)) {}
  Value(double d) : type_(Type::Double), d_(d) {}
  Value(const char* s) : type_(Type::Str), s_(s) {}
  Value(std::string s) : type_(Type::Str), s_(std::move(s)) {}
  Value(Array a) : type_(Type::Arr), a_(std::move(a)) {}
  Value(Object o) : type_(Type::Obj), o_(std::move(o)) {}

  Type type() const { return type_; }
  bool is_null() const { return type_ == Type::Null; }
  bool is_obj() const { return type_ == Type::Obj; }
  bool is_arr() const { return type_ == Type::Arr; }
  bool is_str() const { return type_ == Type::Str; }
  bool is_num() const { return type_ == Type::Int || type_ == Type::Double; }

  bool as_bool(bool dflt = false) const { return type_ == Type::Bool ? b_ : dflt; }
  int64_t as_int(int64_t dflt = 0) const {
    if (type_ == Type::Int) return i_;
    if (type_ == Type::Double) return static_cast<int64_t>(d_);
    return dflt;
  }
  double as_double(double dflt = 0) const {
    if (type_ == Type::Double) return d_;
    if (type_ == Type::Int) return static_cast<double>(i_);
    return dflt;
  }
  const std::string& as_str() const {
    static const std::string empty;
    return type_ == Type::Str ? s_ : empty;
  }
  const Array& as_arr() const {
    static const Array empty;
    return type_ == Type::Arr ? a_ : empty;
  }
  const Object& as_obj() const {
    static const Object empty;
    return type_ == Type::Obj ? o_ : empty;
  }
  Object& obj() {
    if (type_ != Type::Obj) { type_ = Type::Obj; o_.clear(); }
    return o_;
  }
  Array& arr() {
    if (type_ != Type::Arr) { type_ = Type::Arr; a_.clear(); }
    return a_;
  }

  // object field access (missing -> Null value)
  const Value& operator[](const std::string& k) const {
    static const Value null_v;
    if (type_ != Type::Obj) return null_v;
    auto it = o_.find(k);
    return it == o_.end() ? null_v : it->second;
  }
  Value& set(const std::string& k, Value v) {
    obj()[k] = std::move(v);
    return *this;
  }
  bool has(const std::string& k) const {
    return type_ == Type::Obj && o_.count(k) > 0;
  }

  std::string dump() const {
    std::string out;
    dump_to(out);
    return out;
  }

 private:
  void dump_to(std::string& out) const {
    char buf[32];
    switch (type_) {
      case Type::Null: out += "null"; break;
      case Type::Bool: out += b_ ? "true" : "false"; break;
      case Type::Int:
        snprintf(buf, sizeof buf, "%lld", static_cast<long long>(i_));
        out += buf;
        break;
      case Type::Double:
        snprintf(buf, sizeof buf, "%.17g", d_);
        // keep the double-ness on round trip: "%.17g" renders integral
        // values (incl. -0) without '.', which would re-parse as Int
        if (!strpbrk(buf, ".eEnN")) strcat(buf, ".0");
        out += buf;
        break;
      case Type::Str: dump_str(s_, out); break;
      case Type::Arr: {
        out += '[';
        bool first = true;
        for (const auto& v : a_) {
          if (!first) out += ',';
          first = false;
          v.dump_to(out);
        }
        out += ']';
        break;
      }
      case Type::Obj: {
        out += '{';
        bool first = true;
        for (const auto& kv : o_) {
          if (!first) out += ',';
          first = false;
          dump_str(kv.first, out);
          out += ':';
          kv.second.dump_to(out);
        }
        out += '}';
        break;
      }
    }
  }

  static void dump_str(const std::string& s, std::string& out) {
    out += '"';
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
            out += static_cast<char>(c);
          }
      }
    }
    out += '"';
  }

  Type type_;
  bool b_ = false;
  int64_t i_ = 0;
  double d_ = 0;
  std::string s_;
  Array a_;
  Object o_;
};

class ParseError : public std::runtime_error {
 public:
  explicit ParseError(const std::string& m) : std::runtime_error("json: " + m) {}
};

class Parser {
 public:
  explicit Parser(const std::string& text) : t_(text) {}

  Value parse() {
    Value v = value();
    ws();
    if (pos_ != t_.size()) throw ParseError("trailing data");
    return v;
  }

 private:
  void ws() {
    while (pos_ < t_.size() &&
           (t_[pos_] == ' ' || t_[pos_] == '\t' || t_[pos_] == '\n' || t_[pos_] == '\r'))
      pos_++;
  }
  char peek() {
    if (pos_ >= t_.size()) throw ParseError("unexpected end");
    return t_[pos_];
  }
  char next() {
    char c = peek();
    pos_++;
    return c;
  }
  void expect(char c) {
    if (next() != c) throw ParseError(std::string("expected '") + c + "'");
  }
  bool consume_lit(const char* lit) {
    size_t n = strlen(lit);
    if (t_.compare(pos_, n, lit) == 0) {
      pos_ += n;
      return true;
    }
    return false;
  }

  Value value() {
    ws();
    char c = peek();
    if (c == '{') return object();
    if (c == '[') return array();
    if (c == '"') return Value(string());
    if (c == 't') { if (consume_lit("true")) return Value(true); throw ParseError("bad literal"); }
    if (c == 'f') { if (consume_lit("false")) return Value(false); throw ParseError("bad literal"); }
    if (c == 'n') { if (consume_lit("null")) return Value(nullptr); throw ParseError("bad literal"); }
    return number();
  }

  Value object() {
    expect('{');
    Object o;
    ws();
    if (peek() == '}') { next(); return Value(std::move(o)); }
    while (true) {
      ws();
      std::string k = string();
      ws();
      expect(':');
      o[std::move(k)] = value();
      ws();
      char c = next();
      if (c == '}') break;
      if (c != ',') throw ParseError("expected ',' or '}'");
    }
    return Value(std::move(o));
  }

  Value array() {
    expect('[');
    Array a;
    ws();
    if (peek() == ']') { next(); return Value(std::move(a)); }
    while (true) {
      a.push_back(value());
      ws();
      char c = next();
      if (c == ']') break;
      if (c != ',') throw ParseError("expected ',' or ']'");
    }
    return Value(std::move(a));
  }

  std::string string() {
    expect('"');
    std::string out;
    while (true) {
      char c = next();
      if (c == '"') break;
      if (c == '\\') {
        char e = next();
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
            unsigned cp = hex4();
            if (cp >= 0xD800 && cp <= 0xDBFF) {   // surrogate pair
              if (next() != '\\' || next() != 'u') throw ParseError("bad surrogate");
              unsigned lo = hex4();
              cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
            }
            append_utf8(cp, out);
            break;
          }
          default: throw ParseError("bad escape");
        }
      } else {
        out += c;
      }
    }
    return out;
  }

  unsigned hex4() {
    unsigned v = 0;
    for (int i = 0; i < 4; i++) {
      char c = next();
      v <<= 4;
      if (c >= '0' && c <= '9') v |= c - '0';
      else if (c >= 'a' && c <= 'f') v |= c - 'a' + 10;
      else if (c >= 'A' && c <= 'F') v |= c - 'A' + 10;
      else throw ParseError("bad hex");
    }
    return v;
  }

  static void append_utf8(unsigned cp, std::string& out) {
    if (cp < 0x80) out += static_cast<char>(cp);
    else if (cp < 0x800) {
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

  Value number() {
    size_t start = pos_;
    if (peek() == '-') next();
    bool is_double = false;
    while (pos_ < t_.size()) {
      char c = t_[pos_];
      if ((c >= '0' && c <= '9')) pos_++;
      else if (c == '.' || c == 'e' || c == 'E' || c == '+' || c == '-') {
        is_double = true;
        pos_++;
      } else break;
    }
    std::string tok = t_.substr(start, pos_ - start);
    if (tok.empty() || tok == "-") throw ParseError("bad number");
    if (is_double) return Value(strtod(tok.c_str(), nullptr));
    return Value(static_cast<int64_t>(strtoll(tok.c_str(), nullptr, 10)));
  }

  const std::string& t_;
  size_t pos_ = 0;
};

inline Value parse(const std::string& text) { return Parser(text).parse(); }

}  // namespace mj
