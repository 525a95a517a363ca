// Minimal JSON parser/serializer for the native extender fast path.
//
// Scope: exactly what the kube-scheduler extender wire protocol needs —
// objects, arrays, strings (with escapes incl. \uXXXX), numbers, booleans,
// null. DOM-style into a small variant type. No external dependencies (this
// image has no nlohmann/rapidjson and no network to fetch one).
#pragma once

#include <cmath>
#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace egsjson {

class Value;
using Object = std::map<std::string, Value>;
using Array = std::vector<Value>;

class Value {
 public:
  enum class Type { Null, Bool, Int, Double, String, Array, Object };

  Value() : type_(Type::Null) {}
  Value(bool b) : type_(Type::Bool), bool_(b) {}
  Value(int64_t i) : type_(Type::Int), int_(i) {}
  Value(int i) : type_(Type::Int), int_(i) {}
  Value(double d) : type_(Type::Double), dbl_(d) {}
  Value(const char* s) : type_(Type::String), str_(s) {}
  Value(std::string s) : type_(Type::String), str_(std::move(s)) {}
  Value(Array a) : type_(Type::Array), arr_(std::make_shared<Array>(std::move(a))) {}
  Value(Object o) : type_(Type::Object), obj_(std::make_shared<Object>(std::move(o))) {}

  static Value make_object() { return Value(Object{}); }
  static Value make_array() { return Value(Array{}); }

  Type type() const { return type_; }
  bool is_null() const { return type_ == Type::Null; }
  bool is_object() const { return type_ == Type::Object; }
  bool is_array() const { return type_ == Type::Array; }
  bool is_string() const { return type_ == Type::String; }
  bool is_number() const { return type_ == Type::Int || type_ == Type::Double; }

  bool as_bool() const { return type_ == Type::Bool ? bool_ : false; }
  int64_t as_int() const {
    if (type_ == Type::Int) return int_;
    if (type_ == Type::Double) return static_cast<int64_t>(dbl_);
    return 0;
  }
  double as_double() const {
    if (type_ == Type::Double) return dbl_;
    if (type_ == Type::Int) return static_cast<double>(int_);
    return 0.0;
  }
  const std::string& as_string() const {
    static const std::string kEmpty;
    return type_ == Type::String ? str_ : kEmpty;
  }

  const Array& as_array() const {
    static const Array kEmpty;
    return type_ == Type::Array && arr_ ? *arr_ : kEmpty;
  }
  Array& mutable_array() {
    if (type_ != Type::Array) *this = make_array();
    return *arr_;
  }
  const Object& as_object() const {
    static const Object kEmpty;
    return type_ == Type::Object && obj_ ? *obj_ : kEmpty;
  }
  Object& mutable_object() {
    if (type_ != Type::Object) *this = make_object();
    return *obj_;
  }

  // Path lookup helpers (missing -> Null value).
  const Value& get(const std::string& key) const {
    static const Value kNull;
    if (type_ != Type::Object || !obj_) return kNull;
    auto it = obj_->find(key);
    return it == obj_->end() ? kNull : it->second;
  }
  void set(const std::string& key, Value v) {
    mutable_object()[key] = std::move(v);
  }

 private:
  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double dbl_ = 0.0;
  std::string str_;
  std::shared_ptr<Array> arr_;
  std::shared_ptr<Object> obj_;
};

// ---------------------------------------------------------------- parsing

class ParseError : public std::runtime_error {
 public:
  using std::runtime_error::runtime_error;
};

namespace detail {

// Recursion guard: the parser descends once per nesting level, so without a
// bound a body of 100k '[' characters would overflow the C stack (found by
// adversarial review; Python's json raises RecursionError at a similar
// depth). Kubernetes objects nest ~10 deep.
constexpr int kMaxDepth = 256;

class Parser {
 public:
  Parser(const char* data, size_t len) : p_(data), end_(data + len) {}

  Value parse() {
    skip_ws();
    Value v = parse_value();
    skip_ws();
    if (p_ != end_) throw ParseError("trailing characters after JSON value");
    return v;
  }

 private:
  void skip_ws() {
    while (p_ != end_ && (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' || *p_ == '\r'))
      ++p_;
  }
  char peek() {
    if (p_ == end_) throw ParseError("unexpected end of input");
    return *p_;
  }
  char next() {
    if (p_ == end_) throw ParseError("unexpected end of input");
    return *p_++;
  }
  void expect(const char* lit) {
    while (*lit) {
      if (p_ == end_ || *p_++ != *lit++) throw ParseError("invalid literal");
    }
  }

  Value parse_value() {
    struct DepthGuard {
      int& d;
      explicit DepthGuard(int& d_) : d(d_) {
        if (++d > kMaxDepth) throw ParseError("nesting too deep");
      }
      ~DepthGuard() { --d; }
    } guard(depth_);
    switch (peek()) {
      case '{': return parse_object();
      case '[': return parse_array();
      case '"': return Value(parse_string());
      case 't': expect("true"); return Value(true);
      case 'f': expect("false"); return Value(false);
      case 'n': expect("null"); return Value();
      default: return parse_number();
    }
  }

  Value parse_object() {
    next();  // {
    Object obj;
    skip_ws();
    if (peek() == '}') {
      next();
      return Value(std::move(obj));
    }
    for (;;) {
      skip_ws();
      if (peek() != '"') throw ParseError("expected object key");
      std::string key = parse_string();
      skip_ws();
      if (next() != ':') throw ParseError("expected ':'");
      skip_ws();
      obj.emplace(std::move(key), parse_value());
      skip_ws();
      char c = next();
      if (c == '}') break;
      if (c != ',') throw ParseError("expected ',' or '}'");
    }
    return Value(std::move(obj));
  }

  Value parse_array() {
    next();  // [
    Array arr;
    skip_ws();
    if (peek() == ']') {
      next();
      return Value(std::move(arr));
    }
    for (;;) {
      skip_ws();
      arr.push_back(parse_value());
      skip_ws();
      char c = next();
      if (c == ']') break;
      if (c != ',') throw ParseError("expected ',' or ']'");
    }
    return Value(std::move(arr));
  }

  std::string parse_string() {
    next();  // "
    std::string out;
    for (;;) {
      char c = next();
      if (c == '"') break;
      if (c == '\\') {
        char e = next();
        switch (e) {
          case '"': out += '"'; break;
          case '\\': out += '\\'; break;
          case '/': out += '/'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'u': {
            unsigned cp = parse_hex4();
            if (cp >= 0xD800 && cp <= 0xDBFF) {  // surrogate pair
              if (p_ + 1 < end_ && p_[0] == '\\' && p_[1] == 'u') {
                p_ += 2;
                unsigned lo = parse_hex4();
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
              }
            }
            append_utf8(out, cp);
            break;
          }
          default: throw ParseError("invalid escape");
        }
      } else {
        out += c;
      }
    }
    return out;
  }

  unsigned parse_hex4() {
    unsigned v = 0;
    for (int i = 0; i < 4; ++i) {
      char c = next();
      v <<= 4;
      if (c >= '0' && c <= '9') v |= c - '0';
      else if (c >= 'a' && c <= 'f') v |= c - 'a' + 10;
      else if (c >= 'A' && c <= 'F') v |= c - 'A' + 10;
      else throw ParseError("invalid \\u escape");
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

  Value parse_number() {
    const char* start = p_;
    if (peek() == '-') next();
    bool is_double = false;
    while (p_ != end_) {
      char c = *p_;
      if (c >= '0' && c <= '9') {
        ++p_;
      } else if (c == '.' || c == 'e' || c == 'E' || c == '+' || c == '-') {
        is_double = true;
        ++p_;
      } else {
        break;
      }
    }
    std::string num(start, p_ - start);
    if (num.empty() || num == "-") throw ParseError("invalid number");
    try {
      if (!is_double) return Value(static_cast<int64_t>(std::stoll(num)));
      return Value(std::stod(num));
    } catch (const std::exception&) {
      throw ParseError("number out of range");
    }
  }

  const char* p_;
  const char* end_;
  int depth_ = 0;
};

}  // namespace detail

inline Value parse(const std::string& s) {
  return detail::Parser(s.data(), s.size()).parse();
}
inline Value parse(const char* data, size_t len) {
  return detail::Parser(data, len).parse();
}

// ------------------------------------------------------------- serializing

inline void dump_to(const Value& v, std::string& out) {
  switch (v.type()) {
    case Value::Type::Null: out += "null"; break;
    case Value::Type::Bool: out += v.as_bool() ? "true" : "false"; break;
    case Value::Type::Int: out += std::to_string(v.as_int()); break;
    case Value::Type::Double: {
      double d = v.as_double();
      if (std::isfinite(d)) {
        char buf[32];
        snprintf(buf, sizeof(buf), "%.12g", d);
        out += buf;
      } else {
        out += "null";
      }
      break;
    }
    case Value::Type::String: {
      out += '"';
      for (char c : v.as_string()) {
        switch (c) {
          case '"': out += "\\\""; break;
          case '\\': out += "\\\\"; break;
          case '\n': out += "\\n"; break;
          case '\r': out += "\\r"; break;
          case '\t': out += "\\t"; break;
          case '\b': out += "\\b"; break;
          case '\f': out += "\\f"; break;
          default:
            if (static_cast<unsigned char>(c) < 0x20) {
              char buf[8];
              snprintf(buf, sizeof(buf), "\\u%04x", c);
              out += buf;
            } else {
              out += c;
            }
        }
      }
      out += '"';
      break;
    }
    case Value::Type::Array: {
      out += '[';
      bool first = true;
      for (const auto& e : v.as_array()) {
        if (!first) out += ',';
        first = false;
        dump_to(e, out);
      }
      out += ']';
      break;
    }
    case Value::Type::Object: {
      out += '{';
      bool first = true;
      for (const auto& [k, e] : v.as_object()) {
        if (!first) out += ',';
        first = false;
        dump_to(Value(k), out);
        out += ':';
        dump_to(e, out);
      }
      out += '}';
      break;
    }
  }
}

inline std::string dump(const Value& v) {
  std::string out;
  out.reserve(256);
  dump_to(v, out);
  return out;
}

}  // namespace egsjson
