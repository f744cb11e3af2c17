// Minimal JSON parser/serializer for the operator (no third-party JSON
// library ships in the build image). Supports the subset the Kubernetes
// API uses: objects, arrays, strings (with escapes), numbers, bool, null.
#pragma once

#include <map>
#include <memory>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace psjson {

class Value;
using ValuePtr = std::shared_ptr<Value>;

enum class Type { Null, Bool, Number, String, Array, Object };

class Value {
 public:
  Type type = Type::Null;
  bool b = false;
  double num = 0;
  std::string str;
  std::vector<ValuePtr> arr;
  std::map<std::string, ValuePtr> obj;

  static ValuePtr make(Type t) {
    auto v = std::make_shared<Value>();
    v->type = t;
    return v;
  }
  static ValuePtr of(const std::string& s) {
    auto v = make(Type::String);
    v->str = s;
    return v;
  }
  static ValuePtr of(const char* s) { return of(std::string(s)); }
  static ValuePtr of(double d) {
    auto v = make(Type::Number);
    v->num = d;
    return v;
  }
  static ValuePtr of(int i) { return of((double)i); }
  static ValuePtr of(bool x) {
    auto v = make(Type::Bool);
    v->b = x;
    return v;
  }
  static ValuePtr object() { return make(Type::Object); }
  static ValuePtr array() { return make(Type::Array); }

  bool is_object() const { return type == Type::Object; }
  bool is_array() const { return type == Type::Array; }

  // navigation helpers (nullptr-safe)
  ValuePtr get(const std::string& key) const {
    if (type != Type::Object) return nullptr;
    auto it = obj.find(key);
    return it == obj.end() ? nullptr : it->second;
  }
  std::string get_str(const std::string& key,
                      const std::string& dflt = "") const {
    auto v = get(key);
    return (v && v->type == Type::String) ? v->str : dflt;
  }
  double get_num(const std::string& key, double dflt = 0) const {
    auto v = get(key);
    return (v && v->type == Type::Number) ? v->num : dflt;
  }
  bool get_bool(const std::string& key, bool dflt = false) const {
    auto v = get(key);
    return (v && v->type == Type::Bool) ? v->b : dflt;
  }
  void set(const std::string& key, ValuePtr v) { obj[key] = v; }
  void set(const std::string& key, const std::string& s) { obj[key] = of(s); }
  void set(const std::string& key, const char* s) { obj[key] = of(s); }
  void set(const std::string& key, double d) { obj[key] = of(d); }
  void set(const std::string& key, int i) { obj[key] = of(i); }
  void set(const std::string& key, bool x) { obj[key] = of(x); }
  void push(ValuePtr v) { arr.push_back(v); }
};

// ---------------------------------------------------------------------------
class Parser {
 public:
  explicit Parser(const std::string& s) : s_(s) {}

  ValuePtr parse() {
    skip_ws();
    ValuePtr v = parse_value();
    return v;
  }

 private:
  const std::string& s_;
  size_t i_ = 0;

  [[noreturn]] void fail(const std::string& msg) {
    throw std::runtime_error("json parse error at " + std::to_string(i_) +
                             ": " + msg);
  }
  char peek() {
    if (i_ >= s_.size()) fail("eof");
    return s_[i_];
  }
  char next() {
    char c = peek();
    i_++;
    return c;
  }
  void skip_ws() {
    while (i_ < s_.size() &&
           (s_[i_] == ' ' || s_[i_] == '\t' || s_[i_] == '\n' ||
            s_[i_] == '\r'))
      i_++;
  }
  bool consume(const std::string& lit) {
    if (s_.compare(i_, lit.size(), lit) == 0) {
      i_ += lit.size();
      return true;
    }
    return false;
  }

  ValuePtr parse_value() {
    skip_ws();
    char c = peek();
    if (c == '{') return parse_object();
    if (c == '[') return parse_array();
    if (c == '"') return Value::of(parse_string());
    if (consume("true")) return Value::of(true);
    if (consume("false")) return Value::of(false);
    if (consume("null")) return Value::make(Type::Null);
    return parse_number();
  }

  std::string parse_string() {
    if (next() != '"') fail("expected string");
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
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'u': {
            if (i_ + 4 > s_.size()) fail("bad \\u");
            unsigned cp = std::stoul(s_.substr(i_, 4), nullptr, 16);
            i_ += 4;
            // UTF-8 encode the BMP code point (surrogates passed through
            // naively; the k8s API rarely emits them)
            if (cp < 0x80) {
              out += (char)cp;
            } else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: fail("bad escape");
        }
      } else {
        out += c;
      }
    }
    return out;
  }

  ValuePtr parse_number() {
    size_t start = i_;
    while (i_ < s_.size() &&
           (isdigit((unsigned char)s_[i_]) || s_[i_] == '-' ||
            s_[i_] == '+' || s_[i_] == '.' || s_[i_] == 'e' ||
            s_[i_] == 'E'))
      i_++;
    if (start == i_) fail("expected number");
    return Value::of(std::stod(s_.substr(start, i_ - start)));
  }

  ValuePtr parse_object() {
    next();  // {
    auto v = Value::object();
    skip_ws();
    if (peek() == '}') {
      next();
      return v;
    }
    while (true) {
      skip_ws();
      std::string key = parse_string();
      skip_ws();
      if (next() != ':') fail("expected :");
      v->obj[key] = parse_value();
      skip_ws();
      char c = next();
      if (c == '}') break;
      if (c != ',') fail("expected , or }");
    }
    return v;
  }

  ValuePtr parse_array() {
    next();  // [
    auto v = Value::array();
    skip_ws();
    if (peek() == ']') {
      next();
      return v;
    }
    while (true) {
      v->arr.push_back(parse_value());
      skip_ws();
      char c = next();
      if (c == ']') break;
      if (c != ',') fail("expected , or ]");
    }
    return v;
  }
};

inline ValuePtr parse(const std::string& s) { return Parser(s).parse(); }

inline void dump_to(const ValuePtr& v, std::ostringstream& o) {
  if (!v) {
    o << "null";
    return;
  }
  switch (v->type) {
    case Type::Null: o << "null"; break;
    case Type::Bool: o << (v->b ? "true" : "false"); break;
    case Type::Number: {
      double d = v->num;
      if (d == (long long)d)
        o << (long long)d;
      else
        o << d;
      break;
    }
    case Type::String: {
      o << '"';
      for (char c : v->str) {
        switch (c) {
          case '"': o << "\\\""; break;
          case '\\': o << "\\\\"; break;
          case '\n': o << "\\n"; break;
          case '\r': o << "\\r"; break;
          case '\t': o << "\\t"; break;
          default:
            if ((unsigned char)c < 0x20) {
              char buf[8];
              snprintf(buf, sizeof(buf), "\\u%04x", c);
              o << buf;
            } else {
              o << c;
            }
        }
      }
      o << '"';
      break;
    }
    case Type::Array: {
      o << '[';
      for (size_t i = 0; i < v->arr.size(); i++) {
        if (i) o << ',';
        dump_to(v->arr[i], o);
      }
      o << ']';
      break;
    }
    case Type::Object: {
      o << '{';
      bool first = true;
      for (auto& kv : v->obj) {
        if (!first) o << ',';
        first = false;
        o << '"' << kv.first << "\":";
        dump_to(kv.second, o);
      }
      o << '}';
      break;
    }
  }
}

inline std::string dump(const ValuePtr& v) {
  std::ostringstream o;
  dump_to(v, o);
  return o.str();
}

}  // namespace psjson
