// Minimal JSON parser — enough for HF config.json and safetensors headers.
// Objects, arrays, strings (with \u escapes collapsed naively), numbers,
// booleans, null.  No external deps; header-only.
#pragma once
#include <cstdint>
#include <cstdlib>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace minijson {

struct Value;
using ValuePtr = std::shared_ptr<Value>;

struct Value {
  enum Kind { Null, Bool, Num, Str, Arr, Obj } kind = Null;
  bool b = false;
  double num = 0;
  std::string str;
  std::vector<ValuePtr> arr;
  std::vector<std::pair<std::string, ValuePtr>> obj;  // preserves order

  bool is_null() const { return kind == Null; }
  const ValuePtr get(const std::string &k) const {
    if (kind != Obj) return nullptr;
    for (auto &p : obj)
      if (p.first == k) return p.second;
    return nullptr;
  }
  double num_or(double d) const { return kind == Num ? num : d; }
  bool bool_or(bool d) const { return kind == Bool ? b : d; }
  const std::string &str_or(const std::string &d) const {
    return kind == Str ? str : d;
  }
};

class Parser {
 public:
  explicit Parser(const char *s, size_t n) : p_(s), end_(s + n) {}
  ValuePtr parse() {
    skip_ws();
    ValuePtr v = parse_value();
    return v;
  }

 private:
  const char *p_, *end_;
  [[noreturn]] void fail(const char *msg) {
    throw std::runtime_error(std::string("json: ") + msg);
  }
  void skip_ws() {
    while (p_ < end_ && (*p_ == ' ' || *p_ == '\n' || *p_ == '\t' || *p_ == '\r'))
      ++p_;
  }
  char peek() {
    if (p_ >= end_) fail("unexpected end");
    return *p_;
  }
  char next() {
    if (p_ >= end_) fail("unexpected end");
    return *p_++;
  }
  ValuePtr parse_value() {
    skip_ws();
    char c = peek();
    switch (c) {
      case '{': return parse_obj();
      case '[': return parse_arr();
      case '"': {
        auto v = std::make_shared<Value>();
        v->kind = Value::Str;
        v->str = parse_string();
        return v;
      }
      case 't': case 'f': {
        auto v = std::make_shared<Value>();
        v->kind = Value::Bool;
        if (c == 't') { expect("true"); v->b = true; }
        else { expect("false"); v->b = false; }
        return v;
      }
      case 'n': {
        expect("null");
        return std::make_shared<Value>();
      }
      default: return parse_num();
    }
  }
  void expect(const char *lit) {
    for (const char *q = lit; *q; ++q)
      if (next() != *q) fail("bad literal");
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
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {  // keep raw codepoint bytes as '?' for BMP escapes
            char hex[5] = {0};
            for (int i = 0; i < 4; ++i) hex[i] = next();
            unsigned cp = (unsigned)strtoul(hex, nullptr, 16);
            if (cp < 0x80) out += (char)cp;
            else out += '?';
            break;
          }
          default: out += e; break;
        }
      } else {
        out += c;
      }
    }
    return out;
  }
  ValuePtr parse_num() {
    const char *start = p_;
    while (p_ < end_ && (*p_ == '-' || *p_ == '+' || *p_ == '.' ||
                         *p_ == 'e' || *p_ == 'E' || (*p_ >= '0' && *p_ <= '9')))
      ++p_;
    if (p_ == start) fail("bad number");
    auto v = std::make_shared<Value>();
    v->kind = Value::Num;
    v->num = strtod(std::string(start, p_).c_str(), nullptr);
    return v;
  }
  ValuePtr parse_obj() {
    next();  // '{'
    auto v = std::make_shared<Value>();
    v->kind = Value::Obj;
    skip_ws();
    if (peek() == '}') { next(); return v; }
    while (true) {
      skip_ws();
      std::string key = parse_string();
      skip_ws();
      if (next() != ':') fail("expected ':'");
      v->obj.emplace_back(key, parse_value());
      skip_ws();
      char c = next();
      if (c == '}') break;
      if (c != ',') fail("expected ',' or '}'");
    }
    return v;
  }
  ValuePtr parse_arr() {
    next();  // '['
    auto v = std::make_shared<Value>();
    v->kind = Value::Arr;
    skip_ws();
    if (peek() == ']') { next(); return v; }
    while (true) {
      v->arr.push_back(parse_value());
      skip_ws();
      char c = next();
      if (c == ']') break;
      if (c != ',') fail("expected ',' or ']'");
    }
    return v;
  }
};

inline ValuePtr parse(const std::string &s) {
  Parser p(s.data(), s.size());
  return p.parse();
}

}  // namespace minijson
