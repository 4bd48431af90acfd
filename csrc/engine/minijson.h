// Minimal JSON parser for the C++ engine (voice configs + safetensors
// headers).  No external deps (this image ships no nlohmann/json).
// Supports the full JSON grammar; numbers are stored as double, object
// member order is not preserved.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace minijson {

struct Value;
using ValuePtr = std::shared_ptr<Value>;

struct Value {
  enum Type { Null, Bool, Number, String, Array, Object } type = Null;
  bool b = false;
  double num = 0.0;
  std::string str;
  std::vector<ValuePtr> arr;
  std::map<std::string, ValuePtr> obj;

  bool is_null() const { return type == Null; }
  double as_num(double dflt = 0.0) const { return type == Number ? num : dflt; }
  long as_int(long dflt = 0) const {
    return type == Number ? (long)num : dflt;
  }
  bool as_bool(bool dflt = false) const { return type == Bool ? b : dflt; }
  const std::string& as_str() const { return str; }
  const Value* get(const std::string& key) const {
    auto it = obj.find(key);
    return it == obj.end() ? nullptr : it->second.get();
  }
  // nested lookup with default
  double num_at(const std::string& k, double dflt) const {
    const Value* v = get(k);
    return v ? v->as_num(dflt) : dflt;
  }
  std::string str_at(const std::string& k, const std::string& dflt) const {
    const Value* v = get(k);
    return v && v->type == String ? v->str : dflt;
  }
};

class Parser {
 public:
  explicit Parser(const std::string& text) : s_(text) {}

  ValuePtr parse() {
    ValuePtr v = parse_value();
    skip_ws();
    if (pos_ != s_.size()) throw std::runtime_error("json: trailing data");
    return v;
  }

 private:
  const std::string& s_;
  size_t pos_ = 0;

  void skip_ws() {
    while (pos_ < s_.size() &&
           (s_[pos_] == ' ' || s_[pos_] == '\t' || s_[pos_] == '\n' ||
            s_[pos_] == '\r'))
      ++pos_;
  }
  char peek() {
    skip_ws();
    if (pos_ >= s_.size()) throw std::runtime_error("json: eof");
    return s_[pos_];
  }
  void expect(char c) {
    if (peek() != c)
      throw std::runtime_error(std::string("json: expected ") + c);
    ++pos_;
  }

  ValuePtr parse_value() {
    char c = peek();
    auto v = std::make_shared<Value>();
    if (c == '{') {
      v->type = Value::Object;
      ++pos_;
      if (peek() == '}') { ++pos_; return v; }
      while (true) {
        std::string key = parse_string_raw();
        expect(':');
        v->obj[key] = parse_value();
        char d = peek();
        ++pos_;
        if (d == '}') break;
        if (d != ',') throw std::runtime_error("json: bad object");
      }
    } else if (c == '[') {
      v->type = Value::Array;
      ++pos_;
      if (peek() == ']') { ++pos_; return v; }
      while (true) {
        v->arr.push_back(parse_value());
        char d = peek();
        ++pos_;
        if (d == ']') break;
        if (d != ',') throw std::runtime_error("json: bad array");
      }
    } else if (c == '"') {
      v->type = Value::String;
      v->str = parse_string_raw();
    } else if (c == 't') {
      require("true"); v->type = Value::Bool; v->b = true;
    } else if (c == 'f') {
      require("false"); v->type = Value::Bool; v->b = false;
    } else if (c == 'n') {
      require("null"); v->type = Value::Null;
    } else {
      v->type = Value::Number;
      size_t end = pos_;
      while (end < s_.size() &&
             (isdigit((unsigned char)s_[end]) || s_[end] == '-' ||
              s_[end] == '+' || s_[end] == '.' || s_[end] == 'e' ||
              s_[end] == 'E'))
        ++end;
      v->num = std::stod(s_.substr(pos_, end - pos_));
      pos_ = end;
    }
    return v;
  }

  void require(const char* word) {
    size_t n = strlen(word);
    if (s_.compare(pos_, n, word) != 0)
      throw std::runtime_error("json: bad literal");
    pos_ += n;
  }

  std::string parse_string_raw() {
    expect('"');
    std::string out;
    while (pos_ < s_.size()) {
      char c = s_[pos_++];
      if (c == '"') return out;
      if (c == '\\') {
        if (pos_ >= s_.size()) break;
        char e = s_[pos_++];
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
            if (pos_ + 4 > s_.size())
              throw std::runtime_error("json: bad \\u");
            unsigned cp = std::stoul(s_.substr(pos_, 4), nullptr, 16);
            pos_ += 4;
            // surrogate pair
            if (cp >= 0xD800 && cp <= 0xDBFF && pos_ + 6 <= s_.size() &&
                s_[pos_] == '\\' && s_[pos_ + 1] == 'u') {
              unsigned lo = std::stoul(s_.substr(pos_ + 2, 4), nullptr, 16);
              if (lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                pos_ += 6;
              }
            }
            // utf-8 encode
            if (cp < 0x80) out += (char)cp;
            else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else if (cp < 0x10000) {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xF0 | (cp >> 18));
              out += (char)(0x80 | ((cp >> 12) & 0x3F));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: throw std::runtime_error("json: bad escape");
        }
      } else {
        out += c;
      }
    }
    throw std::runtime_error("json: unterminated string");
  }
};

inline ValuePtr parse(const std::string& text) { return Parser(text).parse(); }

}  // namespace minijson
