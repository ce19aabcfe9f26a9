// Minimal JSON parser for Parseable's catalog metadata (stream.json +
// manifest.json). Supports the full JSON grammar the serde serializer emits;
// numbers are kept as both double and int64 views.
#pragma once
#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace gpuq {

struct JValue;
using JPtr = std::shared_ptr<JValue>;

struct JValue {
  enum Kind { NUL, BOOL, INT, DBL, STR, ARR, OBJ } kind = NUL;
  bool b = false;
  int64_t i = 0;
  double d = 0;
  std::string s;
  std::vector<JPtr> arr;
  std::map<std::string, JPtr> obj;

  bool has(const std::string& k) const { return kind == OBJ && obj.count(k); }
  const JValue& at(const std::string& k) const {
    auto it = obj.find(k);
    if (it == obj.end()) throw std::runtime_error("json: missing key " + k);
    return *it->second;
  }
  const JValue* get(const std::string& k) const {
    auto it = obj.find(k);
    return it == obj.end() ? nullptr : it->second.get();
  }
  int64_t as_i64() const { return kind == DBL ? (int64_t)d : i; }
  double as_f64() const { return kind == INT ? (double)i : d; }
};

class JsonParser {
 public:
  explicit JsonParser(const std::string& text) : p_(text.data()), end_(p_ + text.size()) {}
  JPtr parse() {
    JPtr v = value();
    ws();
    return v;
  }

 private:
  const char *p_, *end_;
  void ws() { while (p_ < end_ && (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' || *p_ == '\r')) p_++; }
  char peek() {
    ws();
    if (p_ >= end_) throw std::runtime_error("json: eof");
    return *p_;
  }
  void expect(char c) {
    if (peek() != c) throw std::runtime_error(std::string("json: expected ") + c);
    p_++;
  }
  JPtr value() {
    char c = peek();
    auto v = std::make_shared<JValue>();
    if (c == '{') {
      v->kind = JValue::OBJ;
      p_++;
      if (peek() == '}') { p_++; return v; }
      for (;;) {
        std::string k = str();
        expect(':');
        v->obj.emplace(std::move(k), value());
        char n = peek();
        p_++;
        if (n == '}') return v;
        if (n != ',') throw std::runtime_error("json: bad object");
      }
    }
    if (c == '[') {
      v->kind = JValue::ARR;
      p_++;
      if (peek() == ']') { p_++; return v; }
      for (;;) {
        v->arr.push_back(value());
        char n = peek();
        p_++;
        if (n == ']') return v;
        if (n != ',') throw std::runtime_error("json: bad array");
      }
    }
    if (c == '"') { v->kind = JValue::STR; v->s = str(); return v; }
    if (c == 't') { v->kind = JValue::BOOL; v->b = true; p_ += 4; return v; }
    if (c == 'f') { v->kind = JValue::BOOL; v->b = false; p_ += 5; return v; }
    if (c == 'n') { p_ += 4; return v; }
    // number
    const char* start = p_;
    bool is_float = false;
    if (*p_ == '-') p_++;
    while (p_ < end_ && ((*p_ >= '0' && *p_ <= '9') || *p_ == '.' || *p_ == 'e' ||
                         *p_ == 'E' || *p_ == '+' || *p_ == '-')) {
      if (*p_ == '.' || *p_ == 'e' || *p_ == 'E') is_float = true;
      p_++;
    }
    std::string num(start, p_ - start);
    if (is_float) { v->kind = JValue::DBL; v->d = std::stod(num); }
    else { v->kind = JValue::INT; v->i = std::stoll(num); }
    return v;
  }
  std::string str() {
    expect('"');
    std::string out;
    while (p_ < end_) {
      char c = *p_++;
      if (c == '"') return out;
      if (c == '\\') {
        if (p_ >= end_) break;
        char e = *p_++;
        switch (e) {
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (p_ + 4 > end_) throw std::runtime_error("json: bad \\u");
            unsigned cp = (unsigned)std::stoul(std::string(p_, 4), nullptr, 16);
            p_ += 4;
            // minimal UTF-8 encode (no surrogate pairs expected in catalog data)
            if (cp < 0x80) out += (char)cp;
            else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: out += e;
        }
      } else {
        out += c;
      }
    }
    throw std::runtime_error("json: unterminated string");
  }
};

}  // namespace gpuq
