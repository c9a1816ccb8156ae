/* json.hpp — minimal recursive-descent JSON parser (header-only).
 * Covers the engine-config / index-params JSON the Gamma C ABI receives
 * (reference: gamma_api.cc:36-70, gamma_index_ivfpq.h:1065 ModelParams
 * Parse). Written from scratch; not a general-purpose library. */
#pragma once
#include <cctype>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <string>
#include <vector>

namespace gjson {

struct Value {
  enum Type { NUL, BOOL, NUM, STR, ARR, OBJ } type = NUL;
  bool b = false;
  double num = 0;
  std::string str;
  std::vector<Value> arr;
  std::map<std::string, Value> obj;

  bool has(const std::string &k) const { return obj.count(k) > 0; }
  const Value *get(const std::string &k) const {
    auto it = obj.find(k);
    return it == obj.end() ? nullptr : &it->second;
  }
  bool get_int(const std::string &k, int &out) const {
    const Value *v = get(k);
    if (!v || v->type != NUM) return false;
    out = (int)v->num;
    return true;
  }
  bool get_str(const std::string &k, std::string &out) const {
    const Value *v = get(k);
    if (!v || v->type != STR) return false;
    out = v->str;
    return true;
  }
};

class Parser {
 public:
  bool parse(const char *s, size_t n, Value &out) {
    p_ = s; end_ = s + n;
    skip();
    if (!value(out)) return false;
    skip();
    return p_ == end_;
  }

 private:
  const char *p_, *end_;
  void skip() {
    while (p_ < end_ && (*p_ == ' ' || *p_ == '\t' || *p_ == '\n' ||
                         *p_ == '\r'))
      p_++;
  }
  bool lit(const char *s, size_t n) {
    if ((size_t)(end_ - p_) < n || strncmp(p_, s, n)) return false;
    p_ += n;
    return true;
  }
  bool value(Value &v) {
    if (p_ >= end_) return false;
    switch (*p_) {
      case '{': return object(v);
      case '[': return array(v);
      case '"': v.type = Value::STR; return string(v.str);
      case 't': v.type = Value::BOOL; v.b = true; return lit("true", 4);
      case 'f': v.type = Value::BOOL; v.b = false; return lit("false", 5);
      case 'n': v.type = Value::NUL; return lit("null", 4);
      default: return number(v);
    }
  }
  bool string(std::string &out) {
    if (*p_ != '"') return false;
    p_++;
    out.clear();
    while (p_ < end_ && *p_ != '"') {
      if (*p_ == '\\' && p_ + 1 < end_) {
        p_++;
        switch (*p_) {
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'u': {
            if (end_ - p_ < 5) return false;
            char hex[5] = {p_[1], p_[2], p_[3], p_[4], 0};
            unsigned cp = (unsigned)strtoul(hex, nullptr, 16);
            /* BMP only, encoded as UTF-8 */
            if (cp < 0x80) out += (char)cp;
            else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            p_ += 4;
            break;
          }
          default: out += *p_;
        }
        p_++;
      } else {
        out += *p_++;
      }
    }
    if (p_ >= end_) return false;
    p_++; /* closing quote */
    return true;
  }
  bool number(Value &v) {
    char *e = nullptr;
    v.num = strtod(p_, &e);
    if (e == p_ || e > end_) return false;
    v.type = Value::NUM;
    p_ = e;
    return true;
  }
  bool array(Value &v) {
    v.type = Value::ARR;
    p_++;
    skip();
    if (p_ < end_ && *p_ == ']') { p_++; return true; }
    for (;;) {
      Value item;
      skip();
      if (!value(item)) return false;
      v.arr.push_back(std::move(item));
      skip();
      if (p_ >= end_) return false;
      if (*p_ == ',') { p_++; continue; }
      if (*p_ == ']') { p_++; return true; }
      return false;
    }
  }
  bool object(Value &v) {
    v.type = Value::OBJ;
    p_++;
    skip();
    if (p_ < end_ && *p_ == '}') { p_++; return true; }
    for (;;) {
      skip();
      std::string key;
      if (p_ >= end_ || !string(key)) return false;
      skip();
      if (p_ >= end_ || *p_ != ':') return false;
      p_++;
      skip();
      Value item;
      if (!value(item)) return false;
      v.obj[key] = std::move(item);
      skip();
      if (p_ >= end_) return false;
      if (*p_ == ',') { p_++; continue; }
      if (*p_ == '}') { p_++; return true; }
      return false;
    }
  }
};

inline bool parse(const std::string &s, Value &out) {
  return Parser().parse(s.data(), s.size(), out);
}

/* tiny writer for status/config responses */
inline std::string escape(const std::string &s) {
  std::string o;
  for (char c : s) {
    if (c == '"' || c == '\\') { o += '\\'; o += c; }
    else if (c == '\n') o += "\\n";
    else o += c;
  }
  return o;
}

}  // namespace gjson
