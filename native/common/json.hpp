// json.hpp — minimal dynamic JSON value for the MI355X-native gpu-pruner.
//
// Kubernetes objects (incl. the Notebook / InferenceService CRDs) are handled
// as dynamic JSON throughout, per SURVEY.md §2.2: the reference carries 31k
// lines of generated CRD bindings (resources/src/{notebook,inferenceservice}.rs)
// of which only metadata + two patch paths are ever touched.
//
// Self-contained, no external deps. Objects keep sorted key order (std::map),
// which makes serialized output deterministic — useful for tests and for
// Prometheus label rendering.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>
#include <cmath>
#include <cstdio>
#include <cstring>

namespace jsn {

class Value;
using Array = std::vector<Value>;
using Object = std::map<std::string, Value>;

enum class Type : uint8_t { Null, Bool, Int, Double, String, Array, Object };

class Value {
public:
  Value() : type_(Type::Null) {}
  Value(std::nullptr_t) : type_(Type::Null) {}
  Value(bool b) : type_(Type::Bool), bool_(b) {}
  Value(int i) : type_(Type::Int), int_(i) {}
  Value(int64_t i) : type_(Type::Int), int_(i) {}
  Value(uint64_t i) : type_(Type::Int), int_(static_cast<int64_t>(i)) {}
  Value(double d) : type_(Type::Double), dbl_(d) {}
  Value(const char* s) : type_(Type::String), str_(std::make_shared<std::string>(s)) {}
  Value(std::string s) : type_(Type::String), str_(std::make_shared<std::string>(std::move(s))) {}
  Value(Array a) : type_(Type::Array), arr_(std::make_shared<Array>(std::move(a))) {}
  Value(Object o) : type_(Type::Object), obj_(std::make_shared<Object>(std::move(o))) {}

  static Value array() { return Value(Array{}); }
  static Value object() { return Value(Object{}); }

  Type type() const { return type_; }
  bool is_null() const { return type_ == Type::Null; }
  bool is_bool() const { return type_ == Type::Bool; }
  bool is_num() const { return type_ == Type::Int || type_ == Type::Double; }
  bool is_string() const { return type_ == Type::String; }
  bool is_array() const { return type_ == Type::Array; }
  bool is_object() const { return type_ == Type::Object; }

  bool as_bool(bool dflt = false) const { return is_bool() ? bool_ : dflt; }
  int64_t as_int(int64_t dflt = 0) const {
    if (type_ == Type::Int) return int_;
    if (type_ == Type::Double) return static_cast<int64_t>(dbl_);
    return dflt;
  }
  double as_double(double dflt = 0.0) const {
    if (type_ == Type::Double) return dbl_;
    if (type_ == Type::Int) return static_cast<double>(int_);
    return dflt;
  }
  const std::string& as_string() const {
    static const std::string empty;
    return is_string() ? *str_ : empty;
  }
  std::string as_string_or(const std::string& dflt) const {
    return is_string() ? *str_ : dflt;
  }

  // "truthy" in the minijinja sense: used by the query builder for optional
  // template context fields ({% if args.namespace %} semantics).
  bool truthy() const {
    switch (type_) {
      case Type::Null: return false;
      case Type::Bool: return bool_;
      case Type::Int: return int_ != 0;
      case Type::Double: return dbl_ != 0.0;
      case Type::String: return !str_->empty();
      case Type::Array: return !arr_->empty();
      case Type::Object: return !obj_->empty();
    }
    return false;
  }

  // ---- array access ----
  Array& arr() { ensure(Type::Array); return *arr_; }
  const Array& arr() const { ensure(Type::Array); return *arr_; }
  void push_back(Value v) { ensure(Type::Array); arr_->push_back(std::move(v)); }
  size_t size() const {
    if (type_ == Type::Array) return arr_->size();
    if (type_ == Type::Object) return obj_->size();
    return 0;
  }

  // ---- object access ----
  Object& obj() { ensure(Type::Object); return *obj_; }
  const Object& obj() const { ensure(Type::Object); return *obj_; }

  // Mutating index: auto-vivifies nulls into objects (like serde_json::json! building).
  Value& operator[](const std::string& key) {
    if (type_ == Type::Null) { type_ = Type::Object; obj_ = std::make_shared<Object>(); }
    ensure(Type::Object);
    return (*obj_)[key];
  }
  Value& operator[](size_t i) { ensure(Type::Array); return (*arr_)[i]; }
  const Value& operator[](size_t i) const { ensure(Type::Array); return (*arr_)[i]; }

  bool contains(const std::string& key) const {
    return type_ == Type::Object && obj_->count(key) > 0;
  }

  // Const path lookup; returns a shared Null for missing keys / wrong types.
  const Value& get(const std::string& key) const {
    static const Value null_v;
    if (type_ != Type::Object) return null_v;
    auto it = obj_->find(key);
    return it == obj_->end() ? null_v : it->second;
  }
  // Deep path lookup: v.at({"metadata","name"}).
  const Value& at(std::initializer_list<const char*> path) const {
    const Value* cur = this;
    for (const char* k : path) cur = &cur->get(k);
    return *cur;
  }

  bool operator==(const Value& o) const {
    if (type_ != o.type_) {
      // int/double cross-compare
      if (is_num() && o.is_num()) return as_double() == o.as_double();
      return false;
    }
    switch (type_) {
      case Type::Null: return true;
      case Type::Bool: return bool_ == o.bool_;
      case Type::Int: return int_ == o.int_;
      case Type::Double: return dbl_ == o.dbl_;
      case Type::String: return *str_ == *o.str_;
      case Type::Array: return *arr_ == *o.arr_;
      case Type::Object: return *obj_ == *o.obj_;
    }
    return false;
  }
  bool operator!=(const Value& o) const { return !(*this == o); }

  std::string dump(int indent = -1) const {
    std::string out;
    write(out, indent, 0);
    return out;
  }

  // RFC 7386 JSON merge patch (what kube "Merge" PatchParams applies):
  // objects merge recursively, null deletes, everything else replaces.
  void merge_patch(const Value& patch) {
    if (!patch.is_object() || !is_object()) { *this = patch; return; }
    // copy-on-write safety: detach before mutating
    obj_ = std::make_shared<Object>(*obj_);
    for (const auto& [k, v] : patch.obj()) {
      if (v.is_null()) {
        obj_->erase(k);
      } else if (v.is_object() && contains(k) && (*obj_)[k].is_object()) {
        (*obj_)[k].merge_patch(v);
      } else {
        (*obj_)[k] = v;
      }
    }
  }

private:
  void ensure(Type t) const {
    if (type_ != t) throw std::runtime_error("jsn::Value: wrong type access");
  }

  static void escape_into(std::string& out, const std::string& s) {
    out += '"';
    for (unsigned char c : s) {
      switch (c) {
        case '"': out += "\\\""; break;
        case '\\': out += "\\\\"; break;
        case '\b': out += "\\b"; break;
        case '\f': out += "\\f"; break;
        case '\n': out += "\\n"; break;
        case '\r': out += "\\r"; break;
        case '\t': out += "\\t"; break;
        default:
          if (c < 0x20) {
            char buf[8];
            std::snprintf(buf, sizeof buf, "\\u%04x", c);
            out += buf;
          } else {
            out += static_cast<char>(c);
          }
      }
    }
    out += '"';
  }

  void write(std::string& out, int indent, int depth) const {
    auto nl = [&](int d) {
      if (indent >= 0) {
        out += '\n';
        out.append(static_cast<size_t>(indent) * d, ' ');
      }
    };
    switch (type_) {
      case Type::Null: out += "null"; break;
      case Type::Bool: out += bool_ ? "true" : "false"; break;
      case Type::Int: out += std::to_string(int_); break;
      case Type::Double: {
        if (std::isfinite(dbl_)) {
          char buf[32];
          std::snprintf(buf, sizeof buf, "%.17g", dbl_);
          // trim to shortest round-trip-ish representation
          double rt;
          for (int prec = 1; prec <= 17; prec++) {
            std::snprintf(buf, sizeof buf, "%.*g", prec, dbl_);
            std::sscanf(buf, "%lf", &rt);
            if (rt == dbl_) break;
          }
          out += buf;
        } else {
          out += "null";  // JSON has no NaN/Inf
        }
        break;
      }
      case Type::String: escape_into(out, *str_); break;
      case Type::Array: {
        out += '[';
        bool first = true;
        for (const auto& v : *arr_) {
          if (!first) out += ',';
          first = false;
          nl(depth + 1);
          v.write(out, indent, depth + 1);
        }
        if (!first) nl(depth);
        out += ']';
        break;
      }
      case Type::Object: {
        out += '{';
        bool first = true;
        for (const auto& [k, v] : *obj_) {
          if (!first) out += ',';
          first = false;
          nl(depth + 1);
          escape_into(out, k);
          out += indent >= 0 ? ": " : ":";
          v.write(out, indent, depth + 1);
        }
        if (!first) nl(depth);
        out += '}';
        break;
      }
    }
  }

  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double dbl_ = 0.0;
  std::shared_ptr<std::string> str_;
  std::shared_ptr<Array> arr_;
  std::shared_ptr<Object> obj_;
};

// ------------------------------- parser ------------------------------------

class ParseError : public std::runtime_error {
public:
  ParseError(const std::string& msg, size_t at)
      : std::runtime_error(msg + " at offset " + std::to_string(at)), pos(at) {}
  size_t pos;
};

namespace detail {

class Parser {
public:
  Parser(const char* s, size_t n) : s_(s), n_(n) {}

  Value parse() {
    Value v = value();
    skip_ws();
    if (p_ != n_) throw ParseError("trailing data", p_);
    return v;
  }

private:
  void skip_ws() {
    while (p_ < n_ && (s_[p_] == ' ' || s_[p_] == '\t' || s_[p_] == '\n' || s_[p_] == '\r')) p_++;
  }
  char peek() {
    if (p_ >= n_) throw ParseError("unexpected end of input", p_);
    return s_[p_];
  }
  char next() {
    char c = peek();
    p_++;
    return c;
  }
  void expect(const char* lit) {
    size_t len = std::strlen(lit);
    if (p_ + len > n_ || std::memcmp(s_ + p_, lit, len) != 0)
      throw ParseError(std::string("expected '") + lit + "'", p_);
    p_ += len;
  }

  Value value() {
    skip_ws();
    switch (peek()) {
      case '{': return object();
      case '[': return array();
      case '"': return Value(string());
      case 't': expect("true"); return Value(true);
      case 'f': expect("false"); return Value(false);
      case 'n': expect("null"); return Value(nullptr);
      default: return number();
    }
  }

  Value object() {
    next();  // {
    Object o;
    skip_ws();
    if (peek() == '}') { next(); return Value(std::move(o)); }
    while (true) {
      skip_ws();
      if (peek() != '"') throw ParseError("expected object key", p_);
      std::string key = string();
      skip_ws();
      if (next() != ':') throw ParseError("expected ':'", p_ - 1);
      o[std::move(key)] = value();
      skip_ws();
      char c = next();
      if (c == '}') break;
      if (c != ',') throw ParseError("expected ',' or '}'", p_ - 1);
    }
    return Value(std::move(o));
  }

  Value array() {
    next();  // [
    Array a;
    skip_ws();
    if (peek() == ']') { next(); return Value(std::move(a)); }
    while (true) {
      a.push_back(value());
      skip_ws();
      char c = next();
      if (c == ']') break;
      if (c != ',') throw ParseError("expected ',' or ']'", p_ - 1);
    }
    return Value(std::move(a));
  }

  std::string string() {
    next();  // "
    std::string out;
    while (true) {
      if (p_ >= n_) throw ParseError("unterminated string", p_);
      char c = s_[p_++];
      if (c == '"') break;
      if (c == '\\') {
        if (p_ >= n_) throw ParseError("unterminated escape", p_);
        char e = s_[p_++];
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
            unsigned cp = hex4();
            if (cp >= 0xD800 && cp <= 0xDBFF) {  // surrogate pair
              if (p_ + 1 < n_ && s_[p_] == '\\' && s_[p_ + 1] == 'u') {
                p_ += 2;
                unsigned lo = hex4();
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
              }
            }
            append_utf8(out, cp);
            break;
          }
          default: throw ParseError("bad escape", p_ - 1);
        }
      } else {
        out += c;
      }
    }
    return out;
  }

  unsigned hex4() {
    if (p_ + 4 > n_) throw ParseError("bad \\u escape", p_);
    unsigned v = 0;
    for (int i = 0; i < 4; i++) {
      char c = s_[p_++];
      v <<= 4;
      if (c >= '0' && c <= '9') v |= static_cast<unsigned>(c - '0');
      else if (c >= 'a' && c <= 'f') v |= static_cast<unsigned>(c - 'a' + 10);
      else if (c >= 'A' && c <= 'F') v |= static_cast<unsigned>(c - 'A' + 10);
      else throw ParseError("bad hex digit", p_ - 1);
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

  Value number() {
    size_t start = p_;
    if (peek() == '-') p_++;
    bool is_double = false;
    while (p_ < n_) {
      char c = s_[p_];
      if (c >= '0' && c <= '9') { p_++; }
      else if (c == '.' || c == 'e' || c == 'E' || c == '+' || c == '-') { is_double = true; p_++; }
      else break;
    }
    if (p_ == start) throw ParseError("invalid number", p_);
    std::string tok(s_ + start, p_ - start);
    if (!is_double) {
      errno = 0;
      char* end = nullptr;
      long long v = std::strtoll(tok.c_str(), &end, 10);
      if (errno == 0 && end && *end == '\0') return Value(static_cast<int64_t>(v));
    }
    char* end = nullptr;
    double d = std::strtod(tok.c_str(), &end);
    if (!end || *end != '\0') throw ParseError("invalid number", start);
    return Value(d);
  }

  const char* s_;
  size_t n_;
  size_t p_ = 0;
};

}  // namespace detail

inline Value parse(const std::string& s) { return detail::Parser(s.data(), s.size()).parse(); }
inline Value parse(const char* s) { return detail::Parser(s, std::strlen(s)).parse(); }

}  // namespace jsn
