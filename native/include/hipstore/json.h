// Minimal JSON value/parser/serializer for hipstored's RPC plane.
//
// hipstored speaks the same JSON-RPC 2.0 wire protocol as the reference's
// SPDK daemon (reference lib/jsonrpc/jsonrpc_server.c), so the Go/Python
// clients (reference pkg/spdk/client.go) work unchanged. No third-party
// JSON library is assumed in this image; this header is the whole story.

#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace hipstore {

class Json;
using JsonArray = std::vector<Json>;
// std::map keeps object keys ordered -> deterministic serialization.
using JsonObject = std::map<std::string, Json>;

class Json {
 public:
  enum class Type { Null, Bool, Int, Double, String, Array, Object };

  Json() : type_(Type::Null) {}
  Json(std::nullptr_t) : type_(Type::Null) {}
  Json(bool b) : type_(Type::Bool), bool_(b) {}
  Json(int v) : type_(Type::Int), int_(v) {}
  Json(int64_t v) : type_(Type::Int), int_(v) {}
  Json(uint64_t v) : type_(Type::Int), int_(static_cast<int64_t>(v)) {}
  Json(double v) : type_(Type::Double), double_(v) {}
  Json(const char* s) : type_(Type::String), str_(s) {}
  Json(std::string s) : type_(Type::String), str_(std::move(s)) {}
  Json(JsonArray a) : type_(Type::Array), arr_(std::move(a)) {}
  Json(JsonObject o) : type_(Type::Object), obj_(std::move(o)) {}

  Type type() const { return type_; }
  bool is_null() const { return type_ == Type::Null; }
  bool is_object() const { return type_ == Type::Object; }
  bool is_array() const { return type_ == Type::Array; }
  bool is_string() const { return type_ == Type::String; }
  bool is_number() const { return type_ == Type::Int || type_ == Type::Double; }
  bool is_int() const { return type_ == Type::Int; }
  bool is_bool() const { return type_ == Type::Bool; }

  bool as_bool() const { check(Type::Bool); return bool_; }
  int64_t as_int() const {
    if (type_ == Type::Double) return static_cast<int64_t>(double_);
    check(Type::Int);
    return int_;
  }
  double as_double() const {
    if (type_ == Type::Int) return static_cast<double>(int_);
    check(Type::Double);
    return double_;
  }
  const std::string& as_string() const { check(Type::String); return str_; }
  const JsonArray& as_array() const { check(Type::Array); return arr_; }
  JsonArray& as_array() { check(Type::Array); return arr_; }
  const JsonObject& as_object() const { check(Type::Object); return obj_; }
  JsonObject& as_object() { check(Type::Object); return obj_; }

  // Object convenience: get(key) -> nullptr when absent.
  const Json* get(const std::string& key) const {
    if (type_ != Type::Object) return nullptr;
    auto it = obj_.find(key);
    return it == obj_.end() ? nullptr : &it->second;
  }
  // Typed getters with defaults, for RPC params.
  std::string get_string(const std::string& key, const std::string& dflt = "") const {
    const Json* v = get(key);
    return (v && v->is_string()) ? v->as_string() : dflt;
  }
  int64_t get_int(const std::string& key, int64_t dflt = 0) const {
    const Json* v = get(key);
    return (v && v->is_number()) ? v->as_int() : dflt;
  }
  bool get_bool(const std::string& key, bool dflt = false) const {
    const Json* v = get(key);
    return (v && v->is_bool()) ? v->as_bool() : dflt;
  }
  bool has(const std::string& key) const { return get(key) != nullptr; }

  std::string dump() const;

  // Parse exactly one JSON value from [begin, end); on success sets
  // *consumed to the bytes eaten (incl. trailing whitespace) and returns
  // true. Returns false when the buffer holds only an incomplete value
  // (caller should read more bytes). Throws JsonError on malformed input.
  static bool parse_some(const char* begin, const char* end, Json* out,
                         size_t* consumed);
  static Json parse(const std::string& text);

 private:
  void check(Type t) const {
    if (type_ != t) throw std::runtime_error("json: wrong type access");
  }

  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double double_ = 0;
  std::string str_;
  JsonArray arr_;
  JsonObject obj_;
};

class JsonError : public std::runtime_error {
 public:
  explicit JsonError(const std::string& what) : std::runtime_error(what) {}
};

}  // namespace hipstore
