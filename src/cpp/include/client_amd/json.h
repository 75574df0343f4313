// Minimal JSON value + parser + writer for the KServe-v2 wire schema.
// The reference uses TritonJson (rapidjson) from its common repo; this
// environment vendors nothing, so the codec is written here directly.
// Supports: null, bool, int64/double, string, array, object. Parsing is
// a single-pass recursive descent over a string_view.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

namespace client_amd {

class Json;
using JsonArray = std::vector<Json>;
using JsonObject = std::map<std::string, Json>;

class Json {
 public:
  enum class Type { Null, Bool, Int, Double, String, Array, Object };

  Json() : type_(Type::Null) {}
  Json(bool b) : type_(Type::Bool), bool_(b) {}
  Json(int64_t i) : type_(Type::Int), int_(i) {}
  Json(int i) : type_(Type::Int), int_(i) {}
  Json(uint64_t i) : type_(Type::Int), int_((int64_t)i) {}
  Json(double d) : type_(Type::Double), dbl_(d) {}
  Json(const char* s) : type_(Type::String), str_(s) {}
  Json(std::string s) : type_(Type::String), str_(std::move(s)) {}
  Json(JsonArray a) : type_(Type::Array), arr_(std::move(a)) {}
  Json(JsonObject o) : type_(Type::Object), obj_(std::move(o)) {}

  Type type() const { return type_; }
  bool IsNull() const { return type_ == Type::Null; }
  bool IsObject() const { return type_ == Type::Object; }
  bool IsArray() const { return type_ == Type::Array; }
  bool IsString() const { return type_ == Type::String; }
  bool IsNumber() const { return type_ == Type::Int || type_ == Type::Double; }
  bool IsBool() const { return type_ == Type::Bool; }

  bool AsBool() const { return bool_; }
  int64_t AsInt() const {
    return type_ == Type::Double ? (int64_t)dbl_ : int_;
  }
  double AsDouble() const { return type_ == Type::Int ? (double)int_ : dbl_; }
  const std::string& AsString() const { return str_; }
  const JsonArray& AsArray() const { return arr_; }
  JsonArray& AsArray() { return arr_; }
  const JsonObject& AsObject() const { return obj_; }
  JsonObject& AsObject() { return obj_; }

  // object access; returns Null json for missing keys
  const Json& operator[](const std::string& key) const {
    static const Json null_json;
    auto it = obj_.find(key);
    return it == obj_.end() ? null_json : it->second;
  }
  Json& Set(const std::string& key, Json v) {
    if (type_ != Type::Object) { type_ = Type::Object; }
    obj_[key] = std::move(v);
    return *this;
  }
  bool Has(const std::string& key) const {
    return type_ == Type::Object && obj_.count(key) > 0;
  }

  std::string Dump() const;
  static Json Parse(const std::string& text);
  static Json Parse(const char* begin, size_t len);

 private:
  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double dbl_ = 0.0;
  std::string str_;
  JsonArray arr_;
  JsonObject obj_;
};

}  // namespace client_amd
