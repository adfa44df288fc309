// JSON5 value model, parser and JSON serializer.
//
// ContainerPilot configs are JSON5 (comments, unquoted keys, trailing
// commas, single quotes); Consul payloads and the /status endpoint are
// plain JSON. One value type serves both.
// Behavioral parity: /root/reference/config/config.go:184-232 (parse +
// pretty syntax errors with line/col caret).
#pragma once

#include <cstdint>
#include <memory>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

namespace cpilot {

class Json;
using JsonArray = std::vector<Json>;
// insertion-ordered object (configs are small; linear lookup is fine)
using JsonObject = std::vector<std::pair<std::string, Json>>;

class Json {
 public:
  enum class Type { Null, Bool, Int, Double, String, Array, Object };

  Json() : type_(Type::Null) {}
  Json(std::nullptr_t) : type_(Type::Null) {}
  Json(bool b) : type_(Type::Bool), bool_(b) {}
  Json(int i) : type_(Type::Int), int_(i) {}
  Json(int64_t i) : type_(Type::Int), int_(i) {}
  Json(double d) : type_(Type::Double), dbl_(d) {}
  Json(const char* s) : type_(Type::String), str_(s) {}
  Json(std::string s) : type_(Type::String), str_(std::move(s)) {}
  Json(JsonArray a) : type_(Type::Array), arr_(std::move(a)) {}
  Json(JsonObject o) : type_(Type::Object), obj_(std::move(o)) {}

  Type type() const { return type_; }
  bool isNull() const { return type_ == Type::Null; }
  bool isBool() const { return type_ == Type::Bool; }
  bool isInt() const { return type_ == Type::Int; }
  bool isDouble() const { return type_ == Type::Double; }
  bool isNumber() const { return isInt() || isDouble(); }
  bool isString() const { return type_ == Type::String; }
  bool isArray() const { return type_ == Type::Array; }
  bool isObject() const { return type_ == Type::Object; }

  bool boolean() const { return bool_; }
  int64_t asInt() const { return isDouble() ? (int64_t)dbl_ : int_; }
  double asDouble() const { return isInt() ? (double)int_ : dbl_; }
  const std::string& str() const { return str_; }
  const JsonArray& array() const { return arr_; }
  JsonArray& array() { return arr_; }
  const JsonObject& object() const { return obj_; }
  JsonObject& object() { return obj_; }

  // object helpers
  const Json* find(const std::string& key) const {
    if (type_ != Type::Object) return nullptr;
    for (auto& kv : obj_)
      if (kv.first == key) return &kv.second;
    return nullptr;
  }
  void set(const std::string& key, Json v) {
    for (auto& kv : obj_) {
      if (kv.first == key) {
        kv.second = std::move(v);
        return;
      }
    }
    obj_.emplace_back(key, std::move(v));
  }

  // Serialize as strict JSON (Infinity/NaN become null).
  std::string dump() const;

 private:
  Type type_;
  bool bool_ = false;
  int64_t int_ = 0;
  double dbl_ = 0;
  std::string str_;
  JsonArray arr_;
  JsonObject obj_;
};

struct JsonParseError : std::runtime_error {
  JsonParseError(std::string msg, size_t off)
      : std::runtime_error(std::move(msg)), offset(off) {}
  size_t offset;  // byte offset into the input
};

// Parse a JSON5 document. Throws JsonParseError.
Json parseJson5(const std::string& text);

// Format a parse error the way the reference does: line:col plus the
// offending line with a caret (config/config.go:198-232).
std::string formatParseError(const std::string& text, const JsonParseError& err);

}  // namespace cpilot
