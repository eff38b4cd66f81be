// Minimal JSON value + parser + serializer for the host plane.
//
// Counterpart of the reference's serde_json usage (the reference host plane
// is Rust; this rebuild is native C++ since no Rust toolchain ships in the
// image — SURVEY.md §0 decision point).
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <sstream>
#include <stdexcept>
#include <string>
#include <variant>
#include <vector>

namespace hs {

class Json;
using JsonArray = std::vector<Json>;
// std::map keeps key order stable for deterministic OpenAPI output
using JsonObject = std::map<std::string, Json>;

class Json {
 public:
  using Value = std::variant<std::nullptr_t, bool, double, std::string,
                             JsonArray, JsonObject>;

  Json() : v_(nullptr) {}
  Json(std::nullptr_t) : v_(nullptr) {}
  Json(bool b) : v_(b) {}
  Json(int i) : v_(static_cast<double>(i)) {}
  Json(long i) : v_(static_cast<double>(i)) {}
  Json(size_t i) : v_(static_cast<double>(i)) {}
  Json(double d) : v_(d) {}
  Json(const char* s) : v_(std::string(s)) {}
  Json(std::string s) : v_(std::move(s)) {}
  Json(JsonArray a) : v_(std::move(a)) {}
  Json(JsonObject o) : v_(std::move(o)) {}

  static Json object() { return Json(JsonObject{}); }
  static Json array() { return Json(JsonArray{}); }

  bool is_null() const { return std::holds_alternative<std::nullptr_t>(v_); }
  bool is_bool() const { return std::holds_alternative<bool>(v_); }
  bool is_number() const { return std::holds_alternative<double>(v_); }
  bool is_string() const { return std::holds_alternative<std::string>(v_); }
  bool is_array() const { return std::holds_alternative<JsonArray>(v_); }
  bool is_object() const { return std::holds_alternative<JsonObject>(v_); }

  bool as_bool(bool dflt = false) const {
    return is_bool() ? std::get<bool>(v_) : dflt;
  }
  double as_number(double dflt = 0) const {
    return is_number() ? std::get<double>(v_) : dflt;
  }
  long as_int(long dflt = 0) const {
    return is_number() ? static_cast<long>(std::get<double>(v_)) : dflt;
  }
  const std::string& as_string() const {
    static const std::string empty;
    return is_string() ? std::get<std::string>(v_) : empty;
  }
  std::string as_string(const std::string& dflt) const {
    return is_string() ? std::get<std::string>(v_) : dflt;
  }

  JsonArray& arr() { return std::get<JsonArray>(v_); }
  const JsonArray& arr() const { return std::get<JsonArray>(v_); }
  JsonObject& obj() { return std::get<JsonObject>(v_); }
  const JsonObject& obj() const { return std::get<JsonObject>(v_); }

  bool contains(const std::string& k) const {
    return is_object() && obj().count(k) > 0;
  }
  // object access; creates on mutable access
  Json& operator[](const std::string& k) {
    if (!is_object()) v_ = JsonObject{};
    return obj()[k];
  }
  const Json& at(const std::string& k) const {
    static const Json null_json;
    if (!is_object()) return null_json;
    auto it = obj().find(k);
    return it == obj().end() ? null_json : it->second;
  }
  const Json& at(size_t i) const {
    static const Json null_json;
    if (!is_array() || i >= arr().size()) return null_json;
    return arr()[i];
  }
  void erase(const std::string& k) {
    if (is_object()) obj().erase(k);
  }
  // dotted-path lookup ("modules.llm-gateway.config")
  const Json& path(const std::string& dotted) const;

  void push_back(Json j) {
    if (!is_array()) v_ = JsonArray{};
    arr().push_back(std::move(j));
  }
  size_t size() const {
    if (is_array()) return arr().size();
    if (is_object()) return obj().size();
    return 0;
  }

  std::string dump(int indent = -1) const;
  static Json parse(const std::string& text);  // throws std::runtime_error

  // deep merge: other's keys override, objects merge recursively
  void merge_from(const Json& other);

 private:
  Value v_;
};

}  // namespace hs
