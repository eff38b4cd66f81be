#include "json_schema.h"

#include <string>

namespace hs {

namespace {

std::string schema_err_impl(const Json& sch, const Json& v,
                       const std::string& path) {
  if (!sch.is_object()) return "";
  if (sch.contains("$ref")) return "";
  const Json& typ = sch.at("type");
  auto type_ok = [&](const std::string& t) {
    if (t == "object") return v.is_object();
    if (t == "array") return v.is_array();
    if (t == "string") return v.is_string();
    if (t == "number") return v.is_number();
    if (t == "integer")
      return v.is_number() && v.as_number() == (double)(long long)v.as_number();
    if (t == "boolean") return v.is_bool();
    if (t == "null") return v.is_null();
    return true;
  };
  if (typ.is_string() && !type_ok(typ.as_string()))
    return path + ": expected type " + typ.as_string();
  if (typ.is_array()) {
    bool any = false;
    for (auto& t : typ.arr())
      if (type_ok(t.as_string())) any = true;
    if (!any) return path + ": type not in allowed set";
  }
  if (sch.at("enum").is_array()) {
    bool any = false;
    for (auto& e : sch.at("enum").arr())
      if (e.dump() == v.dump()) any = true;
    if (!any) return path + ": value not in enum";
  }
  if (sch.contains("const") && sch.at("const").dump() != v.dump())
    return path + ": value != const";
  if (v.is_object()) {
    const Json& req = sch.at("required");
    if (req.is_array())
      for (auto& r : req.arr())
        if (!v.contains(r.as_string()))
          return path + "." + r.as_string() + ": required";
    const Json& props = sch.at("properties");
    if (props.is_object()) {
      for (auto& [k, psch] : props.obj())
        if (v.contains(k)) {
          std::string e = schema_err_impl(psch, v.at(k), path + "." + k);
          if (!e.empty()) return e;
        }
      const Json& ap = sch.at("additionalProperties");
      if (ap.is_bool() && !ap.as_bool())
        for (auto& [k, _] : v.obj())
          if (!props.contains(k))
            return path + "." + k + ": additional property not allowed";
    }
  }
  if (v.is_array() && sch.at("items").is_object())
    for (size_t i = 0; i < v.size(); ++i) {
      std::string e = schema_err_impl(sch.at("items"), v.at(i),
                                 path + "[" + std::to_string(i) + "]");
      if (!e.empty()) return e;
    }
  if (v.is_string()) {
    if (sch.contains("minLength") &&
        v.as_string().size() < (size_t)sch.at("minLength").as_int())
      return path + ": shorter than minLength";
    if (sch.contains("maxLength") &&
        v.as_string().size() > (size_t)sch.at("maxLength").as_int())
      return path + ": longer than maxLength";
  }
  if (v.is_number()) {
    if (sch.contains("minimum") &&
        v.as_number() < sch.at("minimum").as_number())
      return path + ": below minimum";
    if (sch.contains("maximum") &&
        v.as_number() > sch.at("maximum").as_number())
      return path + ": above maximum";
  }
  return "";
}



}  // namespace

std::string json_schema_err(const Json& schema, const Json& value,
                            const std::string& path) {
  return schema_err_impl(schema, value, path);
}

}  // namespace hs
