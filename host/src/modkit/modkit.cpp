#include "modkit.h"

#include <algorithm>
#include <unistd.h>
#include <cstdlib>
#include <fstream>
#include <set>
#include <sstream>

#include "../util/log.h"
#include "../util/yaml.h"

namespace hs {

void respond_problem(ResponseWriter& w, const Problem& p,
                     const std::string& instance) {
  w.respond(p.status, "application/problem+json",
            p.to_json(instance).dump());
}

Json SecurityContext::to_json() const {
  Json j = Json::object();
  j["subject_id"] = subject_id;
  j["subject_type"] = subject_type;
  j["subject_tenant_id"] = tenant_id;
  Json sc = Json::array();
  for (auto& s : scopes) sc.push_back(s);
  j["token_scopes"] = sc;
  return j;
}

SecurityContext SecurityContext::from_json(const Json& j) {
  SecurityContext c;
  c.subject_id = j.at("subject_id").as_string();
  c.subject_type = j.at("subject_type").as_string("user");
  c.tenant_id = j.at("subject_tenant_id").as_string();
  for (auto& s : j.at("token_scopes").is_array() ? j.at("token_scopes").arr()
                                                 : JsonArray{})
    c.scopes.push_back(s.as_string());
  return c;
}

// ----------------------------------------------------------- RestRegistry

static std::vector<std::string> split_path(const std::string& p) {
  std::vector<std::string> out;
  size_t s = 1;   // skip leading '/'
  while (s <= p.size()) {
    size_t e = p.find('/', s);
    out.push_back(p.substr(s, e == std::string::npos ? std::string::npos
                                                     : e - s));
    if (e == std::string::npos) break;
    s = e + 1;
  }
  if (!out.empty() && out.back().empty()) out.pop_back();
  return out;
}

void RestRegistry::register_op(OperationSpec spec, HttpHandler handler) {
  if (!spec.authenticated && !spec.is_public)
    throw std::runtime_error("operation " + spec.method + " " + spec.path +
                             " declares neither .authenticated() nor "
                             ".public() — refusing to register");
  Route r{std::move(spec), std::move(handler), {}};
  r.segments = split_path(r.spec.path);
  routes_.push_back(std::move(r));
}

const Route* RestRegistry::match(const std::string& method,
                                 const std::string& path,
                                 std::map<std::string, std::string>& params,
                                 bool* path_exists) const {
  auto segs = split_path(path);
  if (path_exists) *path_exists = false;
  const Route* best = nullptr;
  for (const auto& r : routes_) {
    const bool wildcard = !r.segments.empty() &&
        r.segments.back().rfind("{*", 0) == 0;
    if (wildcard ? segs.size() < r.segments.size() - 1
                 : r.segments.size() != segs.size())
      continue;
    std::map<std::string, std::string> p;
    bool ok = true;
    const size_t fixed = r.segments.size() - (wildcard ? 1 : 0);
    for (size_t i = 0; i < fixed && ok; ++i) {
      const std::string& rs = r.segments[i];
      if (rs.size() >= 2 && rs.front() == '{' && rs.back() == '}')
        p[rs.substr(1, rs.size() - 2)] = i < segs.size() ? segs[i] : "";
      else if (i >= segs.size() || rs != segs[i]) ok = false;
    }
    if (ok && wildcard) {
      std::string restpath;
      for (size_t i = fixed; i < segs.size(); ++i) {
        if (!restpath.empty()) restpath += '/';
        restpath += segs[i];
      }
      const std::string& w = r.segments.back();
      p[w.substr(2, w.size() - 3)] = restpath;
    }
    if (!ok) continue;
    if (path_exists) *path_exists = true;
    if (r.spec.method == method) {
      params = std::move(p);
      best = &r;
      break;
    }
  }
  return best;
}

Json RestRegistry::build_openapi(const std::string& title,
                                 const std::string& version,
                                 const std::string& description) const {
  Json doc = Json::object();
  doc["openapi"] = "3.1.0";
  Json info = Json::object();
  info["title"] = title;
  info["version"] = version;
  if (!description.empty()) info["description"] = description;
  doc["info"] = info;
  Json paths = Json::object();
  for (const auto& r : routes_) {
    const auto& s = r.spec;
    Json op = Json::object();
    op["operationId"] = s.operation_id.empty()
        ? (s.method + "_" + s.path) : s.operation_id;
    if (!s.summary.empty()) op["summary"] = s.summary;
    if (!s.tags.empty()) {
      Json t = Json::array();
      for (auto& tag : s.tags) t.push_back(tag);
      op["tags"] = t;
    }
    Json responses = Json::object();
    for (auto& [code, desc] : s.responses) {
      Json rr = Json::object();
      rr["description"] = desc;
      if (code == 200 && !s.response_schema.is_null()) {
        Json media = Json::object();
        media["schema"] = s.response_schema;
        Json content = Json::object();
        content[s.sse ? "text/event-stream" : "application/json"] = media;
        rr["content"] = content;
      }
      responses[std::to_string(code)] = rr;
    }
    if (responses.size() == 0) {
      Json rr = Json::object();
      rr["description"] = "OK";
      responses["200"] = rr;
    }
    // standard problem responses (reference .standard_errors())
    for (int code : {400, 401, 403, 404, 429, 500}) {
      std::string k = std::to_string(code);
      if (!responses.contains(k)) {
        Json rr = Json::object();
        rr["description"] = "Problem";
        Json media = Json::object();
        Json ref = Json::object();
        ref["$ref"] = "#/components/schemas/Problem";
        media["schema"] = ref;
        Json content = Json::object();
        content["application/problem+json"] = media;
        rr["content"] = content;
        responses[k] = rr;
      }
    }
    op["responses"] = responses;
    if (!s.odata_filter_fields.empty()) {
      Json ext = Json::object();
      Json fields = Json::array();
      for (auto& f : s.odata_filter_fields) fields.push_back(f);
      ext["allowedFields"] = fields;
      op["x-odata-filter"] = ext;
    }
    if (!s.request_schema.is_null()) {
      Json media = Json::object();
      media["schema"] = s.request_schema;
      Json content = Json::object();
      content["application/json"] = media;
      Json rb = Json::object();
      rb["required"] = true;
      rb["content"] = content;
      op["requestBody"] = rb;
    }
    if (s.authenticated) {
      Json sec = Json::array();
      Json bearer = Json::object();
      bearer["bearerAuth"] = Json::array();
      sec.push_back(bearer);
      op["security"] = sec;
    } else {
      // explicit empty security = deliberately public (the auth stance
      // is always visible in the spec, mirroring the typestate builder)
      op["security"] = Json::array();
    }
    // path params
    Json parms = Json::array();
    for (auto& seg : r.segments) {
      if (seg.size() >= 2 && seg.front() == '{') {
        Json p = Json::object();
        p["name"] = seg.substr(1, seg.size() - 2);
        p["in"] = "path";
        p["required"] = true;
        Json sch = Json::object();
        sch["type"] = "string";
        p["schema"] = sch;
        parms.push_back(p);
      }
    }
    if (parms.size() > 0) op["parameters"] = parms;
    std::string m = s.method;
    std::transform(m.begin(), m.end(), m.begin(), ::tolower);
    paths[s.path][m] = op;
  }
  doc["paths"] = paths;
  Json comp = Json::object();
  Json schemas = Json::object();
  {
    Json prob = Json::object();
    prob["type"] = "object";
    Json props = Json::object();
    for (const char* f : {"type", "title", "detail", "instance", "code"}) {
      Json t = Json::object();
      t["type"] = "string";
      props[f] = t;
    }
    Json st = Json::object();
    st["type"] = "integer";
    props["status"] = st;
    prob["properties"] = props;
    schemas["Problem"] = prob;
  }
  for (auto& [name, sch] : schemas_) schemas[name] = sch;
  Json secSchemes = Json::object();
  Json bearer = Json::object();
  bearer["type"] = "http";
  bearer["scheme"] = "bearer";
  secSchemes["bearerAuth"] = bearer;
  comp["schemas"] = schemas;
  comp["securitySchemes"] = secSchemes;
  doc["components"] = comp;
  return doc;
}

// --------------------------------------------------------- ModuleRegistry

std::vector<std::shared_ptr<Module>> ModuleRegistry::sorted() const {
  std::map<std::string, std::shared_ptr<Module>> by_name;
  for (auto& m : modules_) by_name[m->name()] = m;
  std::vector<std::shared_ptr<Module>> out;
  std::set<std::string> done, visiting;
  std::function<void(const std::shared_ptr<Module>&)> visit =
      [&](const std::shared_ptr<Module>& m) {
        if (done.count(m->name())) return;
        if (visiting.count(m->name()))
          throw std::runtime_error("module dependency cycle at " + m->name());
        visiting.insert(m->name());
        for (auto& d : m->deps()) {
          auto it = by_name.find(d);
          if (it == by_name.end())
            throw std::runtime_error("module " + m->name() +
                                     " depends on missing module " + d);
          visit(it->second);
        }
        visiting.erase(m->name());
        done.insert(m->name());
        out.push_back(m);
      };
  for (auto& m : modules_) visit(m);
  return out;
}

// ----------------------------------------------------------------- config

Json load_app_config(const std::string& yaml_path,
                     const std::map<std::string, std::string>& cli) {
  // defaults
  Json cfg = Json::object();
  cfg["server"]["home_dir"] = "~/.hyperspot";
  cfg["logging"]["default"]["console_level"] = "info";
  // YAML layer
  if (!yaml_path.empty()) {
    std::ifstream f(yaml_path);
    if (!f) throw std::runtime_error("cannot open config " + yaml_path);
    std::stringstream ss;
    ss << f.rdbuf();
    cfg.merge_from(yaml_parse(ss.str()));
  }
  // env layer: APP__SECTION__KEY=value (reference Env("APP__","__"))
  for (char** e = ::environ; *e; ++e) {
    std::string kv = *e;
    if (kv.rfind("APP__", 0) != 0) continue;
    size_t eq = kv.find('=');
    if (eq == std::string::npos) continue;
    std::string key = kv.substr(5, eq - 5);
    std::string val = kv.substr(eq + 1);
    Json* cur = &cfg;
    size_t s = 0;
    while (true) {
      size_t sep = key.find("__", s);
      std::string part = key.substr(
          s, sep == std::string::npos ? std::string::npos : sep - s);
      std::transform(part.begin(), part.end(), part.begin(), ::tolower);
      // dashed keys (module names): _DASH_ spells "-" since env names
      // cannot carry dashes (APP__MODULES__LLM_DASH_GATEWAY__...)
      size_t dp;
      while ((dp = part.find("_dash_")) != std::string::npos)
        part = part.substr(0, dp) + "-" + part.substr(dp + 6);
      if (sep == std::string::npos) {
        // scalar inference like YAML
        Json v = yaml_parse(part + ": " + val).at(part);
        (*cur)[part] = v;
        break;
      }
      cur = &(*cur)[part];
      s = sep + 2;
    }
  }
  // CLI layer (dotted keys)
  for (auto& [k, v] : cli) {
    Json* cur = &cfg;
    size_t s = 0;
    while (true) {
      size_t dot = k.find('.', s);
      std::string part = k.substr(
          s, dot == std::string::npos ? std::string::npos : dot - s);
      if (dot == std::string::npos) {
        Json parsed = yaml_parse(part + ": " + v).at(part);
        (*cur)[part] = parsed;
        break;
      }
      cur = &(*cur)[part];
      s = dot + 1;
    }
  }
  return cfg;
}

}  // namespace hs
