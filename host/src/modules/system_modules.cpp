#include "system_modules.h"

#include "llm_gateway.h"

#include "../modkit/db.h"

#include "../modkit/auth.h"

#include <signal.h>
#include <sys/stat.h>
#include <sys/sysinfo.h>
#include <sys/wait.h>
#include <unistd.h>

#include <chrono>

#include <cstring>
#include <fstream>

#include "../util/log.h"

namespace hs {

static SecurityContext sec_of(HttpRequest& req) {
  return SecurityContext::from_json(req.extensions.at("security"));
}

// ------------------------------------------------------- tenant-resolver

namespace {

class StaticTenantResolver : public TenantResolverClient {
 public:
  explicit StaticTenantResolver(const Json& cfg) {
    // config shape of static-tr-plugin (config/static-tenants.yaml:30-47):
    // {vendor, priority, tenants: [{id, name, status, type, parent_id?}]}
    const Json& ts = cfg.at("tenants");
    if (ts.is_array()) {
      for (const auto& t : ts.arr()) {
        Tenant tn{t.at("id").as_string(), t.at("name").as_string(),
                  t.at("status").as_string("active"),
                  t.at("type").as_string("tenant"),
                  t.at("parent_id").as_string()};
        tenants_[tn.id] = tn;
      }
    }
    if (tenants_.empty()) {
      Tenant root{kDefaultTenantId, "Default Tenant", "active", "root", ""};
      tenants_[root.id] = root;
    }
  }
  std::optional<Tenant> get_tenant(const std::string& id) override {
    auto it = tenants_.find(id);
    if (it == tenants_.end()) return std::nullopt;
    return it->second;
  }
  std::vector<Tenant> get_tenants() override {
    std::vector<Tenant> out;
    for (auto& [_, t] : tenants_) out.push_back(t);
    return out;
  }
  std::vector<Tenant> get_ancestors(const std::string& id) override {
    std::vector<Tenant> out;
    auto cur = get_tenant(id);
    while (cur && !cur->parent_id.empty()) {
      cur = get_tenant(cur->parent_id);
      if (cur) out.push_back(*cur);
    }
    return out;
  }
  std::vector<Tenant> get_descendants(const std::string& id) override {
    std::vector<Tenant> out;
    for (auto& [_, t] : tenants_) {
      for (auto a = get_tenant(t.id); a;) {
        if (a->parent_id == id) { out.push_back(t); break; }
        a = a->parent_id.empty() ? std::nullopt : get_tenant(a->parent_id);
      }
    }
    return out;
  }
  bool is_ancestor(const std::string& a, const std::string& b) override {
    for (auto& t : get_ancestors(b))
      if (t.id == a) return true;
    return false;
  }

 private:
  std::map<std::string, Tenant> tenants_;
};

}  // namespace

void TenantResolverModule::init(ModuleCtx& ctx) {
  ctx.hub->register_client<TenantResolverClient>(
      "tenant-resolver",
      std::make_shared<StaticTenantResolver>(ctx.config));
}

// -------------------------------------------------------- authn-resolver

namespace {

// static tokens + optional JWT validation (modkit-auth equivalent;
// config.jwt: {hs256_secret | rs256_public_pem, issuer?, audience?,
// tenant_claim?, leeway_s?})
class StaticAuthnResolver : public AuthnResolverClient {
 public:
  explicit StaticAuthnResolver(const Json& cfg) {
    const Json& jw = cfg.at("jwt");
    if (jw.is_object()) {
      jwt_.hs256_secret = jw.at("hs256_secret").as_string("");
      jwt_.rs256_public_pem = jw.at("rs256_public_pem").as_string("");
      jwt_.issuer = jw.at("issuer").as_string("");
      jwt_.audience = jw.at("audience").as_string("");
      jwt_.tenant_claim = jw.at("tenant_claim").as_string("tid");
      jwt_.leeway_s = (int)jw.at("leeway_s").as_int(30);
      // real token-validation plane: JWKS endpoint / OIDC discovery
      // (libs/modkit-auth/src/providers/jwks.rs, oauth2/discovery.rs)
      const std::string jwks_uri = jw.at("jwks_uri").as_string("");
      const std::string disc =
          jw.at("oidc_discovery_url").as_string("");
      if (!jwks_uri.empty() || !disc.empty()) {
        auto cache = std::make_shared<JwksCache>();
        cache->jwks_uri = jwks_uri;
        cache->discovery_url = disc;
        cache->ttl_s = (int)jw.at("jwks_ttl_s").as_int(300);
        cache->tls_ca_file = jw.at("tls_ca_file").as_string("");
        jwt_.jwks = cache;
      }
    }
    // static-authn-plugin shape (plugins/static-authn-plugin/src/config.rs):
    // {tokens: [{token, subject_id, subject_tenant_id, subject_type?,
    //            scopes?: []}]}
    const Json& toks = cfg.at("tokens");
    if (toks.is_array()) {
      for (const auto& t : toks.arr()) {
        SecurityContext c;
        c.subject_id = t.at("subject_id").as_string();
        c.tenant_id = t.at("subject_tenant_id").as_string(kDefaultTenantId);
        c.subject_type = t.at("subject_type").as_string("user");
        if (t.at("scopes").is_array())
          for (auto& s : t.at("scopes").arr())
            c.scopes.push_back(s.as_string());
        map_[t.at("token").as_string()] = c;
      }
    }
  }
  std::optional<SecurityContext> authenticate(
      const std::string& bearer) override {
    auto it = map_.find(bearer);
    if (it != map_.end()) return it->second;
    if (jwt_.configured()) return jwt_.validate(bearer);
    return std::nullopt;
  }

 private:
  std::map<std::string, SecurityContext> map_;
  JwtValidator jwt_;
};

}  // namespace

void AuthnResolverModule::init(ModuleCtx& ctx) {
  ctx.hub->register_client<AuthnResolverClient>(
      "authn-resolver", std::make_shared<StaticAuthnResolver>(ctx.config));
}

// -------------------------------------------------------- authz-resolver

namespace {

// Static PDP: allow actions inside the subject's own tenant (or its
// descendants), deny cross-tenant.  Returned tenant_scope is the AccessScope
// the PEP (handlers) must filter rows by; empty scope = deny-all.
class StaticAuthzResolver : public AuthzResolverClient {
 public:
  explicit StaticAuthzResolver(ClientHub* hub) : hub_(hub) {}
  EvaluationResponse evaluate(const EvaluationRequest& r) override {
    EvaluationResponse resp;
    if (r.subject.subject_type == "anonymous") {
      resp.deny_reason = "anonymous subject";
      return resp;
    }
    const std::string& own = r.subject.tenant_id;
    if (r.tenant_id.empty() || r.tenant_id == own) {
      resp.allow = true;
      resp.tenant_scope = {own};
      return resp;
    }
    auto tr = hub_->get<TenantResolverClient>("tenant-resolver");
    if (tr && tr->is_ancestor(own, r.tenant_id)) {
      resp.allow = true;
      resp.tenant_scope = {r.tenant_id};
      return resp;
    }
    resp.deny_reason = "cross-tenant access denied";
    return resp;
  }

 private:
  ClientHub* hub_;
};

}  // namespace

void AuthzResolverModule::init(ModuleCtx& ctx) {
  ctx.hub->register_client<AuthzResolverClient>(
      "authz-resolver", std::make_shared<StaticAuthzResolver>(ctx.hub));
}

// -------------------------------------------------------- types-registry

namespace {

class InMemoryTypesRegistry : public TypesRegistryClient {
 public:
  std::optional<Json> get(const std::string& gts_id) override {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = entities_.find(gts_id);
    if (it == entities_.end()) return std::nullopt;
    return it->second;
  }
  std::vector<Json> list(const std::string& filter) override {
    std::vector<Json> out;
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& [id, e] : entities_) {
      if (!filter.empty()) {
        std::string pre = filter;
        bool wild = !pre.empty() && pre.back() == '*';
        if (wild) pre.pop_back();
        if (wild ? id.rfind(pre, 0) != 0 : id != filter) continue;
      }
      out.push_back(e);
    }
    return out;
  }
  int register_entities(const Json& ents) override {
    int n = 0;
    std::lock_guard<std::mutex> lk(mu_);
    for (const auto& e : ents.arr()) {
      std::string id = e.at("gts_id").as_string();
      if (id.empty()) throw Problem::bad_request("entity without gts_id");
      entities_[id] = e;
      ++n;
    }
    return n;
  }

 private:
  std::mutex mu_;
  std::map<std::string, Json> entities_;   // gts_id -> entity
};

std::shared_ptr<InMemoryTypesRegistry> g_types;

}  // namespace

void TypesRegistryModule::init(ModuleCtx& ctx) {
  g_types = std::make_shared<InMemoryTypesRegistry>();
  ctx.hub->register_client<TypesRegistryClient>("types-registry", g_types);
  // plugin discovery backbone: a single-flight cached selector resolving
  // a plugin TYPE id to the first registered instance whose gts_id
  // extends it (reference GtsPluginSelector over types-registry)
  auto types = g_types;
  ctx.hub->register_client<GtsPluginSelector>(
      "plugin-selector",
      std::make_shared<GtsPluginSelector>(
          [types](const std::string& plugin_type) -> std::string {
            auto hits = types->list(plugin_type + "*");
            if (hits.empty())
              throw std::runtime_error("no instance registered for " +
                                       plugin_type);
            return hits[0].at("gts_id").as_string();
          }));
}

void TypesRegistryModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  // REST surface per reference: /types-registry/v1/entities{,/{gts_id}}
  OperationSpec list;
  list.method = "GET";
  list.path = "/types-registry/v1/entities";
  list.operation_id = "types_registry_list";
  list.summary = "List GTS entities (wildcard filter via ?filter=)";
  list.authenticated = true;
  list.tags = {"types-registry"};
  rest.register_op(list, [](HttpRequest& rq, ResponseWriter& w) {
    std::string filter = rq.query.count("filter") ? rq.query["filter"] : "";
    Json items = Json::array();
    for (auto& e : g_types->list(filter)) items.push_back(e);
    Json out = Json::object();
    out["items"] = items;
    w.respond(200, "application/json", out.dump());
  });

  OperationSpec reg;
  reg.method = "POST";
  reg.path = "/types-registry/v1/entities";
  reg.operation_id = "types_registry_register";
  reg.summary = "Batch-register GTS types/instances";
  reg.authenticated = true;
  reg.allowed_content_types = {"application/json"};
  reg.tags = {"types-registry"};
  ClientHub* hub = ctx.hub;
  rest.register_op(reg, [hub](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON body"); }
    const Json& ents = body.at("entities");
    if (!ents.is_array())
      throw Problem::bad_request("'entities' array required");
    const int n = g_types->register_entities(ents);
    // new instances may shadow cached plugin selections
    if (auto sel = hub->get<GtsPluginSelector>("plugin-selector"))
      sel->invalidate();
    Json out = Json::object();
    out["registered"] = n;
    w.respond(201, "application/json", out.dump());
  });

  OperationSpec get;
  get.method = "GET";
  get.path = "/types-registry/v1/entities/{gts_id}";
  get.operation_id = "types_registry_get";
  get.authenticated = true;
  get.tags = {"types-registry"};
  rest.register_op(get, [](HttpRequest& rq, ResponseWriter& w) {
    auto e = g_types->get(rq.path_params["gts_id"]);
    if (!e) throw Problem::not_found("no such entity");
    w.respond(200, "application/json", e->dump());
  });
}

// -------------------------------------------------------- nodes-registry

void NodesRegistryModule::init(ModuleCtx& ctx) {}

Json NodesRegistryModule::node_info() const {
  Json n = Json::object();
  char host[256] = {0};
  gethostname(host, sizeof host - 1);
  n["hostname"] = host;
  n["os"] = "linux";
  n["num_cpus"] = (long)sysconf(_SC_NPROCESSORS_ONLN);
  struct sysinfo si{};
  if (sysinfo(&si) == 0) {
    n["mem_total_bytes"] = (double)si.totalram * si.mem_unit;
    n["mem_free_bytes"] = (double)si.freeram * si.mem_unit;
  }
  // GPU inventory: AMD KFD topology (the reference uses NVML enumeration,
  // libs/modkit-node-info/src/gpu_collector_linux.rs:22; MI355X-native is
  // the KFD sysfs tree)
  Json gpus = Json::array();
  for (int i = 0; i < 64; ++i) {
    std::string base = "/sys/class/kfd/kfd/topology/nodes/" +
                       std::to_string(i);
    std::ifstream f(base + "/properties");
    if (!f) break;
    std::string key;
    long simd = 0, uid = 0;
    Json g = Json::object();
    std::string line;
    while (std::getline(f, line)) {
      auto sp = line.find(' ');
      if (sp == std::string::npos) continue;
      std::string k = line.substr(0, sp);
      long v = atol(line.c_str() + sp + 1);
      if (k == "simd_count") simd = v;
      if (k == "unique_id") uid = v;
      if (k == "gfx_target_version") g["gfx_target_version"] = v;
    }
    if (simd > 0) {   // CPU nodes have simd_count 0
      g["index"] = (long)gpus.size();
      g["simd_count"] = simd;
      g["unique_id"] = uid;
      gpus.push_back(g);
    }
  }
  n["gpus"] = gpus;
  return n;
}

void NodesRegistryModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  OperationSpec nodes;
  nodes.method = "GET";
  nodes.path = "/nodes-registry/v1/nodes";
  nodes.operation_id = "nodes_list";
  nodes.summary = "Node inventory";
  nodes.authenticated = true;
  nodes.tags = {"nodes-registry"};
  rest.register_op(nodes, [this](HttpRequest& rq, ResponseWriter& w) {
    Json out = Json::object();
    Json items = Json::array();
    Json self = node_info();
    self["id"] = "local";
    items.push_back(self);
    out["items"] = items;
    w.respond(200, "application/json", out.dump());
  });
  OperationSpec sysinfo_op;
  sysinfo_op.method = "GET";
  sysinfo_op.path = "/nodes-registry/v1/nodes/{id}/sysinfo";
  sysinfo_op.operation_id = "nodes_sysinfo";
  sysinfo_op.authenticated = true;
  sysinfo_op.tags = {"nodes-registry"};
  rest.register_op(sysinfo_op, [this](HttpRequest& rq, ResponseWriter& w) {
    if (rq.path_params["id"] != "local")
      throw Problem::not_found("unknown node");
    w.respond(200, "application/json", node_info().dump());
  });
  // syscap: hardware capabilities incl. the GPU inventory (the reference
  // collects GpuInfo via NVML; the MI355X-native equivalent enumerates
  // amdgpu devices from sysfs — no vendor daemon needed)
  OperationSpec syscap_op;
  syscap_op.method = "GET";
  syscap_op.path = "/nodes-registry/v1/nodes/{id}/syscap";
  syscap_op.operation_id = "nodes_syscap";
  syscap_op.authenticated = true;
  syscap_op.tags = {"nodes-registry"};
  rest.register_op(syscap_op, [this](HttpRequest& rq, ResponseWriter& w) {
    if (rq.path_params["id"] != "local")
      throw Problem::not_found("unknown node");
    Json out = node_info();
    Json gpus = Json::array();
    for (int card = 0; card < 16; ++card) {
      const std::string base =
          "/sys/class/drm/card" + std::to_string(card) + "/device/";
      std::ifstream vf(base + "vendor");
      std::string vendor;
      if (!(vf >> vendor)) continue;
      if (vendor != "0x1002") continue;            // AMD
      Json g = Json::object();
      g["index"] = (long)gpus.size();
      std::ifstream df(base + "device");
      std::string dev;
      if (df >> dev) g["device_id"] = dev;
      std::ifstream mf(base + "mem_info_vram_total");
      long long vram = 0;
      if (mf >> vram) g["vram_mb"] = (long)(vram / (1024 * 1024));
      g["vendor"] = "AMD";
      gpus.push_back(g);
    }
    out["gpus"] = gpus;
    out["gpu_count"] = (long)gpus.size();
    w.respond(200, "application/json", out.dump());
  });
}

// -------------------------------------------------------- model-registry

namespace {

class LocalModelRegistry : public ModelRegistryClient {
 public:
  LocalModelRegistry(const Json& cfg, const std::string& serving_model,
                     std::unique_ptr<Db> db)
      : db_(std::move(db)) {
    // models from config: modules.model-registry.config.models:
    // [{canonical_id, name, architecture, context_window, managed...}]
    const Json& ms = cfg.at("models");
    if (ms.is_array())
      for (auto& m : ms.arr())
        models_[m.at("canonical_id").as_string()] = m;
    if (models_.empty()) {
      std::vector<std::string> ids = {"local::llama3-8b",
                                      "local::llama3-70b",
                                      "local::mixtral-8x7b"};
      // whatever the llm-gateway actually serves is always registered
      if (!serving_model.empty())
        ids.push_back("local::" + serving_model);
      for (const std::string& cid : ids) {
        if (models_.count(cid)) continue;
        Json m = Json::object();
        m["canonical_id"] = cid;
        m["provider_slug"] = "local";
        m["provider_model_id"] = cid.substr(cid.find("::") + 2);
        m["managed"] = true;
        m["architecture"] =
            cid.find("mixtral") != std::string::npos ? "mistral" : "llama";
        m["format"] = "safetensors";
        m["lifecycle_status"] = "production";
        Json caps = Json::object();
        caps["text_input"] = true;
        caps["text_output"] = true;
        caps["streaming"] = true;
        m["capabilities"] = caps;
        m["context_window"] = 8192;
        models_[cid] = m;
      }
    }
  }
  std::optional<Json> get_tenant_model(const std::string& tenant,
                                       const std::string& canonical)
      override {
    // resolution (reference PRD.md:11,:298-306): aliases first (tenant
    // alias shadows a parent/global alias), then canonical id split on
    // FIRST "::"; bare ids default to "local::".  Tenant-registered
    // models shadow globals by canonical id (PRD.md:181-188).
    std::string name = canonical;
    if (name.find("::") == std::string::npos) {
      if (auto a = alias_lookup(tenant, name)) name = *a;
    }
    std::string cid = name.find("::") == std::string::npos
        ? "local::" + name : name;
    std::optional<Json> m = db_model(tenant, cid);
    if (!m) m = db_model("", cid);
    if (!m) {
      auto it = models_.find(cid);
      if (it == models_.end()) return std::nullopt;
      m = it->second;
    }
    (*m)["approval"] = approval(tenant, cid);
    return m;
  }
  std::vector<Json> list_tenant_models(const std::string& tenant) override {
    std::vector<Json> out;
    std::map<std::string, Json> merged = models_;   // global defaults
    {
      std::lock_guard<std::mutex> lk(db_->mu());
      for (auto& r : db_->query(
               "SELECT spec FROM models WHERE tenant_id IN ('', ?) "
               "ORDER BY tenant_id",   // tenant rows overwrite globals
               {DbValue::S(tenant)})) {
        try {
          Json m = Json::parse(r.at("spec").as_string());
          merged[m.at("canonical_id").as_string()] = m;
        } catch (...) {}
      }
    }
    for (auto& [_, m] : merged) out.push_back(m);
    return out;
  }

  // tenant-scoped model registration with shadow-by-canonical-id
  void put_model(const std::string& tenant, const Json& spec) {
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query(
        "INSERT INTO models (tenant_id, canonical_id, spec) VALUES "
        "(?,?,?) ON CONFLICT(tenant_id, canonical_id) DO UPDATE SET "
        "spec=excluded.spec",
        {DbValue::S(tenant),
         DbValue::S(spec.at("canonical_id").as_string()),
         DbValue::S(spec.dump())});
  }

  // aliases (PRD.md:298-306): name -> canonical, tenant shadows global
  void put_alias(const std::string& tenant, const std::string& name,
                 const std::string& cid) {
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query(
        "INSERT INTO aliases (tenant_id, name, canonical_id) VALUES "
        "(?,?,?) ON CONFLICT(tenant_id, name) DO UPDATE SET "
        "canonical_id=excluded.canonical_id",
        {DbValue::S(tenant), DbValue::S(name), DbValue::S(cid)});
  }
  bool del_alias(const std::string& tenant, const std::string& name) {
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query("DELETE FROM aliases WHERE tenant_id=? AND name=?",
               {DbValue::S(tenant), DbValue::S(name)});
    return db_->changes() > 0;
  }
  std::optional<std::string> alias_lookup(const std::string& tenant,
                                          const std::string& name) {
    std::lock_guard<std::mutex> lk(db_->mu());
    for (const char* scope : {"t", "g"}) {
      auto r = db_->query(
          "SELECT canonical_id FROM aliases WHERE tenant_id=? AND name=?",
          {DbValue::S(scope[0] == 't' ? tenant : ""), DbValue::S(name)});
      if (!r.empty()) return r[0].at("canonical_id").as_string();
    }
    return std::nullopt;
  }

  // ModelApproval states pending->approved|rejected->revoked
  // (model-registry PRD.md:225-253); default approved so single-tenant
  // deployments need no workflow.  Persisted (survives host restart).
  std::string approval(const std::string& tenant, const std::string& cid) {
    std::lock_guard<std::mutex> lk(db_->mu());
    auto r = db_->query(
        "SELECT status FROM approvals WHERE tenant_id=? AND "
        "canonical_id=?",
        {DbValue::S(tenant), DbValue::S(cid)});
    return r.empty() ? "approved" : r[0].at("status").as_string();
  }
  void set_approval(const std::string& tenant, const std::string& cid,
                    const std::string& st) {
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query(
        "INSERT INTO approvals (tenant_id, canonical_id, status) VALUES "
        "(?,?,?) ON CONFLICT(tenant_id, canonical_id) DO UPDATE SET "
        "status=excluded.status",
        {DbValue::S(tenant), DbValue::S(cid), DbValue::S(st)});
  }

 private:
  std::optional<Json> db_model(const std::string& tenant,
                               const std::string& cid) {
    std::lock_guard<std::mutex> lk(db_->mu());
    auto r = db_->query(
        "SELECT spec FROM models WHERE tenant_id=? AND canonical_id=?",
        {DbValue::S(tenant), DbValue::S(cid)});
    if (r.empty()) return std::nullopt;
    try { return Json::parse(r[0].at("spec").as_string()); }
    catch (...) { return std::nullopt; }
  }
  std::unique_ptr<Db> db_;
  std::map<std::string, Json> models_;
};

std::shared_ptr<LocalModelRegistry> g_registry;

}  // namespace

void ModelRegistryModule::init(ModuleCtx& ctx) {
  std::string file = ctx.full_config
                         .path("modules.model-registry.database.file")
                         .as_string("");
  if (file.empty()) {
    std::string home = ctx.home_dir;
    if (!home.empty() && home[0] == '~') {
      const char* h = getenv("HOME");
      home = std::string(h ? h : "/tmp") + home.substr(1);
    }
    mkdir(home.c_str(), 0755);
    file = home + "/model-registry.db";
  }
  auto db = std::make_unique<Db>(file);
  db->migrate("model-registry", {
      {"0001_approvals",
       "CREATE TABLE approvals (tenant_id TEXT NOT NULL, canonical_id "
       "TEXT NOT NULL, status TEXT NOT NULL, UNIQUE (tenant_id, "
       "canonical_id))"},
      {"0002_aliases",
       "CREATE TABLE aliases (tenant_id TEXT NOT NULL, name TEXT NOT "
       "NULL, canonical_id TEXT NOT NULL, UNIQUE (tenant_id, name))"},
      {"0003_models",
       "CREATE TABLE models (tenant_id TEXT NOT NULL, canonical_id TEXT "
       "NOT NULL, spec TEXT NOT NULL, UNIQUE (tenant_id, "
       "canonical_id))"},
  });
  g_registry = std::make_shared<LocalModelRegistry>(
      ctx.config,
      ctx.full_config.path("modules.llm-gateway.config.model")
          .as_string(""),
      std::move(db));
  ctx.hub->register_client<ModelRegistryClient>("model-registry",
                                                g_registry);
}

void ModelRegistryModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  {
    // ProviderHealth (reference PRD:280-294): discovery-level status of
    // the in-node serving providers (engine workers), fed by the
    // llm-gateway watchdog probes through the hub
    OperationSpec op;
    op.method = "GET";
    op.path = "/model-registry/v1/providers/health";
    op.operation_id = "providers_health";
    op.summary = "Discovery-level provider health";
    op.authenticated = true;
    op.tags = {"model-registry"};
    ClientHub* hub = ctx.hub;
    rest.register_op(op, [hub](HttpRequest& rq, ResponseWriter& w) {
      (void)rq;
      auto ph = hub->get<ProviderHealthClient>("llm-gateway");
      Json out = ph ? ph->provider_health() : Json::object();
      if (!out.contains("items")) out["items"] = Json::array();
      w.respond(200, "application/json", out.dump());
    });
  }
  OperationSpec list;
  list.method = "GET";
  list.path = "/model-registry/v1/models";
  list.operation_id = "models_list";
  list.summary = "List tenant-visible models";
  list.authenticated = true;
  list.tags = {"model-registry"};
  rest.register_op(list, [](HttpRequest& rq, ResponseWriter& w) {
    Json items = Json::array();
    for (auto& m : g_registry->list_tenant_models(
             sec_of(rq).tenant_id))
      items.push_back(m);
    Json out = Json::object();
    out["items"] = items;
    w.respond(200, "application/json", out.dump());
  });
  OperationSpec get;
  get.method = "GET";
  get.path = "/model-registry/v1/models/{canonical_id}";
  get.operation_id = "models_get";
  get.authenticated = true;
  get.tags = {"model-registry"};
  rest.register_op(get, [](HttpRequest& rq, ResponseWriter& w) {
    auto m = g_registry->get_tenant_model(sec_of(rq).tenant_id,
                                          rq.path_params["canonical_id"]);
    if (!m) throw Problem::not_found("model_not_found");
    w.respond(200, "application/json", m->dump());
  });
  // approval workflow (PRD.md:225-253): approve | reject | revoke
  OperationSpec ap;
  ap.method = "POST";
  ap.path = "/model-registry/v1/models/{canonical_id}/approval";
  ap.operation_id = "models_set_approval";
  ap.summary = "Set this tenant's approval state for a model";
  ap.authenticated = true;
  ap.allowed_content_types = {"application/json"};
  ap.tags = {"model-registry"};
  rest.register_op(ap, [](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON body"); }
    const std::string st = body.at("status").as_string();
    if (st != "approved" && st != "pending" && st != "rejected" &&
        st != "revoked")
      throw Problem::bad_request(
          "status must be approved|pending|rejected|revoked");
    const std::string cid = rq.path_params["canonical_id"];
    auto sec = sec_of(rq);
    if (!g_registry->get_tenant_model(sec.tenant_id, cid))
      throw Problem::not_found("model_not_found");
    g_registry->set_approval(sec.tenant_id,
                             cid.find("::") == std::string::npos
                                 ? "local::" + cid : cid, st);
    Json out = Json::object();
    out["canonical_id"] = cid;
    out["status"] = st;
    w.respond(200, "application/json", out.dump());
  });

  // tenant model registration w/ shadow-by-canonical-id (PRD.md:181-188)
  OperationSpec post;
  post.method = "POST";
  post.path = "/model-registry/v1/models";
  post.operation_id = "models_register";
  post.summary = "Register/shadow a model for this tenant";
  post.authenticated = true;
  post.allowed_content_types = {"application/json"};
  post.tags = {"model-registry"};
  rest.register_op(post, [](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON body"); }
    const std::string cid = body.at("canonical_id").as_string();
    if (cid.find("::") == std::string::npos)
      throw Problem::bad_request(
          "canonical_id must be {provider_slug}::{model_id}");
    g_registry->put_model(sec_of(rq).tenant_id, body);
    w.respond(201, "application/json", body.dump());
  });

  // aliases: name -> canonical_id with tenant->global shadowing
  // (PRD.md:298-306)
  OperationSpec aput;
  aput.method = "PUT";
  aput.path = "/model-registry/v1/aliases/{name}";
  aput.operation_id = "alias_put";
  aput.authenticated = true;
  aput.allowed_content_types = {"application/json"};
  aput.tags = {"model-registry"};
  rest.register_op(aput, [](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON body"); }
    const std::string cid = body.at("canonical_id").as_string();
    if (cid.empty())
      throw Problem::bad_request("'canonical_id' required");
    g_registry->put_alias(sec_of(rq).tenant_id, rq.path_params["name"],
                          cid);
    w.respond(204, "application/json", "");
  });
  OperationSpec aget;
  aget.method = "GET";
  aget.path = "/model-registry/v1/aliases/{name}";
  aget.operation_id = "alias_get";
  aget.authenticated = true;
  aget.tags = {"model-registry"};
  rest.register_op(aget, [](HttpRequest& rq, ResponseWriter& w) {
    auto cid = g_registry->alias_lookup(sec_of(rq).tenant_id,
                                        rq.path_params["name"]);
    if (!cid) throw Problem::not_found("no such alias");
    Json out = Json::object();
    out["name"] = rq.path_params["name"];
    out["canonical_id"] = *cid;
    w.respond(200, "application/json", out.dump());
  });
  OperationSpec adel;
  adel.method = "DELETE";
  adel.path = "/model-registry/v1/aliases/{name}";
  adel.operation_id = "alias_delete";
  adel.authenticated = true;
  adel.tags = {"model-registry"};
  rest.register_op(adel, [](HttpRequest& rq, ResponseWriter& w) {
    if (!g_registry->del_alias(sec_of(rq).tenant_id,
                               rq.path_params["name"]))
      throw Problem::not_found("no such alias");
    w.respond(204, "application/json", "");
  });
}

// ------------------------------------------------------------- credstore

namespace {

// sqlite-backed tenant-scoped secret store with sharing modes
// private|tenant|shared and hierarchical inheritance
// (reference modules/credstore/docs/DESIGN.md:295-339):
//   private  — owner(subject)-only, scoped per owner (never conflicts
//              across owners)
//   tenant   — tenant-wide (the default)
//   shared   — visible to DESCENDANT tenants via ancestor resolution;
//              child tenants may shadow by creating the same reference
// Inaccessible secrets are ALWAYS 404 (anti-enumeration, DESIGN.md:336);
// REST GET returns metadata (sharing, owner_tenant_id, is_inherited),
// never the value — the OAGW credential injector is the only consumer
// of secret material (documented deviation, stricter than the spec).
class SqlCredStore : public CredStoreClient {
 public:
  SqlCredStore(std::unique_ptr<Db> db, ClientHub* hub)
      : db_(std::move(db)), hub_(hub) {}

  struct Meta {
    std::string value, sharing, owner_tenant;
    bool is_inherited = false;
  };

  std::optional<Meta> resolve(const std::string& tenant,
                              const std::string& subject,
                              const std::string& ref) {
    {
      std::lock_guard<std::mutex> lk(db_->mu());
      // 1) caller's own private secret
      if (!subject.empty()) {
        auto r = db_->query(
            "SELECT value, sharing, tenant_id FROM secrets WHERE "
            "tenant_id=? AND scope_owner=? AND ref=?",
            {DbValue::S(tenant), DbValue::S(subject), DbValue::S(ref)});
        if (!r.empty()) return mk(r[0], false);
      }
      // 2) tenant-wide secret in the caller's tenant
      auto r = db_->query(
          "SELECT value, sharing, tenant_id FROM secrets WHERE "
          "tenant_id=? AND scope_owner='' AND ref=?",
          {DbValue::S(tenant), DbValue::S(ref)});
      if (!r.empty()) return mk(r[0], false);
    }
    // 3) hierarchical: nearest ancestor's `shared` secret
    auto tr = hub_ ? hub_->get<TenantResolverClient>("tenant-resolver")
                   : nullptr;
    if (tr) {
      for (auto& anc : tr->get_ancestors(tenant)) {
        std::lock_guard<std::mutex> lk(db_->mu());
        auto r = db_->query(
            "SELECT value, sharing, tenant_id FROM secrets WHERE "
            "tenant_id=? AND scope_owner='' AND ref=? AND "
            "sharing='shared'",
            {DbValue::S(anc.id), DbValue::S(ref)});
        if (!r.empty()) return mk(r[0], true);
      }
    }
    return std::nullopt;
  }

  // create-only (POST): false when the reference already exists in the
  // same scope (per-owner for private, per-tenant otherwise)
  bool create(const std::string& tenant, const std::string& subject,
              const std::string& ref, const std::string& value,
              const std::string& sharing) {
    const std::string owner = sharing == "private" ? subject : "";
    std::lock_guard<std::mutex> lk(db_->mu());
    auto r = db_->query(
        "SELECT 1 FROM secrets WHERE tenant_id=? AND scope_owner=? AND "
        "ref=?",
        {DbValue::S(tenant), DbValue::S(owner), DbValue::S(ref)});
    if (!r.empty()) return false;
    db_->query(
        "INSERT INTO secrets (tenant_id, scope_owner, ref, value, "
        "sharing, updated_at) VALUES (?,?,?,?,?,datetime('now'))",
        {DbValue::S(tenant), DbValue::S(owner), DbValue::S(ref),
         DbValue::S(value), DbValue::S(sharing)});
    return true;
  }

  // upsert (PUT): updates the caller's secret in-scope, creating it
  // with the given sharing when absent
  void upsert(const std::string& tenant, const std::string& subject,
              const std::string& ref, const std::string& value,
              const std::string& sharing) {
    const std::string owner = sharing == "private" ? subject : "";
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query(
        "INSERT INTO secrets (tenant_id, scope_owner, ref, value, "
        "sharing, updated_at) VALUES (?,?,?,?,?,datetime('now')) "
        "ON CONFLICT(tenant_id, scope_owner, ref) DO UPDATE SET "
        "value=excluded.value, sharing=excluded.sharing, "
        "updated_at=excluded.updated_at",
        {DbValue::S(tenant), DbValue::S(owner), DbValue::S(ref),
         DbValue::S(value), DbValue::S(sharing)});
  }

  // delete own-scope only: an INHERITED secret is not deletable from a
  // child tenant (404 — the child shadows instead)
  bool remove(const std::string& tenant, const std::string& subject,
              const std::string& ref) {
    std::lock_guard<std::mutex> lk(db_->mu());
    db_->query(
        "DELETE FROM secrets WHERE tenant_id=? AND ref=? AND "
        "(scope_owner='' OR scope_owner=?)",
        {DbValue::S(tenant), DbValue::S(ref), DbValue::S(subject)});
    return db_->changes() > 0;
  }

  // ---- CredStoreClient (OAGW injector: tenant-plane resolution) ----
  std::optional<std::string> get(const std::string& tenant,
                                 const std::string& ref) override {
    auto m = resolve(tenant, "", ref);
    if (!m) return std::nullopt;
    return m->value;
  }
  void put(const std::string& tenant, const std::string& ref,
           const std::string& value) override {
    upsert(tenant, "", ref, value, "tenant");
  }
  bool del(const std::string& tenant, const std::string& ref) override {
    return remove(tenant, "", ref);
  }

 private:
  static Meta mk(const DbRow& r, bool inherited) {
    Meta m;
    m.value = r.at("value").as_string();
    m.sharing = r.at("sharing").as_string();
    m.owner_tenant = r.at("tenant_id").as_string();
    m.is_inherited = inherited;
    return m;
  }
  std::unique_ptr<Db> db_;
  ClientHub* hub_;
};

std::shared_ptr<SqlCredStore> g_creds;

}  // namespace

void CredStoreModule::init(ModuleCtx& ctx) {
  std::string file = ctx.full_config
                         .path("modules.credstore.database.file")
                         .as_string("");
  if (file.empty()) {
    std::string home = ctx.home_dir;
    if (!home.empty() && home[0] == '~') {
      const char* h = getenv("HOME");
      home = std::string(h ? h : "/tmp") + home.substr(1);
    }
    mkdir(home.c_str(), 0755);
    file = home + "/credstore.db";
  }
  auto db = std::make_unique<Db>(file);
  db->migrate("credstore", {
      {"0001_secrets",
       "CREATE TABLE secrets ("
       "  tenant_id TEXT NOT NULL,"
       "  scope_owner TEXT NOT NULL DEFAULT '',"   // subject for private
       "  ref TEXT NOT NULL,"
       "  value TEXT NOT NULL,"
       "  sharing TEXT NOT NULL DEFAULT 'tenant',"
       "  updated_at TEXT NOT NULL,"
       "  UNIQUE (tenant_id, scope_owner, ref))"},
  });
  g_creds = std::make_shared<SqlCredStore>(std::move(db), ctx.hub);
  ctx.hub->register_client<CredStoreClient>("credstore", g_creds);
}

void CredStoreModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  // REST per reference DESIGN.md:283-339; tenant always from SecurityCtx;
  // inaccessible secrets are ALWAYS 404 (anti-enumeration).
  auto sharing_of = [](const Json& body) -> std::string {
    const std::string sh = body.at("sharing").as_string("tenant");
    if (sh != "private" && sh != "tenant" && sh != "shared")
      throw Problem::bad_request(
          "'sharing' must be private|tenant|shared");
    return sh;
  };

  OperationSpec post;
  post.method = "POST";
  post.path = "/credstore/v1/secrets";
  post.operation_id = "credstore_create";
  post.summary = "Create a secret (409 if the reference exists in scope)";
  post.authenticated = true;
  post.allowed_content_types = {"application/json"};
  post.tags = {"credstore"};
  rest.register_op(post, [sharing_of](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON"); }
    const std::string ref = body.at("reference").as_string();
    if (ref.empty() || !body.at("value").is_string())
      throw Problem::bad_request("'reference' and 'value' required");
    auto sec = sec_of(rq);
    if (!g_creds->create(sec.tenant_id, sec.subject_id, ref,
                         body.at("value").as_string(), sharing_of(body)))
      throw Problem{409, "Conflict", "about:blank",
                    "secret already exists in this scope", "conflict"};
    w.respond(201, "application/json", "{\"created\":true}");
  });

  OperationSpec put;
  put.method = "PUT";
  put.path = "/credstore/v1/secrets/{ref}";
  put.operation_id = "credstore_put";
  put.authenticated = true;
  put.allowed_content_types = {"application/json"};
  put.tags = {"credstore"};
  rest.register_op(put, [sharing_of](HttpRequest& rq, ResponseWriter& w) {
    Json body;
    try { body = Json::parse(rq.body); }
    catch (...) { throw Problem::bad_request("invalid JSON"); }
    if (!body.at("value").is_string())
      throw Problem::bad_request("'value' string required");
    auto sec = sec_of(rq);
    g_creds->upsert(sec.tenant_id, sec.subject_id, rq.path_params["ref"],
                    body.at("value").as_string(), sharing_of(body));
    w.respond(204, "application/json", "");
  });

  OperationSpec get;
  get.method = "GET";
  get.path = "/credstore/v1/secrets/{ref}";
  get.operation_id = "credstore_get";
  get.authenticated = true;
  get.tags = {"credstore"};
  rest.register_op(get, [](HttpRequest& rq, ResponseWriter& w) {
    auto sec = sec_of(rq);
    auto m = g_creds->resolve(sec.tenant_id, sec.subject_id,
                              rq.path_params["ref"]);
    if (!m) throw Problem::not_found();   // never reveal existence
    // write-only store: values are consumed by the OAGW credential
    // injector, never read back over REST
    Json meta = Json::object();
    meta["owner_tenant_id"] = m->owner_tenant;
    meta["sharing"] = m->sharing;
    meta["is_inherited"] = m->is_inherited;
    Json out = Json::object();
    out["ref"] = rq.path_params["ref"];
    out["exists"] = true;
    out["value_length"] = (long)m->value.size();
    out["metadata"] = meta;
    w.respond(200, "application/json", out.dump());
  });

  OperationSpec del;
  del.method = "DELETE";
  del.path = "/credstore/v1/secrets/{ref}";
  del.operation_id = "credstore_delete";
  del.authenticated = true;
  del.tags = {"credstore"};
  rest.register_op(del, [](HttpRequest& rq, ResponseWriter& w) {
    auto sec = sec_of(rq);
    if (!g_creds->remove(sec.tenant_id, sec.subject_id,
                         rq.path_params["ref"]))
      throw Problem::not_found();
    w.respond(204, "application/json", "");
  });
}

// ----------------------------------------------------------- file-storage

namespace {

std::string sanitize_rel_path(const std::string& p) {
  if (p.empty() || p.find("..") != std::string::npos || p.front() == '/')
    throw Problem::bad_request("invalid file name");
  return p;
}

void mkdirs(const std::string& path) {
  std::string cur;
  for (size_t i = 0; i < path.size(); ++i) {
    if (path[i] == '/' && !cur.empty()) mkdir(cur.c_str(), 0755);
    cur += path[i];
  }
  mkdir(cur.c_str(), 0755);
}

class LocalFileStorage : public FileStorageClient {
 public:
  explicit LocalFileStorage(std::string root) : root_(std::move(root)) {}
  std::string root_for(const std::string& tenant) override {
    std::string dir = root_ + "/" + tenant;
    mkdirs(dir);
    return dir;
  }

 private:
  std::string root_;
};

std::shared_ptr<LocalFileStorage> g_files;

}  // namespace

void FileStorageModule::init(ModuleCtx& ctx) {
  std::string home = ctx.home_dir;
  if (!home.empty() && home[0] == '~') {
    const char* h = getenv("HOME");
    home = std::string(h ? h : "/tmp") + home.substr(1);
  }
  root_ = ctx.config.at("root").as_string(home + "/file-storage");
  mkdirs(root_);
  g_files = std::make_shared<LocalFileStorage>(root_);
  ctx.hub->register_client<FileStorageClient>("file-storage", g_files);
}

void FileStorageModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  OperationSpec put;
  put.method = "PUT";
  put.path = "/file-storage/v1/files/{*name}";
  put.operation_id = "files_put";
  put.summary = "Store a file (checkpoints, media)";
  put.authenticated = true;
  put.tags = {"file-storage"};
  rest.register_op(put, [](HttpRequest& rq, ResponseWriter& w) {
    std::string rel = sanitize_rel_path(rq.path_params["name"]);
    std::string dir = g_files->root_for(sec_of(rq).tenant_id);
    std::string full = dir + "/" + rel;
    auto slash = full.rfind('/');
    mkdirs(full.substr(0, slash));
    std::ofstream f(full, std::ios::binary | std::ios::trunc);
    if (!f) throw Problem{500, "Internal Server Error", "about:blank",
                          "cannot write file", ""};
    f.write(rq.body.data(), (long)rq.body.size());
    Json meta = Json::object();
    meta["name"] = rel;
    meta["size"] = rq.body.size();
    meta["mime"] = rq.header("content-type", "application/octet-stream");
    w.respond(201, "application/json", meta.dump());
  });

  OperationSpec get;
  get.method = "GET";
  get.path = "/file-storage/v1/files/{*name}";
  get.operation_id = "files_get";
  get.authenticated = true;
  get.tags = {"file-storage"};
  rest.register_op(get, [](HttpRequest& rq, ResponseWriter& w) {
    std::string rel = sanitize_rel_path(rq.path_params["name"]);
    std::string full =
        g_files->root_for(sec_of(rq).tenant_id) + "/" + rel;
    std::ifstream f(full, std::ios::binary);
    if (!f) throw Problem::not_found();
    std::stringstream ss;
    ss << f.rdbuf();
    w.respond(200, "application/octet-stream", ss.str());
  });

  // metadata query (PRD "Get file metadata": pre-fetch validation of
  // size/type without transferring the bytes)
  OperationSpec md;
  md.method = "GET";
  md.path = "/file-storage/v1/metadata/{*name}";
  md.operation_id = "files_metadata";
  md.summary = "File metadata (size, mtime) without the body";
  md.authenticated = true;
  md.tags = {"file-storage"};
  rest.register_op(md, [](HttpRequest& rq, ResponseWriter& w) {
    std::string rel = sanitize_rel_path(rq.path_params["name"]);
    std::string full =
        g_files->root_for(sec_of(rq).tenant_id) + "/" + rel;
    struct stat st{};
    if (stat(full.c_str(), &st) != 0 || !S_ISREG(st.st_mode))
      throw Problem::not_found();
    Json meta = Json::object();
    meta["name"] = rel;
    meta["size"] = (long)st.st_size;
    meta["modified_at"] = (double)st.st_mtime;
    w.respond(200, "application/json", meta.dump());
  });

  OperationSpec del;
  del.method = "DELETE";
  del.path = "/file-storage/v1/files/{*name}";
  del.operation_id = "files_delete";
  del.authenticated = true;
  del.tags = {"file-storage"};
  rest.register_op(del, [](HttpRequest& rq, ResponseWriter& w) {
    std::string rel = sanitize_rel_path(rq.path_params["name"]);
    std::string full =
        g_files->root_for(sec_of(rq).tenant_id) + "/" + rel;
    if (unlink(full.c_str()) != 0) throw Problem::not_found();
    w.respond(204, "application/json", "");
  });
}

// ---------------------------------------------------- module-orchestrator

static double mono_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch()).count();
}

void ModuleOrchestratorModule::init(ModuleCtx& ctx) {
  // collect modules declared runtime.type: oop (A.6 envelope)
  const Json& mods = ctx.full_config.at("modules");
  if (mods.is_object()) {
    for (auto& [mname, mcfg] : mods.obj()) {
      if (mcfg.path("runtime.type").as_string() != "oop") continue;
      const Json& ex = mcfg.path("runtime.execution");
      OopSpec sp;
      sp.name = mname;
      sp.exe = ex.at("executable_path").as_string();
      if (ex.at("args").is_array())
        for (auto& a : ex.at("args").arr())
          sp.args.push_back(a.as_string());
      if (ex.at("environment").is_object())
        for (auto& [k, v] : ex.at("environment").obj())
          sp.env[k] = v.as_string();
      sp.config = mcfg.at("config");
      oop_specs_.push_back(std::move(sp));
    }
  }
  const std::string bind = ctx.full_config
      .path("modules.api-gateway.config.bind_addr")
      .as_string("127.0.0.1:8087");
  directory_endpoint_ = "http://" + bind + "/module-orchestrator/v1";

  // hub DirectoryClient so in-process modules can call OoP services
  // (the module objects outlive the hub: both owned by the host)
  struct HubDir : DirectoryClient {
    ModuleOrchestratorModule* mo;
    std::string resolve(const std::string& n) override {
      return mo->resolve_endpoint(n);
    }
  };
  auto dir = std::make_shared<HubDir>();
  dir->mo = this;
  ctx.hub->register_client<DirectoryClient>("module-orchestrator", dir);
}

std::string ModuleOrchestratorModule::resolve_endpoint(
    const std::string& name) {
  const double now = mono_s();
  std::lock_guard<std::mutex> lk(inst_mu_);
  const Instance* best = nullptr;
  for (auto& [id, in] : instances_)
    if (in.name == name && (now - in.last_heartbeat) < 15.0 &&
        (!best || in.last_heartbeat > best->last_heartbeat))
      best = &in;
  return best ? best->endpoint : "";
}

void ModuleOrchestratorModule::start(ModuleCtx& ctx) {
  for (auto& sp : oop_specs_) {
    int pfd[2];
    if (pipe(pfd) != 0) continue;
    pid_t pid = fork();
    if (pid == 0) {
      setpgid(0, 0);
      dup2(pfd[1], 1);
      dup2(pfd[1], 2);
      close(pfd[0]);
      close(pfd[1]);
      // reference host_runtime.rs:56-59 env contract
      setenv("MODKIT_MODULE_CONFIG", sp.config.dump().c_str(), 1);
      setenv("MODKIT_DIRECTORY_ENDPOINT", directory_endpoint_.c_str(), 1);
      setenv("MODKIT_MODULE_NAME", sp.name.c_str(), 1);
      for (auto& [k, v] : sp.env) setenv(k.c_str(), v.c_str(), 1);
      std::vector<std::string> args = {sp.exe};
      for (auto& a : sp.args) args.push_back(a);
      std::vector<char*> argv;
      for (auto& a : args) argv.push_back(const_cast<char*>(a.c_str()));
      argv.push_back(nullptr);
      execvp(argv[0], argv.data());
      _exit(127);
    }
    close(pfd[1]);
    Child c;
    c.name = sp.name;
    c.pid = pid;
    c.out_fd = pfd[0];
    // log forwarder: child stdout/stderr lines -> host log
    c.fwd = std::thread([fd = pfd[0], mname = sp.name] {
      std::string buf;
      char tmp[4096];
      while (true) {
        ssize_t r = read(fd, tmp, sizeof tmp);
        if (r <= 0) break;
        buf.append(tmp, (size_t)r);
        size_t nl;
        while ((nl = buf.find('\n')) != std::string::npos) {
          LOG_INFO(("oop:" + mname).c_str(), "%s",
                   buf.substr(0, nl).c_str());
          buf.erase(0, nl + 1);
        }
      }
      close(fd);
    });
    children_.push_back(std::move(c));
    LOG_INFO("module-orchestrator", "spawned oop module %s pid=%d",
             sp.name.c_str(), pid);
  }
}

void ModuleOrchestratorModule::stop(ModuleCtx& ctx) {
  for (auto& c : children_) {
    if (c.pid > 0) {
      kill(c.pid, SIGTERM);
      int st = 0;
      for (int i = 0; i < 30 && waitpid(c.pid, &st, WNOHANG) == 0; ++i)
        usleep(100000);
      if (waitpid(c.pid, &st, WNOHANG) == 0) {
        kill(c.pid, SIGKILL);
        waitpid(c.pid, &st, 0);
      }
    }
    if (c.fwd.joinable()) c.fwd.join();
  }
  children_.clear();
}

void ModuleOrchestratorModule::register_rest(ModuleCtx& ctx,
                                             RestRegistry& rest) {
  OperationSpec list;
  list.method = "GET";
  list.path = "/module-orchestrator/v1/modules";
  list.operation_id = "orchestrator_modules";
  list.summary = "Registered module instances";
  list.authenticated = true;
  list.tags = {"module-orchestrator"};
  rest.register_op(list, [this](HttpRequest& rq, ResponseWriter& w) {
    Json items = Json::array();
    if (module_infos_.is_array())
      for (auto& info : module_infos_.arr()) {
        Json m = info;
        m["status"] = "running";
        items.push_back(m);
      }
    for (auto& c : children_) {
      Json m = Json::object();
      m["name"] = c.name;
      int st = 0;
      m["status"] = (c.pid > 0 && waitpid(c.pid, &st, WNOHANG) == 0)
                        ? "running" : "exited";
      m["runtime"] = "oop";
      m["pid"] = (long)c.pid;
      items.push_back(m);
    }
    Json out = Json::object();
    out["items"] = items;
    w.respond(200, "application/json", out.dump());
  });

  // DirectoryService semantics over REST (proto/directory/v1:
  // RegisterInstance / Heartbeat / Resolve / List)
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/module-orchestrator/v1/instances/register";
    op.operation_id = "register_instance";
    op.summary = "Register an OoP module instance";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"module-orchestrator"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      const std::string name = body.at("name").as_string();
      if (name.empty()) throw Problem::bad_request("'name' is required");
      std::lock_guard<std::mutex> lk(inst_mu_);
      Instance in;
      in.id = name + "-" + std::to_string(++inst_ctr_);
      in.name = name;
      in.endpoint = body.at("endpoint").as_string();
      in.meta = body.at("meta");
      in.last_heartbeat = mono_s();
      instances_[in.id] = in;
      Json out = Json::object();
      out["id"] = in.id;
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/module-orchestrator/v1/instances/{id}/heartbeat";
    op.operation_id = "instance_heartbeat";
    op.summary = "Instance liveness heartbeat";
    op.authenticated = true;
    op.tags = {"module-orchestrator"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      std::lock_guard<std::mutex> lk(inst_mu_);
      auto it = instances_.find(rq.path_params.at("id"));
      if (it == instances_.end())
        throw Problem::not_found("no such instance");
      it->second.last_heartbeat = mono_s();
      w.respond(204, "application/json", "");
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/module-orchestrator/v1/instances";
    op.operation_id = "list_instances";
    op.summary = "Registered instances with liveness";
    op.authenticated = true;
    op.tags = {"module-orchestrator"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      Json items = Json::array();
      const double now = mono_s();
      std::lock_guard<std::mutex> lk(inst_mu_);
      for (auto& [id, in] : instances_) {
        Json m = Json::object();
        m["id"] = id;
        m["name"] = in.name;
        m["endpoint"] = in.endpoint;
        m["alive"] = (now - in.last_heartbeat) < 15.0;
        if (!in.meta.is_null()) m["meta"] = in.meta;
        items.push_back(m);
      }
      Json out = Json::object();
      out["items"] = items;
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/module-orchestrator/v1/instances/resolve/{name}";
    op.operation_id = "resolve_instance";
    op.summary = "Resolve a live instance endpoint by module name";
    op.authenticated = true;
    op.tags = {"module-orchestrator"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      const std::string name = rq.path_params.at("name");
      const double now = mono_s();
      std::lock_guard<std::mutex> lk(inst_mu_);
      const Instance* best = nullptr;
      for (auto& [id, in] : instances_)
        if (in.name == name && (now - in.last_heartbeat) < 15.0 &&
            (!best || in.last_heartbeat > best->last_heartbeat))
          best = &in;
      if (!best) throw Problem::not_found("no live instance of " + name);
      Json out = Json::object();
      out["id"] = best->id;
      out["endpoint"] = best->endpoint;
      w.respond(200, "application/json", out.dump());
    });
  }
}

}  // namespace hs
