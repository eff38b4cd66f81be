#include "oagw.h"

#include "../http/client.h"

#include <algorithm>
#include "../util/log.h"
#include "system_modules.h"

namespace hs {

namespace {

SecurityContext sec_of(HttpRequest& req) {
  return SecurityContext::from_json(req.extensions.at("security"));
}

// validate the Upstream model shape (oagw-sdk/src/models.rs:272-284)
void validate_upstream(const Json& u) {
  if (!u.at("alias").is_string() || u.at("alias").as_string().empty())
    throw Problem::bad_request("'alias' required");
  const Json& eps = u.path("server.endpoints");
  if (!eps.is_array() || eps.size() == 0)
    throw Problem::bad_request("'server.endpoints' required");
  for (const auto& e : eps.arr()) {
    const std::string scheme = e.at("scheme").as_string("http");
    if (scheme != "http" && scheme != "https")
      throw Problem::bad_request("unsupported scheme '" + scheme +
                                 "' (http/https only)");
    if (!e.at("host").is_string())
      throw Problem::bad_request("endpoint host required");
  }
}

}  // namespace

void OagwModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  // config-seeded upstreams (the reference's InMemoryCredentialResolver
  // pattern: config `upstreams:` list for static deployments)
  const Json& ups = ctx.config.at("upstreams");
  if (ups.is_array()) {
    for (auto u : ups.arr()) {
      std::string id = "up-" + std::to_string(next_id_++);
      u["id"] = id;
      if (!u.contains("tenant_id")) u["tenant_id"] = kDefaultTenantId;
      if (!u.contains("enabled")) u["enabled"] = true;
      upstreams_[id] = u;
    }
  }
}

Json* OagwModule::find_upstream(const std::string& tenant,
                                const std::string& alias) {
  for (auto& [id, u] : upstreams_) {
    if (u.at("alias").as_string() == alias &&
        u.at("enabled").as_bool(true) &&
        (u.at("tenant_id").as_string() == tenant ||
         u.at("tenant_id").as_string() == kDefaultTenantId))
      return &u;
  }
  return nullptr;
}

static double mono_s_oagw() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

bool OagwModule::breaker_admit(const std::string& key, int threshold,
                               double open_s, bool* probe) {
  *probe = false;
  if (threshold <= 0) return true;          // breaker disabled
  std::lock_guard<std::mutex> lk(mu_);
  Breaker& b = breakers_[key];
  if (!b.open) return true;
  const double now = mono_s_oagw();
  if (now - b.opened_at < open_s) return false;       // still cooling
  if (b.probing) return false;              // one trial at a time
  b.probing = true;                         // half-open: this call probes
  *probe = true;
  return true;
}

void OagwModule::breaker_report(const std::string& key, bool ok) {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = breakers_.find(key);
  if (it == breakers_.end()) {
    if (!ok) breakers_[key].fails = 1;
    return;
  }
  Breaker& b = it->second;
  if (ok) {
    b.fails = 0;
    b.open = false;
    b.probing = false;
    return;
  }
  b.probing = false;
  if (b.open) {                             // failed half-open probe
    b.opened_at = mono_s_oagw();
    return;
  }
  b.fails++;
  // threshold check happens at the call site (config-dependent)
}

void OagwModule::proxy(HttpRequest& req, ResponseWriter& w) {
  SecurityContext sec = sec_of(req);
  std::string alias = req.path_params["alias"];
  std::string suffix = req.path_params["path"];
  Json up;
  {
    std::lock_guard<std::mutex> lk(mu_);
    Json* u = find_upstream(sec.tenant_id, alias);
    if (!u)
      throw Problem{404, "Not Found", "about:blank",
                    "no upstream with alias '" + alias + "'",
                    "upstream_not_found"};
    up = *u;
    // per-upstream rate limit (token-bucket default,
    // oagw-sdk RateLimitConfig)
    const Json& rl = up.at("rate_limit");
    if (rl.is_object()) {
      auto& b = limiters_[up.at("id").as_string()];
      if (!b)
        b = std::make_unique<TokenBucket>(rl.at("sustained").as_number(50),
                                          rl.at("burst").as_number(100));
      if (!b->try_acquire())
        throw Problem{429, "Too Many Requests", "about:blank",
                      "upstream rate limit", "rate_limited"};
    }
  }
  // route matching (oagw-sdk Route: match_rules{methods, path prefix},
  // priority desc, enabled — models.rs:258-268): when routes exist for
  // this upstream, the request must match one; a matched route's
  // rate_limit applies on top of the upstream's
  {
    std::lock_guard<std::mutex> lk(mu_);
    std::vector<const Json*> rts;
    for (auto& [rid, r] : routes_) {
      if (r.at("upstream_id").as_string() != up.at("id").as_string())
        continue;
      if (!r.at("enabled").as_bool(true)) continue;
      rts.push_back(&r);
    }
    if (!rts.empty()) {
      std::sort(rts.begin(), rts.end(), [](const Json* a, const Json* b) {
        return a->at("priority").as_int(0) > b->at("priority").as_int(0);
      });
      const Json* hit = nullptr;
      for (const Json* r : rts) {
        const Json& mrs = r->at("match_rules");
        const Json& methods = mrs.at("methods");
        bool mok = !methods.is_array() || methods.size() == 0;
        if (!mok)
          for (auto& mm : methods.arr())
            if (mm.as_string() == req.method) mok = true;
        if (!mok) continue;
        const std::string pfx = mrs.at("path").as_string("");
        if (!pfx.empty() &&
            ("/" + suffix).rfind(pfx, 0) != 0)
          continue;
        hit = r;
        break;
      }
      if (!hit)
        throw Problem{404, "Not Found", "about:blank",
                      "no route matches " + req.method + " /" + suffix,
                      "route_not_found"};
      const Json& rl = hit->at("rate_limit");
      if (rl.is_object()) {
        auto& b = limiters_["route:" + hit->at("id").as_string()];
        if (!b)
          b = std::make_unique<TokenBucket>(
              rl.at("sustained").as_number(50),
              rl.at("burst").as_number(100));
        if (!b->try_acquire())
          throw Problem{429, "Too Many Requests", "about:blank",
                        "route rate limit", "rate_limited"};
      }
    }
  }

  const Json& ep = up.path("server.endpoints").arr()[0];
  const std::string host = ep.at("host").as_string();
  const std::string scheme = ep.at("scheme").as_string("http");
  const int port = (int)ep.at("port").as_int(scheme == "https" ? 443 : 80);
  TlsOpts tls;
  tls.enable = scheme == "https";
  tls.ca_file = up.path("tls.ca_file").as_string("");
  tls.verify = up.path("tls.verify").as_bool(true);

  // header policy: drop hop-by-hop + authorization (never forwarded),
  // pass the rest (reference src/infra/proxy/headers.rs allow-list idea)
  std::map<std::string, std::string> fwd;
  for (auto& [k, v] : req.headers) {
    if (k == "host" || k == "connection" || k == "authorization" ||
        k == "content-length" || k == "transfer-encoding")
      continue;
    fwd[k] = v;
  }
  // auth plugin: apikey header injection, value inline or from credstore
  const Json& auth = up.at("auth");
  if (auth.is_object() &&
      auth.at("plugin_type").as_string() == "apikey") {
    std::string header = auth.path("config.header").as_string("x-api-key");
    std::string value = auth.path("config.value").as_string("");
    std::string ref = auth.path("config.credential_ref").as_string("");
    if (!ref.empty()) {
      auto cs = hub_->get<CredStoreClient>("credstore");
      auto v = cs ? cs->get(sec.tenant_id, ref) : std::nullopt;
      if (!v)
        throw Problem{502, "Bad Gateway", "about:blank",
                      "credential_ref not resolvable", "provider_error"};
      value = *v;
    }
    if (!value.empty()) fwd[header] = value;
  }

  std::string target = "/" + suffix;
  if (req.target.find('?') != std::string::npos)
    target += req.target.substr(req.target.find('?'));

  // circuit breaker (ADR-0004 infrastructure layer): consecutive
  // connect/5xx failures open the endpoint; after `open_ms` ONE
  // half-open probe decides recovery.  Config per upstream:
  //   circuit: {failure_threshold: 5, open_ms: 30000}   (0 disables)
  const int cb_threshold =
      (int)up.path("circuit.failure_threshold").as_int(5);
  const double cb_open_s =
      up.path("circuit.open_ms").as_number(30000) / 1000.0;
  const std::string cb_key =
      alias + "|" + host + ":" + std::to_string(port);
  bool cb_probe = false;
  if (!breaker_admit(cb_key, cb_threshold, cb_open_s, &cb_probe))
    throw Problem{503, "Service Unavailable", "about:blank",
                  "upstream circuit is open (cooling down)",
                  "circuit_open"};
  auto cb_result = [&](bool ok) {
    if (cb_threshold <= 0) return;
    breaker_report(cb_key, ok);
    if (!ok && !cb_probe) {
      std::lock_guard<std::mutex> lk(mu_);
      Breaker& b = breakers_[cb_key];
      if (!b.open && b.fails >= cb_threshold) {
        b.open = true;
        b.opened_at = mono_s_oagw();
      }
    }
  };

  // SSE pass-through (reference oagw: no total timeout so SSE can
  // stream, src/infra/proxy/service.rs:43-49): when the client asks for
  // an event stream, chunks are forwarded as they arrive
  if (req.header("accept").find("text/event-stream") != std::string::npos) {
    bool started = false;
    int upstatus = 200;
    std::string upct = "text/event-stream";
    auto r = http_request(
        host, port, req.method, target, fwd, req.body, 10000,
        [&](const char* p, size_t n) {
          if (!started) {
            w.begin_stream(upstatus, upct, {{"x-oagw-upstream", alias}});
            started = true;
          }
          return w.write_chunk(std::string(p, n));
        },
        [&](const ClientResponse& hr) {
          upstatus = hr.status;
          auto it = hr.headers.find("content-type");
          if (it != hr.headers.end()) upct = it->second;
        },
        &tls);
    if (started) {
      cb_result(true);
      w.end_stream();
      return;
    }
    if (!r) {
      cb_result(false);
      throw Problem{502, "Bad Gateway", "about:blank",
                    "upstream connect failed", "provider_error"};
    }
    cb_result(r->status < 500);
    // empty-bodied response: fall through with the buffered result
    w.respond(r->status, upct, "", {{"x-oagw-upstream", alias}});
    return;
  }

  // forward (buffered)
  auto resp = http_request(host, port, req.method, target, fwd, req.body,
                           10000, nullptr, nullptr, &tls);
  if (!resp) {
    cb_result(false);
    throw Problem{502, "Bad Gateway", "about:blank",
                  "upstream connect failed", "provider_error"};
  }
  cb_result(resp->status < 500);
  std::string ct = resp->headers.count("content-type")
      ? resp->headers["content-type"] : "application/octet-stream";
  std::vector<std::pair<std::string, std::string>> hdrs;
  for (auto& [k, v] : resp->headers) {
    if (k == "content-type" || k == "content-length" ||
        k == "connection" || k == "transfer-encoding")
      continue;
    hdrs.push_back({k, v});
  }
  hdrs.push_back({"x-oagw-upstream", alias});
  w.respond(resp->status, ct, resp->body, hdrs);
}

void OagwModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  auto crud = [this, &rest](const std::string& kind,
                            std::map<std::string, Json>* store,
                            bool validate) {
    OperationSpec list;
    list.method = "GET";
    list.path = "/oagw/v1/" + kind;
    list.operation_id = "oagw_" + kind + "_list";
    list.authenticated = true;
    list.tags = {"oagw"};
    rest.register_op(list, [this, store](HttpRequest& rq,
                                         ResponseWriter& w) {
      // ListQuery top/skip (oagw-sdk models.rs:293-303)
      long top = rq.query.count("top") ? atol(rq.query["top"].c_str()) : 50;
      long skip = rq.query.count("skip") ? atol(rq.query["skip"].c_str()) : 0;
      SecurityContext sec = sec_of(rq);
      Json items = Json::array();
      std::lock_guard<std::mutex> lk(mu_);
      long i = 0;
      for (auto& [id, u] : *store) {
        if (u.at("tenant_id").as_string() != sec.tenant_id &&
            u.at("tenant_id").as_string() != kDefaultTenantId)
          continue;
        if (i++ < skip) continue;
        if ((long)items.size() >= top) break;
        items.push_back(u);
      }
      Json out = Json::object();
      out["items"] = items;
      w.respond(200, "application/json", out.dump());
    });

    OperationSpec create;
    create.method = "POST";
    create.path = "/oagw/v1/" + kind;
    create.operation_id = "oagw_" + kind + "_create";
    create.authenticated = true;
    create.allowed_content_types = {"application/json"};
    create.tags = {"oagw"};
    rest.register_op(create, [this, store, validate, kind](
                                 HttpRequest& rq, ResponseWriter& w) {
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON"); }
      if (validate) validate_upstream(body);
      SecurityContext sec = sec_of(rq);
      std::lock_guard<std::mutex> lk(mu_);
      std::string id = kind.substr(0, 2) + "-" + std::to_string(next_id_++);
      body["id"] = id;
      body["tenant_id"] = sec.tenant_id;
      if (!body.contains("enabled")) body["enabled"] = true;
      (*store)[id] = body;
      w.respond(201, "application/json", body.dump());
    });

    OperationSpec get;
    get.method = "GET";
    get.path = "/oagw/v1/" + kind + "/{id}";
    get.operation_id = "oagw_" + kind + "_get";
    get.authenticated = true;
    get.tags = {"oagw"};
    rest.register_op(get, [this, store](HttpRequest& rq,
                                        ResponseWriter& w) {
      SecurityContext sec = sec_of(rq);
      std::lock_guard<std::mutex> lk(mu_);
      auto it = store->find(rq.path_params["id"]);
      if (it == store->end() ||
          (it->second.at("tenant_id").as_string() != sec.tenant_id &&
           it->second.at("tenant_id").as_string() != kDefaultTenantId))
        throw Problem::not_found();
      w.respond(200, "application/json", it->second.dump());
    });

    OperationSpec del;
    del.method = "DELETE";
    del.path = "/oagw/v1/" + kind + "/{id}";
    del.operation_id = "oagw_" + kind + "_delete";
    del.authenticated = true;
    del.tags = {"oagw"};
    rest.register_op(del, [this, store](HttpRequest& rq,
                                        ResponseWriter& w) {
      SecurityContext sec = sec_of(rq);
      std::lock_guard<std::mutex> lk(mu_);
      auto it = store->find(rq.path_params["id"]);
      if (it == store->end() ||
          it->second.at("tenant_id").as_string() != sec.tenant_id)
        throw Problem::not_found();
      store->erase(it);
      w.respond(204, "application/json", "");
    });
  };
  crud("upstreams", &upstreams_, true);
  crud("routes", &routes_, false);

  OperationSpec px;
  px.method = "POST";
  px.path = "/oagw/v1/proxy/{alias}/{*path}";
  px.operation_id = "oagw_proxy_post";
  px.summary = "Proxy a request through a configured upstream";
  px.authenticated = true;
  px.tags = {"oagw"};
  rest.register_op(px, [this](HttpRequest& rq, ResponseWriter& w) {
    proxy(rq, w);
  });
  OperationSpec pxg = px;
  pxg.method = "GET";
  pxg.operation_id = "oagw_proxy_get";
  rest.register_op(pxg, [this](HttpRequest& rq, ResponseWriter& w) {
    proxy(rq, w);
  });
}

}  // namespace hs
