#include "api_gateway.h"

#include "../modkit/json_schema.h"

#include "../modkit/telemetry.h"

#include <random>

#include "../util/log.h"

namespace hs {

bool TokenBucket::try_acquire() {
  std::lock_guard<std::mutex> lk(mu_);
  auto now = std::chrono::steady_clock::now();
  double dt = std::chrono::duration<double>(now - last_).count();
  last_ = now;
  tokens_ = std::min(burst_, tokens_ + dt * rps_);
  if (tokens_ >= 1.0) {
    tokens_ -= 1.0;
    return true;
  }
  return false;
}

static std::string gen_request_id() {
  static std::atomic<uint64_t> ctr{0};
  static std::random_device rd;
  char buf[40];
  snprintf(buf, sizeof buf, "%08x-%04x-%012llx", rd(),
           unsigned(rd() & 0xffff),
           (unsigned long long)ctr.fetch_add(1));
  return buf;
}

void ApiGatewayModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  const Json& c = ctx.config;
  bind_addr_ = c.at("bind_addr").as_string(bind_addr_);
  enable_docs_ = c.at("enable_docs").as_bool(true);
  cors_enabled_ = c.at("cors_enabled").as_bool(false);
  cors_cfg_ = c.at("cors");
  auth_disabled_ = c.at("auth_disabled").as_bool(false);
  if (auth_disabled_)
    LOG_WARN("api-gateway",
             "authentication is DISABLED - requests run as the default "
             "tenant (single-user/on-prem mode)");
  openapi_title_ = c.path("openapi.title").as_string(openapi_title_);
  openapi_version_ = c.path("openapi.version").as_string(openapi_version_);
  openapi_desc_ = c.path("openapi.description").as_string("");
  default_rl_.rps = c.path("defaults.rate_limit.rps").as_number(50);
  default_rl_.burst = c.path("defaults.rate_limit.burst").as_number(100);
  default_rl_.in_flight =
      (int)c.path("defaults.rate_limit.in_flight").as_int(64);
  body_limit_ =
      (size_t)c.path("defaults.body_limit_bytes").as_int(16 * 1024 * 1024);
  if (c.path("license.features").is_array())
    for (auto& f : c.path("license.features").arr())
      licensed_features_.push_back(f.as_string());
}

void ApiGatewayModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  // /health /healthz live on the gateway itself (reference src/web.rs:23)
  OperationSpec health;
  health.method = "GET";
  health.path = "/health";
  health.operation_id = "health";
  health.summary = "Detailed health JSON";
  health.is_public = true;
  health.tags = {"system"};
  rest.register_op(health, [this](HttpRequest& rq, ResponseWriter& w) {
    Json j = Json::object();
    j["status"] = "ok";
    j["version"] = "0.1.0";
    j["uptime_seconds"] = std::chrono::duration<double>(
        std::chrono::steady_clock::now() - start_time_).count();
    j["requests_total"] = (long)req_counter_.load();
    w.respond(200, "application/json", j.dump());
  });
  OperationSpec hz;
  hz.method = "GET";
  hz.path = "/healthz";
  hz.operation_id = "healthz";
  hz.is_public = true;
  hz.tags = {"system"};
  rest.register_op(hz, [](HttpRequest& rq, ResponseWriter& w) {
    w.respond(200, "text/plain", "ok");
  });
}

void ApiGatewayModule::start(ModuleCtx& ctx) {
  buckets_.clear();
  in_flight_.clear();
  for (const auto& r : rest_.routes()) {
    RateLimitCfg rl = r.spec.rate_limit.value_or(default_rl_);
    buckets_.push_back(std::make_unique<TokenBucket>(rl.rps, rl.burst));
    in_flight_.push_back(std::make_unique<std::atomic<int>>(0));
  }
  server_ = std::make_unique<HttpServer>(
      bind_addr_, [this](HttpRequest& rq, ResponseWriter& w) {
        handle(rq, w);
      }, body_limit_);
  if (!server_->start())
    throw std::runtime_error("api-gateway: cannot bind " + bind_addr_);
  LOG_INFO("api-gateway", "listening on %s (port %d)", bind_addr_.c_str(),
           server_->port());
}

void ApiGatewayModule::stop(ModuleCtx& ctx) {
  if (server_) server_->stop();
}

static const char* kDocsPage = R"HTML(<!doctype html>
<html><head><meta charset="utf-8"><title>API Documentation</title>
<style>
body{font-family:system-ui,sans-serif;margin:2rem;max-width:960px}
h1{font-size:1.4rem} .op{border:1px solid #ddd;border-radius:6px;margin:.5rem 0;padding:.6rem}
.m{display:inline-block;min-width:4rem;font-weight:700;padding:.1rem .4rem;border-radius:4px;color:#fff;text-align:center}
.GET{background:#2f8132}.POST{background:#1a6fb5}.DELETE{background:#b03030}.PUT{background:#b07c1a}.PATCH{background:#7c1ab0}
code{background:#f4f4f4;padding:.1rem .3rem;border-radius:3px}
pre{background:#f8f8f8;padding:.6rem;border-radius:6px;overflow:auto;max-height:320px}
.auth{float:right;color:#888;font-size:.85rem}
</style></head><body>
<h1 id="t">API Documentation</h1>
<p>OpenAPI document: <a href="/openapi.json">/openapi.json</a></p>
<div id="ops">loading…</div>
<script>
fetch('/openapi.json').then(r=>r.json()).then(d=>{
 document.getElementById('t').textContent=d.info.title+' '+d.info.version;
 const out=[];
 for(const [p,methods] of Object.entries(d.paths||{}))
  for(const [m,op] of Object.entries(methods)){
   const auth=op.security?'&#128274; bearer':'public';
   let body='';
   if(op.requestBody){body='<pre>'+JSON.stringify(op.requestBody.content['application/json'].schema,null,1)+'</pre>';}
   out.push(`<div class="op"><span class="auth">${auth}</span><span class="m ${m.toUpperCase()}">${m.toUpperCase()}</span> <code>${p}</code><div>${op.summary||''}</div>${body}</div>`);
  }
 document.getElementById('ops').innerHTML=out.join('');
});
</script></body></html>)HTML";

void ApiGatewayModule::handle(HttpRequest& req, ResponseWriter& w) {
  const auto t0 = std::chrono::steady_clock::now();
  // 1. request id (propagate inbound x-request-id)
  req.request_id = req.header("x-request-id");
  if (req.request_id.empty()) req.request_id = gen_request_id();
  req_counter_++;

  // 1b. W3C trace context: parse inbound traceparent
  // (00-<trace32>-<span16>-<flags>), mint this hop's span, expose both to
  // handlers (extensions.trace) and echo traceparent downstream — the
  // reference's http_request span parenting (api-gateway module.rs:274-330)
  {
    std::string tp = req.header("traceparent");
    std::string trace_id, parent_span;
    if (tp.size() >= 55 && tp[2] == '-' && tp[35] == '-' && tp[52] == '-') {
      trace_id = tp.substr(3, 32);
      parent_span = tp.substr(36, 16);
    }
    auto hex = [](uint64_t v, int w) {
      char b[32];
      snprintf(b, sizeof b, "%0*llx", w, (unsigned long long)v);
      return std::string(b);
    };
    static std::atomic<uint64_t> span_ctr{0x1a2b};
    uint64_t sp = (uint64_t)time(nullptr) << 20 ^ (++span_ctr * 0x9e3779b9);
    if (trace_id.empty())
      trace_id = hex(sp ^ 0xdeadbeefcafe1234ull, 16) + hex(sp * 31, 16);
    Json tr = Json::object();
    tr["trace_id"] = trace_id;
    tr["span_id"] = hex(sp, 16);
    if (!parent_span.empty()) tr["parent_span_id"] = parent_span;
    req.extensions["trace"] = tr;
  }

  // CORS preflight short-circuit + response headers
  std::vector<std::pair<std::string, std::string>> cors_headers;
  if (cors_enabled_) {
    std::string origin = req.header("origin", "*");
    cors_headers = {
        {"access-control-allow-origin",
         cors_cfg_.at("allowed_origins").is_array()
             ? origin : "*"},
        {"access-control-allow-headers", "authorization, content-type"},
        {"access-control-allow-methods",
         "GET, POST, PUT, PATCH, DELETE, OPTIONS"}};
    if (req.method == "OPTIONS") {
      w.respond(204, "text/plain", "", cors_headers);
      return;
    }
  }

  const uint64_t span_start_ns = (uint64_t)
      std::chrono::duration_cast<std::chrono::nanoseconds>(
          std::chrono::system_clock::now().time_since_epoch()).count();
  dispatch(req, w);
  auto dt = std::chrono::duration<double, std::milli>(
      std::chrono::steady_clock::now() - t0).count();
  // http_request span record (OTel semantic fields): logged + exported
  // over OTLP/HTTP when tracing.otlp_endpoint is configured
  LOG_DEBUG("http", "%s %s -> done in %.2fms rid=%s trace=%s span=%s",
            req.method.c_str(), req.path.c_str(), dt,
            req.request_id.c_str(),
            req.extensions.path("trace.trace_id").as_string().c_str(),
            req.extensions.path("trace.span_id").as_string().c_str());
  auto& exp = TraceExporter::instance();
  if (exp.enabled()) {
    SpanRecord sp;
    sp.trace_id = req.extensions.path("trace.trace_id").as_string();
    sp.span_id = req.extensions.path("trace.span_id").as_string();
    sp.parent_span_id =
        req.extensions.path("trace.parent_span_id").as_string();
    sp.name = "http_request";
    sp.start_ns = span_start_ns;
    sp.end_ns = span_start_ns + (uint64_t)(dt * 1e6);
    const int st = w.status();
    sp.status_code = st >= 500 ? 2 : 1;
    sp.attrs = {{"http.request.method", req.method},
                {"url.path", req.path},
                {"http.response.status_code", std::to_string(st)},
                {"http.request.id", req.request_id}};
    exp.record(std::move(sp));
  }
}

void ApiGatewayModule::dispatch(HttpRequest& req, ResponseWriter& w) {
  const std::string rid_hdr = req.request_id;
  const std::string tp_hdr =
      "00-" + req.extensions.path("trace.trace_id").as_string() + "-" +
      req.extensions.path("trace.span_id").as_string() + "-01";
  const std::vector<std::pair<std::string, std::string>> rid_headers = {
      {"x-request-id", rid_hdr}, {"traceparent", tp_hdr}};
  w.default_headers = rid_headers;

  // built-ins outside the registry: docs + openapi
  if (req.method == "GET" && req.path == "/openapi.json") {
    w.respond(200, "application/json",
              rest_.build_openapi(openapi_title_, openapi_version_,
                                  openapi_desc_).dump(1), rid_headers);
    return;
  }
  if (req.method == "GET" && (req.path == "/docs" || req.path == "/docs/")) {
    if (!enable_docs_) {
      respond_problem(w, Problem::not_found(), req.path);
      return;
    }
    w.respond(200, "text/html; charset=utf-8", kDocsPage, rid_headers);
    return;
  }

  std::map<std::string, std::string> params;
  bool path_exists = false;
  const Route* route = rest_.match(req.method, req.path, params,
                                   &path_exists);
  if (!route) {
    Problem p = path_exists
        ? Problem{405, "Method Not Allowed", "about:blank", "", ""}
        : Problem::not_found();
    respond_problem(w, p, req.path);
    return;
  }
  req.path_params = std::move(params);
  const size_t ridx = size_t(route - rest_.routes().data());

  // MIME validation (415)
  if (!route->spec.allowed_content_types.empty() && !req.body.empty()) {
    std::string ct = req.header("content-type");
    std::string base = ct.substr(0, ct.find(';'));
    bool ok = false;
    for (auto& a : route->spec.allowed_content_types)
      if (a == base) { ok = true; break; }
    if (!ok) {
      respond_problem(w, {415, "Unsupported Media Type", "about:blank",
                          "content-type '" + base + "' not allowed", ""},
                      req.path);
      return;
    }
  }

  // license validation (reference src/middleware/license_validation.rs:
  // route specs declare required features; the gateway config grants them)
  if (!route->spec.license_features.empty()) {
    for (auto& need : route->spec.license_features) {
      bool have = false;
      for (auto& f : licensed_features_)
        if (f == need) { have = true; break; }
      if (!have) {
        respond_problem(w, {403, "Forbidden", "about:blank",
                            "license feature '" + need + "' not granted",
                            "license_required"}, req.path);
        return;
      }
    }
  }

  // rate limit: token bucket + in-flight cap (reference
  // src/middleware/rate_limit.rs:53-86)
  if (!buckets_[ridx]->try_acquire()) {
    respond_problem(w, {429, "Too Many Requests", "about:blank",
                        "rate limit exceeded", "rate_limited"}, req.path);
    return;
  }
  struct InFlightGuard {
    std::atomic<int>* c;
    ~InFlightGuard() { if (c) c->fetch_sub(1); }
  } guard{nullptr};
  {
    RateLimitCfg rl = route->spec.rate_limit.value_or(default_rl_);
    int cur = in_flight_[ridx]->fetch_add(1);
    guard.c = in_flight_[ridx].get();
    if (cur >= rl.in_flight) {
      respond_problem(w, {429, "Too Many Requests", "about:blank",
                          "too many in-flight requests", "rate_limited"},
                      req.path);
      return;
    }
  }

  // authn (reference src/middleware/auth.rs:197)
  SecurityContext sec = SecurityContext::anonymous();
  if (route->spec.authenticated) {
    if (auth_disabled_) {
      sec = SecurityContext::default_ctx();
    } else {
      std::string auth = req.header("authorization");
      if (auth.rfind("Bearer ", 0) != 0) {
        respond_problem(w, Problem::unauthorized("missing bearer token"),
                        req.path);
        return;
      }
      auto client = hub_->get<AuthnResolverClient>("authn-resolver");
      if (!client) {
        respond_problem(w, {503, "Service Unavailable", "about:blank",
                            "authn resolver not available", ""}, req.path);
        return;
      }
      auto ctx = client->authenticate(auth.substr(7));
      if (!ctx) {
        respond_problem(w, Problem::unauthorized("invalid token"), req.path);
        return;
      }
      sec = *ctx;
    }
  }
  req.extensions["security"] = sec.to_json();

  // declared request schema => automatic body validation (the runtime
  // analog of the reference's typed OperationBuilder request bodies);
  // non-JSON/empty bodies are left to the handler's own checks
  if (route->spec.request_schema.is_object() && !req.body.empty()) {
    Json body;
    bool parsed = true;
    try { body = Json::parse(req.body); } catch (...) { parsed = false; }
    if (parsed) {
      const std::string err =
          json_schema_err(route->spec.request_schema, body, "body");
      if (!err.empty()) {
        respond_problem(w, {400, "Bad Request", "about:blank",
                            "request validation: " + err,
                            "validation_error"}, req.path);
        return;
      }
    }
  }

  // error-mapping boundary around the handler
  try {
    route->handler(req, w);
  } catch (const Problem& p) {
    if (!w.started()) respond_problem(w, p, req.path);
  } catch (const std::exception& e) {
    LOG_ERROR("api-gateway", "handler error on %s: %s", req.path.c_str(),
              e.what());
    if (!w.started())
      respond_problem(w, {500, "Internal Server Error", "about:blank",
                          "", ""}, req.path);
  }
}

}  // namespace hs
