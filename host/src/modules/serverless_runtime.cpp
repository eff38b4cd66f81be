#include "serverless_runtime.h"

#include "../modkit/json_schema.h"

#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <ctime>
#include <sstream>

#include "../util/log.h"
#include "llm_gateway.h"       // AdmissionClient, ChatInvoker
#include "system_modules.h"

namespace hs {

namespace {

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

std::string now_iso() {
  time_t t = time(nullptr);
  char buf[32];
  strftime(buf, sizeof buf, "%Y-%m-%dT%H:%M:%SZ", gmtime(&t));
  return buf;
}

// ---- per-tenant admission (token bucket + in-flight cap) ----
// noisy-neighbor isolation NFR, reference PRD.md:944-957
class StaticAdmission : public AdmissionClient {
 public:
  explicit StaticAdmission(const Json& cfg) {
    max_concurrent_ =
        (int)cfg.path("limits.max_concurrent_per_tenant").as_int(64);
    rps_ = cfg.path("limits.rps_per_tenant").as_number(100);
    burst_ = cfg.path("limits.burst_per_tenant").as_number(200);
  }
  std::string admit(const std::string& tenant) override {
    std::lock_guard<std::mutex> lk(mu_);
    auto& st = tenants_[tenant];
    auto now = std::chrono::steady_clock::now();
    if (st.last.time_since_epoch().count() == 0) st.tokens = burst_;
    double dt = std::chrono::duration<double>(now - st.last).count();
    st.last = now;
    st.tokens = std::min(burst_, st.tokens + dt * rps_);
    if (st.in_flight >= max_concurrent_) return "rate_limited";
    if (st.tokens < 1.0) return "rate_limited";
    st.tokens -= 1.0;
    st.in_flight++;
    return "";
  }
  void release(const std::string& tenant) override {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = tenants_.find(tenant);
    if (it != tenants_.end() && it->second.in_flight > 0)
      it->second.in_flight--;
  }

 private:
  struct State {
    int in_flight = 0;
    double tokens = 0;
    std::chrono::steady_clock::time_point last{};
  };
  std::mutex mu_;
  std::map<std::string, State> tenants_;
  int max_concurrent_;
  double rps_, burst_;
};

// ADR:1068-1087 allowed-transition table
bool allowed_transition(const std::string& from, const std::string& to) {
  static const std::map<std::string, std::vector<std::string>> t = {
      {"queued", {"running", "canceled"}},
      {"running", {"succeeded", "failed", "suspended", "canceled"}},
      {"suspended", {"running", "canceled", "failed"}},
      {"failed", {"queued", "compensating", "dead_lettered"}},
      {"canceled", {"compensating"}},
      {"compensating", {"compensated", "dead_lettered"}},
  };
  auto it = t.find(from);
  return it != t.end() &&
         std::find(it->second.begin(), it->second.end(), to) !=
             it->second.end();
}

bool is_terminal(const std::string& s) {
  return s == "succeeded" || s == "compensated" || s == "dead_lettered" ||
         s == "canceled" || s == "failed";
}

// validation per implementation kinds (ADR:549-600)
Json validate_entrypoint(const Json& body) {
  Json issues = Json::array();
  auto issue = [&](const std::string& m) {
    Json i = Json::object();
    i["message"] = m;
    issues.push_back(i);
  };
  if (body.at("name").as_string().empty()) issue("'name' is required");
  const std::string et = body.at("entrypoint_type").as_string();
  if (et != "function" && et != "workflow")
    issue("'entrypoint_type' must be function|workflow");
  const Json& impl = body.at("implementation");
  if (!impl.is_object()) {
    issue("'implementation' is required");
    return issues;
  }
  if (impl.at("adapter").as_string().empty())
    issue("implementation.adapter is required");
  const std::string kind = impl.at("kind").as_string();
  if (kind == "code") {
    if (impl.path("code.language").as_string().empty() ||
        impl.path("code.source").as_string().empty())
      issue("code implementation needs code.language + code.source");
  } else if (kind == "workflow_spec") {
    const Json& steps = impl.path("workflow.steps");
    if (!steps.is_array() || steps.size() == 0)
      issue("workflow_spec needs workflow.steps[]");
    else
      for (auto& st : steps.arr()) {
        if (st.at("name").as_string().empty() ||
            st.at("op").as_string().empty())
          issue("every step needs name + op");
        const Json& wh = st.at("when");
        if (!wh.is_null()) {
          if (!wh.is_object() || wh.at("field").as_string().empty())
            issue("step.when needs {field, op, value}");
          else {
            const std::string wop = wh.at("op").as_string("exists");
            if (wop != "eq" && wop != "ne" && wop != "exists" &&
                wop != "gt" && wop != "lt")
              issue("step.when.op must be eq|ne|exists|gt|lt");
          }
        }
        if (!st.at("args").is_null() && !st.at("args").is_object())
          issue("step.args must be an object");
        if (!st.at("output_to").is_null() &&
            !st.at("output_to").is_string())
          issue("step.output_to must be a string");
      }
    if (et != "workflow") issue("workflow_spec requires entrypoint_type "
                                "workflow");
  } else if (kind == "adapter_ref") {
    // adapter-provided definition; nothing further to validate locally
  } else {
    issue("implementation.kind must be code|workflow_spec|adapter_ref");
  }
  const Json& rp = body.at("retry_policy");
  if (rp.is_object() && rp.at("max_attempts").as_int(1) < 1)
    issue("retry_policy.max_attempts must be >= 1");
  const Json& io = body.at("io_schema");
  if (!io.is_null() && !io.is_object())
    issue("io_schema must be an object (params/returns/errors)");
  if (io.is_object())
    for (const char* k : {"params", "returns"}) {
      const Json& s = io.at(k);
      if (!s.is_null() && !s.is_object())
        issue(std::string("io_schema.") + k +
              " must be a JSON Schema object, gts $ref, or null (void)");
    }
  return issues;
}

// ---- schedule expressions (ADR:2038-2164: cron | ISO-8601 interval,
// evaluated in UTC — `timezone` other than UTC is rejected at create) ----

// parse one cron field ("*", "a", "a-b", "*/n", "a-b/n", lists) into an
// allow-set over [lo, hi]
bool cron_field(const std::string& f, int lo, int hi,
                std::vector<bool>& out) {
  out.assign((size_t)hi + 1, false);
  size_t pos = 0;
  while (pos <= f.size()) {
    size_t comma = f.find(',', pos);
    std::string part = f.substr(
        pos, comma == std::string::npos ? std::string::npos : comma - pos);
    if (part.empty()) return false;
    int step = 1;
    size_t slash = part.find('/');
    bool stepped = slash != std::string::npos;
    if (stepped) {
      step = atoi(part.c_str() + slash + 1);
      part = part.substr(0, slash);
      if (step < 1) return false;
    }
    int a = lo, b = hi;
    if (part != "*") {
      size_t dash = part.find('-');
      if (dash != std::string::npos) {
        a = atoi(part.substr(0, dash).c_str());
        b = atoi(part.substr(dash + 1).c_str());
      } else {
        a = atoi(part.c_str());
        b = stepped ? hi : a;   // "5/10" = every 10 starting at 5
      }
    }
    if (a < lo || b > hi || a > b) return false;
    for (int v = a; v <= b; v += step) out[(size_t)v] = true;
    if (comma == std::string::npos) break;
    pos = comma + 1;
  }
  return true;
}

// next fire strictly after `after`, or 0 on parse failure / no match
// within 366 days.  5 fields: min hour day-of-month month day-of-week.
time_t cron_next(const std::string& expr, time_t after) {
  std::istringstream ss(expr);
  std::string f[5], extra;
  for (int i = 0; i < 5; ++i)
    if (!(ss >> f[i])) return 0;
  if (ss >> extra) return 0;
  std::vector<bool> mi, ho, dom, mon, dow;
  if (!cron_field(f[0], 0, 59, mi) || !cron_field(f[1], 0, 23, ho) ||
      !cron_field(f[2], 1, 31, dom) || !cron_field(f[3], 1, 12, mon) ||
      !cron_field(f[4], 0, 6, dow))
    return 0;
  const bool dom_any = f[2] == "*", dow_any = f[4] == "*";
  time_t t = after - after % 60 + 60;
  for (int i = 0; i < 366 * 24 * 60; ++i, t += 60) {
    tm g{};
    gmtime_r(&t, &g);
    if (!mon[(size_t)g.tm_mon + 1] || !mi[(size_t)g.tm_min] ||
        !ho[(size_t)g.tm_hour])
      continue;
    const bool dmatch = dom[(size_t)g.tm_mday];
    const bool wmatch = dow[(size_t)g.tm_wday];
    // vixie-cron rule: both day fields restricted -> either may match
    if (dom_any && dow_any ? true
        : dom_any          ? wmatch
        : dow_any          ? dmatch
                           : (dmatch || wmatch))
      return t;
  }
  return 0;
}

// ISO-8601 duration (PT1H, P1D, PT90S, P1DT12H...) or plain seconds
double interval_seconds(const std::string& s) {
  if (s.empty()) return 0;
  if (s[0] != 'P') {
    char* end = nullptr;
    double v = strtod(s.c_str(), &end);
    return end && *end == '\0' ? v : 0;
  }
  double total = 0, num = 0;
  bool in_time = false, have = false;
  for (size_t i = 1; i < s.size(); ++i) {
    char c = s[i];
    if (c == 'T') {
      in_time = true;
      continue;
    }
    if (isdigit((unsigned char)c) || c == '.') {
      size_t j = i;
      while (j < s.size() &&
             (isdigit((unsigned char)s[j]) || s[j] == '.'))
        j++;
      num = atof(s.substr(i, j - i).c_str());
      have = true;
      i = j - 1;
      continue;
    }
    double mult = 0;
    if (c == 'W' && !in_time) mult = 604800;
    else if (c == 'D' && !in_time) mult = 86400;
    else if (c == 'H' && in_time) mult = 3600;
    else if (c == 'M') mult = in_time ? 60 : 2592000;
    else if (c == 'S' && in_time) mult = 1;
    else return 0;
    if (!have) return 0;
    total += num * mult;
    have = false;
  }
  return total;
}

time_t parse_iso(const std::string& s) {
  tm g{};
  if (!strptime(s.c_str(), "%Y-%m-%dT%H:%M:%S", &g)) return 0;
  return timegm(&g);
}

std::string iso_of(time_t t) {
  char buf[32];
  tm g{};
  gmtime_r(&t, &g);
  strftime(buf, sizeof buf, "%Y-%m-%dT%H:%M:%SZ", &g);
  return buf;
}

// next occurrence strictly after `after` per the expression, 0 = invalid
time_t sched_next(const std::string& kind, const std::string& value,
                  time_t after) {
  if (kind == "cron") return cron_next(value, after);
  if (kind == "interval") {
    double iv = interval_seconds(value);
    return iv >= 1 ? after + (time_t)iv : 0;
  }
  return 0;
}

// ---- io_schema validation (ADR:131-185; PRD BR-032/BR-037: inputs are
// validated before invocation).  Minimal JSON-Schema subset: type,
// required, properties, additionalProperties:false, items, enum, const,
// minimum/maximum, minLength/maxLength.  `$ref` (gts:// types) passes —
// resolution against the types registry is out of scope here. ----
// step `when` predicate: {field, op: eq|ne|exists|gt|lt, value}
bool when_matches(const Json& cond, const Json& state) {
  const std::string field = cond.at("field").as_string();
  const std::string op = cond.at("op").as_string("exists");
  const bool has = state.contains(field);
  if (op == "exists") return has;
  if (!has) return false;
  const Json& v = state.at(field);
  const Json& want = cond.at("value");
  if (op == "eq") return v.dump() == want.dump();
  if (op == "ne") return v.dump() != want.dump();
  if (op == "gt") return v.as_number() > want.as_number();
  if (op == "lt") return v.as_number() < want.as_number();
  return false;
}

std::string schema_err(const Json& sch, const Json& v,
                       const std::string& path) {
  return json_schema_err(sch, v, path);
}

SecurityContext sec_of(HttpRequest& rq) {
  return SecurityContext::from_json(rq.extensions.at("security"));
}

Json row_entrypoint(const DbRow& r) {
  Json j;
  try { j = Json::parse(r.at("spec").as_string()); }
  catch (...) { j = Json::object(); }
  j["id"] = r.at("id");
  j["status"] = r.at("status");
  j["version"] = r.at("version");
  j["created_at"] = r.at("created_at");
  j["updated_at"] = r.at("updated_at");
  return j;
}

Json row_invocation(const DbRow& r) {
  Json j = Json::object();
  j["id"] = r.at("id");
  j["entrypoint_id"] = r.at("entrypoint_id");
  j["status"] = r.at("status");
  j["mode"] = r.at("mode");
  const std::string sid = r.at("schedule_id").as_string();
  if (!sid.empty()) j["schedule_id"] = sid;
  j["attempts"] = r.at("attempts");
  j["created_at"] = r.at("created_at");
  j["updated_at"] = r.at("updated_at");
  try { j["input"] = Json::parse(r.at("input").as_string()); }
  catch (...) {}
  const std::string res = r.at("result").as_string();
  if (!res.empty()) {
    try { j["result"] = Json::parse(res); }
    catch (...) { j["result"] = res; }
  }
  const std::string err = r.at("error").as_string();
  if (!err.empty()) j["error"] = err;
  return j;
}

Json row_schedule(const DbRow& r) {
  Json j = Json::object();
  j["schedule_id"] = r.at("id");
  j["tenant_id"] = r.at("tenant_id");
  j["entrypoint_id"] = r.at("entrypoint_id");
  j["name"] = r.at("name");
  j["timezone"] = r.at("timezone");
  Json expr = Json::object();
  expr["kind"] = r.at("expr_kind");
  expr["value"] = r.at("expr_value");
  j["expression"] = expr;
  try {
    j["input_overrides"] = Json::parse(r.at("input_overrides").as_string());
  } catch (...) {}
  j["missed_policy"] = r.at("missed_policy");
  j["status"] = r.at("status");
  for (const char* k : {"next_run_at", "last_run_at"}) {
    const std::string v = r.at(k).as_string();
    if (!v.empty()) j[k] = v;
    else j[k] = Json();
  }
  j["created_at"] = r.at("created_at");
  j["updated_at"] = r.at("updated_at");
  return j;
}

Json row_trigger(const DbRow& r) {
  Json j = Json::object();
  j["trigger_id"] = r.at("id");
  j["tenant_id"] = r.at("tenant_id");
  j["event_type_id"] = r.at("event_type_id");
  const std::string f = r.at("event_filter_query").as_string();
  if (!f.empty()) j["event_filter_query"] = f;
  j["entrypoint_id"] = r.at("entrypoint_id");
  j["status"] = r.at("status");
  try { j["dead_letter_queue"] = Json::parse(r.at("dlq").as_string()); }
  catch (...) {}
  j["created_at"] = r.at("created_at");
  j["updated_at"] = r.at("updated_at");
  return j;
}

// gts-style event type match: exact id, or a trigger pattern ending in
// '*' matches by prefix (x-gts-ref wildcard convention)
bool event_type_matches(const std::string& pattern,
                        const std::string& event_type) {
  if (!pattern.empty() && pattern.back() == '*')
    return event_type.rfind(pattern.substr(0, pattern.size() - 1), 0)
           == 0;
  return pattern == event_type;
}

}  // namespace

void ServerlessRuntimeModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  ctx.hub->register_client<AdmissionClient>(
      "serverless-runtime", std::make_shared<StaticAdmission>(ctx.config));
  n_executors_ = (int)ctx.config.path("executors").as_int(4);

  std::string file = ctx.full_config
                         .path("modules.serverless-runtime.database.file")
                         .as_string("");
  if (file.empty()) {
    std::string home = ctx.home_dir;
    if (!home.empty() && home[0] == '~') {
      const char* h = getenv("HOME");
      home = std::string(h ? h : "/tmp") + home.substr(1);
    }
    mkdir(home.c_str(), 0755);
    file = home + "/serverless-runtime.db";
  }
  db_ = std::make_unique<Db>(file);
  db_->migrate("serverless-runtime", {
      {"0001_entrypoints",
       "CREATE TABLE entrypoints ("
       "  tenant_id TEXT NOT NULL,"
       "  id TEXT NOT NULL UNIQUE,"
       "  name TEXT NOT NULL,"
       "  entrypoint_type TEXT NOT NULL,"
       "  status TEXT NOT NULL DEFAULT 'draft',"
       "  version INTEGER NOT NULL DEFAULT 1,"
       "  spec TEXT NOT NULL,"
       "  created_at TEXT NOT NULL,"
       "  updated_at TEXT NOT NULL)"},
      {"0002_invocations",
       "CREATE TABLE invocations ("
       "  tenant_id TEXT NOT NULL,"
       "  id TEXT NOT NULL UNIQUE,"
       "  entrypoint_id TEXT NOT NULL,"
       "  status TEXT NOT NULL DEFAULT 'queued',"
       "  mode TEXT NOT NULL DEFAULT 'async',"
       "  attempts INTEGER NOT NULL DEFAULT 0,"
       "  step_index INTEGER NOT NULL DEFAULT 0,"
       "  input TEXT NOT NULL DEFAULT '{}',"
       "  result TEXT NOT NULL DEFAULT '',"
       "  error TEXT NOT NULL DEFAULT '',"
       "  created_at TEXT NOT NULL,"
       "  updated_at TEXT NOT NULL)"},
      {"0003_timeline",
       "CREATE TABLE timeline ("
       "  tenant_id TEXT NOT NULL,"
       "  invocation_id TEXT NOT NULL,"
       "  at TEXT NOT NULL,"
       "  seq INTEGER NOT NULL,"
       "  event_type TEXT NOT NULL,"
       "  status TEXT NOT NULL,"
       "  step_name TEXT NOT NULL DEFAULT '',"
       "  duration_ms INTEGER NOT NULL DEFAULT -1,"
       "  detail TEXT NOT NULL DEFAULT '')"},
      {"0004_schedules",
       "CREATE TABLE schedules ("
       "  tenant_id TEXT NOT NULL,"
       "  id TEXT NOT NULL UNIQUE,"
       "  entrypoint_id TEXT NOT NULL,"
       "  name TEXT NOT NULL,"
       "  timezone TEXT NOT NULL DEFAULT 'UTC',"
       "  expr_kind TEXT NOT NULL,"
       "  expr_value TEXT NOT NULL,"
       "  input_overrides TEXT NOT NULL DEFAULT '{}',"
       "  missed_policy TEXT NOT NULL DEFAULT 'skip',"
       "  status TEXT NOT NULL DEFAULT 'active',"
       "  next_run_at TEXT NOT NULL DEFAULT '',"
       "  last_run_at TEXT NOT NULL DEFAULT '',"
       "  created_at TEXT NOT NULL,"
       "  updated_at TEXT NOT NULL)"},
      {"0005_invocation_schedule",
       "ALTER TABLE invocations ADD COLUMN schedule_id TEXT NOT NULL "
       "DEFAULT ''"},
      {"0007_tenant_policies",
       "CREATE TABLE tenant_policies ("
       "  tenant_id TEXT NOT NULL UNIQUE,"
       "  enabled INTEGER NOT NULL DEFAULT 1,"
       "  quotas TEXT NOT NULL DEFAULT '{}',"
       "  updated_at TEXT NOT NULL)"},
      {"0006_triggers",
       "CREATE TABLE triggers ("
       "  tenant_id TEXT NOT NULL,"
       "  id TEXT NOT NULL UNIQUE,"
       "  event_type_id TEXT NOT NULL,"
       "  event_filter_query TEXT NOT NULL DEFAULT '',"
       "  entrypoint_id TEXT NOT NULL,"
       "  status TEXT NOT NULL DEFAULT 'active',"
       "  dlq TEXT NOT NULL DEFAULT '{}',"
       "  created_at TEXT NOT NULL,"
       "  updated_at TEXT NOT NULL)"},
  });
}

AccessScope ServerlessRuntimeModule::scope_for(const SecurityContext& sec,
                                               const std::string& action,
                                               const std::string& resource) {
  auto pdp = hub_->get<AuthzResolverClient>("authz-resolver");
  if (!pdp) return AccessScope::deny_all();
  EvaluationRequest er;
  er.subject = sec;
  er.action = action;
  er.resource = resource;
  er.tenant_id = sec.tenant_id;
  EvaluationResponse r = pdp->evaluate(er);
  if (!r.allow)
    throw Problem{403, "Forbidden", "about:blank",
                  r.deny_reason.empty() ? "access denied" : r.deny_reason,
                  "pdp_deny"};
  return r.tenant_scope.empty() ? AccessScope::for_tenant(sec.tenant_id)
                                : AccessScope::for_tenants(r.tenant_scope);
}

void ServerlessRuntimeModule::timeline(
    const std::string& tenant, const std::string& inv_id,
    const std::string& event_type, const std::string& status,
    const std::string& step, long long duration_ms,
    const std::string& detail) {
  SecureConn conn(*db_, AccessScope::for_tenant(tenant));
  conn.insert("timeline", {{"invocation_id", DbValue::S(inv_id)},
                           {"at", DbValue::S(now_iso())},
                           {"seq", DbValue::I((long long)(now_s() * 1e6))},
                           {"event_type", DbValue::S(event_type)},
                           {"status", DbValue::S(status)},
                           {"step_name", DbValue::S(step)},
                           {"duration_ms", DbValue::I(duration_ms)},
                           {"detail", DbValue::S(detail)}});
}

bool ServerlessRuntimeModule::transition(SecureConn& conn,
                                         const std::string& id,
                                         const std::string& from,
                                         const std::string& to) {
  if (!allowed_transition(from, to)) return false;
  // CAS on status so two executors / a control call cannot both win
  int n = conn.update("invocations",
                      {{"status", DbValue::S(to)},
                       {"updated_at", DbValue::S(now_iso())}},
                      "id=? AND status=?",
                      {DbValue::S(id), DbValue::S(from)});
  return n == 1;
}

void ServerlessRuntimeModule::enqueue(const std::string& tenant,
                                      const std::string& id) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    queue_.emplace_back(tenant, id);
  }
  cv_.notify_one();
}

void ServerlessRuntimeModule::enqueue_at(double when,
                                         const std::string& tenant,
                                         const std::string& id) {
  {
    std::lock_guard<std::mutex> lk(mu_);
    timers_.push_back({when, tenant, id, "enqueue"});
  }
  cv_.notify_all();
}

// ---- builtin op set (implementation kinds ADR:549-600; the 'code'
// adapter here is a deterministic test runtime — echo/sleep/fail — and
// 'adapter_ref' routes to the in-node LLM engine via the llm-gateway
// client, the MI355X materialisation of "entrypoints are model workers")
Json ServerlessRuntimeModule::run_op(const std::string& op,
                                     const Json& input,
                                     const std::string& tenant,
                                     long long attempts) {
  if (op == "echo") return input;
  if (op == "upper") {
    Json out = input;
    std::string t = input.at("text").as_string();
    for (auto& c : t) c = (char)toupper((unsigned char)c);
    out["text"] = t;
    return out;
  }
  if (op.rfind("sleep:", 0) == 0) {
    long ms = std::min(10000L, atol(op.c_str() + 6));
    usleep((useconds_t)ms * 1000);
    return input;
  }
  if (op.rfind("fail:", 0) == 0) {
    long n = atol(op.c_str() + 5);
    if (attempts <= n)
      throw std::runtime_error("transient failure (attempt " +
                               std::to_string(attempts) + " <= " +
                               std::to_string(n) + ")");
    return input;
  }
  if (op.rfind("error", 0) == 0)
    throw std::runtime_error(op.size() > 6 ? op.substr(6)
                                           : "permanent failure");
  if (op == "llm.chat") {
    auto chat = hub_->get<ChatInvoker>("llm-gateway");
    if (!chat) throw std::runtime_error("llm-gateway adapter unavailable");
    SecurityContext sec;
    sec.tenant_id = tenant;
    sec.subject_id = "serverless-runtime";
    return chat->chat(sec, input);
  }
  throw std::runtime_error("unknown op: " + op);
}

void ServerlessRuntimeModule::run_invocation(const std::string& tenant,
                                             const std::string& id) {
  SecureConn conn(*db_, AccessScope::for_tenant(tenant));
  auto page = conn.select("invocations", "id=?", {DbValue::S(id)}, "id",
                          false, 1, std::nullopt);
  if (page.items.empty()) return;
  DbRow inv = page.items[0];
  std::string status = inv.at("status").as_string();
  if (status == "queued") {
    if (!transition(conn, id, "queued", "running")) return;  // canceled?
    timeline(tenant, id, "started", "running");
  } else if (status != "running") {
    return;          // canceled/terminal while waiting in the queue
  }

  // per-tenant admission: a denied slot re-queues with backpressure
  auto adm = hub_->get<AdmissionClient>("serverless-runtime");
  if (adm) {
    std::string deny = adm->admit(tenant);
    if (!deny.empty()) {
      enqueue_at(now_s() + 0.2, tenant, id);
      return;
    }
  }
  // tenant policy quota: cap concurrently-RUNNING executions (this one
  // just transitioned to running; quota counts others + us)
  {
    long long q = tenant_policy(tenant)
                      .path("quotas.max_concurrent_executions")
                      .as_int(-1);
    if (q >= 0 &&
        count_rows(tenant, "invocations", "status='running'") > q) {
      // scheduling retreat (not a domain transition): back to queued,
      // retried shortly — the invocation is never lost
      conn.update("invocations",
                  {{"status", DbValue::S("queued")},
                   {"updated_at", DbValue::S(now_iso())}},
                  "id=? AND status='running'", {DbValue::S(id)});
      if (adm) adm->release(tenant);
      enqueue_at(now_s() + 0.25, tenant, id);
      return;
    }
  }
  struct Release {
    AdmissionClient* a;
    const std::string& t;
    ~Release() { if (a) a->release(t); }
  } release{adm.get(), tenant};

  long long attempts = inv.at("attempts").as_int(0) + 1;
  conn.update("invocations", {{"attempts", DbValue::I(attempts)}}, "id=?",
              {DbValue::S(id)});

  auto ep_page = conn.select("entrypoints", "id=?",
                             {DbValue::S(inv.at("entrypoint_id")
                                             .as_string())},
                             "id", false, 1, std::nullopt);
  Json input;
  try { input = Json::parse(inv.at("input").as_string()); }
  catch (...) { input = Json::object(); }
  Json spec;
  if (!ep_page.items.empty()) {
    try { spec = Json::parse(ep_page.items[0].at("spec").as_string()); }
    catch (...) {}
  }
  const Json& impl = spec.at("implementation");
  const std::string kind = impl.at("kind").as_string("code");
  const Json& rp = spec.at("retry_policy");
  const long long max_attempts = rp.at("max_attempts").as_int(1);
  const double backoff_ms = rp.at("backoff_ms").as_number(0);
  const double mult = rp.at("backoff_multiplier").as_number(2.0);

  auto fail_routing = [&](const std::string& err,
                          const std::vector<size_t>& completed_steps) {
    conn.update("invocations", {{"error", DbValue::S(err)}}, "id=?",
                {DbValue::S(id)});
    if (!transition(conn, id, "running", "failed")) return;
    timeline(tenant, id, "failed", "failed", "", -1, err);
    if (attempts < max_attempts) {
      // failed → queued (retry with exponential backoff)
      if (transition(conn, id, "failed", "queued")) {
        timeline(tenant, id, "step_retried", "queued", "", -1,
                 "retry " + std::to_string(attempts) + "/" +
                     std::to_string(max_attempts));
        double delay = backoff_ms * std::pow(mult, (double)attempts - 1);
        enqueue_at(now_s() + delay / 1000.0, tenant, id);
      }
      return;
    }
    // retries exhausted: compensation (saga) or DLQ  (ADR:680-977)
    const Json& steps = impl.path("workflow.steps");
    bool any_comp = false;
    for (size_t si : completed_steps)
      if (!steps.at(si).at("compensation").as_string().empty())
        any_comp = true;
    if (any_comp && transition(conn, id, "failed", "compensating")) {
      timeline(tenant, id, "compensation_started", "compensating");
      bool comp_ok = true;
      for (auto it = completed_steps.rbegin();
           it != completed_steps.rend(); ++it) {
        const Json& st = steps.at(*it);
        const std::string comp = st.at("compensation").as_string();
        if (comp.empty()) continue;
        try {
          run_op(comp, input, tenant, 1);
        } catch (const std::exception& e) {
          comp_ok = false;
          timeline(tenant, id, "compensation_failed", "compensating",
                   st.at("name").as_string(), -1, e.what());
          break;
        }
      }
      if (comp_ok && transition(conn, id, "compensating", "compensated"))
        timeline(tenant, id, "compensation_completed", "compensated");
      else if (!comp_ok &&
               transition(conn, id, "compensating", "dead_lettered"))
        timeline(tenant, id, "dead_lettered", "dead_lettered");
    } else if (transition(conn, id, "failed", "dead_lettered")) {
      timeline(tenant, id, "dead_lettered", "dead_lettered", "", -1,
               "no compensation configured");
    }
  };

  try {
    if (kind == "workflow_spec") {
      const Json& steps = impl.path("workflow.steps");
      size_t si = (size_t)inv.at("step_index").as_int(0);
      std::vector<size_t> completed;
      for (size_t k = 0; k < si; ++k) completed.push_back(k);
      for (; si < steps.size(); ++si) {
        // control actions take effect at step boundaries
        std::string ctl;
        {
          std::lock_guard<std::mutex> lk(mu_);
          auto it = control_.find(id);
          if (it != control_.end()) {
            ctl = it->second;
            control_.erase(it);
          }
        }
        if (ctl == "cancel") {
          if (transition(conn, id, "running", "canceled"))
            timeline(tenant, id, "canceled", "canceled");
          done_cv_.notify_all();
          return;
        }
        if (ctl == "suspend") {
          conn.update("invocations",
                      {{"step_index", DbValue::I((long long)si)}}, "id=?",
                      {DbValue::S(id)});
          if (transition(conn, id, "running", "suspended")) {
            timeline(tenant, id, "suspended", "suspended");
            const double sto =
                spec.path("limits.suspension_timeout_ms").as_number(0);
            if (sto > 0) {
              std::lock_guard<std::mutex> lk(mu_);
              timers_.push_back({now_s() + sto / 1000.0, tenant, id,
                                 "suspension_timeout"});
            }
          }
          done_cv_.notify_all();
          return;
        }
        const Json& st = steps.at(si);
        const std::string sname = st.at("name").as_string();
        // conditional step: `when` predicates on the current pipeline
        // state ({field, op: eq|ne|exists|gt|lt, value})
        if (st.at("when").is_object() &&
            !when_matches(st.at("when"), input)) {
          timeline(tenant, id, "step_skipped", "running", sname);
          completed.push_back(si);
          conn.update("invocations",
                      {{"step_index", DbValue::I((long long)si + 1)}},
                      "id=?", {DbValue::S(id)});
          continue;
        }
        timeline(tenant, id, "step_started", "running", sname);
        double t0 = now_s();
        try {
          // per-step args overlay the op input; `output_to` nests the
          // result under a key instead of replacing the pipeline state
          Json op_in = input;
          if (st.at("args").is_object())
            for (auto& [k, v] : st.at("args").obj()) op_in[k] = v;
          Json op_out = run_op(st.at("op").as_string(), op_in, tenant,
                               attempts);
          const std::string out_to = st.at("output_to").as_string();
          if (out_to.empty()) {
            input = op_out;
          } else {
            if (!input.is_object()) input = Json::object();
            input[out_to] = op_out;
          }
        } catch (const std::exception& e) {
          timeline(tenant, id, "step_failed", "running", sname,
                   (long long)((now_s() - t0) * 1000), e.what());
          conn.update("invocations",
                      {{"step_index", DbValue::I(0)}}, "id=?",
                      {DbValue::S(id)});
          fail_routing(e.what(), completed);
          done_cv_.notify_all();
          return;
        }
        timeline(tenant, id, "step_completed", "running", sname,
                 (long long)((now_s() - t0) * 1000));
        completed.push_back(si);
        conn.update("invocations",
                    {{"step_index", DbValue::I((long long)si + 1)}},
                    "id=?", {DbValue::S(id)});
      }
      conn.update("invocations",
                  {{"result", DbValue::S(input.dump())}}, "id=?",
                  {DbValue::S(id)});
    } else if (kind == "adapter_ref") {
      Json out = run_op("llm.chat", input, tenant, attempts);
      conn.update("invocations", {{"result", DbValue::S(out.dump())}},
                  "id=?", {DbValue::S(id)});
    } else {
      const std::string src = impl.path("code.source").as_string("echo");
      Json out = run_op(src, input, tenant, attempts);
      conn.update("invocations", {{"result", DbValue::S(out.dump())}},
                  "id=?", {DbValue::S(id)});
    }
    if (transition(conn, id, "running", "succeeded"))
      timeline(tenant, id, "succeeded", "succeeded");
  } catch (const std::exception& e) {
    fail_routing(e.what(), {});
  }
  done_cv_.notify_all();
}

void ServerlessRuntimeModule::executor_loop() {
  while (!stopping_) {
    std::pair<std::string, std::string> job;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(200),
                   [&] { return !queue_.empty() || stopping_; });
      if (stopping_) return;
      if (queue_.empty()) continue;
      job = queue_.front();
      queue_.pop_front();
    }
    try {
      run_invocation(job.first, job.second);
    } catch (const std::exception& e) {
      LOG_ERROR("serverless", "invocation %s crashed: %s",
                job.second.c_str(), e.what());
    }
  }
}

std::string ServerlessRuntimeModule::create_invocation(
    const std::string& tenant, const std::string& ep_id, const Json& input,
    const std::string& mode, const std::string& schedule_id,
    const std::string& preset_status) {
  SecureConn conn(*db_, AccessScope::for_tenant(tenant));
  const std::string id =
      "inv-" + std::to_string(++ctr_) + "-" +
      std::to_string((long long)(now_s() * 1000) % 100000);
  conn.insert("invocations",
              {{"id", DbValue::S(id)},
               {"entrypoint_id", DbValue::S(ep_id)},
               {"status", DbValue::S(preset_status.empty() ? "queued"
                                                           : preset_status)},
               {"mode", DbValue::S(mode)},
               {"input", DbValue::S(input.dump())},
               {"schedule_id", DbValue::S(schedule_id)},
               {"created_at", DbValue::S(now_iso())},
               {"updated_at", DbValue::S(now_iso())}});
  if (preset_status.empty()) enqueue(tenant, id);
  return id;
}

void ServerlessRuntimeModule::schedule_tick() {
  const time_t now = (time_t)now_s();
  std::vector<DbRow> due;
  {
    std::lock_guard<std::mutex> dblk(db_->mu());
    due = db_->query(
        "SELECT * FROM schedules WHERE status='active' AND "
        "next_run_at != '' AND next_run_at <= ?",
        {DbValue::S(iso_of(now))});
  }
  for (auto& s : due) {
    const std::string tenant = s.at("tenant_id").as_string();
    const std::string sid = s.at("id").as_string();
    const std::string kind = s.at("expr_kind").as_string();
    const std::string value = s.at("expr_value").as_string();
    const std::string policy = s.at("missed_policy").as_string();
    time_t next = parse_iso(s.at("next_run_at").as_string());
    if (next == 0) continue;
    // occurrences to execute per the missed-schedule policy (BR-022):
    //   backfill — one per missed slot; catch_up — exactly one;
    //   skip — one only if the slot is fresh (within a grace window)
    int fires = 0;
    if (policy == "backfill") {
      time_t t = next;
      while (t != 0 && t <= now && fires < 100) {
        fires++;
        t = sched_next(kind, value, t);
      }
      next = t;
    } else {
      fires = (policy == "catch_up" || now - next <= 5) ? 1 : 0;
      next = sched_next(kind, value, now);
    }
    SecureConn conn(*db_, AccessScope::for_tenant(tenant));
    // CAS on next_run_at so two ticks / hosts cannot double-fire
    int won = conn.update(
        "schedules",
        {{"next_run_at", DbValue::S(next ? iso_of(next) : "")},
         {"last_run_at", DbValue::S(iso_of(now))},
         {"updated_at", DbValue::S(now_iso())}},
        "id=? AND next_run_at=?",
        {DbValue::S(sid), DbValue::S(s.at("next_run_at").as_string())});
    if (won != 1) continue;
    if (fires == 0) continue;
    if (!tenant_policy(tenant).at("enabled").as_bool(true))
      continue;              // runtime disabled: cadence advances, no fire
    // resolve the entrypoint; merge its input defaults with overrides
    auto ep = conn.select("entrypoints", "id=?",
                          {DbValue::S(s.at("entrypoint_id").as_string())},
                          "id", false, 1, std::nullopt);
    if (ep.items.empty() ||
        ep.items[0].at("status").as_string() != "active")
      continue;               // paused implicitly while not active
    Json spec;
    try { spec = Json::parse(ep.items[0].at("spec").as_string()); }
    catch (...) {}
    Json input = spec.at("input_defaults").is_object()
                     ? spec.at("input_defaults")
                     : Json::object();
    Json ov;
    try { ov = Json::parse(s.at("input_overrides").as_string()); }
    catch (...) {}
    if (ov.is_object())
      for (auto& [k, v] : ov.obj()) input[k] = v;
    const Json& params = spec.path("io_schema.params");
    std::string verr =
        params.is_object() ? schema_err(params, input, "input") : "";
    for (int i = 0; i < fires; ++i) {
      if (!verr.empty()) {
        // recorded (visible in schedule history), never executed
        std::string id = create_invocation(
            tenant, s.at("entrypoint_id").as_string(), input, "async",
            sid, "failed");
        SecureConn c2(*db_, AccessScope::for_tenant(tenant));
        c2.update("invocations",
                  {{"error", DbValue::S("input validation: " + verr)}},
                  "id=?", {DbValue::S(id)});
      } else {
        create_invocation(tenant, s.at("entrypoint_id").as_string(),
                          input, "async", sid);
      }
    }
  }
}

Json ServerlessRuntimeModule::tenant_policy(const std::string& tenant) {
  Json pol = Json::object();
  pol["tenant_id"] = tenant;
  pol["enabled"] = true;
  pol["quotas"] = Json::object();
  SecureConn conn(*db_, AccessScope::for_tenant(tenant));
  auto page = conn.select("tenant_policies", "", {}, "tenant_id", false,
                          1, std::nullopt);
  if (!page.items.empty()) {
    pol["enabled"] = page.items[0].at("enabled").as_int(1) != 0;
    try {
      pol["quotas"] = Json::parse(page.items[0].at("quotas").as_string());
    } catch (...) {}
    pol["updated_at"] = page.items[0].at("updated_at");
  }
  return pol;
}

long long ServerlessRuntimeModule::count_rows(const std::string& tenant,
                                              const char* table,
                                              const char* extra_where) {
  std::lock_guard<std::mutex> dblk(db_->mu());
  std::string q = std::string("SELECT COUNT(*) AS n FROM ") + table +
                  " WHERE tenant_id=?";
  if (extra_where) q += std::string(" AND ") + extra_where;
  auto rows = db_->query(q, {DbValue::S(tenant)});
  return rows.empty() ? 0 : rows[0].at("n").as_int(0);
}

void ServerlessRuntimeModule::timer_loop() {
  while (!stopping_) {
    std::vector<Timer> due;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(50));
      if (stopping_) return;
      double now = now_s();
      for (auto it = timers_.begin(); it != timers_.end();) {
        if (it->at <= now) {
          due.push_back(*it);
          it = timers_.erase(it);
        } else {
          ++it;
        }
      }
    }
    for (auto& t : due) {
      if (t.kind == "enqueue") {
        enqueue(t.tenant, t.id);
      } else if (t.kind == "suspension_timeout") {
        SecureConn conn(*db_, AccessScope::for_tenant(t.tenant));
        if (transition(conn, t.id, "suspended", "failed")) {
          conn.update("invocations",
                      {{"error", DbValue::S("suspension timeout")}},
                      "id=?", {DbValue::S(t.id)});
          timeline(t.tenant, t.id, "failed", "failed", "", -1,
                   "suspension timeout");
          if (transition(conn, t.id, "failed", "dead_lettered"))
            timeline(t.tenant, t.id, "dead_lettered", "dead_lettered");
          done_cv_.notify_all();
        }
      }
    }
    // scan schedules every ~500 ms (PRD "scheduled execution start skew
    // p95 <= 5 s" NFR — a sub-second scan keeps skew well inside that)
    const double now = now_s();
    if (now - last_sched_scan_ >= 0.5) {
      last_sched_scan_ = now;
      try {
        schedule_tick();
      } catch (const std::exception& e) {
        LOG_ERROR("serverless", "schedule tick failed: %s", e.what());
      }
    }
  }
}

void ServerlessRuntimeModule::start(ModuleCtx& ctx) {
  stopping_ = false;
  // durable-execution recovery (PRD RTO<=30s): queued work re-enters the
  // queue; work that was mid-run when the host died is re-queued (its
  // effects are at-least-once, per the retry contract)
  {
    std::lock_guard<std::mutex> dblk(db_->mu());
    auto rows = db_->query(
        "SELECT tenant_id, id, status FROM invocations WHERE status IN "
        "('queued','running','compensating')", {});
    for (auto& r : rows) {
      const std::string tenant = r.at("tenant_id").as_string();
      const std::string id = r.at("id").as_string();
      const std::string st = r.at("status").as_string();
      if (st == "running") {
        db_->query("UPDATE invocations SET status='queued' WHERE id=?",
                   {DbValue::S(id)});
      } else if (st == "compensating") {
        db_->query(
            "UPDATE invocations SET status='dead_lettered' WHERE id=?",
            {DbValue::S(id)});
        continue;
      }
      std::lock_guard<std::mutex> lk(mu_);
      queue_.emplace_back(tenant, id);
    }
    if (!rows.empty())
      LOG_INFO("serverless", "recovered %zu unfinished invocation(s)",
               rows.size());
  }
  for (int i = 0; i < n_executors_; ++i)
    executors_.emplace_back([this] { executor_loop(); });
  timer_thread_ = std::thread([this] { timer_loop(); });
}

void ServerlessRuntimeModule::stop(ModuleCtx& ctx) {
  stopping_ = true;
  cv_.notify_all();
  for (auto& t : executors_)
    if (t.joinable()) t.join();
  executors_.clear();
  if (timer_thread_.joinable()) timer_thread_.join();
}

void ServerlessRuntimeModule::register_rest(ModuleCtx& ctx,
                                            RestRegistry& rest) {
  const std::vector<std::string> ep_fields = {"id", "name", "status",
                                              "entrypoint_type"};
  const std::vector<std::string> inv_fields = {"id", "entrypoint_id",
                                               "status", "mode",
                                               "schedule_id",
                                               "created_at"};

  auto reg = [&](const char* method, const std::string& path,
                 const std::string& opid,
                 std::function<void(HttpRequest&, ResponseWriter&)> h,
                 const std::vector<std::string>* filter = nullptr) {
    OperationSpec op;
    op.method = method;
    op.path = path;
    op.operation_id = opid;
    op.summary = opid;
    op.authenticated = true;
    op.tags = {"serverless-runtime"};
    if (filter) op.odata_filter_fields = *filter;
    if (std::string(method) == "POST" || std::string(method) == "PUT")
      op.allowed_content_types = {"application/json"};
    rest.register_op(op, std::move(h));
  };

  auto parse_body = [](HttpRequest& rq) -> Json {
    try { return Json::parse(rq.body); }
    catch (...) {
      throw Problem{400, "Bad Request", "about:blank",
                    "invalid JSON body", "validation_error"};
    }
  };

  // ---- entrypoints ----
  reg("POST", "/serverless-runtime/v1/entrypoints", "create_entrypoint",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json body = parse_body(rq);
        Json issues = validate_entrypoint(body);
        if (issues.size())
          throw Problem{400, "Bad Request", "about:blank",
                        issues.at(0).at("message").as_string(),
                        "validation_error"};
        {
          Json pol = tenant_policy(sec.tenant_id);
          if (!pol.at("enabled").as_bool(true))
            throw Problem{403, "Forbidden", "about:blank",
                          "serverless runtime disabled for tenant",
                          "runtime_disabled"};
          long long q = pol.path("quotas.max_definitions").as_int(-1);
          if (q >= 0 && count_rows(sec.tenant_id, "entrypoints") >= q)
            throw Problem{429, "Too Many Requests", "about:blank",
                          "max_definitions quota reached",
                          "quota_exceeded"};
        }
        SecureConn conn(*db_,
                        scope_for(sec, "create",
                                  "serverless-runtime:entrypoints"));
        const std::string id =
            "ep-" + std::to_string(++ctr_) + "-" +
            std::to_string((long long)(now_s() * 1000) % 100000);
        conn.insert("entrypoints",
                    {{"id", DbValue::S(id)},
                     {"name", DbValue::S(body.at("name").as_string())},
                     {"entrypoint_type",
                      DbValue::S(body.at("entrypoint_type").as_string())},
                     {"status", DbValue::S("draft")},
                     {"spec", DbValue::S(body.dump())},
                     {"created_at", DbValue::S(now_iso())},
                     {"updated_at", DbValue::S(now_iso())}});
        auto page = conn.select("entrypoints", "id=?", {DbValue::S(id)},
                                "id", false, 1, std::nullopt);
        w.respond(201, "application/json",
                  row_entrypoint(page.items[0]).dump());
      });

  reg("POST", "/serverless-runtime/v1/entrypoints/validate",
      "validate_entrypoint",
      [parse_body](HttpRequest& rq, ResponseWriter& w) {
        Json issues = validate_entrypoint(parse_body(rq));
        Json out = Json::object();
        out["valid"] = issues.size() == 0;
        out["issues"] = issues;
        w.respond(200, "application/json", out.dump());
      });

  reg("GET", "/serverless-runtime/v1/entrypoints", "list_entrypoints",
      [this, ep_fields](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:entrypoints"));
        std::vector<DbValue> binds;
        std::string where;
        auto fit = rq.query.find("$filter");
        SecureConn::OrderBy ob{{"id", false}};
        try {
          if (fit != rq.query.end())
            where = compile_odata_filter(fit->second, ep_fields, binds);
          auto oit = rq.query.find("$orderby");
          if (oit != rq.query.end())
            ob = parse_odata_orderby(oit->second, ep_fields);
        } catch (const std::exception& e) {
          throw Problem{400, "Bad Request", "about:blank", e.what(),
                        "validation_error"};
        }
        int top = 50;
        auto tit = rq.query.find("$top");
        if (tit != rq.query.end())
          top = std::max(1, std::min(1000, atoi(tit->second.c_str())));
        std::optional<std::string> cursor;
        auto cit = rq.query.find("cursor");
        if (cit != rq.query.end()) cursor = cit->second;
        SecureConn::Page page;
        try {
          page = conn.select("entrypoints", where, binds, ob, top, cursor);
        } catch (const std::exception& e) {
          throw Problem{400, "Bad Request", "about:blank", e.what(),
                        "validation_error"};
        }
        Json items = Json::array();
        for (auto& r : page.items) items.push_back(row_entrypoint(r));
        Json pi = Json::object();
        pi["limit"] = (long)top;
        if (page.next_cursor) pi["next_cursor"] = *page.next_cursor;
        Json out = Json::object();
        out["items"] = items;
        out["page_info"] = pi;
        w.respond(200, "application/json", out.dump());
      }, &ep_fields);

  reg("GET", "/serverless-runtime/v1/entrypoints/{id}", "get_entrypoint",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:entrypoints"));
        auto page = conn.select("entrypoints", "id=?",
                                {DbValue::S(rq.path_params.at("id"))},
                                "id", false, 1, std::nullopt);
        if (page.items.empty())
          throw Problem::not_found("entrypoint not found");
        w.respond(200, "application/json",
                  row_entrypoint(page.items[0]).dump());
      });

  reg("PUT", "/serverless-runtime/v1/entrypoints/{id}",
      "update_entrypoint",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json body = parse_body(rq);
        Json issues = validate_entrypoint(body);
        if (issues.size())
          throw Problem{400, "Bad Request", "about:blank",
                        issues.at(0).at("message").as_string(),
                        "validation_error"};
        SecureConn conn(*db_,
                        scope_for(sec, "update",
                                  "serverless-runtime:entrypoints"));
        const std::string id = rq.path_params.at("id");
        // ADR:2594 — PUT only while draft
        int n = conn.update(
            "entrypoints",
            {{"spec", DbValue::S(body.dump())},
             {"name", DbValue::S(body.at("name").as_string())},
             {"updated_at", DbValue::S(now_iso())}},
            "id=? AND status='draft'", {DbValue::S(id)});
        if (n == 0) {
          auto page = conn.select("entrypoints", "id=?", {DbValue::S(id)},
                                  "id", false, 1, std::nullopt);
          if (page.items.empty())
            throw Problem::not_found("entrypoint not found");
          throw Problem{409, "Conflict", "about:blank",
                        "only draft entrypoints can be updated",
                        "conflict"};
        }
        auto page = conn.select("entrypoints", "id=?", {DbValue::S(id)},
                                "id", false, 1, std::nullopt);
        w.respond(200, "application/json",
                  row_entrypoint(page.items[0]).dump());
      });

  reg("POST", "/serverless-runtime/v1/entrypoints/{id}/status",
      "entrypoint_status",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string action =
            parse_body(rq).at("action").as_string();
        // lifecycle: draft -activate-> active -deprecate-> deprecated;
        // active|deprecated -disable-> disabled -enable-> active
        static const std::map<std::string,
                              std::pair<std::string, std::string>> acts = {
            {"activate", {"draft", "active"}},
            {"deprecate", {"active", "deprecated"}},
            {"enable", {"disabled", "active"}},
        };
        SecureConn conn(*db_,
                        scope_for(sec, "update",
                                  "serverless-runtime:entrypoints"));
        const std::string id = rq.path_params.at("id");
        int n = 0;
        if (action == "disable") {
          n = conn.update("entrypoints",
                          {{"status", DbValue::S("disabled")},
                           {"updated_at", DbValue::S(now_iso())}},
                          "id=? AND status IN ('active','deprecated')",
                          {DbValue::S(id)});
        } else {
          auto it = acts.find(action);
          if (it == acts.end())
            throw Problem{400, "Bad Request", "about:blank",
                          "action must be activate|deprecate|disable|"
                          "enable", "validation_error"};
          n = conn.update("entrypoints",
                          {{"status", DbValue::S(it->second.second)},
                           {"updated_at", DbValue::S(now_iso())}},
                          "id=? AND status=?",
                          {DbValue::S(id), DbValue::S(it->second.first)});
        }
        if (n == 0)
          throw Problem{409, "Conflict", "about:blank",
                        "illegal status action for this entrypoint",
                        "conflict"};
        auto page = conn.select("entrypoints", "id=?", {DbValue::S(id)},
                                "id", false, 1, std::nullopt);
        w.respond(200, "application/json",
                  row_entrypoint(page.items[0]).dump());
      });

  reg("DELETE", "/serverless-runtime/v1/entrypoints/{id}",
      "delete_entrypoint", [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_,
                        scope_for(sec, "delete",
                                  "serverless-runtime:entrypoints"));
        const std::string id = rq.path_params.at("id");
        // draft → hard delete; anything else → archive (ADR:2605)
        int n = conn.remove("entrypoints", "id=? AND status='draft'",
                            {DbValue::S(id)});
        if (n == 0) {
          n = conn.update("entrypoints",
                          {{"status", DbValue::S("archived")},
                           {"updated_at", DbValue::S(now_iso())}},
                          "id=?", {DbValue::S(id)});
          if (n == 0) throw Problem::not_found("entrypoint not found");
        }
        w.respond(204, "application/json", "");
      });

  // ---- invocations ----
  reg("POST", "/serverless-runtime/v1/invocations", "start_invocation",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json body = parse_body(rq);
        const std::string ep_id = body.at("entrypoint_id").as_string();
        if (ep_id.empty())
          throw Problem{400, "Bad Request", "about:blank",
                        "'entrypoint_id' is required", "validation_error"};
        if (!tenant_policy(sec.tenant_id).at("enabled").as_bool(true))
          throw Problem{403, "Forbidden", "about:blank",
                        "serverless runtime disabled for tenant",
                        "runtime_disabled"};
        SecureConn conn(*db_, scope_for(sec, "invoke",
                                        "serverless-runtime:invocations"));
        auto ep = conn.select("entrypoints", "id=?", {DbValue::S(ep_id)},
                              "id", false, 1, std::nullopt);
        if (ep.items.empty())
          throw Problem::not_found("entrypoint not found");
        if (ep.items[0].at("status").as_string() != "active")
          throw Problem{409, "Conflict", "about:blank",
                        "entrypoint is not active", "conflict"};
        // io_schema.params gate (BR-032: inputs validated BEFORE start)
        Json vinput = body.at("input");
        if (vinput.is_null()) vinput = Json::object();
        Json espec;
        try {
          espec = Json::parse(ep.items[0].at("spec").as_string());
        } catch (...) {}
        const Json& params = espec.path("io_schema.params");
        if (params.is_object()) {
          std::string verr = schema_err(params, vinput, "input");
          if (!verr.empty())
            throw Problem{400, "Bad Request", "about:blank",
                          "input validation: " + verr, "invalid_input"};
        }
        if (body.at("dry_run").as_bool(false)) {
          Json out = Json::object();
          out["valid"] = true;
          out["dry_run"] = true;
          w.respond(200, "application/json", out.dump());
          return;
        }
        const std::string mode = body.at("mode").as_string("async");
        const std::string id =
            "inv-" + std::to_string(++ctr_) + "-" +
            std::to_string((long long)(now_s() * 1000) % 100000);
        Json input = body.at("input");
        if (input.is_null()) input = Json::object();
        conn.insert("invocations",
                    {{"id", DbValue::S(id)},
                     {"entrypoint_id", DbValue::S(ep_id)},
                     {"status", DbValue::S("queued")},
                     {"mode", DbValue::S(mode)},
                     {"input", DbValue::S(input.dump())},
                     {"created_at", DbValue::S(now_iso())},
                     {"updated_at", DbValue::S(now_iso())}});
        enqueue(sec.tenant_id, id);
        if (mode == "sync") {
          // wait for a terminal status (NFR: start p95 <= 100 ms)
          const double deadline =
              now_s() + body.at("timeout_ms").as_number(30000) / 1000.0;
          std::unique_lock<std::mutex> lk(mu_);
          while (now_s() < deadline) {
            done_cv_.wait_for(lk, std::chrono::milliseconds(20));
            lk.unlock();
            auto page = conn.select("invocations", "id=?",
                                    {DbValue::S(id)}, "id", false, 1,
                                    std::nullopt);
            const std::string st =
                page.items[0].at("status").as_string();
            if (is_terminal(st) || st == "suspended") {
              w.respond(200, "application/json",
                        row_invocation(page.items[0]).dump());
              return;
            }
            lk.lock();
          }
          lk.unlock();
          auto page = conn.select("invocations", "id=?", {DbValue::S(id)},
                                  "id", false, 1, std::nullopt);
          w.respond(202, "application/json",
                    row_invocation(page.items[0]).dump());
          return;
        }
        auto page = conn.select("invocations", "id=?", {DbValue::S(id)},
                                "id", false, 1, std::nullopt);
        w.respond(202, "application/json",
                  row_invocation(page.items[0]).dump());
      });

  reg("GET", "/serverless-runtime/v1/invocations", "list_invocations",
      [this, inv_fields](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:invocations"));
        std::vector<DbValue> binds;
        std::string where;
        SecureConn::OrderBy ob{{"created_at", true}};
        try {
          auto fit = rq.query.find("$filter");
          if (fit != rq.query.end())
            where = compile_odata_filter(fit->second, inv_fields, binds);
          auto oit = rq.query.find("$orderby");
          if (oit != rq.query.end())
            ob = parse_odata_orderby(oit->second, inv_fields);
        } catch (const std::exception& e) {
          throw Problem{400, "Bad Request", "about:blank", e.what(),
                        "validation_error"};
        }
        int top = 50;
        auto tit = rq.query.find("$top");
        if (tit != rq.query.end())
          top = std::max(1, std::min(1000, atoi(tit->second.c_str())));
        std::optional<std::string> cursor;
        auto cit = rq.query.find("cursor");
        if (cit != rq.query.end()) cursor = cit->second;
        SecureConn::Page page;
        try {
          page = conn.select("invocations", where, binds, ob, top,
                             cursor);
        } catch (const std::exception& e) {
          throw Problem{400, "Bad Request", "about:blank", e.what(),
                        "validation_error"};
        }
        Json items = Json::array();
        for (auto& r : page.items) items.push_back(row_invocation(r));
        Json pi = Json::object();
        pi["limit"] = (long)top;
        if (page.next_cursor) pi["next_cursor"] = *page.next_cursor;
        Json out = Json::object();
        out["items"] = items;
        out["page_info"] = pi;
        w.respond(200, "application/json", out.dump());
      }, &inv_fields);

  reg("GET", "/serverless-runtime/v1/invocations/{id}", "get_invocation",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:invocations"));
        auto page = conn.select("invocations", "id=?",
                                {DbValue::S(rq.path_params.at("id"))},
                                "id", false, 1, std::nullopt);
        if (page.items.empty())
          throw Problem::not_found("invocation not found");
        w.respond(200, "application/json",
                  row_invocation(page.items[0]).dump());
      });

  reg("POST", "/serverless-runtime/v1/invocations/{id}/control",
      "control_invocation",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string action =
            parse_body(rq).at("action").as_string();
        SecureConn conn(*db_, scope_for(sec, "invoke",
                                        "serverless-runtime:invocations"));
        const std::string id = rq.path_params.at("id");
        auto page = conn.select("invocations", "id=?", {DbValue::S(id)},
                                "id", false, 1, std::nullopt);
        if (page.items.empty())
          throw Problem::not_found("invocation not found");
        const DbRow& inv = page.items[0];
        const std::string st = inv.at("status").as_string();
        auto conflict = [&](const std::string& msg) -> void {
          throw Problem{409, "Conflict", "about:blank", msg, "conflict"};
        };
        if (action == "cancel") {
          if (st == "queued") {
            if (transition(conn, id, "queued", "canceled"))
              timeline(sec.tenant_id, id, "canceled", "canceled", "", -1,
                       "canceled before start");
          } else if (st == "running") {
            std::lock_guard<std::mutex> lk(mu_);
            control_[id] = "cancel";       // takes effect at a boundary
          } else if (st == "suspended") {
            if (transition(conn, id, "suspended", "canceled"))
              timeline(sec.tenant_id, id, "canceled", "canceled");
          } else {
            conflict("cannot cancel a " + st + " invocation");
          }
        } else if (action == "suspend") {
          if (st != "running") conflict("suspend requires running");
          std::lock_guard<std::mutex> lk(mu_);
          control_[id] = "suspend";
        } else if (action == "resume") {
          if (!transition(conn, id, "suspended", "running"))
            conflict("resume requires suspended");
          timeline(sec.tenant_id, id, "resumed", "running");
          enqueue(sec.tenant_id, id);
        } else if (action == "retry") {
          if (!transition(conn, id, "failed", "queued"))
            conflict("retry requires failed");
          timeline(sec.tenant_id, id, "step_retried", "queued", "", -1,
                   "manual retry");
          enqueue(sec.tenant_id, id);
        } else if (action == "replay") {
          // ADR:1063 — replay creates a NEW invocation from a terminal
          if (st != "succeeded" && st != "failed" &&
              st != "dead_lettered")
            conflict("replay requires a terminal invocation");
          const std::string nid =
              "inv-" + std::to_string(++ctr_) + "-" +
              std::to_string((long long)(now_s() * 1000) % 100000);
          conn.insert("invocations",
                      {{"id", DbValue::S(nid)},
                       {"entrypoint_id",
                        DbValue::S(inv.at("entrypoint_id").as_string())},
                       {"status", DbValue::S("queued")},
                       {"mode", DbValue::S("async")},
                       {"input",
                        DbValue::S(inv.at("input").as_string())},
                       {"created_at", DbValue::S(now_iso())},
                       {"updated_at", DbValue::S(now_iso())}});
          enqueue(sec.tenant_id, nid);
          auto np = conn.select("invocations", "id=?", {DbValue::S(nid)},
                                "id", false, 1, std::nullopt);
          w.respond(202, "application/json",
                    row_invocation(np.items[0]).dump());
          return;
        } else {
          throw Problem{400, "Bad Request", "about:blank",
                        "action must be cancel|suspend|resume|retry|"
                        "replay", "validation_error"};
        }
        auto np = conn.select("invocations", "id=?", {DbValue::S(id)},
                              "id", false, 1, std::nullopt);
        w.respond(200, "application/json",
                  row_invocation(np.items[0]).dump());
      });

  reg("GET", "/serverless-runtime/v1/invocations/{id}/timeline",
      "invocation_timeline", [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:invocations"));
        auto page = conn.select(
            "timeline", "invocation_id=?",
            {DbValue::S(rq.path_params.at("id"))}, "seq", false, 1000,
            std::nullopt);
        Json items = Json::array();
        for (auto& r : page.items) {
          Json e = Json::object();
          e["at"] = r.at("at");
          e["event_type"] = r.at("event_type");
          e["status"] = r.at("status");
          const std::string sn = r.at("step_name").as_string();
          if (!sn.empty()) e["step_name"] = sn;
          long long d = r.at("duration_ms").as_int(-1);
          if (d >= 0) e["duration_ms"] = (long)d;
          const std::string det = r.at("detail").as_string();
          if (!det.empty()) e["detail"] = det;
          items.push_back(e);
        }
        Json out = Json::object();
        out["items"] = items;
        w.respond(200, "application/json", out.dump());
      });

  // ---- schedules (ADR:2038-2164 + Schedule API table ADR:2884-2893;
  // ':pause'/':resume' actions mounted as subresource segments per this
  // host's route grammar).  Expressions are evaluated in UTC. ----
  auto load_schedule = [this](SecureConn& conn, const std::string& id) {
    auto page = conn.select("schedules", "id=?", {DbValue::S(id)}, "id",
                            false, 1, std::nullopt);
    if (page.items.empty())
      throw Problem::not_found("schedule not found");
    return page.items[0];
  };

  auto parse_schedule = [parse_body](HttpRequest& rq) {
    Json b = parse_body(rq);
    if (b.at("name").as_string().empty())
      throw Problem{400, "Bad Request", "about:blank",
                    "'name' is required", "validation_error"};
    if (b.at("entrypoint_id").as_string().empty())
      throw Problem{400, "Bad Request", "about:blank",
                    "'entrypoint_id' is required", "validation_error"};
    const std::string tz = b.at("timezone").as_string("UTC");
    if (tz != "UTC")
      throw Problem{400, "Bad Request", "about:blank",
                    "only timezone 'UTC' is supported",
                    "validation_error"};
    const std::string kind = b.path("expression.kind").as_string();
    const std::string value = b.path("expression.value").as_string();
    if (sched_next(kind, value, (time_t)now_s()) == 0)
      throw Problem{400, "Bad Request", "about:blank",
                    "expression must be a valid cron (5 fields) or "
                    "ISO-8601 interval >= 1s",
                    "validation_error"};
    const std::string mp = b.at("missed_policy").as_string("skip");
    if (mp != "skip" && mp != "catch_up" && mp != "backfill")
      throw Problem{400, "Bad Request", "about:blank",
                    "missed_policy must be skip|catch_up|backfill",
                    "validation_error"};
    return b;
  };

  reg("POST", "/serverless-runtime/v1/schedules", "create_schedule",
      [this, parse_schedule, load_schedule](HttpRequest& rq,
                                            ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json b = parse_schedule(rq);
        {
          Json pol = tenant_policy(sec.tenant_id);
          long long q = pol.path("quotas.max_schedules").as_int(-1);
          if (q >= 0 && count_rows(sec.tenant_id, "schedules") >= q)
            throw Problem{429, "Too Many Requests", "about:blank",
                          "max_schedules quota reached",
                          "quota_exceeded"};
        }
        SecureConn conn(*db_, scope_for(sec, "create",
                                        "serverless-runtime:schedules"));
        auto ep = conn.select(
            "entrypoints", "id=?",
            {DbValue::S(b.at("entrypoint_id").as_string())}, "id", false,
            1, std::nullopt);
        if (ep.items.empty())
          throw Problem::not_found("entrypoint not found");
        const std::string id =
            "sch-" + std::to_string(++ctr_) + "-" +
            std::to_string((long long)(now_s() * 1000) % 100000);
        const std::string kind = b.path("expression.kind").as_string();
        const std::string value = b.path("expression.value").as_string();
        time_t next = sched_next(kind, value, (time_t)now_s());
        Json ov = b.at("input_overrides");
        if (!ov.is_object()) ov = Json::object();
        conn.insert(
            "schedules",
            {{"id", DbValue::S(id)},
             {"entrypoint_id",
              DbValue::S(b.at("entrypoint_id").as_string())},
             {"name", DbValue::S(b.at("name").as_string())},
             {"timezone", DbValue::S("UTC")},
             {"expr_kind", DbValue::S(kind)},
             {"expr_value", DbValue::S(value)},
             {"input_overrides", DbValue::S(ov.dump())},
             {"missed_policy",
              DbValue::S(b.at("missed_policy").as_string("skip"))},
             {"status", DbValue::S("active")},
             {"next_run_at", DbValue::S(iso_of(next))},
             {"created_at", DbValue::S(now_iso())},
             {"updated_at", DbValue::S(now_iso())}});
        w.respond(201, "application/json",
                  row_schedule(load_schedule(conn, id)).dump());
      });

  reg("GET", "/serverless-runtime/v1/schedules", "list_schedules",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:schedules"));
        auto page = conn.select("schedules", "", {}, "id", false, 1000,
                                std::nullopt);
        Json items = Json::array();
        for (auto& r : page.items) items.push_back(row_schedule(r));
        Json out = Json::object();
        out["items"] = items;
        w.respond(200, "application/json", out.dump());
      });

  reg("GET", "/serverless-runtime/v1/schedules/{id}", "get_schedule",
      [this, load_schedule](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:schedules"));
        w.respond(200, "application/json",
                  row_schedule(load_schedule(conn,
                                             rq.path_params.at("id")))
                      .dump());
      });

  reg("PUT", "/serverless-runtime/v1/schedules/{id}", "update_schedule",
      [this, parse_schedule, load_schedule](HttpRequest& rq,
                                            ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json b = parse_schedule(rq);
        SecureConn conn(*db_, scope_for(sec, "update",
                                        "serverless-runtime:schedules"));
        const std::string id = rq.path_params.at("id");
        const std::string kind = b.path("expression.kind").as_string();
        const std::string value = b.path("expression.value").as_string();
        Json ov = b.at("input_overrides");
        if (!ov.is_object()) ov = Json::object();
        int n = conn.update(
            "schedules",
            {{"name", DbValue::S(b.at("name").as_string())},
             {"expr_kind", DbValue::S(kind)},
             {"expr_value", DbValue::S(value)},
             {"input_overrides", DbValue::S(ov.dump())},
             {"missed_policy",
              DbValue::S(b.at("missed_policy").as_string("skip"))},
             {"next_run_at",
              DbValue::S(iso_of(sched_next(kind, value,
                                           (time_t)now_s())))},
             {"updated_at", DbValue::S(now_iso())}},
            "id=?", {DbValue::S(id)});
        if (n == 0) throw Problem::not_found("schedule not found");
        w.respond(200, "application/json",
                  row_schedule(load_schedule(conn, id)).dump());
      });

  auto sched_status = [this, load_schedule](HttpRequest& rq,
                                            ResponseWriter& w,
                                            const std::string& from_csv,
                                            const std::string& to) {
    auto sec = sec_of(rq);
    SecureConn conn(*db_, scope_for(sec, "update",
                                    "serverless-runtime:schedules"));
    const std::string id = rq.path_params.at("id");
    std::vector<std::pair<std::string, DbValue>> vals = {
        {"status", DbValue::S(to)},
        {"updated_at", DbValue::S(now_iso())}};
    if (to == "active") {
      // recompute from now: a paused schedule never backfills its gap
      auto row = load_schedule(conn, id);
      time_t next = sched_next(row.at("expr_kind").as_string(),
                               row.at("expr_value").as_string(),
                               (time_t)now_s());
      vals.emplace_back("next_run_at", DbValue::S(iso_of(next)));
    }
    int n = conn.update("schedules", vals,
                        "id=? AND status IN (" + from_csv + ")",
                        {DbValue::S(id)});
    if (n == 0) {
      load_schedule(conn, id);   // 404 if absent
      throw Problem{409, "Conflict", "about:blank",
                    "illegal schedule state for this action", "conflict"};
    }
    w.respond(200, "application/json",
              row_schedule(load_schedule(conn, id)).dump());
  };

  reg("POST", "/serverless-runtime/v1/schedules/{id}/pause",
      "pause_schedule",
      [sched_status](HttpRequest& rq, ResponseWriter& w) {
        sched_status(rq, w, "'active'", "paused");
      });

  reg("POST", "/serverless-runtime/v1/schedules/{id}/resume",
      "resume_schedule",
      [sched_status](HttpRequest& rq, ResponseWriter& w) {
        sched_status(rq, w, "'paused'", "active");
      });

  reg("DELETE", "/serverless-runtime/v1/schedules/{id}", "delete_schedule",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "delete",
                                        "serverless-runtime:schedules"));
        int n = conn.remove("schedules", "id=?",
                            {DbValue::S(rq.path_params.at("id"))});
        if (n == 0) throw Problem::not_found("schedule not found");
        w.respond(204, "application/json", "");
      });

  // ---- tenant runtime policy + quota usage (ADR Tenant Runtime
  // Policy / Quota Usage APIs; quotas gate the create/start paths) ----
  reg("GET", "/serverless-runtime/v1/tenants/{tenant_id}/runtime-policy",
      "get_runtime_policy", [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string t = rq.path_params.at("tenant_id");
        scope_for(sec, "read", "serverless-runtime:policy");
        if (t != sec.tenant_id)
          throw Problem::forbidden("cross-tenant policy access");
        w.respond(200, "application/json", tenant_policy(t).dump());
      });

  reg("PUT", "/serverless-runtime/v1/tenants/{tenant_id}/runtime-policy",
      "put_runtime_policy",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string t = rq.path_params.at("tenant_id");
        scope_for(sec, "update", "serverless-runtime:policy");
        if (t != sec.tenant_id)
          throw Problem::forbidden("cross-tenant policy access");
        Json b = parse_body(rq);
        Json q = b.at("quotas");
        if (!q.is_null() && !q.is_object())
          throw Problem{400, "Bad Request", "about:blank",
                        "quotas must be an object", "validation_error"};
        if (q.is_object())
          for (const char* k : {"max_concurrent_executions",
                                "max_definitions", "max_schedules",
                                "max_triggers"})
            if (q.contains(k) && q.at(k).as_int(-1) < 0)
              throw Problem{400, "Bad Request", "about:blank",
                            std::string(k) + " must be >= 0",
                            "validation_error"};
        SecureConn conn(*db_, AccessScope::for_tenant(t));
        int n = conn.update(
            "tenant_policies",
            {{"enabled", DbValue::I(b.at("enabled").as_bool(true) ? 1
                                                                  : 0)},
             {"quotas",
              DbValue::S(q.is_object() ? q.dump() : "{}")},
             {"updated_at", DbValue::S(now_iso())}},
            "tenant_id=?", {DbValue::S(t)});
        if (n == 0)
          conn.insert("tenant_policies",
                      {{"enabled",
                        DbValue::I(b.at("enabled").as_bool(true) ? 1
                                                                 : 0)},
                       {"quotas",
                        DbValue::S(q.is_object() ? q.dump() : "{}")},
                       {"updated_at", DbValue::S(now_iso())}});
        w.respond(200, "application/json", tenant_policy(t).dump());
      });

  reg("GET", "/serverless-runtime/v1/tenants/{tenant_id}/usage",
      "get_quota_usage", [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string t = rq.path_params.at("tenant_id");
        scope_for(sec, "read", "serverless-runtime:policy");
        if (t != sec.tenant_id)
          throw Problem::forbidden("cross-tenant usage access");
        Json pol = tenant_policy(t);
        Json cur = Json::object();
        cur["definitions"] = (long)count_rows(t, "entrypoints");
        cur["schedules"] = (long)count_rows(t, "schedules");
        cur["triggers"] = (long)count_rows(t, "triggers");
        cur["concurrent_executions"] = (long)count_rows(
            t, "invocations", "status IN ('queued','running')");
        Json out = Json::object();
        out["tenant_id"] = t;
        out["timestamp"] = now_iso();
        out["current"] = cur;
        out["quotas"] = pol.at("quotas");
        w.respond(200, "application/json", out.dump());
      });

  reg("GET", "/serverless-runtime/v1/tenants/{tenant_id}/usage/history",
      "get_quota_usage_history",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        const std::string t = rq.path_params.at("tenant_id");
        scope_for(sec, "read", "serverless-runtime:policy");
        if (t != sec.tenant_id)
          throw Problem::forbidden("cross-tenant usage access");
        // executions per UTC hour over the last 24h, computed from the
        // durable invocation rows (no separate snapshot store)
        Json items = Json::array();
        {
          std::lock_guard<std::mutex> dblk(db_->mu());
          auto rows = db_->query(
              "SELECT substr(created_at, 1, 13) AS hour, COUNT(*) AS n "
              "FROM invocations WHERE tenant_id=? "
              "GROUP BY hour ORDER BY hour DESC LIMIT 24",
              {DbValue::S(t)});
          for (auto& r : rows) {
            Json it = Json::object();
            it["hour"] = r.at("hour").as_string() + ":00Z";
            it["executions"] = (long)r.at("n").as_int(0);
            items.push_back(it);
          }
        }
        Json out = Json::object();
        out["tenant_id"] = t;
        out["items"] = items;
        w.respond(200, "application/json", out.dump());
      });

  // ---- triggers (event-driven mechanism, ADR:2194-2290 + Trigger API
  // table; the EventBroker is deliberately minimal here: POST /events is
  // the in-node publish endpoint — the reference leaves broker, filter
  // syntax and DLQ management TBD.  event_filter_query takes the same
  // {field, op, value} predicate as workflow `when`; dead_letter_queue
  // config is stored, delivery failures follow the entrypoint's
  // retry_policy into the dead_lettered state) ----
  auto load_trigger = [this](SecureConn& conn, const std::string& id) {
    auto page = conn.select("triggers", "id=?", {DbValue::S(id)}, "id",
                            false, 1, std::nullopt);
    if (page.items.empty())
      throw Problem::not_found("trigger not found");
    return page.items[0];
  };

  auto parse_trigger = [parse_body](HttpRequest& rq) {
    Json b = parse_body(rq);
    if (b.at("event_type_id").as_string().empty())
      throw Problem{400, "Bad Request", "about:blank",
                    "'event_type_id' is required", "validation_error"};
    if (b.at("entrypoint_id").as_string().empty())
      throw Problem{400, "Bad Request", "about:blank",
                    "'entrypoint_id' is required", "validation_error"};
    const std::string st = b.at("status").as_string("active");
    if (st != "active" && st != "paused" && st != "disabled")
      throw Problem{400, "Bad Request", "about:blank",
                    "status must be active|paused|disabled",
                    "validation_error"};
    const Json& f = b.at("event_filter_query");
    if (!f.is_null() && !f.is_string())
      throw Problem{400, "Bad Request", "about:blank",
                    "event_filter_query must be a string",
                    "validation_error"};
    return b;
  };

  reg("POST", "/serverless-runtime/v1/triggers", "create_trigger",
      [this, parse_trigger, load_trigger](HttpRequest& rq,
                                          ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json b = parse_trigger(rq);
        {
          Json pol = tenant_policy(sec.tenant_id);
          long long q = pol.path("quotas.max_triggers").as_int(-1);
          if (q >= 0 && count_rows(sec.tenant_id, "triggers") >= q)
            throw Problem{429, "Too Many Requests", "about:blank",
                          "max_triggers quota reached",
                          "quota_exceeded"};
        }
        SecureConn conn(*db_, scope_for(sec, "create",
                                        "serverless-runtime:triggers"));
        auto ep = conn.select(
            "entrypoints", "id=?",
            {DbValue::S(b.at("entrypoint_id").as_string())}, "id",
            false, 1, std::nullopt);
        if (ep.items.empty())
          throw Problem::not_found("entrypoint not found");
        const std::string id =
            "trg-" + std::to_string(++ctr_) + "-" +
            std::to_string((long long)(now_s() * 1000) % 100000);
        Json dlq = b.at("dead_letter_queue");
        if (!dlq.is_object()) dlq = Json::object();
        conn.insert(
            "triggers",
            {{"id", DbValue::S(id)},
             {"event_type_id",
              DbValue::S(b.at("event_type_id").as_string())},
             {"event_filter_query",
              DbValue::S(b.at("event_filter_query").as_string(""))},
             {"entrypoint_id",
              DbValue::S(b.at("entrypoint_id").as_string())},
             {"status", DbValue::S(b.at("status").as_string("active"))},
             {"dlq", DbValue::S(dlq.dump())},
             {"created_at", DbValue::S(now_iso())},
             {"updated_at", DbValue::S(now_iso())}});
        w.respond(201, "application/json",
                  row_trigger(load_trigger(conn, id)).dump());
      });

  reg("GET", "/serverless-runtime/v1/triggers", "list_triggers",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:triggers"));
        auto page = conn.select("triggers", "", {}, "id", false, 1000,
                                std::nullopt);
        Json items = Json::array();
        for (auto& r : page.items) items.push_back(row_trigger(r));
        Json out = Json::object();
        out["items"] = items;
        w.respond(200, "application/json", out.dump());
      });

  reg("GET", "/serverless-runtime/v1/triggers/{id}", "get_trigger",
      [this, load_trigger](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:triggers"));
        w.respond(200, "application/json",
                  row_trigger(load_trigger(conn,
                                           rq.path_params.at("id")))
                      .dump());
      });

  reg("PUT", "/serverless-runtime/v1/triggers/{id}", "update_trigger",
      [this, parse_trigger, load_trigger](HttpRequest& rq,
                                          ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json b = parse_trigger(rq);
        SecureConn conn(*db_, scope_for(sec, "update",
                                        "serverless-runtime:triggers"));
        const std::string id = rq.path_params.at("id");
        Json dlq = b.at("dead_letter_queue");
        if (!dlq.is_object()) dlq = Json::object();
        int n = conn.update(
            "triggers",
            {{"event_type_id",
              DbValue::S(b.at("event_type_id").as_string())},
             {"event_filter_query",
              DbValue::S(b.at("event_filter_query").as_string(""))},
             {"entrypoint_id",
              DbValue::S(b.at("entrypoint_id").as_string())},
             {"status", DbValue::S(b.at("status").as_string("active"))},
             {"dlq", DbValue::S(dlq.dump())},
             {"updated_at", DbValue::S(now_iso())}},
            "id=?", {DbValue::S(id)});
        if (n == 0) throw Problem::not_found("trigger not found");
        w.respond(200, "application/json",
                  row_trigger(load_trigger(conn, id)).dump());
      });

  reg("DELETE", "/serverless-runtime/v1/triggers/{id}", "delete_trigger",
      [this](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "delete",
                                        "serverless-runtime:triggers"));
        int n = conn.remove("triggers", "id=?",
                            {DbValue::S(rq.path_params.at("id"))});
        if (n == 0) throw Problem::not_found("trigger not found");
        w.respond(204, "application/json", "");
      });

  reg("POST", "/serverless-runtime/v1/events", "publish_event",
      [this, parse_body](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        Json b = parse_body(rq);
        const std::string et = b.at("event_type_id").as_string();
        if (et.empty())
          throw Problem{400, "Bad Request", "about:blank",
                        "'event_type_id' is required",
                        "validation_error"};
        Json payload = b.at("payload");
        if (payload.is_null()) payload = Json::object();
        if (!tenant_policy(sec.tenant_id).at("enabled").as_bool(true))
          throw Problem{403, "Forbidden", "about:blank",
                        "serverless runtime disabled for tenant",
                        "runtime_disabled"};
        SecureConn conn(*db_, scope_for(sec, "invoke",
                                        "serverless-runtime:events"));
        auto page = conn.select("triggers", "status='active'", {}, "id",
                                false, 1000, std::nullopt);
        Json fired = Json::array();
        for (auto& t : page.items) {
          if (!event_type_matches(t.at("event_type_id").as_string(), et))
            continue;
          const std::string fq = t.at("event_filter_query").as_string();
          if (!fq.empty()) {
            Json cond;
            try { cond = Json::parse(fq); } catch (...) { continue; }
            if (cond.is_object() && !when_matches(cond, payload))
              continue;
          }
          // entrypoint must be active (same gate as schedules)
          auto ep = conn.select(
              "entrypoints", "id=? AND status='active'",
              {DbValue::S(t.at("entrypoint_id").as_string())}, "id",
              false, 1, std::nullopt);
          if (ep.items.empty()) continue;
          const std::string id = create_invocation(
              sec.tenant_id, t.at("entrypoint_id").as_string(), payload,
              "async", "");
          Json f = Json::object();
          f["trigger_id"] = t.at("id");
          f["invocation_id"] = id;
          fired.push_back(f);
        }
        Json out = Json::object();
        out["fired"] = fired;
        w.respond(202, "application/json", out.dump());
      });

  reg("GET", "/serverless-runtime/v1/schedules/{id}/history",
      "schedule_history",
      [this, load_schedule](HttpRequest& rq, ResponseWriter& w) {
        auto sec = sec_of(rq);
        SecureConn conn(*db_, scope_for(sec, "read",
                                        "serverless-runtime:schedules"));
        const std::string id = rq.path_params.at("id");
        load_schedule(conn, id);   // 404 if absent
        auto page = conn.select("invocations", "schedule_id=?",
                                {DbValue::S(id)},
                                SecureConn::OrderBy{{"created_at", true}},
                                200, std::nullopt);
        Json items = Json::array();
        for (auto& r : page.items) items.push_back(row_invocation(r));
        Json out = Json::object();
        out["items"] = items;
        w.respond(200, "application/json", out.dump());
      });
}

}  // namespace hs
