#include "serverless_runtime.h"

#include <sys/stat.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cmath>

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
      for (auto& st : steps.arr())
        if (st.at("name").as_string().empty() ||
            st.at("op").as_string().empty())
          issue("every step needs name + op");
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
  return issues;
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
        timeline(tenant, id, "step_started", "running", sname);
        double t0 = now_s();
        try {
          input = run_op(st.at("op").as_string(), input, tenant, attempts);
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
        SecureConn conn(*db_, scope_for(sec, "invoke",
                                        "serverless-runtime:invocations"));
        auto ep = conn.select("entrypoints", "id=?", {DbValue::S(ep_id)},
                              "id", false, 1, std::nullopt);
        if (ep.items.empty())
          throw Problem::not_found("entrypoint not found");
        if (ep.items[0].at("status").as_string() != "active")
          throw Problem{409, "Conflict", "about:blank",
                        "entrypoint is not active", "conflict"};
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
}

}  // namespace hs
