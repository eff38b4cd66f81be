#include "llm_gateway.h"

#include <poll.h>
#include <signal.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <sys/wait.h>
#include <unistd.h>

#include <cstring>

#include "../util/log.h"
#include "system_modules.h"

namespace hs {

// ------------------------------------------------------ serverless-runtime

namespace {

class StaticAdmission : public AdmissionClient {
 public:
  explicit StaticAdmission(const Json& cfg) {
    max_concurrent_ = (int)cfg.path("limits.max_concurrent_per_tenant")
                          .as_int(64);
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

}  // namespace

void ServerlessRuntimeModule::init(ModuleCtx& ctx) {
  ctx.hub->register_client<AdmissionClient>(
      "serverless-runtime", std::make_shared<StaticAdmission>(ctx.config));
}

// ------------------------------------------------------------- EngineConn

EngineConn::EngineConn(const std::string& socket_path) {
  fd_ = socket(AF_UNIX, SOCK_STREAM, 0);
  if (fd_ < 0) return;
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, socket_path.c_str(), sizeof addr.sun_path - 1);
  if (connect(fd_, (sockaddr*)&addr, sizeof addr) != 0) {
    close(fd_);
    fd_ = -1;
  }
}

EngineConn::~EngineConn() {
  if (fd_ >= 0) close(fd_);
}

bool EngineConn::send_json(const Json& j) {
  std::string line = j.dump() + "\n";
  const char* p = line.data();
  size_t n = line.size();
  while (n) {
    ssize_t w = ::send(fd_, p, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    p += w;
    n -= size_t(w);
  }
  return true;
}

std::optional<Json> EngineConn::read_json(int timeout_ms) {
  while (true) {
    size_t nl = buf_.find('\n');
    if (nl != std::string::npos) {
      std::string line = buf_.substr(0, nl);
      buf_.erase(0, nl + 1);
      if (line.empty()) continue;
      try { return Json::parse(line); }
      catch (...) { return std::nullopt; }
    }
    struct pollfd pf{fd_, POLLIN, 0};
    int pr = poll(&pf, 1, timeout_ms);
    if (pr <= 0) return std::nullopt;
    char tmp[8192];
    ssize_t r = recv(fd_, tmp, sizeof tmp, 0);
    if (r <= 0) return std::nullopt;
    buf_.append(tmp, size_t(r));
  }
}

// ---------------------------------------------------------- LlmGateway

void LlmGatewayModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  model_ = ctx.config.at("model").as_string(model_);
  socket_path_ = ctx.config.at("worker_socket").as_string(socket_path_);
  auto_start_ = ctx.config.at("auto_start_worker").as_bool(true);
  python_ = ctx.config.at("python").as_string("python3");
  worker_cfg_ = ctx.config.at("worker");
}

bool LlmGatewayModule::worker_ready() {
  if (ready_) return true;
  EngineConn c(socket_path_);
  if (!c.ok()) return false;
  Json q = Json::object();
  q["type"] = "info";
  if (!c.send_json(q)) return false;
  auto r = c.read_json(3000);
  if (r && r->at("ready").as_bool()) {
    ready_ = true;
    return true;
  }
  return false;
}

void LlmGatewayModule::spawn_worker() {
  pid_t pid = fork();
  if (pid == 0) {
    setpgid(0, 0);
    std::vector<std::string> args = {
        python_, "-m", "hyperspot.serving.worker",
        "--uds", socket_path_, "--model", model_};
    if (worker_cfg_.is_object()) {
      if (worker_cfg_.contains("max_num_seqs")) {
        args.push_back("--max-num-seqs");
        args.push_back(std::to_string(
            worker_cfg_.at("max_num_seqs").as_int(256)));
      }
      if (worker_cfg_.at("eager").as_bool(false))
        args.push_back("--eager");
      if (worker_cfg_.contains("device")) {
        args.push_back("--device");
        args.push_back(worker_cfg_.at("device").as_string());
      }
      if (worker_cfg_.contains("num_gpu_blocks")) {
        args.push_back("--num-gpu-blocks");
        args.push_back(std::to_string(
            worker_cfg_.at("num_gpu_blocks").as_int(0)));
      }
      if (worker_cfg_.contains("tp")) {
        args.push_back("--tp");
        args.push_back(std::to_string(worker_cfg_.at("tp").as_int(1)));
      }
    }
    std::vector<char*> argv;
    for (auto& a : args) argv.push_back(const_cast<char*>(a.c_str()));
    argv.push_back(nullptr);
    execvp(argv[0], argv.data());
    _exit(127);
  }
  worker_pid_ = pid;
  LOG_INFO("llm-gateway", "spawned engine worker pid=%d model=%s sock=%s",
           pid, model_.c_str(), socket_path_.c_str());
}

void LlmGatewayModule::start(ModuleCtx& ctx) {
  if (auto_start_) {
    unlink(socket_path_.c_str());
    spawn_worker();
  }
}

void LlmGatewayModule::stop(ModuleCtx& ctx) {
  if (worker_pid_ > 0) {
    kill(worker_pid_, SIGTERM);
    int st = 0;
    for (int i = 0; i < 50; ++i) {
      if (waitpid(worker_pid_, &st, WNOHANG) == worker_pid_) {
        worker_pid_ = -1;
        break;
      }
      usleep(100000);
    }
    if (worker_pid_ > 0) {
      kill(worker_pid_, SIGKILL);
      waitpid(worker_pid_, &st, 0);
    }
  }
}

// Build a stream_chunk.v1-shaped SSE event
// (llm-gateway-sdk/schemas/core/stream_chunk.v1.schema.json)
static std::string sse_chunk(const std::string& id, const std::string& model,
                             const Json& delta,
                             const std::string& finish_reason = "",
                             const Json& usage = Json()) {
  Json c = Json::object();
  c["id"] = id;
  c["model"] = model;
  c["delta"] = delta;
  if (!finish_reason.empty()) c["finish_reason"] = finish_reason;
  if (!usage.is_null()) c["usage"] = usage;
  return "data: " + c.dump() + "\n\n";
}

void LlmGatewayModule::chat_handler(HttpRequest& req, ResponseWriter& w) {
  Json body;
  try { body = Json::parse(req.body); }
  catch (...) {
    throw Problem{400, "Bad Request", "about:blank", "invalid JSON body",
                  "validation_error"};
  }
  // request.v1 schema: required model + messages (Appendix B)
  const std::string model = body.at("model").as_string();
  if (model.empty() || !body.at("messages").is_array() ||
      body.at("messages").size() == 0)
    throw Problem{400, "Bad Request", "about:blank",
                  "'model' and non-empty 'messages' are required",
                  "validation_error"};
  const bool stream = body.at("stream").as_bool(false);

  SecurityContext sec =
      SecurityContext::from_json(req.extensions.at("security"));

  // model resolution via model-registry (DESIGN.md:317-346)
  auto reg = hub_->get<ModelRegistryClient>("model-registry");
  auto resolved = reg ? reg->get_tenant_model(sec.tenant_id, model)
                      : std::nullopt;
  if (!resolved)
    throw Problem{404, "Not Found", "about:blank",
                  "model '" + model + "' not found or not approved",
                  "model_not_found"};

  // per-tenant admission (serverless-runtime quota machinery)
  auto adm = hub_->get<AdmissionClient>("serverless-runtime");
  std::string deny = adm ? adm->admit(sec.tenant_id) : "";
  if (!deny.empty())
    throw Problem{429, "Too Many Requests", "about:blank",
                  "tenant quota exceeded", deny};
  struct Release {
    AdmissionClient* a;
    std::string t;
    ~Release() { if (a) a->release(t); }
  } rel{adm.get(), sec.tenant_id};

  if (!worker_ready())
    throw Problem{503, "Service Unavailable", "about:blank",
                  "inference engine is not ready", "provider_error"};

  const std::string rid = "chat-" + std::to_string(req_ctr_.fetch_add(1));
  EngineConn conn(socket_path_);
  if (!conn.ok())
    throw Problem{503, "Service Unavailable", "about:blank",
                  "engine connection failed", "provider_error"};
  Json wreq = Json::object();
  wreq["type"] = "chat";
  wreq["id"] = rid;
  wreq["model"] = resolved->at("provider_model_id").as_string();
  wreq["messages"] = body.at("messages");
  Json params = body.at("params");
  if (params.is_null()) params = Json::object();
  // OpenAI-style top-level sampling fields accepted additively (the v1
  // schema has none — Appendix B note)
  for (const char* f : {"temperature", "top_p", "top_k", "max_tokens",
                        "seed"})
    if (body.contains(f)) params[f] = body.at(f);
  wreq["params"] = params;
  if (!conn.send_json(wreq))
    throw Problem{502, "Bad Gateway", "about:blank", "engine write failed",
                  "provider_error"};

  if (!stream) {
    std::string text;
    Json usage;
    std::string finish = "stop";
    while (true) {
      auto msg = conn.read_json();
      if (!msg)
        throw Problem{504, "Gateway Timeout", "about:blank",
                      "engine timed out", "provider_timeout"};
      const std::string ev = msg->at("event").as_string();
      if (ev == "delta") text += msg->at("text").as_string();
      else if (ev == "done") {
        usage = msg->at("usage");
        finish = msg->at("finish_reason").as_string("stop");
        break;
      } else if (ev == "error") {
        throw Problem{502, "Bad Gateway", "about:blank",
                      msg->at("message").as_string(), "provider_error"};
      }
    }
    // response.v1: content[] parts + usage + model_used (required)
    Json part = Json::object();
    part["type"] = "text";
    part["text"] = text;
    Json content = Json::array();
    content.push_back(part);
    Json resp = Json::object();
    resp["content"] = content;
    resp["usage"] = usage;
    resp["model_used"] = resolved->at("canonical_id").as_string();
    resp["fallback_used"] = false;
    (void)finish;
    w.respond(200, "application/json", resp.dump(),
              {{"x-request-id", req.request_id}});
    return;
  }

  // SSE stream per DESIGN.md:289-311: role chunk, delta chunks, final
  // finish_reason+usage chunk, then data: [DONE]
  w.begin_stream(200, "text/event-stream",
                 {{"x-request-id", req.request_id}});
  const std::string canonical = resolved->at("canonical_id").as_string();
  Json role_delta = Json::object();
  role_delta["role"] = "assistant";
  w.write_chunk(sse_chunk(rid, canonical, role_delta));
  bool client_gone = false;
  while (true) {
    auto msg = conn.read_json();
    if (!msg) {
      w.write_chunk("data: {\"error\":\"provider_timeout\"}\n\n");
      break;
    }
    const std::string ev = msg->at("event").as_string();
    if (ev == "delta") {
      Json d = Json::object();
      d["content"] = msg->at("text").as_string();
      if (!w.write_chunk(sse_chunk(rid, canonical, d))) {
        client_gone = true;    // abort generation server-side
        Json ab = Json::object();
        ab["type"] = "abort";
        ab["id"] = rid;
        conn.send_json(ab);
        break;
      }
    } else if (ev == "done") {
      w.write_chunk(sse_chunk(rid, canonical, Json::object(),
                              msg->at("finish_reason").as_string("stop"),
                              msg->at("usage")));
      break;
    } else if (ev == "error") {
      w.write_chunk("data: {\"error\":\"provider_error\"}\n\n");
      break;
    }
  }
  if (!client_gone) w.write_chunk("data: [DONE]\n\n");
  w.end_stream();
}

void LlmGatewayModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  Json req_schema = Json::object();
  req_schema["type"] = "object";
  {
    Json props = Json::object();
    Json s = Json::object(); s["type"] = "string";
    props["model"] = s;
    Json msgs = Json::object();
    msgs["type"] = "array";
    props["messages"] = msgs;
    Json b = Json::object(); b["type"] = "boolean";
    props["stream"] = b;
    Json n = Json::object(); n["type"] = "number";
    props["temperature"] = n;
    props["top_p"] = n;
    Json i = Json::object(); i["type"] = "integer";
    props["top_k"] = i;
    props["max_tokens"] = i;
    req_schema["properties"] = props;
    Json required = Json::array();
    required.push_back("model");
    required.push_back("messages");
    req_schema["required"] = required;
  }
  auto handler = [this](HttpRequest& rq, ResponseWriter& w) {
    chat_handler(rq, w);
  };
  // canonical module route + unversioned OpenAI-style alias (Appendix B
  // endpoint-paths note)
  for (const char* p : {"/llm-gateway/v1/chat/completions",
                        "/v1/chat/completions"}) {
    OperationSpec op;
    op.method = "POST";
    op.path = p;
    op.operation_id = std::string("chat_completions") +
        (p[1] == 'v' ? "_alias" : "");
    op.summary = "Chat completion (sync or SSE stream)";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    op.request_schema = req_schema;
    op.responses[200] = "completion response or SSE stream";
    op.sse = true;
    rest.register_op(op, handler);
  }

  OperationSpec status;
  status.method = "GET";
  status.path = "/llm-gateway/v1/status";
  status.operation_id = "llm_status";
  status.summary = "Engine worker status";
  status.authenticated = true;
  status.tags = {"llm-gateway"};
  rest.register_op(status, [this](HttpRequest& rq, ResponseWriter& w) {
    Json out = Json::object();
    out["model"] = model_;
    out["worker_ready"] = worker_ready();
    if (ready_) {
      EngineConn c(socket_path_);
      Json q = Json::object();
      q["type"] = "info";
      if (c.ok() && c.send_json(q)) {
        auto r = c.read_json(3000);
        if (r) out["engine"] = *r;
      }
    }
    w.respond(200, "application/json", out.dump());
  });
}

}  // namespace hs
