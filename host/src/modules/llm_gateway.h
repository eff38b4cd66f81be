// llm-gateway — the serving-plane API module.
//
// The reference ships this module as specification only
// (modules/llm-gateway/docs/DESIGN.md; 32 GTS schemas).  Here it is real:
// REST + SSE per the spec contract, backed by the in-node MI355X inference
// engine (hyperspot.serving.worker, one process per GPU) over a Unix
// socket — the reference's OoP module pattern with the
// provider call replaced by the engine (SURVEY.md §3.3 note).
#pragma once

#include <atomic>
#include <mutex>
#include <thread>

#include "../modkit/modkit.h"

namespace hs {

// serverless-runtime admission client (per-tenant quotas / noisy-neighbor
// isolation — reference modules/serverless-runtime/docs/PRD.md:944-957)
struct AdmissionClient {
  virtual ~AdmissionClient() = default;
  // empty = admitted; else deny problem code (rate_limited/budget_exceeded)
  virtual std::string admit(const std::string& tenant) = 0;
  virtual void release(const std::string& tenant) = 0;
};

class ServerlessRuntimeModule : public Module {
 public:
  std::string name() const override { return "serverless-runtime"; }
  void init(ModuleCtx& ctx) override;
};

// Blocking JSON-lines client for one worker request over UDS.
class EngineConn {
 public:
  explicit EngineConn(const std::string& socket_path);
  ~EngineConn();
  bool ok() const { return fd_ >= 0; }
  bool send_json(const Json& j);
  // reads one newline-terminated JSON message; empty on EOF/error
  std::optional<Json> read_json(int timeout_ms = 120000);

 private:
  int fd_ = -1;
  std::string buf_;
};

class LlmGatewayModule : public Module {
 public:
  std::string name() const override { return "llm-gateway"; }
  std::vector<std::string> deps() const override {
    return {"model-registry", "serverless-runtime"};
  }
  bool is_stateful() const override { return true; }

  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
  void start(ModuleCtx& ctx) override;
  void stop(ModuleCtx& ctx) override;

  const std::string& socket_path() const { return socket_path_; }
  bool worker_ready();

 private:
  void chat_handler(HttpRequest& req, ResponseWriter& w);
  void spawn_worker();

  ClientHub* hub_ = nullptr;
  std::string model_ = "llama3-8b";
  std::string socket_path_ = "/tmp/hyperspot-llm.sock";
  bool auto_start_ = true;
  std::string python_ = "python3";
  Json worker_cfg_;
  pid_t worker_pid_ = -1;
  std::atomic<bool> ready_{false};
  std::atomic<uint64_t> req_ctr_{0};
};

}  // namespace hs
