// OAGW — outbound API gateway (egress path for provider calls).
//
// Reference: modules/system/oagw (15.6k LoC Rust).  Control plane:
// tenant-scoped Upstream/Route CRUD (oagw-sdk/src/models.rs shapes).
// Data plane: /oagw/v1/proxy/{alias}/{*path} — resolve -> auth plugin
// (credential injection from credstore) -> rate limit -> forward (no
// redirects = SSRF guard; streaming pass-through for SSE).
#pragma once

#include <mutex>

#include "../modkit/modkit.h"
#include "api_gateway.h"

namespace hs {

class OagwModule : public Module {
 public:
  std::string name() const override { return "oagw"; }
  std::vector<std::string> deps() const override {
    return {"types-registry", "credstore"};
  }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

 private:
  void proxy(HttpRequest& req, ResponseWriter& w);
  Json* find_upstream(const std::string& tenant, const std::string& alias);

  ClientHub* hub_ = nullptr;
  std::mutex mu_;
  std::map<std::string, Json> upstreams_;   // id -> upstream
  std::map<std::string, Json> routes_;      // id -> route
  std::map<std::string, std::unique_ptr<TokenBucket>> limiters_;
  uint64_t next_id_ = 1;

  // per-endpoint circuit breaker (ADR-0004 infrastructure layer:
  // reactive fast-fail with automatic half-open recovery; the
  // llm-gateway's worker watchdog/ready-lease is the business layer)
  struct Breaker {
    int fails = 0;              // consecutive failures while closed
    bool open = false;
    double opened_at = 0;       // monotonic seconds
    bool probing = false;       // one half-open trial in flight
  };
  std::map<std::string, Breaker> breakers_;   // key: alias|host:port
  // returns false => fast-fail (circuit open, not yet cool);
  // *probe set when this call is the single half-open trial
  bool breaker_admit(const std::string& key, int threshold, double open_s,
                     bool* probe);
  void breaker_report(const std::string& key, bool ok);
};

}  // namespace hs
