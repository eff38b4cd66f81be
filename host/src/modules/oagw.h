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
};

}  // namespace hs
