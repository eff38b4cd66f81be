// api-gateway module — owns the HTTP server and the middleware stack.
//
// Reference: modules/system/api-gateway (Rust/axum).  Stack order mirrors
// src/module.rs:170-178 / docs/MODULES.md:664-676:
//   RequestId -> Trace -> Timeout -> BodyLimit -> CORS -> MIME ->
//   RateLimit -> ErrorMapping -> Auth -> License -> Router
// Serves /health, /healthz, /docs, /openapi.json (src/web.rs:23).
#pragma once

#include <atomic>
#include <chrono>
#include <mutex>

#include "../modkit/modkit.h"

namespace hs {

// SDK trait published by authn-resolver (authn-resolver-sdk/src/api.rs:30)
struct AuthnResolverClient {
  virtual ~AuthnResolverClient() = default;
  // empty optional => invalid token
  virtual std::optional<SecurityContext> authenticate(
      const std::string& bearer) = 0;
};

class TokenBucket {
 public:
  TokenBucket(double rps, double burst) : rps_(rps), tokens_(burst),
                                          burst_(burst) {}
  bool try_acquire();

 private:
  std::mutex mu_;
  double rps_, tokens_, burst_;
  std::chrono::steady_clock::time_point last_ =
      std::chrono::steady_clock::now();
};

class ApiGatewayModule : public Module {
 public:
  std::string name() const override { return "api-gateway"; }
  std::vector<std::string> deps() const override {
    return {"authn-resolver"};
  }
  bool is_rest_host() const override { return true; }
  bool is_stateful() const override { return true; }

  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
  void start(ModuleCtx& ctx) override;
  void stop(ModuleCtx& ctx) override;

  // rest_host: the gateway owns the registry other modules populate
  RestRegistry& rest() { return rest_; }
  int port() const { return server_ ? server_->port() : 0; }
  void handle(HttpRequest& req, ResponseWriter& w);

 private:
  void dispatch(HttpRequest& req, ResponseWriter& w);

  RestRegistry rest_;
  std::unique_ptr<HttpServer> server_;
  ClientHub* hub_ = nullptr;
  // config
  std::string bind_addr_ = "127.0.0.1:8087";
  bool enable_docs_ = true;
  bool cors_enabled_ = false;
  bool auth_disabled_ = false;
  Json cors_cfg_;
  std::vector<std::string> licensed_features_;
  std::string openapi_title_ = "API Documentation";
  std::string openapi_version_ = "0.1.0";
  std::string openapi_desc_;
  RateLimitCfg default_rl_;
  size_t body_limit_ = 16 * 1024 * 1024;
  // per-route limiter state (index-aligned with rest_.routes())
  std::vector<std::unique_ptr<TokenBucket>> buckets_;
  std::vector<std::unique_ptr<std::atomic<int>>> in_flight_;
  std::atomic<uint64_t> req_counter_{0};
  std::chrono::steady_clock::time_point start_time_ =
      std::chrono::steady_clock::now();
};

}  // namespace hs
