// users-info — the reference's canonical full-module blueprint
// (examples/modkit/users-info): SDK client in the hub, domain entities on
// the secure ORM (tenant-scoped users + shared cities), migrations, OData
// cursor pagination, PDP-gated writes, and an SSE event stream at
// /users-info/v1/users/events (SseBroadcaster wiring, module.rs:47-57).
#pragma once

#include "../modkit/db.h"
#include "../modkit/modkit.h"
#include "../modkit/sse.h"

namespace hs {

// SDK trait registered in the ClientHub (users-info-sdk/src/api.rs shape)
struct UsersInfoClient {
  virtual ~UsersInfoClient() = default;
  virtual std::optional<Json> get_user(const std::string& tenant,
                                       const std::string& id) = 0;
  virtual long count_users(const std::string& tenant) = 0;
};

class UsersInfoModule : public Module {
 public:
  std::string name() const override { return "users-info"; }
  std::vector<std::string> deps() const override {
    return {"authz-resolver"};
  }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

  std::unique_ptr<Db> db_;

 private:
  AccessScope scope_for(const SecurityContext& sec,
                        const std::string& action);
  ClientHub* hub_ = nullptr;
  SseBroadcaster events_;
};

}  // namespace hs
