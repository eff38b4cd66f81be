// simple-user-settings — tenant/user-scoped settings CRUD on the secure
// DB layer; the reference's demonstration module for secure ORM + PDP
// (modules/simple-user-settings; users-info test matrix:
// tenant scoping / resource scoping / PDP deny / cursor pagination).
#pragma once

#include "../modkit/db.h"
#include "../modkit/modkit.h"

namespace hs {

class UserSettingsModule : public Module {
 public:
  std::string name() const override { return "simple-user-settings"; }
  std::vector<std::string> deps() const override {
    return {"authz-resolver"};
  }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

 private:
  AccessScope scope_for(const SecurityContext& sec,
                        const std::string& action);
  std::unique_ptr<Db> db_;
  ClientHub* hub_ = nullptr;
};

}  // namespace hs
