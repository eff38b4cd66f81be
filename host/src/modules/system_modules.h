// System modules (control plane): authn/authz/tenant resolvers with static
// plugins, types-registry (GTS store), nodes-registry, model-registry,
// credstore, file-storage.
//
// Reference crates: modules/system/{authn-resolver,authz-resolver,
// tenant-resolver,types-registry,nodes-registry}, spec-only
// modules/{model-registry,credstore,file-storage}.
#pragma once

#include <mutex>
#include <thread>

#include "../modkit/modkit.h"
#include "api_gateway.h"

namespace hs {

// ---- tenant-resolver (static-tr-plugin: YAML tenant tree,
//      config/static-tenants.yaml shape) ----
struct Tenant {
  std::string id, name, status, type, parent_id;
};

struct TenantResolverClient {
  virtual ~TenantResolverClient() = default;
  virtual std::optional<Tenant> get_tenant(const std::string& id) = 0;
  virtual std::vector<Tenant> get_tenants() = 0;
  virtual std::vector<Tenant> get_ancestors(const std::string& id) = 0;
  virtual std::vector<Tenant> get_descendants(const std::string& id) = 0;
  virtual bool is_ancestor(const std::string& a, const std::string& b) = 0;
};

class TenantResolverModule : public Module {
 public:
  std::string name() const override { return "tenant-resolver"; }
  void init(ModuleCtx& ctx) override;
};

// ---- authn-resolver (static-authn-plugin: token -> identity map) ----
class AuthnResolverModule : public Module {
 public:
  std::string name() const override { return "authn-resolver"; }
  void init(ModuleCtx& ctx) override;
};

// ---- authz-resolver (PDP; static plugin = allow within own tenant) ----
struct EvaluationRequest {
  SecurityContext subject;
  std::string action;       // read | write | admin
  std::string resource;     // e.g. "llm:chat"
  std::string tenant_id;
};
struct EvaluationResponse {
  bool allow = false;
  std::string deny_reason;
  std::vector<std::string> tenant_scope;   // row-level AccessScope
};
struct AuthzResolverClient {
  virtual ~AuthzResolverClient() = default;
  virtual EvaluationResponse evaluate(const EvaluationRequest& r) = 0;
};

class AuthzResolverModule : public Module {
 public:
  std::string name() const override { return "authz-resolver"; }
  std::vector<std::string> deps() const override {
    return {"tenant-resolver"};
  }
  void init(ModuleCtx& ctx) override;
};

// ---- types-registry (GTS entity store; plugin discovery backbone) ----
// in-process client (reference types-registry-sdk/src/api.rs:22)
struct TypesRegistryClient {
  virtual ~TypesRegistryClient() = default;
  virtual std::optional<Json> get(const std::string& gts_id) = 0;
  // '*'-suffix wildcard listing
  virtual std::vector<Json> list(const std::string& filter) = 0;
  virtual int register_entities(const Json& entities) = 0;
};

class TypesRegistryModule : public Module {
 public:
  std::string name() const override { return "types-registry"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
};

// ---- nodes-registry (node inventory incl. GPU info) ----
class NodesRegistryModule : public Module {
 public:
  std::string name() const override { return "nodes-registry"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

 private:
  Json node_info() const;
};

// ---- model-registry (canonical {provider}::{model} resolution,
//      tenant approvals — reference modules/model-registry/docs/PRD.md) ----
struct ModelRegistryClient {
  virtual ~ModelRegistryClient() = default;
  // canonical id "local::llama3-8b"; empty optional = not found/approved
  virtual std::optional<Json> get_tenant_model(const std::string& tenant,
                                               const std::string& canonical)
      = 0;
  virtual std::vector<Json> list_tenant_models(const std::string& tenant) = 0;
};

class ModelRegistryModule : public Module {
 public:
  std::string name() const override { return "model-registry"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
};

// ---- credstore (tenant-scoped secrets; always-404 on inaccessible) ----
struct CredStoreClient {
  virtual ~CredStoreClient() = default;
  virtual std::optional<std::string> get(const std::string& tenant,
                                         const std::string& ref) = 0;
  virtual void put(const std::string& tenant, const std::string& ref,
                   const std::string& value) = 0;
  virtual bool del(const std::string& tenant, const std::string& ref) = 0;
};

class CredStoreModule : public Module {
 public:
  std::string name() const override { return "credstore"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
};

// ---- file-storage (binary store; serves model checkpoints for registry
//      hot-swap — reference modules/file-storage/docs/PRD.md) ----
struct FileStorageClient {
  virtual ~FileStorageClient() = default;
  virtual std::string root_for(const std::string& tenant) = 0;
};

class FileStorageModule : public Module {
 public:
  std::string name() const override { return "file-storage"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

 private:
  std::string root_;
};

// ---- module-orchestrator (module/worker instance visibility;
//      reference modules/system/module-orchestrator DirectoryService) ----
// module-orchestrator: module listing + the OoP (out-of-process) module
// runtime — spawn children declared as modules.<name>.runtime.type: oop
// (reference A.6 envelope: execution.{executable_path,args,environment}),
// pass config by env (MODKIT_MODULE_CONFIG) and the directory endpoint
// (MODKIT_DIRECTORY_ENDPOINT), forward child stdout into the host log
// (backends/log_forwarder.rs), track register/heartbeat liveness
// (proto/directory/v1 semantics over REST).
// in-process service-discovery client (reference: module-orchestrator
// registers a DirectoryClient in the hub, src/module.rs:80-88, so other
// modules consume OoP services without going through REST)
class DirectoryClient {
 public:
  virtual ~DirectoryClient() = default;
  // live endpoint (e.g. "http://127.0.0.1:port") for a module name,
  // "" when no instance has heartbeated recently
  virtual std::string resolve(const std::string& name) = 0;
};

class ModuleOrchestratorModule : public Module {
 public:
  std::string name() const override { return "module-orchestrator"; }
  bool is_stateful() const override { return true; }
  void init(ModuleCtx& ctx) override;
  void start(ModuleCtx& ctx) override;
  void stop(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
  // freshest live instance endpoint ("" if none) — DirectoryClient impl
  std::string resolve_endpoint(const std::string& name);
  // ModuleManager parity (reference runtime/module_manager.rs): the
  // orchestrator tracks module INSTANCES with capabilities + mounted
  // endpoints, not just names
  void set_modules(Json infos) { module_infos_ = std::move(infos); }

 private:
  struct OopSpec {
    std::string name, exe;
    std::vector<std::string> args;
    std::map<std::string, std::string> env;
    Json config;
  };
  struct Child {
    std::string name;
    pid_t pid = -1;
    int out_fd = -1;
    std::thread fwd;
  };
  struct Instance {
    std::string id, name, endpoint;
    double last_heartbeat = 0;
    Json meta;
  };

  Json module_infos_;   // [{name, deps, stateful, endpoints[]}]
  std::vector<OopSpec> oop_specs_;
  std::vector<Child> children_;
  std::string directory_endpoint_;
  std::mutex inst_mu_;
  std::map<std::string, Instance> instances_;
  uint64_t inst_ctr_ = 0;
};

}  // namespace hs
