// ModKit — the module runtime of the host plane.
//
// Native C++ re-creation of the reference's Rust modkit (libs/modkit):
//   - Module capability contracts      (reference src/contracts.rs:37-135)
//   - registry + topo-sorted phases    (src/registry.rs:310, runtime/host_runtime.rs:717)
//   - ClientHub type-erased DI         (src/client_hub.rs:123)
//   - OperationSpec/OpenAPI registry   (src/api/operation_builder.rs:196, openapi_registry.rs:83)
//   - RFC-9457 Problem                 (libs/modkit-errors/src/problem.rs:41)
//   - SecurityContext / AccessScope    (libs/modkit-security/src/context.rs:23)
#pragma once

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <functional>
#include <future>
#include <thread>
#include <typeinfo>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <vector>

#include "../http/http.h"
#include "../util/json.h"

namespace hs {

// ---------------------------------------------------------------- problems
struct Problem {
  int status = 500;
  std::string title = "Internal Server Error";
  std::string type = "about:blank";
  std::string detail;
  std::string code;      // stable machine-readable error code

  Json to_json(const std::string& instance = "") const {
    Json j = Json::object();
    j["type"] = type;
    j["title"] = title;
    j["status"] = status;
    if (!detail.empty()) j["detail"] = detail;
    if (!code.empty()) j["code"] = code;
    if (!instance.empty()) j["instance"] = instance;
    return j;
  }
  static Problem not_found(const std::string& detail = "") {
    return {404, "Not Found", "about:blank", detail, ""};
  }
  static Problem bad_request(const std::string& detail = "") {
    return {400, "Bad Request", "about:blank", detail, ""};
  }
  static Problem unauthorized(const std::string& detail = "") {
    return {401, "Unauthorized", "about:blank", detail, ""};
  }
  static Problem forbidden(const std::string& detail = "") {
    return {403, "Forbidden", "about:blank", detail, ""};
  }
};

void respond_problem(ResponseWriter& w, const Problem& p,
                     const std::string& instance = "");

// ---------------------------------------------------- security (PDP/PEP)
// DEFAULT_TENANT_ID must match the reference's modkit-security constant
// (config/static-tenants.yaml:36 comments pin this UUID).
inline const char* kDefaultTenantId = "00000000-df51-5b42-9538-d2b56b7ee953";
inline const char* kDefaultSubjectId = "00000000-7f5e-5e2f-a26c-6c08aa2eaa4a";

struct SecurityContext {
  std::string subject_id;
  std::string subject_type = "user";   // user | service | anonymous
  std::string tenant_id;
  std::vector<std::string> scopes;

  static SecurityContext anonymous() {
    return {"", "anonymous", "", {}};
  }
  static SecurityContext default_ctx() {   // auth-disabled mode
    return {kDefaultSubjectId, "user", kDefaultTenantId, {}};
  }
  Json to_json() const;
  static SecurityContext from_json(const Json& j);
};

// ------------------------------------------------------------- operations
struct RateLimitCfg {
  double rps = 50;
  double burst = 100;
  int in_flight = 64;
};

struct OperationSpec {
  std::string method;        // GET/POST/...
  std::string path;          // /module/v1/things/{id}
  std::string operation_id;
  std::string summary;
  std::vector<std::string> tags;
  bool authenticated = false;  // one of authenticated/is_public must be set
  bool is_public = false;
  std::optional<RateLimitCfg> rate_limit;
  std::vector<std::string> allowed_content_types;   // 415 enforcement
  Json request_schema;        // JSON schema (null = none)
  std::map<int, std::string> responses;             // status -> description
  Json response_schema;       // schema of the 200 response
  std::vector<std::string> license_features;
  // OData vendor extension: filterable fields advertised in the OpenAPI
  // doc as x-odata-filter.allowedFields (operation_builder.rs:227-239)
  std::vector<std::string> odata_filter_fields;
  bool sse = false;
};

struct Route {
  OperationSpec spec;
  HttpHandler handler;
  std::vector<std::string> segments;   // pre-split path
};

// Typed route collector passed to modules during the `rest` phase —
// the OperationBuilder equivalent: register() refuses specs that don't
// declare an auth stance (typestate in the reference; runtime check here).
class RestRegistry {
 public:
  void register_op(OperationSpec spec, HttpHandler handler);
  const std::vector<Route>& routes() const { return routes_; }
  // returns nullptr if no match; fills params
  const Route* match(const std::string& method, const std::string& path,
                     std::map<std::string, std::string>& params,
                     bool* path_exists = nullptr) const;
  Json build_openapi(const std::string& title, const std::string& version,
                     const std::string& description) const;
  void add_schema(const std::string& name, Json schema) {
    schemas_[name] = std::move(schema);
  }

 private:
  std::vector<Route> routes_;
  std::map<std::string, Json> schemas_;
};

// -------------------------------------------------------------- ClientHub
// Type-erased inter-module DI: modules publish SDK client objects under an
// interface name; consumers resolve them without direct dependencies.
// Typed DI hub (reference libs/modkit/src/client_hub.rs:123).  Besides
// plain interface-keyed clients it supports SCOPED registrations keyed
// by (interface, ClientScope) where ClientScope is a GTS instance id
// (client_hub.rs:155-195): get_scoped falls back to the unscoped
// registration when no scope-specific client exists.
class ClientHub {
 public:
  // keys are (interface TYPE, name): one module may publish several
  // client interfaces under its own name (e.g. llm-gateway registers
  // ChatInvoker AND ProviderHealthClient) — keying by name alone would
  // silently type-confuse the unchecked static cast below
  template <typename T>
  void register_client(const std::string& iface, std::shared_ptr<T> impl) {
    std::lock_guard<std::mutex> lk(mu_);
    clients_[key<T>(iface)] = std::static_pointer_cast<void>(impl);
  }
  template <typename T>
  void register_scoped(const std::string& iface, const std::string& scope,
                       std::shared_ptr<T> impl) {
    std::lock_guard<std::mutex> lk(mu_);
    clients_[key<T>(iface + "\x1f" + scope)] =
        std::static_pointer_cast<void>(impl);
  }
  template <typename T>
  std::shared_ptr<T> get(const std::string& iface) const {
    std::lock_guard<std::mutex> lk(mu_);
    auto it = clients_.find(key<T>(iface));
    if (it == clients_.end()) return nullptr;
    return std::static_pointer_cast<T>(it->second);
  }
  template <typename T>
  std::shared_ptr<T> get_scoped(const std::string& iface,
                                const std::string& scope) const {
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = clients_.find(key<T>(iface + "\x1f" + scope));
      if (it != clients_.end())
        return std::static_pointer_cast<T>(it->second);
    }
    return get<T>(iface);          // fallback: unscoped registration
  }
  template <typename T>
  bool has(const std::string& iface) const {
    std::lock_guard<std::mutex> lk(mu_);
    return clients_.count(key<T>(iface)) > 0;
  }

 private:
  template <typename T>
  static std::string key(const std::string& iface) {
    return std::string(typeid(T).name()) + "\x1e" + iface;
  }
  mutable std::mutex mu_;
  std::map<std::string, std::shared_ptr<void>> clients_;
};

// Lifecycle state machine for stateful runnables (reference
// libs/modkit/src/lifecycle.rs:33,:74 WithLifecycle<T: Runnable>):
// Stopped -> Starting -> Running -> Stopping -> Stopped, with a
// ReadySignal the entry function fires once it is serving, and a
// bounded stop (cancel flag -> join with timeout; an overrunning
// runnable is detached and reported, the thread analog of the
// reference's stop_timeout kill ladder).
class WithLifecycle {
 public:
  enum class State { Stopped, Starting, Running, Stopping };
  // entry(cancel, ready): run until `cancel` is set; call ready() once
  // serving (await_ready semantics)
  using Runnable = std::function<void(std::atomic<bool>& cancel,
                                      std::function<void()> ready)>;

  explicit WithLifecycle(Runnable r, int ready_timeout_ms = 30000,
                         int stop_timeout_ms = 30000)
      : runnable_(std::move(r)), ready_timeout_ms_(ready_timeout_ms),
        stop_timeout_ms_(stop_timeout_ms) {}
  ~WithLifecycle() { stop(); }

  State state() const { return state_.load(); }

  // false if not Stopped, or the runnable missed the ready deadline
  bool start() {
    State want = State::Stopped;
    if (!state_.compare_exchange_strong(want, State::Starting))
      return false;
    // shared with the runnable thread: an overrunning runnable is
    // detached at stop(), so everything it touches must outlive *this
    auto inner = std::make_shared<Inner>();
    inner_ = inner;
    auto fn = runnable_;
    thread_ = std::thread([inner, fn] {
      fn(inner->cancel, [inner] { inner->ready = true; });
    });
    const auto deadline = std::chrono::steady_clock::now() +
                          std::chrono::milliseconds(ready_timeout_ms_);
    while (!inner->ready &&
           std::chrono::steady_clock::now() < deadline)
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    if (!inner->ready) {
      state_ = State::Running;   // so stop() proceeds
      stop();
      return false;
    }
    state_ = State::Running;
    return true;
  }

  // true if the runnable unwound within stop_timeout
  bool stop() {
    State s = state_.load();
    if (s == State::Stopped || s == State::Stopping) return true;
    state_ = State::Stopping;
    auto inner = inner_;
    if (inner) inner->cancel = true;
    bool clean = true;
    if (thread_.joinable()) {
      // bounded join: hand the thread to a waiter so an overrunning
      // runnable can be detached without dangling into *this
      auto done = std::make_shared<std::atomic<bool>>(false);
      std::thread waiter(
          [t = std::move(thread_), done]() mutable {
            t.join();
            *done = true;
          });
      auto deadline = std::chrono::steady_clock::now() +
                      std::chrono::milliseconds(stop_timeout_ms_);
      while (!*done && std::chrono::steady_clock::now() < deadline)
        std::this_thread::sleep_for(std::chrono::milliseconds(5));
      if (*done) {
        waiter.join();
      } else {
        clean = false;           // runnable overran its stop budget
        waiter.detach();
      }
    }
    state_ = State::Stopped;
    return clean;
  }

 private:
  struct Inner {
    std::atomic<bool> cancel{false};
    std::atomic<bool> ready{false};
  };
  Runnable runnable_;
  int ready_timeout_ms_, stop_timeout_ms_;
  std::atomic<State> state_{State::Stopped};
  std::shared_ptr<Inner> inner_;
  std::thread thread_;
};

// Single-flight cached plugin-instance resolution (reference
// libs/modkit/src/plugins/mod.rs:14,:44 GtsPluginSelector): the first
// caller runs the resolver (a types-registry lookup), concurrent
// callers wait on the same future, later callers hit the cache until
// invalidate().
class GtsPluginSelector {
 public:
  using Resolver = std::function<std::string(const std::string&)>;
  explicit GtsPluginSelector(Resolver r) : resolver_(std::move(r)) {}

  std::string select(const std::string& plugin_type) {
    std::shared_ptr<std::promise<std::string>> leader;
    std::shared_future<std::string> fut;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = cache_.find(plugin_type);
      if (it != cache_.end()) {
        fut = it->second;
      } else {
        leader = std::make_shared<std::promise<std::string>>();
        fut = leader->get_future().share();
        cache_[plugin_type] = fut;
      }
    }
    if (leader) {
      try {
        leader->set_value(resolver_(plugin_type));
      } catch (...) {
        {
          std::lock_guard<std::mutex> lk(mu_);
          cache_.erase(plugin_type);      // failed lookups are not cached
        }
        leader->set_exception(std::current_exception());
      }
    }
    return fut.get();
  }

  void invalidate(const std::string& plugin_type = "") {
    std::lock_guard<std::mutex> lk(mu_);
    if (plugin_type.empty()) cache_.clear();
    else cache_.erase(plugin_type);
  }

 private:
  Resolver resolver_;
  std::mutex mu_;
  std::map<std::string, std::shared_future<std::string>> cache_;
};

// ---------------------------------------------------------------- modules
class ModuleCtx {
 public:
  Json config;          // modules.<name>.config
  Json full_config;     // whole AppConfig (read-only use)
  ClientHub* hub = nullptr;
  std::string home_dir;
  bool* cancel = nullptr;   // set true on shutdown
};

class Module {
 public:
  virtual ~Module() = default;
  virtual std::string name() const = 0;
  virtual std::vector<std::string> deps() const { return {}; }
  // capabilities
  virtual bool is_rest_host() const { return false; }
  virtual bool is_stateful() const { return false; }
  // phases (reference host_runtime.rs: pre_init/db/init/post_init/rest/
  // start/stop; grpc+oop handled by the owning modules directly)
  virtual void init(ModuleCtx& ctx) {}
  virtual void post_init(ModuleCtx& ctx) {}
  virtual void register_rest(ModuleCtx& ctx, RestRegistry& rest) {}
  virtual void start(ModuleCtx& ctx) {}
  virtual void stop(ModuleCtx& ctx) {}
};

class ModuleRegistry {
 public:
  void add(std::shared_ptr<Module> m) { modules_.push_back(std::move(m)); }
  // topo-sort by deps; throws on cycle/missing dep
  std::vector<std::shared_ptr<Module>> sorted() const;
  const std::vector<std::shared_ptr<Module>>& all() const { return modules_; }

 private:
  std::vector<std::shared_ptr<Module>> modules_;
};

// --------------------------------------------------------------- config
// Layered config: struct defaults -> YAML file -> APP__SECTION__KEY env ->
// CLI overrides (reference libs/modkit/src/bootstrap/config/mod.rs:270-283).
Json load_app_config(const std::string& yaml_path,
                     const std::map<std::string, std::string>& cli_overrides);

}  // namespace hs
