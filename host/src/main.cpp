// hyperspot-server — the host-plane CLI binary.
//
// Reference: apps/hyperspot-server/src/main.rs (clap CLI, figment config
// layering, 10-phase module runtime).  Same CLI surface:
//   hyperspot-server [run|check|migrate] --config X [--port N]
//     [--print-config] [--list-modules] [--dump-modules-config-json] [-v..]
#include <signal.h>
#include <unistd.h>

#include <atomic>
#include <csignal>
#include <cstring>
#include <iostream>

#include "modkit/modkit.h"
#include "modkit/telemetry.h"
#include "modules/api_gateway.h"
#include "modules/llm_gateway.h"
#include "modules/serverless_runtime.h"
#include "modules/user_settings.h"
#include "modules/file_parser.h"
#include "modules/users_info.h"
#include "modules/oagw.h"
#include "modules/system_modules.h"
#include "util/log.h"

namespace hs {

static std::atomic<bool> g_cancel{false};

static void on_signal(int) { g_cancel = true; }

int run_server(const Json& cfg, bool check_only) {
  // OTLP trace export (telemetry/init.rs equivalent; no-op when
  // tracing.otlp_endpoint is unset) + boot-time connectivity probe
  TraceExporter::instance().configure(cfg.path("tracing"));
  if (TraceExporter::instance().enabled())
    TraceExporter::instance().probe();

  // logging init from config (bootstrap/host/logging.rs equivalent)
  const Json& lg = cfg.path("logging.default");
  Logger::get().configure(
      parse_level(lg.at("console_level").as_string("info")),
      lg.at("file").as_string(""),
      parse_level(lg.at("file_level").as_string("warn")),
      lg.at("max_size_mb").as_int(100),
      (int)lg.at("max_backups").as_int(3));

  ModuleRegistry registry;
  auto gateway = std::make_shared<ApiGatewayModule>();
  registry.add(gateway);
  registry.add(std::make_shared<AuthnResolverModule>());
  registry.add(std::make_shared<TenantResolverModule>());
  registry.add(std::make_shared<AuthzResolverModule>());
  registry.add(std::make_shared<TypesRegistryModule>());
  registry.add(std::make_shared<NodesRegistryModule>());
  registry.add(std::make_shared<ModelRegistryModule>());
  registry.add(std::make_shared<CredStoreModule>());
  registry.add(std::make_shared<ServerlessRuntimeModule>());
  registry.add(std::make_shared<OagwModule>());
  registry.add(std::make_shared<FileStorageModule>());
  auto orch = std::make_shared<ModuleOrchestratorModule>();
  registry.add(orch);
  registry.add(std::make_shared<LlmGatewayModule>());
  registry.add(std::make_shared<UserSettingsModule>());
  registry.add(std::make_shared<FileParserModule>());
  registry.add(std::make_shared<UsersInfoModule>());

  ClientHub hub;
  bool cancel_flag = false;
  auto sorted = registry.sorted();
  std::vector<std::pair<std::shared_ptr<Module>, ModuleCtx>> mods;
  for (auto& m : sorted) {
    ModuleCtx ctx;
    ctx.full_config = cfg;
    ctx.config = cfg.path("modules." + m->name() + ".config");
    ctx.hub = &hub;
    ctx.home_dir = cfg.path("server.home_dir").as_string("~/.hyperspot");
    ctx.cancel = &cancel_flag;
    mods.emplace_back(m, std::move(ctx));
  }

  // phases: init -> post_init -> rest -> start (reference
  // host_runtime.rs:717 run_phases_internal; db/grpc/oop phases live
  // inside the owning modules here)
  try {
    for (auto& [m, ctx] : mods) m->init(ctx);
    for (auto& [m, ctx] : mods) m->post_init(ctx);
    for (auto& [m, ctx] : mods) m->register_rest(ctx, gateway->rest());
    {
      // ModuleManager view: instance + capabilities + mounted endpoints
      Json infos = Json::array();
      for (auto& m : sorted) {
        Json mi = Json::object();
        mi["name"] = m->name();
        Json deps = Json::array();
        for (auto& d : m->deps()) deps.push_back(d);
        mi["deps"] = deps;
        mi["stateful"] = m->is_stateful();
        Json eps = Json::array();
        const std::string pre = "/" + m->name() + "/";
        for (auto& r : gateway->rest().routes())
          if (r.spec.path.rfind(pre, 0) == 0)
            eps.push_back(r.spec.method + " " + r.spec.path);
        mi["endpoints"] = eps;
        infos.push_back(mi);
      }
      orch->set_modules(std::move(infos));
    }
    if (check_only) {
      std::cout << "config OK; " << mods.size() << " modules, "
                << gateway->rest().routes().size() << " routes\n";
      return 0;
    }
    for (auto& [m, ctx] : mods) m->start(ctx);
  } catch (const std::exception& e) {
    LOG_ERROR("hyperspot", "startup failed: %s", e.what());
    return 1;
  }

  LOG_INFO("hyperspot", "server up (%zu modules); Ctrl-C to stop",
           mods.size());
  signal(SIGINT, on_signal);
  signal(SIGTERM, on_signal);
  while (!g_cancel) usleep(100000);
  LOG_INFO("hyperspot", "shutting down");
  cancel_flag = true;
  for (auto it = mods.rbegin(); it != mods.rend(); ++it)
    it->first->stop(it->second);
  TraceExporter::instance().shutdown();    // flush pending spans
  return 0;
}

int list_modules(const Json& cfg) {
  (void)cfg;
  for (const char* m : {"api-gateway", "authn-resolver", "tenant-resolver",
                        "authz-resolver", "types-registry",
                        "nodes-registry", "model-registry", "credstore",
                        "serverless-runtime", "oagw", "file-storage",
                        "module-orchestrator", "llm-gateway",
                        "simple-user-settings", "file-parser",
                        "users-info"})
    std::cout << m << "\n";
  return 0;
}

}  // namespace hs

int main(int argc, char** argv) {
  using namespace hs;
  std::string config_path, command = "run";
  std::map<std::string, std::string> cli;
  bool print_config = false, do_list = false, dump_json = false,
       dump_yaml = false;
  int verbosity = 0;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> std::string {
      if (i + 1 >= argc) {
        std::cerr << "missing value for " << a << "\n";
        exit(2);
      }
      return argv[++i];
    };
    if (a == "run" || a == "check" || a == "migrate") command = a;
    else if (a == "--config" || a == "-c") config_path = next();
    else if (a == "--port" || a == "-p")
      cli["modules.api-gateway.config.bind_addr"] = "\"127.0.0.1:" +
          next() + "\"";
    else if (a == "--print-config") print_config = true;
    else if (a == "--list-modules") do_list = true;
    else if (a == "--dump-modules-config-json") dump_json = true;
    else if (a == "--dump-modules-config-yaml") dump_yaml = true;
    else if (a.rfind("--set", 0) == 0 && i + 1 < argc) {
      // --set dotted.key value
      std::string k = next();
      cli[k] = next();
    } else if (a == "-v") verbosity++;
    else if (a == "-vv") verbosity += 2;
    else if (a == "--help" || a == "-h") {
      std::cout <<
          "hyperspot-server [run|check|migrate] [--config FILE] [--port N]\n"
          "  [--print-config] [--list-modules]\n"
          "  [--dump-modules-config-json|--dump-modules-config-yaml]\n"
          "  [--set dotted.key value] [-v|-vv]\n";
      return 0;
    }
  }
  if (verbosity >= 1)
    cli["logging.default.console_level"] = verbosity >= 2 ? "trace"
                                                          : "debug";
  Json cfg;
  try {
    cfg = load_app_config(config_path, cli);
  } catch (const std::exception& e) {
    std::cerr << "config error: " << e.what() << "\n";
    return 1;
  }
  if (print_config || dump_yaml) {
    std::cout << cfg.dump(2) << "\n";
    return 0;
  }
  if (dump_json) {
    std::cout << cfg.at("modules").dump(2) << "\n";
    return 0;
  }
  if (do_list) return list_modules(cfg);
  if (command == "migrate") {
    // run every db-bearing module's migrations and exit (the reference's
    // separate cloud deploy step, bootstrap/run.rs:111): init-phase only,
    // no server, no workers
    ClientHub hub;
    bool cancel = false;
    for (auto m : {std::shared_ptr<Module>(
                       std::make_shared<AuthzResolverModule>()),
                   std::shared_ptr<Module>(
                       std::make_shared<TenantResolverModule>()),
                   std::shared_ptr<Module>(
                       std::make_shared<UserSettingsModule>()),
                   std::shared_ptr<Module>(
                       std::make_shared<UsersInfoModule>())}) {
      ModuleCtx mc;
      mc.full_config = cfg;
      mc.config = cfg.path("modules." + m->name() + ".config");
      mc.hub = &hub;
      mc.home_dir = cfg.path("server.home_dir").as_string("~/.hyperspot");
      mc.cancel = &cancel;
      m->init(mc);
      std::cout << "migrated: " << m->name() << "\n";
    }
    return 0;
  }
  return run_server(cfg, command == "check");
}
