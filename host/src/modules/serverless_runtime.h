// serverless-runtime — per-tenant admission control (the inference
// scheduler's quota plane) PLUS the full serverless domain model the
// reference specifies but never implements:
//
//   entrypoints (functions + workflows, draft→active→deprecated→disabled
//   lifecycle), invocations with the ADR status state machine
//   (queued→running→succeeded|failed|suspended|canceled →
//   retry/compensating/compensated/dead_lettered), retry_policy with
//   backoff, saga compensation for workflow steps, timeline events, and
//   durable execution state in modkit-db so invocations survive a host
//   restart (RTO/RPO NFRs).
//
// Contract: /root/reference/modules/serverless-runtime/docs/
//   ADR_DOMAIN_MODEL_AND_APIS.md:48-64 (entrypoints), :549-600
//   (implementation kinds), :1030-1087 (state machine + transition
//   table), :1233-1290 (timeline events), :2599-2636 (REST surface);
//   PRD.md:38-46 (durability NFRs), :913-927 (latency NFRs).
// Deviation: ADR ":action" suffixes (entrypoints:validate, {id}:status,
// {id}:control) are mounted as subresource segments (/validate,
// /{id}/status, /{id}/control) to match this host's route grammar.
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <thread>

#include "../modkit/db.h"
#include "../modkit/modkit.h"

namespace hs {

class ServerlessRuntimeModule : public Module {
 public:
  std::string name() const override { return "serverless-runtime"; }
  bool is_stateful() const override { return true; }

  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;
  void start(ModuleCtx& ctx) override;
  void stop(ModuleCtx& ctx) override;

 private:
  // ---- execution plane ----
  void executor_loop();
  void timer_loop();
  // fire due schedules (cron/interval triggers, PRD BR-007/BR-022)
  void schedule_tick();
  // tenant runtime policy (ADR Tenant Runtime Policy API): stored row
  // or defaults; quotas gate create/start paths
  Json tenant_policy(const std::string& tenant);
  long long count_rows(const std::string& tenant, const char* table,
                       const char* extra_where = nullptr);
  std::string create_invocation(const std::string& tenant,
                                const std::string& ep_id, const Json& input,
                                const std::string& mode,
                                const std::string& schedule_id,
                                const std::string& preset_status = "");
  void run_invocation(const std::string& tenant, const std::string& id);
  // one op of a builtin/step implementation; throws on failure
  Json run_op(const std::string& op, const Json& input,
              const std::string& tenant, long long attempts);
  void enqueue(const std::string& tenant, const std::string& id);
  void enqueue_at(double when, const std::string& tenant,
                  const std::string& id);
  void timeline(const std::string& tenant, const std::string& inv_id,
                const std::string& event_type, const std::string& status,
                const std::string& step = "", long long duration_ms = -1,
                const std::string& detail = "");
  // guarded status transition per the ADR table; false if illegal
  bool transition(SecureConn& conn, const std::string& id,
                  const std::string& from, const std::string& to);
  AccessScope scope_for(const SecurityContext& sec,
                        const std::string& action,
                        const std::string& resource);

  ClientHub* hub_ = nullptr;
  std::unique_ptr<Db> db_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::pair<std::string, std::string>> queue_;  // tenant, id
  // delayed retries / suspension timeouts: (ready_at, tenant, id, kind)
  struct Timer {
    double at;
    std::string tenant, id, kind;
  };
  std::vector<Timer> timers_;
  std::map<std::string, std::string> control_;   // id -> cancel|suspend
  std::vector<std::thread> executors_;
  std::thread timer_thread_;
  std::atomic<bool> stopping_{false};
  int n_executors_ = 4;
  std::atomic<uint64_t> ctr_{0};
  double last_sched_scan_ = 0;   // timer-thread only
  // sync-mode waiters notified on terminal status
  std::condition_variable done_cv_;
};

}  // namespace hs
