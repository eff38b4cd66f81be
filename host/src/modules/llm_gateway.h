// llm-gateway — the serving-plane API module.
//
// The reference ships this module as specification only
// (modules/llm-gateway/docs/DESIGN.md; 32 GTS schemas).  Here it is real:
// REST + SSE per the spec contract, backed by the in-node MI355X inference
// engine (hyperspot.serving.worker, one process per GPU) over a Unix
// socket — the reference's OoP module pattern with the
// provider call replaced by the engine (SURVEY.md §3.3 note).
//
// Surface (DESIGN.md:262-270): POST /chat/completions (sync + SSE),
// POST /embeddings, POST/GET/DELETE /jobs{,/{id}}, POST/GET
// /batches{,/{id}}; plus usage tracking (PRD.md:224-232) and a
// Prometheus /metrics endpoint (SURVEY.md §5.5 gap the build must fill).
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <thread>

#include "../modkit/db.h"
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

// blocking chat invoker registered by llm-gateway for in-process
// consumers (serverless-runtime adapter_ref entrypoints — the spec's
// "entrypoints are model workers" reading)
// discovery-level provider health (reference model-registry PRD:280-294
// ProviderHealth): llm-gateway registers this; model-registry serves it
// at /model-registry/v1/providers/health
struct ProviderHealthClient {
  virtual ~ProviderHealthClient() = default;
  virtual Json provider_health() = 0;
};

struct ChatInvoker {
  virtual ~ChatInvoker() = default;
  virtual Json chat(const SecurityContext& sec, const Json& body) = 0;
};

// Blocking JSON-lines client for one worker request over UDS.
class EngineConn {
 public:
  explicit EngineConn(const std::string& socket_path);
  ~EngineConn();
  bool ok() const { return fd_ >= 0; }
  bool eof() const { return eof_; }
  bool send_json(const Json& j);
  // reads one newline-terminated JSON message; empty on EOF/error OR
  // timeout (eof() distinguishes the two)
  std::optional<Json> read_json(int timeout_ms = 120000);

 private:
  int fd_ = -1;
  bool eof_ = false;
  std::string buf_;
};

// Multiplexed engine channel: ONE UDS connection per worker carries all
// chat streams (worker "attach_mux" mode).  The worker writes one
// batched line per engine step; the reader thread here demultiplexes it
// into per-request sinks consumed by the SSE/blocking chat handlers.
// This keeps the worker's GPU loop at one json-encode + one send per
// step instead of per-token fan-out to thousands of connections.
// Incremental byte-level detokenizer — C++ mirror of the worker's
// StreamDetokenizer (hyperspot/serving/tokenizer.py): byte tokens are
// id-4, BOS=1/EOS=2 are silent, other ids render as "<id>"; only
// complete UTF-8 sequences are emitted, invalid bytes become U+FFFD.
// Running it gateway-side keeps the worker's batch lines to bare
// token ids (no per-token Python text work on the GPU host process).
class ByteDetok {
 public:
  std::string push(long tok);
  std::string flush();

 private:
  std::string buf_;              // pending (possibly incomplete) UTF-8
};

struct MuxSink {
  std::mutex mu;
  std::condition_variable cv;
  std::deque<Json> q;            // delta/done/error events
  bool dead = false;             // channel lost before completion
  ByteDetok detok;               // owned by the mux reader thread
};

class MuxClient {
 public:
  explicit MuxClient(const std::string& socket_path);
  ~MuxClient();
  bool ok() const { return attached_ && !closed_; }
  // register the request id and queue the chat request (micro-batched:
  // a flusher thread ships pending submissions as ONE chat_batch line
  // every ~1 ms, so the worker's Python reader — which competes with
  // the GPU stepping thread for the GIL — admits a whole burst per
  // line; per-line submission measured ~30 req/s on a busy worker,
  // starving a 1024-stream ramp); null on failure
  std::shared_ptr<MuxSink> submit(const Json& wreq);
  void abort(const std::string& rid);    // stop generation + drop sink
  void remove(const std::string& rid);   // drop sink (request finished)
  // wait for the next event on a sink; nullopt on timeout or dead
  static std::optional<Json> next_event(MuxSink& s, int timeout_ms);

 private:
  void reader_loop();
  void flusher_loop();
  void fail_all();
  std::unique_ptr<EngineConn> conn_;
  std::atomic<bool> attached_{false};
  std::atomic<bool> closed_{false};
  std::atomic<bool> stop_{false};
  std::mutex mu_;                // guards writes + sinks map + sub_q_
  std::condition_variable sub_cv_;
  std::vector<Json> sub_q_;      // pending chat submissions
  std::map<std::string, std::shared_ptr<MuxSink>> sinks_;
  std::thread reader_;
  std::thread flusher_;
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
  bool worker_ready();                      // any worker ready
  // in-process blocking chat (ChatInvoker registration target)
  Json invoke_chat(const SecurityContext& sec, const Json& body);

 private:
  // async job state machine (DESIGN.md job/batch schemas; statuses
  // queued|running|succeeded|failed|cancelled)
  struct Job {
    std::string id, tenant, batch_id;
    std::string status = "queued";
    Json request;
    Json result;
    std::string error;       // problem code on failure
    double created_at = 0, finished_at = 0;
  };
  struct Batch {
    std::string id, tenant;
    std::vector<std::string> job_ids;
    double created_at = 0;
  };
  struct TenantUsage {
    uint64_t input_tokens = 0, output_tokens = 0, requests = 0;
  };

  void chat_handler(HttpRequest& req, ResponseWriter& w);
  void embeddings_handler(HttpRequest& req, ResponseWriter& w);
  void spawn_worker();

  // shared pipeline pieces
  Json resolve_model(const SecurityContext& sec, const std::string& model);
  void check_budget(const std::string& tenant);
  void record_usage(const std::string& tenant, const Json& usage);
  // blocking non-stream chat against the engine; throws Problem
  Json run_chat_blocking(const Json& body, const Json& resolved,
                         const std::string& rid);
  Json run_chat_with_fallback(const SecurityContext& sec, const Json& body,
                              const std::string& rid);

  // jobs
  std::shared_ptr<Job> submit_job(const SecurityContext& sec, Json body,
                                  const std::string& batch_id = "");
  Json job_json(const Job& j) const;
  void job_loop();

  ClientHub* hub_ = nullptr;
  std::string model_ = "llama3-8b";
  std::string socket_path_ = "/tmp/hyperspot-llm.sock";
  bool auto_start_ = true;
  std::string python_ = "python3";
  Json worker_cfg_;

  // data-parallel worker fleet: one engine process per GPU
  // (worker.count / worker.devices config); requests go to the
  // least-loaded ready worker
  struct Worker {
    int index = 0;
    int device = 0;
    std::string socket;
    pid_t pid = -1;
    std::atomic<bool> ready{false};
    std::atomic<int> in_flight{0};
    std::mutex mux_mu;
    std::shared_ptr<MuxClient> mux;   // serving channel (lazy, respawn-safe)
    // discovery-level ProviderHealth (reference model-registry PRD:280-294:
    // can the gateway reach the worker? NOT routing health, which is the
    // breaker's job).  Fed by the watchdog's periodic info round-trips.
    std::mutex health_mu;
    std::deque<double> probe_ms;      // recent probe latencies (<=64)
    int consec_fail = 0, consec_ok = 0;
    double last_check = 0, last_success = 0;   // unix seconds
    std::string last_error;
  };
  // the worker's mux channel, (re)connecting if stale; null if down
  std::shared_ptr<MuxClient> ensure_mux(Worker& wk);
  struct Lease {
    Worker* w = nullptr;
    ~Lease() { if (w) w->in_flight--; }
  };
  std::vector<std::unique_ptr<Worker>> workers_;
  Worker* pick_worker();                    // least-loaded ready worker
  Json provider_health();                   // ProviderHealth snapshot
  // pick + connect with failover: a dead-but-marked-ready worker is
  // demoted (watchdog respawns it) and the next one is tried
  Worker* pick_live(std::unique_ptr<EngineConn>& conn);
  bool probe_worker(Worker& wk);
  void spawn_one(Worker& wk);
  std::atomic<uint64_t> req_ctr_{0};

  // jobs/batches: in-memory hot path + write-through sqlite journal so
  // queued/running jobs survive a host restart (reference: serverless
  // durable-execution NFR; the spec's async jobs outlive the process)
  std::unique_ptr<Db> jobs_db_;
  void persist_job(const Job& j);
  void persist_batch(const Batch& b);
  void load_jobs();
  std::mutex jobs_mu_;
  std::condition_variable jobs_cv_;
  std::deque<std::string> job_queue_;
  std::map<std::string, std::shared_ptr<Job>> jobs_;
  std::map<std::string, Batch> batches_;
  std::vector<std::thread> job_threads_;
  std::thread watchdog_;
  std::atomic<bool> stopping_{false};
  std::atomic<uint64_t> m_worker_restarts_{0};

  // hook plugin (DESIGN.md:743-766): pre_call may block a request,
  // post_response may block/redact the response (config.hooks.blocklist)
  std::vector<std::string> hook_blocklist_;
  void hook_pre_call(const Json& body);          // throws request_blocked
  bool hook_blocks(const std::string& text) const;

  // timeouts (DESIGN.md:706-741 TTFT + total state machine; 0 = off)
  long ttft_timeout_ms_ = 0;
  long job_ttl_s_ = 3600;
  long total_timeout_ms_ = 0;

  // usage tracker + budget (tokens per tenant; 0 = unlimited)
  std::mutex usage_mu_;
  std::map<std::string, TenantUsage> usage_;
  uint64_t budget_tokens_ = 0;
  std::string license_feature_;

  // metrics
  std::atomic<uint64_t> m_requests_{0}, m_streams_{0}, m_errors_{0};
  std::atomic<uint64_t> m_submits_{0};
  std::atomic<uint64_t> m_input_tokens_{0}, m_output_tokens_{0};
  std::atomic<uint64_t> m_ttft_us_sum_{0}, m_ttft_count_{0};
};

}  // namespace hs
