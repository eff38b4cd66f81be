#include "llm_gateway.h"

#include <poll.h>
#include <signal.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <sys/un.h>
#include <sys/wait.h>
#include <unistd.h>

#include <chrono>
#include <cstring>

#include "../util/log.h"
#include "system_modules.h"

namespace hs {

static double now_s() {
  return std::chrono::duration<double>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

namespace {

// base64 for embedding_response encoding_format=base64 (little-endian f32,
// the OpenAI-compatible convention the schema references)
std::string b64_encode(const unsigned char* data, size_t n) {
  static const char* tbl =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  std::string out;
  out.reserve((n + 2) / 3 * 4);
  for (size_t i = 0; i < n; i += 3) {
    unsigned v = data[i] << 16;
    if (i + 1 < n) v |= data[i + 1] << 8;
    if (i + 2 < n) v |= data[i + 2];
    out.push_back(tbl[(v >> 18) & 63]);
    out.push_back(tbl[(v >> 12) & 63]);
    out.push_back(i + 1 < n ? tbl[(v >> 6) & 63] : '=');
    out.push_back(i + 2 < n ? tbl[v & 63] : '=');
  }
  return out;
}

}  // namespace

// ------------------------------------------------------------- EngineConn

EngineConn::EngineConn(const std::string& socket_path) {
  fd_ = socket(AF_UNIX, SOCK_STREAM, 0);
  if (fd_ < 0) return;
  sockaddr_un addr{};
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, socket_path.c_str(), sizeof addr.sun_path - 1);
  if (connect(fd_, (sockaddr*)&addr, sizeof addr) != 0) {
    close(fd_);
    fd_ = -1;
  }
}

EngineConn::~EngineConn() {
  if (fd_ >= 0) close(fd_);
}

bool EngineConn::send_json(const Json& j) {
  std::string line = j.dump() + "\n";
  const char* p = line.data();
  size_t n = line.size();
  while (n) {
    ssize_t w = ::send(fd_, p, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    p += w;
    n -= size_t(w);
  }
  return true;
}

std::optional<Json> EngineConn::read_json(int timeout_ms) {
  while (true) {
    size_t nl = buf_.find('\n');
    if (nl != std::string::npos) {
      std::string line = buf_.substr(0, nl);
      buf_.erase(0, nl + 1);
      if (line.empty()) continue;
      try { return Json::parse(line); }
      catch (...) { return std::nullopt; }
    }
    struct pollfd pf{fd_, POLLIN, 0};
    int pr = poll(&pf, 1, timeout_ms);
    if (pr <= 0) return std::nullopt;
    char tmp[65536];
    ssize_t r = recv(fd_, tmp, sizeof tmp, 0);
    if (r <= 0) {
      eof_ = true;
      return std::nullopt;
    }
    buf_.append(tmp, size_t(r));
  }
}

// ------------------------------------------------------------- ByteDetok

namespace {

// bytes [0, n) that form complete UTF-8 sequences; -1 byte position of
// the first invalid byte, or n if only an incomplete tail remains
size_t utf8_complete_prefix(const std::string& b, size_t* bad) {
  *bad = std::string::npos;
  size_t i = 0;
  while (i < b.size()) {
    unsigned char c = b[i];
    int need;
    if (c < 0x80) need = 0;
    else if ((c & 0xe0) == 0xc0 && c >= 0xc2) need = 1;
    else if ((c & 0xf0) == 0xe0) need = 2;
    else if ((c & 0xf8) == 0xf0 && c <= 0xf4) need = 3;
    else { *bad = i; return i; }
    if (i + 1 + need > b.size()) return i;     // incomplete tail
    for (int k = 1; k <= need; ++k)
      if ((static_cast<unsigned char>(b[i + k]) & 0xc0) != 0x80) {
        *bad = i;
        return i;
      }
    i += 1 + need;
  }
  return i;
}

// decode with U+FFFD for invalid bytes (python errors="replace")
std::string utf8_replace(const std::string& b) {
  std::string out;
  size_t i = 0;
  while (i < b.size()) {
    size_t bad;
    std::string rest = b.substr(i);
    size_t ok = utf8_complete_prefix(rest, &bad);
    out += rest.substr(0, ok);
    if (ok == rest.size()) break;
    out += "\xEF\xBF\xBD";        // U+FFFD
    i += ok + 1;
  }
  return out;
}

}  // namespace

std::string ByteDetok::push(long tok) {
  if (tok < 4 || tok >= 260) {
    std::string pre = flush();
    if (tok == 1 || tok == 2) return pre;       // BOS/EOS silent
    return pre + "<" + std::to_string(tok) + ">";
  }
  buf_.push_back(static_cast<char>(tok - 4));
  size_t bad;
  size_t ok = utf8_complete_prefix(buf_, &bad);
  if (ok == buf_.size()) {                      // fully valid
    std::string out;
    out.swap(buf_);
    return out;
  }
  if (ok > 0) {                                 // emit valid prefix
    std::string out = buf_.substr(0, ok);
    buf_.erase(0, ok);
    // python re-checks the remainder next push; if the head byte is
    // invalid it stays until 4 bytes accumulate — mirror that
    return out;
  }
  if (buf_.size() >= 4) return flush();         // invalid, give up
  return "";
}

std::string ByteDetok::flush() {
  std::string out = utf8_replace(buf_);
  buf_.clear();
  return out;
}

// ------------------------------------------------------------- MuxClient

MuxClient::MuxClient(const std::string& socket_path) {
  conn_ = std::make_unique<EngineConn>(socket_path);
  if (!conn_->ok()) { closed_ = true; return; }
  Json a = Json::object();
  a["type"] = "attach_mux";
  if (!conn_->send_json(a)) { closed_ = true; return; }
  auto r = conn_->read_json(10000);
  if (!r || r->at("event").as_string() != "mux_attached") {
    closed_ = true;
    return;
  }
  attached_ = true;
  reader_ = std::thread([this] { reader_loop(); });
  flusher_ = std::thread([this] { flusher_loop(); });
}

MuxClient::~MuxClient() {
  stop_ = true;
  closed_ = true;
  sub_cv_.notify_all();
  if (flusher_.joinable()) flusher_.join();
  conn_.reset();                 // closes the fd; reader unblocks
  if (reader_.joinable()) reader_.join();
  fail_all();
}

void MuxClient::fail_all() {
  std::lock_guard<std::mutex> lk(mu_);
  for (auto& [rid, s] : sinks_) {
    std::lock_guard<std::mutex> sl(s->mu);
    s->dead = true;
    s->cv.notify_all();
  }
  sinks_.clear();
}

void MuxClient::reader_loop() {
  while (!stop_) {
    auto msg = conn_->read_json(1000);
    if (!msg) {
      if (stop_ || conn_->eof()) break;
      continue;                  // poll timeout, keep waiting
    }
    const std::string ev = msg->at("event").as_string();
    if (ev == "batch") {
      const Json& items = msg->at("items");
      std::lock_guard<std::mutex> lk(mu_);
      for (auto& it : items.arr()) {
        // [rid, token_id] delta | [rid, tok, fin, in, out] final —
        // detokenization happens HERE (C++, off the worker's GPU loop)
        const std::string rid = it.at(0).as_string();
        auto f = sinks_.find(rid);
        if (f == sinks_.end()) continue;       // aborted meanwhile
        auto& s = *f->second;
        const bool fin = it.size() > 2;
        const long tok = it.at(1).as_int(0);
        std::string text = s.detok.push(tok);
        if (fin) text += s.detok.flush();
        {
          std::lock_guard<std::mutex> sl(s.mu);
          Json d = Json::object();
          d["event"] = "delta";
          d["token_id"] = (long)tok;
          d["text"] = text;
          s.q.push_back(std::move(d));
          if (fin) {
            Json u = Json::object();
            u["input_tokens"] = it.at(3);
            u["output_tokens"] = it.at(4);
            Json dn = Json::object();
            dn["event"] = "done";
            dn["finish_reason"] = it.at(2);
            dn["usage"] = u;
            s.q.push_back(std::move(dn));
          }
          s.cv.notify_all();
        }
        if (fin) sinks_.erase(f);
      }
    } else if (ev == "error") {
      const std::string rid = msg->at("id").as_string();
      std::lock_guard<std::mutex> lk(mu_);
      auto f = sinks_.find(rid);
      if (f != sinks_.end()) {
        std::lock_guard<std::mutex> sl(f->second->mu);
        f->second->q.push_back(*msg);
        f->second->cv.notify_all();
        sinks_.erase(f);
      }
    }
  }
  closed_ = true;
  fail_all();
}

std::shared_ptr<MuxSink> MuxClient::submit(const Json& wreq) {
  const std::string rid = wreq.at("id").as_string();
  auto sink = std::make_shared<MuxSink>();
  {
    std::lock_guard<std::mutex> lk(mu_);
    if (closed_) return nullptr;
    sinks_[rid] = sink;
    sub_q_.push_back(wreq);
  }
  sub_cv_.notify_one();
  return sink;
}

void MuxClient::flusher_loop() {
  while (!stop_) {
    std::vector<Json> batch;
    {
      std::unique_lock<std::mutex> lk(mu_);
      sub_cv_.wait_for(lk, std::chrono::milliseconds(1),
                       [&] { return !sub_q_.empty() || stop_; });
      if (stop_ && sub_q_.empty()) return;
      batch.swap(sub_q_);
    }
    if (batch.empty()) continue;
    Json reqs = Json::array();
    for (auto& b : batch) reqs.push_back(std::move(b));
    Json line = Json::object();
    line["type"] = "chat_batch";
    line["reqs"] = std::move(reqs);
    std::lock_guard<std::mutex> lk(mu_);
    if (closed_) return;
    if (!conn_->send_json(line)) {
      closed_ = true;
      return;
    }
  }
}

void MuxClient::abort(const std::string& rid) {
  std::lock_guard<std::mutex> lk(mu_);
  sinks_.erase(rid);
  for (auto it = sub_q_.begin(); it != sub_q_.end(); ++it)
    if (it->at("id").as_string() == rid) {
      sub_q_.erase(it);
      return;                       // never reached the worker
    }
  if (closed_) return;
  Json ab = Json::object();
  ab["type"] = "abort";
  ab["id"] = rid;
  conn_->send_json(ab);
}

void MuxClient::remove(const std::string& rid) {
  std::lock_guard<std::mutex> lk(mu_);
  sinks_.erase(rid);
}

std::optional<Json> MuxClient::next_event(MuxSink& s, int timeout_ms) {
  std::unique_lock<std::mutex> lk(s.mu);
  if (!s.cv.wait_for(lk, std::chrono::milliseconds(timeout_ms),
                     [&] { return !s.q.empty() || s.dead; }))
    return std::nullopt;                      // timeout
  if (s.q.empty()) return std::nullopt;       // dead
  Json out = std::move(s.q.front());
  s.q.pop_front();
  return out;
}

// ---------------------------------------------------------- LlmGateway

namespace {
class GatewayChatInvoker : public ChatInvoker {
 public:
  explicit GatewayChatInvoker(LlmGatewayModule* m) : m_(m) {}
  Json chat(const SecurityContext& sec, const Json& body) override {
    return m_->invoke_chat(sec, body);
  }

 private:
  LlmGatewayModule* m_;
};
}  // namespace

Json LlmGatewayModule::invoke_chat(const SecurityContext& sec,
                                   const Json& body) {
  const std::string rid = "sl-" + std::to_string(++req_ctr_);
  return run_chat_with_fallback(sec, body, rid);
}

void LlmGatewayModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  ctx.hub->register_client<ChatInvoker>(
      "llm-gateway",
      std::make_shared<GatewayChatInvoker>(this));
  struct HubHealth : ProviderHealthClient {
    LlmGatewayModule* m;
    Json provider_health() override { return m->provider_health(); }
  };
  auto hh = std::make_shared<HubHealth>();
  hh->m = this;
  ctx.hub->register_client<ProviderHealthClient>("llm-gateway", hh);
  {
    std::string file = ctx.full_config
                           .path("modules.llm-gateway.database.file")
                           .as_string("");
    if (file.empty()) {
      std::string home = ctx.full_config.path("server.home_dir")
                             .as_string("~/.hyperspot");
      if (!home.empty() && home[0] == '~') {
        const char* h = getenv("HOME");
        home = std::string(h ? h : "/tmp") + home.substr(1);
      }
      mkdir(home.c_str(), 0755);
      file = home + "/llm-gateway.db";
    }
    jobs_db_ = std::make_unique<Db>(file);
    jobs_db_->migrate("llm-gateway", {
        {"0001_jobs",
         "CREATE TABLE jobs (tenant_id TEXT NOT NULL, id TEXT NOT NULL "
         "UNIQUE, batch_id TEXT NOT NULL DEFAULT '', status TEXT NOT "
         "NULL, request TEXT NOT NULL, result TEXT NOT NULL DEFAULT "
         "'', error TEXT NOT NULL DEFAULT '', created_at REAL NOT "
         "NULL, finished_at REAL NOT NULL DEFAULT 0)"},
        {"0002_batches",
         "CREATE TABLE batches (tenant_id TEXT NOT NULL, id TEXT NOT "
         "NULL UNIQUE, job_ids TEXT NOT NULL, created_at REAL NOT "
         "NULL)"},
    });
    load_jobs();
  }
  model_ = ctx.config.at("model").as_string(model_);
  socket_path_ = ctx.config.at("worker_socket").as_string(socket_path_);
  auto_start_ = ctx.config.at("auto_start_worker").as_bool(true);
  python_ = ctx.config.at("python").as_string("python3");
  worker_cfg_ = ctx.config.at("worker");
  budget_tokens_ =
      (uint64_t)ctx.config.path("usage.budget_tokens_per_tenant").as_int(0);
  license_feature_ = ctx.config.at("require_license_feature").as_string("");
  ttft_timeout_ms_ = ctx.config.path("timeouts.ttft_ms").as_int(0);
  job_ttl_s_ = ctx.config.path("jobs.ttl_s").as_int(3600);
  if (ctx.config.path("hooks.blocklist").is_array())
    for (auto& wd : ctx.config.path("hooks.blocklist").arr())
      hook_blocklist_.push_back(wd.as_string());
  total_timeout_ms_ = ctx.config.path("timeouts.total_ms").as_int(0);
  // worker fleet: explicit device list, or count (devices 0..count-1)
  std::vector<int> devices;
  if (worker_cfg_.at("devices").is_array())
    for (auto& d : worker_cfg_.at("devices").arr())
      devices.push_back((int)d.as_int(0));
  else
    for (int i = 0; i < (int)worker_cfg_.at("count").as_int(1); ++i)
      devices.push_back(i);
  if (devices.empty()) devices.push_back(0);
  // a tp>1 worker is itself a torchrun group spanning tp GPUs — it owns
  // the device selection (LOCAL_RANK); don't also shard by fleet index
  if (worker_cfg_.is_object() && worker_cfg_.at("tp").as_int(1) > 1 &&
      devices.size() > 1) {
    LOG_INFO("llm-gateway",
             "worker.tp>1: collapsing fleet to one TP group");
    devices.resize(1);
  }
  for (size_t i = 0; i < devices.size(); ++i) {
    auto w = std::make_unique<Worker>();
    w->index = (int)i;
    w->device = devices[i];
    w->socket = devices.size() == 1
                    ? socket_path_
                    : socket_path_ + "." + std::to_string(i);
    workers_.push_back(std::move(w));
  }
}

bool LlmGatewayModule::probe_worker(Worker& wk) {
  if (wk.ready) return true;
  EngineConn c(wk.socket);
  if (!c.ok()) return false;
  Json q = Json::object();
  q["type"] = "info";
  if (!c.send_json(q)) return false;
  auto r = c.read_json(3000);
  if (r && r->at("ready").as_bool()) {
    wk.ready = true;
    return true;
  }
  return false;
}

bool LlmGatewayModule::worker_ready() {
  for (auto& w : workers_)
    if (probe_worker(*w)) return true;
  return false;
}

LlmGatewayModule::Worker* LlmGatewayModule::pick_live(
    std::unique_ptr<EngineConn>& conn) {
  for (size_t att = 0; att < workers_.size() + 1; ++att) {
    Worker* w = pick_worker();
    if (!w) return nullptr;
    conn = std::make_unique<EngineConn>(w->socket);
    if (conn->ok()) return w;
    LOG_ERROR("llm-gateway", "worker %d connect failed; demoting",
              w->index);
    w->ready = false;          // stale: process died since the last probe
    w->in_flight--;
    conn.reset();
  }
  return nullptr;
}

LlmGatewayModule::Worker* LlmGatewayModule::pick_worker() {
  Worker* best = nullptr;
  for (auto& w : workers_) {
    if (!probe_worker(*w)) continue;
    if (!best || w->in_flight.load() < best->in_flight.load())
      best = w.get();
  }
  if (best) best->in_flight++;
  return best;
}

Json LlmGatewayModule::provider_health() {
  // status rules per reference PRD:289-294: unhealthy at 3+ consecutive
  // failures, healthy at 2+ consecutive successes, degraded between
  Json items = Json::array();
  for (auto& w : workers_) {
    std::lock_guard<std::mutex> hl(w->health_mu);
    Json it = Json::object();
    it["provider_id"] = "local::worker-" + std::to_string(w->index);
    it["device"] = (long)w->device;
    std::string status = "degraded";
    if (w->consec_fail >= 3) status = "unhealthy";
    else if (w->consec_ok >= 2) status = "healthy";
    it["status"] = status;
    Json m = Json::object();
    if (!w->probe_ms.empty()) {
      std::vector<double> v(w->probe_ms.begin(), w->probe_ms.end());
      std::sort(v.begin(), v.end());
      m["latency_p50_ms"] = v[v.size() / 2];
      m["latency_p99_ms"] = v[(v.size() * 99) / 100];
    }
    m["consecutive_failures"] = (long)w->consec_fail;
    m["consecutive_successes"] = (long)w->consec_ok;
    it["metrics"] = m;
    if (w->last_check > 0) it["last_check"] = w->last_check;
    if (w->last_success > 0) it["last_success"] = w->last_success;
    if (!w->last_error.empty()) it["last_error_message"] = w->last_error;
    items.push_back(it);
  }
  Json out = Json::object();
  out["items"] = items;
  return out;
}

std::shared_ptr<MuxClient> LlmGatewayModule::ensure_mux(Worker& wk) {
  std::lock_guard<std::mutex> lk(wk.mux_mu);
  if (wk.mux && wk.mux->ok()) return wk.mux;
  auto m = std::make_shared<MuxClient>(wk.socket);
  if (!m->ok()) return nullptr;
  wk.mux = m;
  return m;
}

void LlmGatewayModule::spawn_one(Worker& wk) {
  pid_t pid = fork();
  if (pid == 0) {
    setpgid(0, 0);
    if (workers_.size() > 1) {
      const std::string dev = std::to_string(wk.device);
      setenv("HIP_VISIBLE_DEVICES", dev.c_str(), 1);
      setenv("CUDA_VISIBLE_DEVICES", dev.c_str(), 1);
    }
    // worker.environment: extra env for the engine process (A.6
    // execution.environment envelope) — the RCCL tuning surface for
    // 8-GPU runs (NCCL_ALGO/NCCL_PROTO/NCCL_MIN_NCHANNELS etc.)
    if (worker_cfg_.is_object() &&
        worker_cfg_.at("environment").is_object())
      for (auto& [k, v] : worker_cfg_.at("environment").obj())
        setenv(k.c_str(), v.as_string().c_str(), 1);
    // tp>1 workers are an SPMD torchrun group (one rank per GPU, RCCL
    // over xGMI); rank 0 owns the socket and broadcasts the op log
    const long tp = worker_cfg_.is_object()
                        ? worker_cfg_.at("tp").as_int(1) : 1;
    std::vector<std::string> args;
    if (tp > 1) {
      args = {python_, "-m", "torch.distributed.run", "--standalone",
              "--local-addr", "127.0.0.1", "--nnodes", "1",
              "--nproc-per-node", std::to_string(tp), "-m",
              "hyperspot.serving.worker"};
    } else {
      args = {python_, "-m", "hyperspot.serving.worker"};
    }
    args.insert(args.end(), {"--uds", wk.socket, "--model", model_});
    if (worker_cfg_.is_object()) {
      if (worker_cfg_.contains("max_num_seqs")) {
        args.push_back("--max-num-seqs");
        args.push_back(std::to_string(
            worker_cfg_.at("max_num_seqs").as_int(256)));
      }
      if (worker_cfg_.at("eager").as_bool(false))
        args.push_back("--eager");
      if (worker_cfg_.contains("device")) {
        args.push_back("--device");
        args.push_back(worker_cfg_.at("device").as_string());
      }
      if (worker_cfg_.contains("num_gpu_blocks")) {
        args.push_back("--num-gpu-blocks");
        args.push_back(std::to_string(
            worker_cfg_.at("num_gpu_blocks").as_int(0)));
      }
      if (worker_cfg_.contains("max_model_len")) {
        args.push_back("--max-model-len");
        args.push_back(std::to_string(
            worker_cfg_.at("max_model_len").as_int(8192)));
      }
      if (worker_cfg_.contains("tp")) {
        args.push_back("--tp");
        args.push_back(std::to_string(worker_cfg_.at("tp").as_int(1)));
      }
      if (worker_cfg_.contains("ep")) {
        args.push_back("--ep");
        args.push_back(std::to_string(worker_cfg_.at("ep").as_int(0)));
      }
      if (worker_cfg_.contains("quant")) {
        args.push_back("--quant");
        args.push_back(worker_cfg_.at("quant").as_string());
      }
      if (worker_cfg_.contains("kv_dtype")) {
        args.push_back("--kv-dtype");
        args.push_back(worker_cfg_.at("kv_dtype").as_string());
      }
    }
    std::vector<char*> argv;
    for (auto& a : args) argv.push_back(const_cast<char*>(a.c_str()));
    argv.push_back(nullptr);
    execvp(argv[0], argv.data());
    _exit(127);
  }
  wk.pid = pid;
  LOG_INFO("llm-gateway",
           "spawned engine worker %d pid=%d device=%d sock=%s", wk.index,
           pid, wk.device, wk.socket.c_str());
}

void LlmGatewayModule::spawn_worker() {
  for (auto& w : workers_) {
    unlink(w->socket.c_str());
    spawn_one(*w);
  }
}

void LlmGatewayModule::start(ModuleCtx& ctx) {
  stopping_ = false;
  if (auto_start_) {
    spawn_worker();
    // failure detection + elastic recovery: reap dead
    // engine workers and respawn them; requests route to the remaining
    // live workers meanwhile
    watchdog_ = std::thread([this] {
      while (!stopping_) {
        for (int i = 0; i < 20 && !stopping_; ++i) usleep(100000);
        if (stopping_) break;
        for (auto& w : workers_) {
          if (w->pid <= 0) continue;
          int st = 0;
          if (waitpid(w->pid, &st, WNOHANG) == w->pid) {
            LOG_ERROR("llm-gateway",
                      "engine worker %d pid=%d died (status=%d); respawn",
                      w->index, w->pid, st);
            w->ready = false;
            m_worker_restarts_++;
            unlink(w->socket.c_str());
            spawn_one(*w);
          }
          // discovery health probe: one info round-trip per cycle
          // (ProviderHealth metrics — latency ring + consec counters)
          const auto t0 = std::chrono::steady_clock::now();
          bool ok = false;
          std::string err;
          {
            EngineConn c(w->socket);
            if (!c.ok()) {
              err = "connect failed";
            } else {
              Json q = Json::object();
              q["type"] = "info";
              if (!c.send_json(q)) {
                err = "send failed";
              } else {
                auto r = c.read_json(3000);
                if (r && r->at("ready").as_bool()) ok = true;
                else err = r ? "not ready" : "info timeout";
              }
            }
          }
          const double ms = std::chrono::duration<double, std::milli>(
              std::chrono::steady_clock::now() - t0).count();
          std::lock_guard<std::mutex> hl(w->health_mu);
          w->last_check = (double)time(nullptr);
          if (ok) {
            w->consec_ok++;
            w->consec_fail = 0;
            w->last_success = w->last_check;
            w->probe_ms.push_back(ms);
            if (w->probe_ms.size() > 64) w->probe_ms.pop_front();
          } else {
            w->consec_fail++;
            w->consec_ok = 0;
            w->last_error = err;
          }
        }
      }
    });
  }
  // small executor pool so batch jobs overlap in the engine's
  // continuous batch instead of serializing
  for (int i = 0; i < 4; ++i)
    job_threads_.emplace_back([this] { job_loop(); });
}

void LlmGatewayModule::stop(ModuleCtx& ctx) {
  {
    std::lock_guard<std::mutex> lk(jobs_mu_);
    stopping_ = true;
  }
  jobs_cv_.notify_all();
  for (auto& t : job_threads_)
    if (t.joinable()) t.join();
  job_threads_.clear();
  if (watchdog_.joinable()) watchdog_.join();
  for (auto& w : workers_) {
    if (w->pid <= 0) continue;
    kill(w->pid, SIGTERM);
    int st = 0;
    for (int i = 0; i < 50; ++i) {
      if (waitpid(w->pid, &st, WNOHANG) == w->pid) {
        w->pid = -1;
        break;
      }
      usleep(100000);
    }
    if (w->pid > 0) {
      kill(w->pid, SIGKILL);
      waitpid(w->pid, &st, 0);
    }
  }
}

// ------------------------------------------------- shared pipeline pieces

Json LlmGatewayModule::resolve_model(const SecurityContext& sec,
                                     const std::string& model) {
  auto reg = hub_->get<ModelRegistryClient>("model-registry");
  auto resolved = reg ? reg->get_tenant_model(sec.tenant_id, model)
                      : std::nullopt;
  if (!resolved)
    throw Problem{404, "Not Found", "about:blank",
                  "model '" + model + "' not found", "model_not_found"};
  if (resolved->at("approval").as_string("approved") != "approved")
    throw Problem{403, "Forbidden", "about:blank",
                  "model '" + model + "' is not approved for this tenant",
                  "model_not_approved"};
  return *resolved;
}

void LlmGatewayModule::check_budget(const std::string& tenant) {
  if (!budget_tokens_) return;
  std::lock_guard<std::mutex> lk(usage_mu_);
  auto it = usage_.find(tenant);
  if (it != usage_.end() &&
      it->second.input_tokens + it->second.output_tokens >= budget_tokens_)
    throw Problem{429, "Too Many Requests", "about:blank",
                  "tenant token budget exhausted", "budget_exceeded"};
}

void LlmGatewayModule::record_usage(const std::string& tenant,
                                    const Json& usage) {
  const uint64_t in = (uint64_t)usage.at("input_tokens").as_int(0);
  const uint64_t out = (uint64_t)usage.at("output_tokens").as_int(0);
  m_input_tokens_ += in;
  m_output_tokens_ += out;
  std::lock_guard<std::mutex> lk(usage_mu_);
  auto& u = usage_[tenant];
  u.input_tokens += in;
  u.output_tokens += out;
  u.requests += 1;
}

Json LlmGatewayModule::run_chat_blocking(const Json& body,
                                         const Json& resolved,
                                         const std::string& rid) {
  const auto t0 = std::chrono::steady_clock::now();
  auto left = [&](long limit_ms) -> int {
    if (!limit_ms) return 120000;
    auto used = std::chrono::duration_cast<std::chrono::milliseconds>(
        std::chrono::steady_clock::now() - t0).count();
    return (int)std::max<long>(1, limit_ms - used);
  };
  Json wreq = Json::object();
  wreq["type"] = "chat";
  wreq["id"] = rid;
  wreq["model"] = resolved.at("provider_model_id").as_string();
  wreq["messages"] = body.at("messages");
  Json params = body.at("params");
  if (params.is_null()) params = Json::object();
  for (const char* f : {"temperature", "top_p", "top_k", "max_tokens",
                        "seed", "ignore_eos", "response_format"})
    if (body.contains(f)) params[f] = body.at(f);
  // reference request schema: `response_schema` asks for structured
  // output — the engine enforces the JSON grammar via constrained
  // decoding (engine/guided.py); the schema body itself is advisory
  if (body.at("response_schema").is_object()) {
    // schema-SHAPED decoding: the engine forces the declared skeleton
    // (guided.SchemaMachine); non-object schemas fall back to plain
    // JSON-grammar enforcement
    params["response_schema"] = body.at("response_schema");
    params["response_format"] = "json";
  } else if (body.contains("response_schema") &&
             !params.contains("response_format")) {
    params["response_format"] = "json";
  }
  if (body.at("tools").is_array() && body.at("tools").size() > 0) {
    wreq["tools"] = body.at("tools");
    if (body.contains("tool_choice"))
      wreq["tool_choice"] = body.at("tool_choice");
    if (body.at("tool_choice").as_string("") == "required")
      // forced call: the engine's ToolCallMachine grammar emits the
      // {"name":..., "arguments":{...}} skeleton byte-exactly
      params["response_format"] = "tool_call";
  }
  wreq["params"] = params;

  // a submit can race a DYING worker (SIGKILL teardown window): before
  // any token arrived the request is trivially retryable on another
  // worker (the sink goes dead, never delivers)
  for (int attempt = 0; attempt < 3; ++attempt) {
    Lease lease{pick_worker()};
    if (!lease.w)
      throw Problem{503, "Service Unavailable", "about:blank",
                    "no engine worker ready", "provider_error"};
    auto mux = ensure_mux(*lease.w);
    if (!mux) {
      lease.w->ready = false;
      continue;
    }
    auto sink = mux->submit(wreq);
    if (!sink) {
      lease.w->ready = false;
      continue;
    }
    std::string text;
    Json usage;
    std::string finish = "stop";
    bool first = true;
    bool retry = false;
    while (true) {
      // TTFT timer until the first delta, then the total timer
      // (DESIGN.md:706-741)
      auto msg = MuxClient::next_event(
          *sink, left(first && ttft_timeout_ms_ ? ttft_timeout_ms_
                                                : total_timeout_ms_));
      if (!msg) {
        bool dead;
        {
          std::lock_guard<std::mutex> sl(sink->mu);
          dead = sink->dead;
        }
        if (first && dead && attempt < 2) {
          // channel lost pre-token: dying-worker race — retry
          lease.w->ready = false;
          retry = true;
          break;
        }
        if (dead)
          throw Problem{502, "Bad Gateway", "about:blank",
                        "engine connection lost mid-generation",
                        "provider_error"};
        mux->abort(rid);
        throw Problem{504, "Gateway Timeout", "about:blank",
                      first ? "no first token within the TTFT budget"
                            : "generation exceeded the total budget",
                      "provider_timeout"};
      }
      const std::string ev = msg->at("event").as_string();
      if (ev == "delta") { first = false; text += msg->at("text").as_string(); }
      else if (ev == "done") {
        usage = msg->at("usage");
        finish = msg->at("finish_reason").as_string("stop");
        break;
      } else if (ev == "error") {
        mux->remove(rid);
        // worker-declared request errors (e.g. prompt exceeds the
        // model's context window) are the CLIENT's fault, not a
        // provider failure
        if (msg->at("code").as_string("") == "validation_error")
          throw Problem{400, "Bad Request", "about:blank",
                        msg->at("message").as_string(),
                        "validation_error"};
        throw Problem{502, "Bad Gateway", "about:blank",
                      msg->at("message").as_string(), "provider_error"};
      }
    }
    if (retry) continue;
    Json content = Json::array();
    bool is_tool_call = false;
    if (body.at("tool_choice").as_string("") == "required" &&
        body.at("tools").is_array() && body.at("tools").size() > 0) {
      // guided decoding forced valid JSON; shape it as a tool call when
      // it parses as {name, arguments} (ToolCall content schema)
      try {
        Json tc = Json::parse(text);
        if (tc.is_object() && !tc.at("name").as_string().empty()) {
          Json call = Json::object();
          call["id"] = "call-" + rid;
          call["name"] = tc.at("name");
          call["arguments"] = tc.at("arguments");
          Json part = Json::object();
          part["type"] = "tool_call";
          part["tool_call"] = call;
          content.push_back(part);
          is_tool_call = true;
          finish = "tool_calls";
        }
      } catch (...) {}
    }
    if (!is_tool_call) {
      Json part = Json::object();
      part["type"] = "text";
      part["text"] = text;
      content.push_back(part);
    }
    Json resp = Json::object();
    resp["content"] = content;
    resp["usage"] = usage;
    resp["model_used"] = resolved.at("canonical_id").as_string();
    resp["fallback_used"] = false;
    resp["finish_reason"] = finish;
    return resp;
  }
  throw Problem{503, "Service Unavailable", "about:blank",
                "engine workers unavailable", "provider_error"};
}

// Build a stream_chunk.v1-shaped SSE event
// (llm-gateway-sdk/schemas/core/stream_chunk.v1.schema.json)
static std::string sse_chunk(const std::string& id, const std::string& model,
                             const Json& delta,
                             const std::string& finish_reason = "",
                             const Json& usage = Json()) {
  Json c = Json::object();
  c["id"] = id;
  c["model"] = model;
  c["delta"] = delta;
  if (!finish_reason.empty()) c["finish_reason"] = finish_reason;
  if (!usage.is_null()) c["usage"] = usage;
  return "data: " + c.dump() + "\n\n";
}

bool LlmGatewayModule::hook_blocks(const std::string& text) const {
  for (auto& w : hook_blocklist_)
    if (!w.empty() && text.find(w) != std::string::npos) return true;
  return false;
}

// pre_call hook: scan inbound message text (DESIGN.md:743-766)
void LlmGatewayModule::hook_pre_call(const Json& body) {
  if (hook_blocklist_.empty()) return;
  const Json& msgs = body.at("messages");
  if (!msgs.is_array()) return;
  for (auto& m : msgs.arr()) {
    const Json& content = m.at("content");
    if (!content.is_array()) continue;
    for (auto& part : content.arr())
      if (hook_blocks(part.at("text").as_string()))
        throw Problem{403, "Forbidden", "about:blank",
                      "request blocked by content policy",
                      "request_blocked"};
  }
}

// fallback chain (DESIGN.md:680-704): on provider_error/timeout try the
// next model in request.fallback.models; mark fallback_used + model_used.
Json LlmGatewayModule::run_chat_with_fallback(const SecurityContext& sec,
                                              const Json& body,
                                              const std::string& rid) {
  std::vector<std::string> chain = {body.at("model").as_string()};
  const Json& fb = body.path("fallback.models");
  if (fb.is_array())
    for (auto& m : fb.arr()) chain.push_back(m.as_string());
  Problem last{502, "Bad Gateway", "about:blank", "no model attempted",
               "provider_error"};
  for (size_t i = 0; i < chain.size(); ++i) {
    Json resolved;
    try {
      resolved = resolve_model(sec, chain[i]);
    } catch (const Problem& p) {
      last = p;
      continue;            // unknown model in the chain: try the next
    }
    try {
      Json resp = run_chat_blocking(body, resolved, rid);
      resp["fallback_used"] = i > 0;
      return resp;
    } catch (const Problem& p) {
      if (p.code != "provider_error" && p.code != "provider_timeout")
        throw;             // validation/budget errors do not fall back
      last = p;
    }
  }
  throw last;
}

void LlmGatewayModule::chat_handler(HttpRequest& req, ResponseWriter& w) {
  m_requests_++;
  Json body;
  try { body = Json::parse(req.body); }
  catch (...) {
    throw Problem{400, "Bad Request", "about:blank", "invalid JSON body",
                  "validation_error"};
  }
  // request.v1 schema: required model + messages (Appendix B)
  const std::string model = body.at("model").as_string();
  if (model.empty() || !body.at("messages").is_array() ||
      body.at("messages").size() == 0)
    throw Problem{400, "Bad Request", "about:blank",
                  "'model' and non-empty 'messages' are required",
                  "validation_error"};
  const bool stream = body.at("stream").as_bool(false);

  SecurityContext sec =
      SecurityContext::from_json(req.extensions.at("security"));

  // async=true: the engine is a sync backend, so the gateway simulates a
  // job (model-registry PRD.md:204-214 sync/async abstraction)
  if (body.at("async").as_bool(false)) {
    auto job = submit_job(sec, body);
    w.respond(202, "application/json", job_json(*job).dump(),
              {{"x-request-id", req.request_id}});
    return;
  }

  // model resolution via model-registry (DESIGN.md:317-346); with a
  // fallback chain, resolution failures are handled per-chain-entry
  hook_pre_call(body);   // may throw request_blocked (DESIGN.md:743-766)
  // tool calling is PASS-THROUGH (reference FR
  // cpt-cf-llm-gateway-fr-tool-calling-v1 / ADR-0002: the gateway
  // converts formats, never executes): tool defs + tool_call/tool_result
  // parts render into the prompt by the worker; `tool_choice:
  // "required"` additionally turns on grammar-constrained JSON decoding
  // so the forced call IS valid JSON.  Media parts stay unsupported.
  if (body.at("tools").is_array())
    for (auto& t : body.at("tools").arr())
      if (t.at("name").as_string().empty() &&
          t.path("function.name").as_string().empty())
        throw Problem{400, "Bad Request", "about:blank",
                      "every tool needs a name", "validation_error"};
  for (auto& m : body.at("messages").arr()) {
    const Json& content = m.at("content");
    if (!content.is_array()) continue;
    for (auto& part : content.arr()) {
      const std::string pt = part.at("type").as_string("text");
      if (pt != "text" && pt != "tool_call" && pt != "tool_result")
        throw Problem{400, "Bad Request", "about:blank",
                      "content part type '" + pt + "' is not supported",
                      "capability_not_supported"};
    }
  }
  const bool has_fb = body.path("fallback.models").is_array();
  Json resolved;
  if (!has_fb) resolved = resolve_model(sec, model);
  check_budget(sec.tenant_id);

  // per-tenant admission (serverless-runtime quota machinery)
  auto adm = hub_->get<AdmissionClient>("serverless-runtime");
  std::string deny = adm ? adm->admit(sec.tenant_id) : "";
  if (!deny.empty())
    throw Problem{429, "Too Many Requests", "about:blank",
                  "tenant quota exceeded", deny};
  struct Release {
    AdmissionClient* a;
    std::string t;
    ~Release() { if (a) a->release(t); }
  } rel{adm.get(), sec.tenant_id};

  if (!worker_ready())
    throw Problem{503, "Service Unavailable", "about:blank",
                  "inference engine is not ready", "provider_error"};

  const std::string rid = "chat-" + std::to_string(req_ctr_.fetch_add(1));

  if (!stream) {
    Json resp = has_fb ? run_chat_with_fallback(sec, body, rid)
                       : run_chat_blocking(body, resolved, rid);
    // post_response hook: generated content can be blocked too
    if (!hook_blocklist_.empty() &&
        resp.at("content").at(0).at("type").as_string("") == "text" &&
        hook_blocks(resp.at("content").at(0).at("text").as_string("")))
      throw Problem{403, "Forbidden", "about:blank",
                    "response blocked by content policy",
                    "response_blocked"};
    record_usage(sec.tenant_id, resp.at("usage"));
    resp.erase("finish_reason");
    w.respond(200, "application/json", resp.dump(),
              {{"x-request-id", req.request_id}});
    return;
  }

  if (has_fb) {
    // streaming + fallback: pick the first RESOLVABLE model up front
    // (mid-stream failover is not possible once bytes are sent)
    std::vector<std::string> chain = {model};
    for (auto& m : body.path("fallback.models").arr())
      chain.push_back(m.as_string());
    bool ok = false;
    for (size_t i = 0; i < chain.size() && !ok; ++i) {
      try {
        resolved = resolve_model(sec, chain[i]);
        ok = true;
      } catch (const Problem&) {}
    }
    if (!ok)
      throw Problem{404, "Not Found", "about:blank",
                    "no model in the fallback chain is available",
                    "model_not_found"};
  }
  m_streams_++;
  Json wreq = Json::object();
  wreq["type"] = "chat";
  wreq["id"] = rid;
  wreq["model"] = resolved.at("provider_model_id").as_string();
  wreq["messages"] = body.at("messages");
  Json params = body.at("params");
  if (params.is_null()) params = Json::object();
  for (const char* f : {"temperature", "top_p", "top_k", "max_tokens",
                        "seed", "ignore_eos", "response_format"})
    if (body.contains(f)) params[f] = body.at(f);
  // reference request schema: `response_schema` asks for structured
  // output — the engine enforces the JSON grammar via constrained
  // decoding (engine/guided.py); the schema body itself is advisory
  if (body.at("response_schema").is_object()) {
    // schema-SHAPED decoding: the engine forces the declared skeleton
    // (guided.SchemaMachine); non-object schemas fall back to plain
    // JSON-grammar enforcement
    params["response_schema"] = body.at("response_schema");
    params["response_format"] = "json";
  } else if (body.contains("response_schema") &&
             !params.contains("response_format")) {
    params["response_format"] = "json";
  }
  if (body.at("tools").is_array() && body.at("tools").size() > 0) {
    wreq["tools"] = body.at("tools");
    if (body.contains("tool_choice"))
      wreq["tool_choice"] = body.at("tool_choice");
    if (body.at("tool_choice").as_string("") == "required")
      // forced call: the engine's ToolCallMachine grammar emits the
      // {"name":..., "arguments":{...}} skeleton byte-exactly
      params["response_format"] = "tool_call";
  }
  wreq["params"] = params;

  // dying-worker race (see run_chat_blocking): obtain the FIRST engine
  // event before committing the SSE response, so a pre-token channel
  // loss can retry on another worker
  std::shared_ptr<MuxClient> mux;
  std::shared_ptr<MuxSink> sink;
  Lease lease;
  std::optional<Json> first_msg;
  for (int attempt = 0; attempt < 3 && !first_msg; ++attempt) {
    if (lease.w) { lease.w->in_flight--; lease.w = nullptr; }
    lease.w = pick_worker();
    if (!lease.w)
      throw Problem{503, "Service Unavailable", "about:blank",
                    "no engine worker ready", "provider_error"};
    mux = ensure_mux(*lease.w);
    if (!mux) {
      lease.w->ready = false;
      continue;
    }
    sink = mux->submit(wreq);
    if (!sink) {
      lease.w->ready = false;
      continue;
    }
    m_submits_++;
    first_msg = MuxClient::next_event(
        *sink, ttft_timeout_ms_ ? (int)ttft_timeout_ms_ : 120000);
    if (!first_msg) {
      bool dead;
      {
        std::lock_guard<std::mutex> sl(sink->mu);
        dead = sink->dead;
      }
      if (!dead && ttft_timeout_ms_) {
        // stop the generation server-side before reporting 504 — the
        // blocking path does the same; otherwise the engine keeps
        // producing until the total-budget guard notices
        mux->abort(rid);
        throw Problem{504, "Gateway Timeout", "about:blank",
                      "no first token within the TTFT budget",
                      "provider_timeout"};
      }
      if (!dead) mux->abort(rid);
      lease.w->ready = false;   // retry
    }
  }
  if (!first_msg)
    throw Problem{503, "Service Unavailable", "about:blank",
                  "engine workers unavailable", "provider_error"};

  // SSE stream per DESIGN.md:289-311: role chunk, delta chunks, final
  // finish_reason+usage chunk, then data: [DONE]
  const auto t0 = std::chrono::steady_clock::now();
  bool first_token = false;
  w.begin_stream(200, "text/event-stream",
                 {{"x-request-id", req.request_id}});
  const std::string canonical = resolved.at("canonical_id").as_string();
  Json role_delta = Json::object();
  role_delta["role"] = "assistant";
  w.write_chunk(sse_chunk(rid, canonical, role_delta));
  bool client_gone = false;
  bool pending_first = true;
  // forced tool call: accumulate the (grammar-guaranteed JSON) text so
  // the FINAL chunk can carry the parsed ToolCall alongside the deltas
  const bool forced_tool =
      body.at("tool_choice").as_string("") == "required" &&
      body.at("tools").is_array() && body.at("tools").size() > 0;
  std::string acc_text;
  while (true) {
    std::optional<Json> msg;
    if (pending_first) {
      msg = first_msg;
      pending_first = false;
    } else {
      msg = MuxClient::next_event(
          *sink, total_timeout_ms_ ? (int)total_timeout_ms_ : 120000);
    }
    if (!msg) {
      mux->abort(rid);              // stop the server-side generation
      w.write_chunk("data: {\"error\":\"provider_timeout\"}\n\n");
      break;
    }
    const std::string ev = msg->at("event").as_string();
    if (ev == "delta") {
      if (!first_token) {
        first_token = true;
        m_ttft_us_sum_ += (uint64_t)
            std::chrono::duration_cast<std::chrono::microseconds>(
                std::chrono::steady_clock::now() - t0).count();
        m_ttft_count_++;
      }
      Json d = Json::object();
      d["content"] = msg->at("text").as_string();
      if (forced_tool) acc_text += msg->at("text").as_string();
      if (!w.write_chunk(sse_chunk(rid, canonical, d))) {
        client_gone = true;    // abort generation server-side
        mux->abort(rid);
        break;
      }
    } else if (ev == "done") {
      record_usage(sec.tenant_id, msg->at("usage"));
      Json final_delta = Json::object();
      std::string finish = msg->at("finish_reason").as_string("stop");
      if (forced_tool) {
        try {
          Json tc = Json::parse(acc_text);
          if (tc.is_object() && !tc.at("name").as_string().empty()) {
            Json call = Json::object();
            call["id"] = "call-" + rid;
            call["name"] = tc.at("name");
            call["arguments"] = tc.at("arguments");
            final_delta["tool_call"] = call;
            finish = "tool_calls";
          }
        } catch (...) {}
      }
      w.write_chunk(sse_chunk(rid, canonical, final_delta, finish,
                              msg->at("usage")));
      break;
    } else if (ev == "error") {
      m_errors_++;
      w.write_chunk("data: {\"error\":\"provider_error\"}\n\n");
      break;
    }
  }
  if (!client_gone) w.write_chunk("data: [DONE]\n\n");
  w.end_stream();
}

// -------------------------------------------------------------- embeddings

void LlmGatewayModule::embeddings_handler(HttpRequest& req,
                                          ResponseWriter& w) {
  m_requests_++;
  Json body;
  try { body = Json::parse(req.body); }
  catch (...) {
    throw Problem{400, "Bad Request", "about:blank", "invalid JSON body",
                  "validation_error"};
  }
  // embedding_request.v1: model + input (string | array of strings)
  const std::string model = body.at("model").as_string();
  const Json& input = body.at("input");
  if (model.empty() || (!input.is_string() && !input.is_array()))
    throw Problem{400, "Bad Request", "about:blank",
                  "'model' and 'input' (string or array) are required",
                  "validation_error"};
  const std::string fmt =
      body.at("encoding_format").as_string("float");
  if (fmt != "float" && fmt != "base64")
    throw Problem{400, "Bad Request", "about:blank",
                  "encoding_format must be float|base64",
                  "validation_error"};

  SecurityContext sec =
      SecurityContext::from_json(req.extensions.at("security"));
  Json resolved = resolve_model(sec, model);
  check_budget(sec.tenant_id);
  if (!worker_ready())
    throw Problem{503, "Service Unavailable", "about:blank",
                  "inference engine is not ready", "provider_error"};

  std::unique_ptr<EngineConn> connp;
  Lease lease{pick_live(connp)};
  if (!lease.w)
    throw Problem{503, "Service Unavailable", "about:blank",
                  "no engine worker ready", "provider_error"};
  EngineConn& conn = *connp;
  Json wreq = Json::object();
  wreq["type"] = "embeddings";
  wreq["input"] = input;
  if (body.contains("dimensions")) wreq["dimensions"] = body.at("dimensions");
  if (!conn.send_json(wreq))
    throw Problem{502, "Bad Gateway", "about:blank", "engine write failed",
                  "provider_error"};
  auto msg = conn.read_json();
  if (!msg)
    throw Problem{504, "Gateway Timeout", "about:blank", "engine timed out",
                  "provider_timeout"};
  if (msg->at("event").as_string() == "error")
    throw Problem{502, "Bad Gateway", "about:blank",
                  msg->at("message").as_string(), "provider_error"};

  record_usage(sec.tenant_id, msg->at("usage"));
  Json data = Json::array();
  const Json& vecs = msg->at("data");
  for (size_t i = 0; i < vecs.size(); ++i) {
    Json item = Json::object();
    item["index"] = (int64_t)i;
    if (fmt == "base64") {
      const Json& v = vecs.at(i);
      std::vector<float> f(v.size());
      for (size_t j = 0; j < v.size(); ++j)
        f[j] = (float)v.at(j).as_number();
      item["embedding"] = b64_encode(
          reinterpret_cast<const unsigned char*>(f.data()),
          f.size() * sizeof(float));
    } else {
      item["embedding"] = vecs.at(i);
    }
    data.push_back(item);
  }
  Json resp = Json::object();
  resp["data"] = data;
  resp["model"] = resolved.at("canonical_id").as_string();
  resp["usage"] = msg->at("usage");
  w.respond(200, "application/json", resp.dump(),
            {{"x-request-id", req.request_id}});
}

// ------------------------------------------------------------------- jobs

Json LlmGatewayModule::job_json(const Job& j) const {
  Json o = Json::object();
  o["id"] = j.id;
  o["status"] = j.status;
  o["created_at"] = j.created_at;
  if (j.finished_at > 0) o["finished_at"] = j.finished_at;
  if (!j.batch_id.empty()) o["batch_id"] = j.batch_id;
  if (j.status == "succeeded") o["result"] = j.result;
  if (j.status == "failed") o["error"] = j.error;
  return o;
}

void LlmGatewayModule::persist_job(const Job& j) {
  if (!jobs_db_) return;
  std::lock_guard<std::mutex> lk(jobs_db_->mu());
  jobs_db_->query(
      "INSERT INTO jobs (tenant_id, id, batch_id, status, request, "
      "result, error, created_at, finished_at) VALUES "
      "(?,?,?,?,?,?,?,?,?) ON CONFLICT(id) DO UPDATE SET "
      "status=excluded.status, result=excluded.result, "
      "error=excluded.error, finished_at=excluded.finished_at",
      {DbValue::S(j.tenant), DbValue::S(j.id), DbValue::S(j.batch_id),
       DbValue::S(j.status), DbValue::S(j.request.dump()),
       DbValue::S(j.result.is_null() ? "" : j.result.dump()),
       DbValue::S(j.error), DbValue::R(j.created_at),
       DbValue::R(j.finished_at)});
}

void LlmGatewayModule::persist_batch(const Batch& b) {
  if (!jobs_db_) return;
  Json ids = Json::array();
  for (auto& id : b.job_ids) ids.push_back(id);
  std::lock_guard<std::mutex> lk(jobs_db_->mu());
  jobs_db_->query(
      "INSERT INTO batches (tenant_id, id, job_ids, created_at) VALUES "
      "(?,?,?,?) ON CONFLICT(id) DO UPDATE SET job_ids=excluded.job_ids",
      {DbValue::S(b.tenant), DbValue::S(b.id), DbValue::S(ids.dump()),
       DbValue::R(b.created_at)});
}

void LlmGatewayModule::load_jobs() {
  // restart recovery: finished jobs restore for result fetches (until
  // TTL pruning), queued/running re-enter the queue (at-least-once)
  std::vector<DbRow> rows, brows;
  {
    std::lock_guard<std::mutex> lk(jobs_db_->mu());
    rows = jobs_db_->query("SELECT * FROM jobs", {});
    brows = jobs_db_->query("SELECT * FROM batches", {});
  }
  uint64_t max_ctr = 0;
  std::lock_guard<std::mutex> lk(jobs_mu_);
  for (auto& r : rows) {
    auto job = std::make_shared<Job>();
    job->id = r.at("id").as_string();
    job->tenant = r.at("tenant_id").as_string();
    job->batch_id = r.at("batch_id").as_string();
    job->status = r.at("status").as_string();
    try { job->request = Json::parse(r.at("request").as_string()); }
    catch (...) {}
    const std::string res = r.at("result").as_string();
    if (!res.empty()) {
      try { job->result = Json::parse(res); } catch (...) {}
    }
    job->error = r.at("error").as_string();
    job->created_at = r.at("created_at").as_number(0);
    job->finished_at = r.at("finished_at").as_number(0);
    if (job->status == "running") job->status = "queued";
    if (job->status == "queued") job_queue_.push_back(job->id);
    jobs_[job->id] = job;
    if (job->id.rfind("job-", 0) == 0)
      max_ctr = std::max(max_ctr,
                         (uint64_t)atoll(job->id.c_str() + 4) + 1);
  }
  for (auto& r : brows) {
    Batch b;
    b.id = r.at("id").as_string();
    b.tenant = r.at("tenant_id").as_string();
    b.created_at = r.at("created_at").as_number(0);
    try {
      // named first: a range-for over parse(...).arr() iterates a dead
      // temporary (caught by the ASan e2e suite, tests/test_host_asan.py)
      Json ids = Json::parse(r.at("job_ids").as_string());
      for (auto& x : ids.arr()) b.job_ids.push_back(x.as_string());
    } catch (...) {}
    if (b.id.rfind("batch-", 0) == 0)
      max_ctr = std::max(max_ctr,
                         (uint64_t)atoll(b.id.c_str() + 6) + 1);
    batches_[b.id] = std::move(b);
  }
  uint64_t cur = req_ctr_.load();
  while (max_ctr > cur && !req_ctr_.compare_exchange_weak(cur, max_ctr))
    ;
  if (!rows.empty())
    LOG_INFO("llm-gateway", "restored %zu job(s), %zu batch(es)",
             rows.size(), brows.size());
}

std::shared_ptr<LlmGatewayModule::Job> LlmGatewayModule::submit_job(
    const SecurityContext& sec, Json body, const std::string& batch_id) {
  auto job = std::make_shared<Job>();
  job->id = "job-" + std::to_string(req_ctr_.fetch_add(1));
  job->tenant = sec.tenant_id;
  job->batch_id = batch_id;
  job->created_at = now_s();
  body.erase("async");
  body.erase("stream");
  job->request = std::move(body);
  {
    std::lock_guard<std::mutex> lk(jobs_mu_);
    jobs_[job->id] = job;
    job_queue_.push_back(job->id);
  }
  persist_job(*job);
  jobs_cv_.notify_one();
  return job;
}

void LlmGatewayModule::job_loop() {
  while (true) {
    std::shared_ptr<Job> job;
    {
      std::unique_lock<std::mutex> lk(jobs_mu_);
      jobs_cv_.wait(lk, [&] { return stopping_ || !job_queue_.empty(); });
      if (stopping_) return;
      // prune expired results so the store stays bounded (job_expired TTL)
      if (job_ttl_s_ > 0) {
        const double now = now_s();
        for (auto it = jobs_.begin(); it != jobs_.end();) {
          if (it->second->finished_at > 0 &&
              now - it->second->finished_at > 2 * (double)job_ttl_s_) {
            if (jobs_db_) {
              std::lock_guard<std::mutex> dlk(jobs_db_->mu());
              jobs_db_->query("DELETE FROM jobs WHERE id=?",
                              {DbValue::S(it->first)});
            }
            it = jobs_.erase(it);
          } else {
            ++it;
          }
        }
      }
      const std::string id = job_queue_.front();
      job_queue_.pop_front();
      auto it = jobs_.find(id);
      if (it == jobs_.end()) continue;
      job = it->second;
      if (job->status != "queued") continue;   // cancelled while queued
      job->status = "running";
    }
    persist_job(*job);
    Json result;
    std::string err;
    try {
      SecurityContext sec;
      sec.tenant_id = job->tenant;
      check_budget(job->tenant);
      if (job->request.path("fallback.models").is_array()) {
        result = run_chat_with_fallback(sec, job->request, job->id);
      } else {
        Json resolved =
            resolve_model(sec, job->request.at("model").as_string());
        result = run_chat_blocking(job->request, resolved, job->id);
      }
    } catch (const Problem& p) {
      err = p.code.empty() ? "provider_error" : p.code;
    } catch (...) {
      err = "provider_error";
    }
    {
      std::lock_guard<std::mutex> lk(jobs_mu_);
      if (err.empty()) {
        job->status = "succeeded";
        result.erase("finish_reason");
        job->result = result;
      } else {
        job->status = "failed";
        job->error = err;
      }
      job->finished_at = now_s();
    }
    persist_job(*job);
    if (err.empty()) record_usage(job->tenant, job->result.at("usage"));
  }
}

// ------------------------------------------------------------------ REST

void LlmGatewayModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  Json req_schema = Json::object();
  req_schema["type"] = "object";
  {
    Json props = Json::object();
    Json s = Json::object(); s["type"] = "string";
    props["model"] = s;
    Json msgs = Json::object();
    msgs["type"] = "array";
    props["messages"] = msgs;
    Json b = Json::object(); b["type"] = "boolean";
    props["stream"] = b;
    props["async"] = b;
    Json n = Json::object(); n["type"] = "number";
    props["temperature"] = n;
    props["top_p"] = n;
    Json i = Json::object(); i["type"] = "integer";
    props["top_k"] = i;
    props["max_tokens"] = i;
    req_schema["properties"] = props;
    Json required = Json::array();
    required.push_back("model");
    required.push_back("messages");
    req_schema["required"] = required;
  }
  auto handler = [this](HttpRequest& rq, ResponseWriter& w) {
    chat_handler(rq, w);
  };
  // canonical module route + unversioned OpenAI-style alias (Appendix B
  // endpoint-paths note)
  for (const char* p : {"/llm-gateway/v1/chat/completions",
                        "/v1/chat/completions"}) {
    OperationSpec op;
    op.method = "POST";
    op.path = p;
    op.operation_id = std::string("chat_completions") +
        (p[1] == 'v' ? "_alias" : "");
    op.summary = "Chat completion (sync, SSE stream, or async job)";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    op.request_schema = req_schema;
    op.responses[200] = "completion response or SSE stream";
    op.responses[202] = "async job accepted";
    op.sse = true;
    if (!license_feature_.empty())
      op.license_features = {license_feature_};
    rest.register_op(op, handler);
  }

  for (const char* p : {"/llm-gateway/v1/embeddings", "/v1/embeddings"}) {
    OperationSpec op;
    op.method = "POST";
    op.path = p;
    op.operation_id = std::string("embeddings") +
        (p[1] == 'v' ? "_alias" : "");
    op.summary = "Embeddings (embedding_request.v1)";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    op.responses[200] = "embedding_response.v1";
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      embeddings_handler(rq, w);
    });
  }

  // ---- jobs (DESIGN.md:262-270 async surface) ----
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/llm-gateway/v1/jobs";
    op.operation_id = "create_job";
    op.summary = "Submit an async chat job";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    op.request_schema = req_schema;
    op.responses[202] = "job.v1";
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      m_requests_++;
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      if (body.at("model").as_string().empty() ||
          !body.at("messages").is_array())
        throw Problem{400, "Bad Request", "about:blank",
                      "'model' and 'messages' are required",
                      "validation_error"};
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      auto job = submit_job(sec, body);
      w.respond(202, "application/json", job_json(*job).dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/llm-gateway/v1/jobs";
    op.operation_id = "list_jobs";
    op.summary = "List this tenant's jobs";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      Json items = Json::array();
      std::lock_guard<std::mutex> lk(jobs_mu_);
      for (auto& [id, j] : jobs_)
        if (j->tenant == sec.tenant_id) items.push_back(job_json(*j));
      Json out = Json::object();
      out["items"] = items;
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/llm-gateway/v1/jobs/{id}";
    op.operation_id = "get_job";
    op.summary = "Job status/result";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      std::lock_guard<std::mutex> lk(jobs_mu_);
      auto it = jobs_.find(rq.path_params.at("id"));
      if (it == jobs_.end() || it->second->tenant != sec.tenant_id)
        throw Problem{404, "Not Found", "about:blank", "no such job",
                      "job_not_found"};
      if (it->second->finished_at > 0 && job_ttl_s_ > 0 &&
          now_s() - it->second->finished_at > job_ttl_s_)
        throw Problem{410, "Gone", "about:blank",
                      "job result expired", "job_expired"};
      w.respond(200, "application/json", job_json(*it->second).dump());
    });
  }
  {
    OperationSpec op;
    op.method = "DELETE";
    op.path = "/llm-gateway/v1/jobs/{id}";
    op.operation_id = "cancel_job";
    op.summary = "Cancel a queued job";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      std::lock_guard<std::mutex> lk(jobs_mu_);
      auto it = jobs_.find(rq.path_params.at("id"));
      if (it == jobs_.end() || it->second->tenant != sec.tenant_id)
        throw Problem{404, "Not Found", "about:blank", "no such job",
                      "job_not_found"};
      if (it->second->status == "queued") {
        it->second->status = "cancelled";
        it->second->finished_at = now_s();
        persist_job(*it->second);
        it->second->finished_at = now_s();
      } else if (it->second->status == "running") {
        // abort propagation: the engine request id IS the job id, so a
        // running job can be cut short server-side (DESIGN timeouts /
        // cancellation machinery)
        for (auto& wk : workers_) {
          if (!wk->ready) continue;
          EngineConn c(wk->socket);
          Json ab = Json::object();
          ab["type"] = "abort";
          ab["id"] = it->first;
          if (c.ok()) c.send_json(ab);
        }
      }
      w.respond(200, "application/json", job_json(*it->second).dump());
    });
  }

  // ---- batches ----
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/llm-gateway/v1/batches";
    op.operation_id = "create_batch";
    op.summary = "Submit a batch of chat requests";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    op.responses[202] = "batch.v1";
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      m_requests_++;
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      const Json& reqs = body.at("requests");
      if (!reqs.is_array() || reqs.size() == 0)
        throw Problem{400, "Bad Request", "about:blank",
                      "'requests' (non-empty array) is required",
                      "validation_error"};
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      Batch b;
      b.id = "batch-" + std::to_string(req_ctr_.fetch_add(1));
      b.tenant = sec.tenant_id;
      b.created_at = now_s();
      for (size_t i = 0; i < reqs.size(); ++i)
        b.job_ids.push_back(submit_job(sec, reqs.at(i), b.id)->id);
      Json out = Json::object();
      out["id"] = b.id;
      out["status"] = "queued";
      out["num_requests"] = (int64_t)b.job_ids.size();
      out["created_at"] = b.created_at;
      persist_batch(b);
      {
        std::lock_guard<std::mutex> lk(jobs_mu_);
        batches_[b.id] = std::move(b);
      }
      w.respond(202, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/llm-gateway/v1/batches/{id}";
    op.operation_id = "get_batch";
    op.summary = "Batch status + per-request results";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      std::lock_guard<std::mutex> lk(jobs_mu_);
      auto it = batches_.find(rq.path_params.at("id"));
      if (it == batches_.end() || it->second.tenant != sec.tenant_id)
        throw Problem{404, "Not Found", "about:blank", "no such batch",
                      "job_not_found"};
      Json items = Json::array();
      size_t done = 0, failed = 0;
      for (auto& id : it->second.job_ids) {
        auto jit = jobs_.find(id);
        if (jit == jobs_.end()) continue;
        items.push_back(job_json(*jit->second));
        const auto& st = jit->second->status;
        if (st == "succeeded") done++;
        else if (st == "failed" || st == "cancelled") failed++;
      }
      Json out = Json::object();
      out["id"] = it->second.id;
      out["created_at"] = it->second.created_at;
      out["num_requests"] = (int64_t)it->second.job_ids.size();
      out["status"] =
          (done + failed == it->second.job_ids.size())
              ? (failed ? "completed_with_errors" : "completed")
              : "in_progress";
      out["jobs"] = items;
      w.respond(200, "application/json", out.dump());
    });
  }

  // ---- usage (PRD.md:224-232 usage tracking) ----
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/llm-gateway/v1/usage";
    op.operation_id = "get_usage";
    op.summary = "This tenant's accumulated usage";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      Json out = Json::object();
      std::lock_guard<std::mutex> lk(usage_mu_);
      auto& u = usage_[sec.tenant_id];
      out["tenant_id"] = sec.tenant_id;
      out["input_tokens"] = (int64_t)u.input_tokens;
      out["output_tokens"] = (int64_t)u.output_tokens;
      out["requests"] = (int64_t)u.requests;
      if (budget_tokens_) {
        out["budget_tokens"] = (int64_t)budget_tokens_;
        out["budget_remaining"] = (int64_t)std::max<int64_t>(
            0, (int64_t)budget_tokens_
                   - (int64_t)(u.input_tokens + u.output_tokens));
      }
      w.respond(200, "application/json", out.dump());
    });
  }

  // ---- Prometheus metrics (the surface the reference
  // lacks; serving KPIs + engine KV occupancy) ----
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/metrics";
    op.operation_id = "metrics";
    op.summary = "Prometheus metrics";
    op.is_public = true;
    op.tags = {"observability"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      std::string t;
      auto add = [&](const char* name, const char* help, const char* type,
                     double v) {
        t += "# HELP " + std::string(name) + " " + help + "\n";
        t += "# TYPE " + std::string(name) + " " + type + "\n";
        t += std::string(name) + " " + std::to_string(v) + "\n";
      };
      add("hyperspot_requests_total", "Gateway requests", "counter",
          (double)m_requests_.load());
      add("hyperspot_sse_streams_total", "SSE streams started", "counter",
          (double)m_streams_.load());
      add("hyperspot_engine_errors_total", "Engine-side errors", "counter",
          (double)m_errors_.load());
      add("hyperspot_worker_restarts_total", "Engine worker respawns",
          "counter", (double)m_worker_restarts_.load());
      add("hyperspot_input_tokens_total", "Prompt tokens", "counter",
          (double)m_input_tokens_.load());
      add("hyperspot_output_tokens_total", "Generated tokens", "counter",
          (double)m_output_tokens_.load());
      if (m_ttft_count_)
        add("hyperspot_ttft_seconds_avg", "Mean stream TTFT", "gauge",
            (double)m_ttft_us_sum_ / 1e6 / (double)m_ttft_count_);
      {
        double running = 0, waiting = 0, total = 0, freeb = 0;
        int live = 0;
        for (auto& wk : workers_) {
          if (!probe_worker(*wk)) continue;
          EngineConn c(wk->socket);
          Json q = Json::object();
          q["type"] = "info";
          if (c.ok() && c.send_json(q)) {
            if (auto r = c.read_json(3000)) {
              live++;
              running += (double)r->at("num_running").as_int(0);
              waiting += (double)r->at("num_waiting").as_int(0);
              total += (double)r->at("kv_blocks_total").as_int(0);
              freeb += (double)r->at("kv_blocks_free").as_int(0);
            }
          }
        }
        add("hyperspot_workers_live", "Ready engine workers", "gauge",
            (double)live);
        if (live) {
          add("hyperspot_engine_running", "Sequences decoding", "gauge",
              running);
          add("hyperspot_engine_waiting", "Sequences queued", "gauge",
              waiting);
          add("hyperspot_kv_blocks_total", "KV pool pages", "gauge",
              total);
          add("hyperspot_kv_blocks_free", "Free KV pages", "gauge", freeb);
          if (total > 0)
            add("hyperspot_kv_occupancy", "KV pool occupancy 0..1",
                "gauge", (total - freeb) / total);
        }
      }
      // per-tenant usage
      {
        std::lock_guard<std::mutex> lk(usage_mu_);
        t += "# HELP hyperspot_tenant_tokens_total Tokens per tenant\n";
        t += "# TYPE hyperspot_tenant_tokens_total counter\n";
        for (auto& [tenant, u] : usage_)
          t += "hyperspot_tenant_tokens_total{tenant=\"" + tenant +
               "\"} " +
               std::to_string(u.input_tokens + u.output_tokens) + "\n";
      }
      w.respond(200, "text/plain; version=0.0.4", t);
    });
  }

  // ---- checkpoint save / live hot-swap (BASELINE config 5:
  // model-registry + file-storage checkpoints; the engine swaps weights
  // in place so the KV pool and captured hipGraphs survive) ----
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/llm-gateway/v1/checkpoints/save";
    op.operation_id = "save_checkpoint";
    op.summary = "Persist current engine weights to file-storage";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      std::string name = body.at("name").as_string("checkpoint");
      auto fs = hub_->get<FileStorageClient>("file-storage");
      if (!fs)
        throw Problem{503, "Service Unavailable", "about:blank",
                      "file-storage unavailable", "provider_error"};
      const std::string path =
          fs->root_for(sec.tenant_id) + "/" + name + ".safetensors";
      if (!worker_ready())
        throw Problem{503, "Service Unavailable", "about:blank",
                      "engine not ready", "provider_error"};
      EngineConn conn(workers_[0]->socket);
      Json wr = Json::object();
      wr["type"] = "save_checkpoint";
      wr["path"] = path;
      if (!conn.ok() || !conn.send_json(wr))
        throw Problem{502, "Bad Gateway", "about:blank",
                      "engine write failed", "provider_error"};
      auto r = conn.read_json(600000);
      if (!r || r->at("event").as_string() != "saved")
        throw Problem{502, "Bad Gateway", "about:blank",
                      r ? r->at("message").as_string() : "save timed out",
                      "provider_error"};
      Json out = Json::object();
      out["path"] = path;
      w.respond(200, "application/json", out.dump());
    });
  }
  {
    OperationSpec op;
    op.method = "POST";
    op.path = "/llm-gateway/v1/checkpoints/swap";
    op.operation_id = "swap_checkpoint";
    op.summary = "Live weight hot-swap (KV pool + graphs survive)";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      const std::string path = body.at("path").as_string();
      if (path.empty()) throw Problem::bad_request("'path' is required");
      if (!worker_ready())
        throw Problem{503, "Service Unavailable", "about:blank",
                      "engine not ready", "provider_error"};
      double secs = 0;
      int swapped = 0;
      for (auto& wk : workers_) {
        if (!probe_worker(*wk)) continue;
        EngineConn conn(wk->socket);
        Json wr = Json::object();
        wr["type"] = "swap";
        wr["checkpoint"] = path;
        if (!conn.ok() || !conn.send_json(wr))
          throw Problem{502, "Bad Gateway", "about:blank",
                        "engine write failed", "provider_error"};
        auto r = conn.read_json(600000);
        if (!r || r->at("event").as_string() != "swapped")
          throw Problem{502, "Bad Gateway", "about:blank",
                        r ? r->at("message").as_string()
                          : "swap timed out",
                        "provider_error"};
        secs = std::max(secs, r->at("seconds").as_number());
        swapped++;
      }
      Json out = Json::object();
      out["checkpoint"] = path;
      out["seconds"] = secs;
      out["workers_swapped"] = (long)swapped;
      w.respond(200, "application/json", out.dump());
    });
  }

  // ---- WS /realtime (DESIGN.md:262-270): JSON messages over RFC-6455.
  // client: {"type":"input_text","text":...,"max_tokens"?:N}
  // server: {"type":"delta","text":...} ... {"type":"done","usage":{...}}
  for (const char* p : {"/llm-gateway/v1/realtime", "/v1/realtime"}) {
    OperationSpec op;
    op.method = "GET";
    op.path = p;
    op.operation_id = std::string("realtime") + (p[1] == 'v' ? "_alias" : "");
    op.summary = "Realtime WebSocket session";
    op.authenticated = true;
    op.tags = {"llm-gateway"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      auto sec = SecurityContext::from_json(rq.extensions.at("security"));
      auto ws = websocket_upgrade(rq, w);
      if (!ws)
        throw Problem{426, "Upgrade Required", "about:blank",
                      "websocket upgrade required", "validation_error"};
      while (true) {
        auto msg = ws->recv_text();
        if (!msg) break;
        Json m;
        try { m = Json::parse(*msg); }
        catch (...) {
          ws->send_text("{\"type\":\"error\",\"code\":"
                        "\"validation_error\"}");
          continue;
        }
        const std::string type = m.at("type").as_string();
        if (type == "session.close") break;
        if (type != "input_text") {
          ws->send_text("{\"type\":\"error\",\"code\":"
                        "\"validation_error\"}");
          continue;
        }
        std::unique_ptr<EngineConn> connp;
        Lease lease{pick_live(connp)};
        if (!lease.w) {
          ws->send_text("{\"type\":\"error\",\"code\":"
                        "\"provider_error\"}");
          continue;
        }
        EngineConn& conn = *connp;
        const std::string rid =
            "rt-" + std::to_string(req_ctr_.fetch_add(1));
        Json wreq = Json::object();
        wreq["type"] = "chat";
        wreq["id"] = rid;
        wreq["model"] = model_;
        Json part = Json::object();
        part["type"] = "text";
        part["text"] = m.at("text").as_string();
        Json content = Json::array();
        content.push_back(part);
        Json um = Json::object();
        um["role"] = "user";
        um["content"] = content;
        Json msgs = Json::array();
        msgs.push_back(um);
        wreq["messages"] = msgs;
        Json params = Json::object();
        if (m.contains("max_tokens")) params["max_tokens"] = m.at("max_tokens");
        if (m.contains("temperature"))
          params["temperature"] = m.at("temperature");
        wreq["params"] = params;
        if (!conn.ok() || !conn.send_json(wreq)) {
          ws->send_text("{\"type\":\"error\",\"code\":"
                        "\"provider_error\"}");
          continue;
        }
        bool gone = false;
        while (true) {
          auto ev = conn.read_json();
          if (!ev) break;
          const std::string e = ev->at("event").as_string();
          if (e == "delta") {
            Json d = Json::object();
            d["type"] = "delta";
            d["text"] = ev->at("text");
            if (!ws->send_text(d.dump())) { gone = true; break; }
          } else if (e == "done") {
            record_usage(sec.tenant_id, ev->at("usage"));
            Json d = Json::object();
            d["type"] = "done";
            d["usage"] = ev->at("usage");
            d["finish_reason"] = ev->at("finish_reason");
            ws->send_text(d.dump());
            break;
          } else if (e == "error") {
            ws->send_text("{\"type\":\"error\",\"code\":"
                          "\"provider_error\"}");
            break;
          }
        }
        if (gone) {
          Json ab = Json::object();
          ab["type"] = "abort";
          ab["id"] = rid;
          conn.send_json(ab);
          break;
        }
      }
      ws->send_close();
    });
  }

  OperationSpec status;
  status.method = "GET";
  status.path = "/llm-gateway/v1/status";
  status.operation_id = "llm_status";
  status.summary = "Engine worker status";
  status.authenticated = true;
  status.tags = {"llm-gateway"};
  rest.register_op(status, [this](HttpRequest& rq, ResponseWriter& w) {
    Json out = Json::object();
    out["model"] = model_;
    out["worker_ready"] = worker_ready();
    out["streams_total"] = (long)m_streams_.load();
    out["submits_total"] = (long)m_submits_.load();
    Json ws_json = Json::array();
    for (auto& wk : workers_) {
      Json wj = Json::object();
      wj["index"] = (long)wk->index;
      wj["device"] = (long)wk->device;
      wj["ready"] = (bool)wk->ready;
      wj["in_flight"] = (long)wk->in_flight.load();
      if (probe_worker(*wk)) {
        EngineConn c(wk->socket);
        Json q = Json::object();
        q["type"] = "info";
        if (c.ok() && c.send_json(q))
          if (auto r = c.read_json(3000)) wj["engine"] = *r;
      }
      ws_json.push_back(wj);
    }
    out["workers"] = ws_json;
    if (ws_json.size() > 0 && ws_json.at(0).contains("engine"))
      out["engine"] = ws_json.at(0).at("engine");
    w.respond(200, "application/json", out.dump());
  });
}

}  // namespace hs
