// OTLP trace export over HTTP/JSON (reference: libs/modkit/src/
// telemetry/init.rs — OTLP exporter with resource attributes and a
// boot-time connectivity probe).  Spans are buffered and flushed by a
// background thread as an ExportTraceServiceRequest JSON document to
// `tracing.otlp_endpoint` (e.g. http://127.0.0.1:4318/v1/traces).
// When unconfigured every call is a no-op (span logs + traceparent
// propagation still happen in the gateway).
#pragma once

#include <atomic>
#include <condition_variable>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../util/json.h"

namespace hs {

struct SpanRecord {
  std::string trace_id;       // 32 hex chars
  std::string span_id;        // 16 hex chars
  std::string parent_span_id; // may be empty
  std::string name;
  uint64_t start_ns = 0;
  uint64_t end_ns = 0;
  int status_code = 0;        // 0 unset, 1 ok, 2 error
  std::vector<std::pair<std::string, std::string>> attrs;
};

class TraceExporter {
 public:
  // cfg: the `tracing` config section
  void configure(const Json& cfg);
  bool enabled() const { return enabled_; }
  void record(SpanRecord span);
  void shutdown();                 // flush + join
  // one-shot connectivity probe (logged, non-fatal)
  bool probe();

  static TraceExporter& instance();

 private:
  void flush_loop();
  void flush(std::vector<SpanRecord> batch);
  std::atomic<bool> enabled_{false};
  std::atomic<bool> stopping_{false};
  std::string host_, path_;
  int port_ = 4318;
  std::string service_name_ = "hyperspot-server";
  int flush_interval_ms_ = 1000;
  size_t batch_max_ = 512;
  std::mutex mu_;
  std::condition_variable cv_;
  std::vector<SpanRecord> buf_;
  std::thread worker_;
};

}  // namespace hs
