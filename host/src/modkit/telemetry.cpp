#include "telemetry.h"

#include "../http/client.h"
#include "../util/log.h"

namespace hs {

namespace {

bool parse_http_url(const std::string& url, std::string* host, int* port,
                    std::string* path) {
  if (url.rfind("http://", 0) != 0) return false;
  std::string rest = url.substr(7);
  size_t slash = rest.find('/');
  std::string hp = slash == std::string::npos ? rest : rest.substr(0, slash);
  *path = slash == std::string::npos ? "/v1/traces" : rest.substr(slash);
  size_t colon = hp.find(':');
  if (colon == std::string::npos) {
    *host = hp;
    *port = 4318;
  } else {
    *host = hp.substr(0, colon);
    *port = atoi(hp.c_str() + colon + 1);
  }
  return !host->empty() && *port > 0;
}

}  // namespace

TraceExporter& TraceExporter::instance() {
  static TraceExporter e;
  return e;
}

void TraceExporter::configure(const Json& cfg) {
  const std::string ep = cfg.at("otlp_endpoint").as_string("");
  if (ep.empty()) return;
  if (!parse_http_url(ep, &host_, &port_, &path_)) {
    LOG_ERROR("telemetry", "bad tracing.otlp_endpoint: %s", ep.c_str());
    return;
  }
  service_name_ = cfg.at("service_name").as_string("hyperspot-server");
  flush_interval_ms_ = (int)cfg.at("flush_interval_ms").as_int(1000);
  batch_max_ = (size_t)cfg.at("batch_max").as_int(512);
  enabled_ = true;
  worker_ = std::thread([this] { flush_loop(); });
  LOG_INFO("telemetry", "OTLP trace export -> http://%s:%d%s",
           host_.c_str(), port_, path_.c_str());
}

bool TraceExporter::probe() {
  if (!enabled_) return false;
  // empty export request: collector reachability check at boot
  // (reference telemetry/init.rs:291 connectivity probe)
  auto r = http_request(host_, port_, "POST", path_,
                        {{"content-type", "application/json"}},
                        "{\"resourceSpans\":[]}", 3000);
  const bool ok = r && r->status >= 200 && r->status < 300;
  LOG_INFO("telemetry", "OTLP probe %s (status %d)",
           ok ? "ok" : "FAILED", r ? r->status : 0);
  return ok;
}

void TraceExporter::record(SpanRecord span) {
  if (!enabled_) return;
  std::lock_guard<std::mutex> lk(mu_);
  if (buf_.size() >= batch_max_ * 4) return;   // backpressure: drop
  buf_.push_back(std::move(span));
  if (buf_.size() >= batch_max_) cv_.notify_one();
}

void TraceExporter::shutdown() {
  if (!enabled_) return;
  stopping_ = true;
  cv_.notify_all();
  if (worker_.joinable()) worker_.join();
  enabled_ = false;
}

void TraceExporter::flush_loop() {
  while (true) {
    std::vector<SpanRecord> batch;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(flush_interval_ms_));
      batch.swap(buf_);
      if (batch.empty() && stopping_) return;
    }
    if (!batch.empty()) flush(std::move(batch));
    if (stopping_) {
      std::lock_guard<std::mutex> lk(mu_);
      if (buf_.empty()) return;
    }
  }
}

void TraceExporter::flush(std::vector<SpanRecord> batch) {
  // OTLP/HTTP JSON ExportTraceServiceRequest
  Json spans = Json::array();
  for (auto& s : batch) {
    Json sp = Json::object();
    sp["traceId"] = s.trace_id;
    sp["spanId"] = s.span_id;
    if (!s.parent_span_id.empty()) sp["parentSpanId"] = s.parent_span_id;
    sp["name"] = s.name;
    sp["kind"] = 2;                       // SPAN_KIND_SERVER
    sp["startTimeUnixNano"] = std::to_string(s.start_ns);
    sp["endTimeUnixNano"] = std::to_string(s.end_ns);
    Json attrs = Json::array();
    for (auto& [k, v] : s.attrs) {
      Json a = Json::object();
      a["key"] = k;
      Json val = Json::object();
      val["stringValue"] = v;
      a["value"] = val;
      attrs.push_back(a);
    }
    sp["attributes"] = attrs;
    Json st = Json::object();
    st["code"] = (long)s.status_code;
    sp["status"] = st;
    spans.push_back(sp);
  }
  Json scope = Json::object();
  scope["name"] = "hyperspot.modkit";
  Json scope_spans = Json::object();
  scope_spans["scope"] = scope;
  scope_spans["spans"] = spans;
  Json res_attr = Json::object();
  res_attr["key"] = "service.name";
  Json res_val = Json::object();
  res_val["stringValue"] = service_name_;
  res_attr["value"] = res_val;
  Json resource = Json::object();
  Json res_attrs = Json::array();
  res_attrs.push_back(res_attr);
  resource["attributes"] = res_attrs;
  Json rs = Json::object();
  rs["resource"] = resource;
  Json ss_arr = Json::array();
  ss_arr.push_back(scope_spans);
  rs["scopeSpans"] = ss_arr;
  Json root = Json::object();
  Json rs_arr = Json::array();
  rs_arr.push_back(rs);
  root["resourceSpans"] = rs_arr;
  auto r = http_request(host_, port_, "POST", path_,
                        {{"content-type", "application/json"}},
                        root.dump(), 3000);
  if (!r || r->status < 200 || r->status >= 300)
    LOG_DEBUG("telemetry", "OTLP export failed (status %d, %zu spans)",
              r ? r->status : 0, batch.size());
}

}  // namespace hs
