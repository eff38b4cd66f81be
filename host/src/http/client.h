// Minimal blocking HTTP/1.1 client for the OAGW data plane
// (reference: reqwest in modules/system/oagw — connect timeout, NO
// redirects [SSRF guard], no total timeout so SSE can stream).
#pragma once

#include <functional>
#include <map>
#include <optional>
#include <string>

namespace hs {

struct ClientResponse {
  int status = 0;
  std::map<std::string, std::string> headers;   // lower-case keys
  std::string body;                             // filled in buffered mode
};

// TLS options for https targets (reference modkit-http rustls layer):
// SNI + hostname verification against a CA bundle; verify=false only
// for explicitly-configured dev upstreams.
struct TlsOpts {
  bool enable = false;
  std::string ca_file;       // PEM bundle; empty = system default paths
  bool verify = true;
};

// on_chunk: called per body chunk in streaming mode; return false to abort.
// on_headers: called once after the status line + headers are parsed (lets
// a proxy begin its own stream before the body arrives — SSE pass-through).
std::optional<ClientResponse> http_request(
    const std::string& host, int port, const std::string& method,
    const std::string& target,
    const std::map<std::string, std::string>& headers,
    const std::string& body, int connect_timeout_ms = 10000,
    const std::function<bool(const char*, size_t)>& on_chunk = nullptr,
    const std::function<void(const ClientResponse&)>& on_headers = nullptr,
    const TlsOpts* tls = nullptr);

}  // namespace hs
