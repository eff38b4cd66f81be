// Native HTTP/1.1 server for the api-gateway module.
//
// The reference host plane serves HTTP with axum/hyper on tokio
// (modules/system/api-gateway); this rebuild is a compact native server:
// accept loop + connection threads, keep-alive, Content-Length bodies,
// chunked streaming responses for SSE (the llm-gateway stream contract,
// reference modules/llm-gateway/docs/DESIGN.md:289-311).
#pragma once

#include <atomic>
#include <functional>
#include <optional>
#include <map>
#include <memory>
#include <string>
#include <thread>
#include <vector>

#include "../util/json.h"

namespace hs {

struct HttpRequest {
  std::string method;
  std::string target;           // raw path?query
  std::string path;
  std::map<std::string, std::string> query;
  std::map<std::string, std::string> headers;   // lower-case keys
  std::map<std::string, std::string> path_params;
  std::string body;
  std::string request_id;
  std::string peer;
  Json extensions = Json::object();   // SecurityContext etc.

  std::string header(const std::string& k, const std::string& dflt = "") const {
    auto it = headers.find(k);
    return it == headers.end() ? dflt : it->second;
  }
};

// Streaming-capable response writer bound to one connection.
class ResponseWriter {
 public:
  explicit ResponseWriter(int fd) : fd_(fd) {}

  // one-shot response
  void respond(int status, const std::string& content_type,
               const std::string& body,
               const std::vector<std::pair<std::string, std::string>>&
                   extra_headers = {});
  // streaming (SSE): chunked transfer encoding
  void begin_stream(int status, const std::string& content_type,
                    const std::vector<std::pair<std::string, std::string>>&
                        extra_headers = {});
  bool write_chunk(const std::string& data);   // false if client went away
  void end_stream();

  bool started() const { return started_; }
  int status() const { return status_; }
  bool keep_alive = true;
  // headers stamped onto every response (request id, traceparent)
  std::vector<std::pair<std::string, std::string>> default_headers;
  int raw_fd() const { return fd_; }
  void mark_started() { started_ = true; }   // used by websocket upgrade

 private:
  bool send_all(const char* p, size_t n);
  int fd_;
  int status_ = 0;
  bool started_ = false;
  bool streaming_ = false;
};

// ------------------------------------------------------------- WebSocket
// Minimal RFC-6455 server side for the llm-gateway /realtime contract:
// handshake (Sec-WebSocket-Accept via SHA-1+base64), text frames, close.
class WsConn {
 public:
  explicit WsConn(int fd) : fd_(fd) {}
  bool send_text(const std::string& payload);
  // one text message (handles masking + fragmentation-free frames);
  // nullopt on close/error/timeout
  std::optional<std::string> recv_text(int timeout_ms = 120000);
  void send_close();

 private:
  int fd_;
  std::string buf_;
};

// Performs the 101 handshake if the request asks for it; returns the
// socket wrapped for frame IO (the HTTP connection loop must stop reusing
// the socket: keep_alive is cleared and the response marked started).
std::optional<WsConn> websocket_upgrade(const HttpRequest& req,
                                        ResponseWriter& w);

using HttpHandler = std::function<void(HttpRequest&, ResponseWriter&)>;

class HttpServer {
 public:
  HttpServer(std::string bind_addr, HttpHandler handler,
             size_t body_limit = 16 * 1024 * 1024);
  ~HttpServer();

  // binds + starts accept thread; returns false on bind failure
  bool start();
  void stop();
  int port() const { return port_; }   // resolved port (0 -> ephemeral)

 private:
  void accept_loop();
  void handle_conn(int fd, std::string peer);

  std::string bind_addr_;
  HttpHandler handler_;
  size_t body_limit_;
  int listen_fd_ = -1;
  int port_ = 0;
  std::atomic<bool> running_{false};
  std::thread accept_thread_;
  std::atomic<int> live_conns_{0};
};

std::string url_decode(const std::string& s);

}  // namespace hs
