#include "http.h"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/sha.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <sstream>

#include "../util/log.h"

namespace hs {

namespace {

const char* status_text(int code) {
  switch (code) {
    case 200: return "OK";
    case 201: return "Created";
    case 204: return "No Content";
    case 400: return "Bad Request";
    case 401: return "Unauthorized";
    case 403: return "Forbidden";
    case 404: return "Not Found";
    case 405: return "Method Not Allowed";
    case 409: return "Conflict";
    case 413: return "Payload Too Large";
    case 415: return "Unsupported Media Type";
    case 422: return "Unprocessable Entity";
    case 429: return "Too Many Requests";
    case 500: return "Internal Server Error";
    case 502: return "Bad Gateway";
    case 503: return "Service Unavailable";
    case 504: return "Gateway Timeout";
    default: return "Unknown";
  }
}

}  // namespace

std::string url_decode(const std::string& s) {
  std::string out;
  for (size_t i = 0; i < s.size(); ++i) {
    if (s[i] == '%' && i + 2 < s.size()) {
      auto hex = [](char c) -> int {
        if (c >= '0' && c <= '9') return c - '0';
        if (c >= 'a' && c <= 'f') return c - 'a' + 10;
        if (c >= 'A' && c <= 'F') return c - 'A' + 10;
        return -1;
      };
      int h = hex(s[i + 1]), l = hex(s[i + 2]);
      if (h >= 0 && l >= 0) {
        out += char(h * 16 + l);
        i += 2;
        continue;
      }
    }
    out += s[i] == '+' ? ' ' : s[i];
  }
  return out;
}

bool ResponseWriter::send_all(const char* p, size_t n) {
  while (n > 0) {
    ssize_t w = ::send(fd_, p, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    p += w;
    n -= size_t(w);
  }
  return true;
}

void ResponseWriter::respond(
    int status, const std::string& content_type, const std::string& body,
    const std::vector<std::pair<std::string, std::string>>& extra) {
  if (started_) return;
  started_ = true;
  status_ = status;
  std::ostringstream h;
  h << "HTTP/1.1 " << status << " " << status_text(status) << "\r\n"
    << "content-type: " << content_type << "\r\n"
    << "content-length: " << body.size() << "\r\n";
  for (auto& [k, v] : default_headers) {
    bool dup = false;
    for (auto& [k2, v2] : extra) if (k2 == k) dup = true;
    if (!dup) h << k << ": " << v << "\r\n";
  }
  for (auto& [k, v] : extra) h << k << ": " << v << "\r\n";
  h << "connection: " << (keep_alive ? "keep-alive" : "close") << "\r\n\r\n";
  std::string head = h.str();
  send_all(head.data(), head.size());
  send_all(body.data(), body.size());
}

void ResponseWriter::begin_stream(
    int status, const std::string& content_type,
    const std::vector<std::pair<std::string, std::string>>& extra) {
  if (started_) return;
  started_ = true;
  status_ = status;
  streaming_ = true;
  std::ostringstream h;
  h << "HTTP/1.1 " << status << " " << status_text(status) << "\r\n"
    << "content-type: " << content_type << "\r\n"
    << "cache-control: no-cache\r\n"
    << "transfer-encoding: chunked\r\n";
  for (auto& [k, v] : default_headers) {
    bool dup = false;
    for (auto& [k2, v2] : extra) if (k2 == k) dup = true;
    if (!dup) h << k << ": " << v << "\r\n";
  }
  for (auto& [k, v] : extra) h << k << ": " << v << "\r\n";
  h << "connection: close\r\n\r\n";
  keep_alive = false;
  std::string head = h.str();
  send_all(head.data(), head.size());
}

bool ResponseWriter::write_chunk(const std::string& data) {
  if (!streaming_ || data.empty()) return true;
  char sz[16];
  int n = snprintf(sz, sizeof sz, "%zx\r\n", data.size());
  if (!send_all(sz, size_t(n))) return false;
  if (!send_all(data.data(), data.size())) return false;
  return send_all("\r\n", 2);
}

void ResponseWriter::end_stream() {
  if (streaming_) send_all("0\r\n\r\n", 5);
  streaming_ = false;
}

HttpServer::HttpServer(std::string bind_addr, HttpHandler handler,
                       size_t body_limit)
    : bind_addr_(std::move(bind_addr)), handler_(std::move(handler)),
      body_limit_(body_limit) {}

HttpServer::~HttpServer() { stop(); }

bool HttpServer::start() {
  std::string host = bind_addr_;
  int port = 80;
  auto colon = bind_addr_.rfind(':');
  if (colon != std::string::npos) {
    host = bind_addr_.substr(0, colon);
    port = atoi(bind_addr_.c_str() + colon + 1);
  }
  listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
  if (listen_fd_ < 0) return false;
  int one = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(uint16_t(port));
  if (host.empty() || host == "0.0.0.0") addr.sin_addr.s_addr = INADDR_ANY;
  else if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1)
    addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
  if (bind(listen_fd_, (sockaddr*)&addr, sizeof addr) != 0) {
    LOG_ERROR("api-gateway", "bind %s failed: %s", bind_addr_.c_str(),
              strerror(errno));
    close(listen_fd_);
    listen_fd_ = -1;
    return false;
  }
  socklen_t alen = sizeof addr;
  getsockname(listen_fd_, (sockaddr*)&addr, &alen);
  port_ = ntohs(addr.sin_port);
  if (listen(listen_fd_, 512) != 0) return false;
  running_ = true;
  accept_thread_ = std::thread([this] { accept_loop(); });
  return true;
}

void HttpServer::stop() {
  if (!running_.exchange(false)) return;
  if (listen_fd_ >= 0) { shutdown(listen_fd_, SHUT_RDWR); close(listen_fd_); }
  if (accept_thread_.joinable()) accept_thread_.join();
  // give in-flight connection threads a moment
  for (int i = 0; i < 100 && live_conns_ > 0; ++i) usleep(10000);
}

void HttpServer::accept_loop() {
  while (running_) {
    sockaddr_in peer{};
    socklen_t plen = sizeof peer;
    int fd = accept(listen_fd_, (sockaddr*)&peer, &plen);
    if (fd < 0) {
      if (!running_) break;
      continue;
    }
    char ip[64];
    inet_ntop(AF_INET, &peer.sin_addr, ip, sizeof ip);
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    struct timeval tv{300, 0};       // idle read timeout
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
    ++live_conns_;
    std::thread([this, fd, ipstr = std::string(ip)] {
      handle_conn(fd, ipstr);
      close(fd);
      --live_conns_;
    }).detach();
  }
}

void HttpServer::handle_conn(int fd, std::string peer) {
  std::string buf;
  char tmp[16384];
  while (running_) {
    // ---- read headers ----
    size_t hdr_end;
    while ((hdr_end = buf.find("\r\n\r\n")) == std::string::npos) {
      if (buf.size() > 64 * 1024) return;     // header flood
      ssize_t r = recv(fd, tmp, sizeof tmp, 0);
      if (r <= 0) return;
      buf.append(tmp, size_t(r));
    }
    HttpRequest req;
    req.peer = peer;
    {
      std::string head = buf.substr(0, hdr_end);
      size_t line_end = head.find("\r\n");
      std::string rl = head.substr(0, line_end);
      size_t sp1 = rl.find(' ');
      size_t sp2 = rl.find(' ', sp1 + 1);
      if (sp1 == std::string::npos || sp2 == std::string::npos) return;
      req.method = rl.substr(0, sp1);
      req.target = rl.substr(sp1 + 1, sp2 - sp1 - 1);
      size_t q = req.target.find('?');
      req.path = url_decode(req.target.substr(0, q));
      if (q != std::string::npos) {
        std::string qs = req.target.substr(q + 1);
        size_t s = 0;
        while (s <= qs.size()) {
          size_t amp = qs.find('&', s);
          std::string kv = qs.substr(
              s, amp == std::string::npos ? std::string::npos : amp - s);
          size_t eq = kv.find('=');
          if (eq != std::string::npos)
            req.query[url_decode(kv.substr(0, eq))] =
                url_decode(kv.substr(eq + 1));
          else if (!kv.empty())
            req.query[url_decode(kv)] = "";
          if (amp == std::string::npos) break;
          s = amp + 1;
        }
      }
      size_t pos = line_end + 2;
      while (pos < head.size()) {
        size_t e = head.find("\r\n", pos);
        if (e == std::string::npos) e = head.size();
        std::string line = head.substr(pos, e - pos);
        size_t c = line.find(':');
        if (c != std::string::npos) {
          std::string k = line.substr(0, c);
          for (auto& ch : k) ch = char(tolower(ch));
          size_t vs = line.find_first_not_of(' ', c + 1);
          req.headers[k] = vs == std::string::npos ? "" : line.substr(vs);
        }
        pos = e + 2;
      }
    }
    // ---- read body ----
    size_t clen = 0;
    auto it = req.headers.find("content-length");
    if (it != req.headers.end()) clen = strtoul(it->second.c_str(), nullptr, 10);
    ResponseWriter w(fd);
    if (clen > body_limit_) {
      w.keep_alive = false;
      w.respond(413, "application/problem+json",
                "{\"type\":\"about:blank\",\"title\":\"Payload Too Large\","
                "\"status\":413}");
      return;
    }
    size_t body_start = hdr_end + 4;
    while (buf.size() - body_start < clen) {
      ssize_t r = recv(fd, tmp, sizeof tmp, 0);
      if (r <= 0) return;
      buf.append(tmp, size_t(r));
    }
    req.body = buf.substr(body_start, clen);
    buf.erase(0, body_start + clen);

    bool ka = req.header("connection") != "close";
    w.keep_alive = ka;
    try {
      handler_(req, w);
    } catch (const std::exception& e) {
      LOG_ERROR("api-gateway", "handler exception: %s", e.what());
      if (!w.started())
        w.respond(500, "application/problem+json",
                  "{\"type\":\"about:blank\",\"title\":\"Internal Server "
                  "Error\",\"status\":500}");
    }
    if (!w.started())
      w.respond(404, "application/problem+json",
                "{\"type\":\"about:blank\",\"title\":\"Not Found\","
                "\"status\":404}");
    if (!w.keep_alive) return;
  }
}



// ------------------------------------------------------------- WebSocket

bool WsConn::send_text(const std::string& p) {
  std::string hdr;
  hdr.push_back((char)0x81);           // FIN + text
  if (p.size() < 126) {
    hdr.push_back((char)p.size());
  } else if (p.size() < 65536) {
    hdr.push_back(126);
    hdr.push_back((char)(p.size() >> 8));
    hdr.push_back((char)(p.size() & 0xff));
  } else {
    hdr.push_back(127);
    for (int i = 7; i >= 0; --i)
      hdr.push_back((char)((p.size() >> (8 * i)) & 0xff));
  }
  std::string f = hdr + p;
  const char* d = f.data();
  size_t n = f.size();
  while (n) {
    ssize_t w = send(fd_, d, n, MSG_NOSIGNAL);
    if (w <= 0) return false;
    d += w;
    n -= (size_t)w;
  }
  return true;
}

void WsConn::send_close() {
  const char f[2] = {(char)0x88, 0};
  send(fd_, f, 2, MSG_NOSIGNAL);
}

std::optional<std::string> WsConn::recv_text(int timeout_ms) {
  while (true) {
    // try to parse one frame from buf_
    if (buf_.size() >= 2) {
      const unsigned char b0 = buf_[0], b1 = buf_[1];
      const int opcode = b0 & 0x0f;
      const bool masked = b1 & 0x80;
      size_t len = b1 & 0x7f;
      size_t off = 2;
      if (len == 126) {
        if (buf_.size() >= 4) {
          len = ((unsigned char)buf_[2] << 8) | (unsigned char)buf_[3];
          off = 4;
        } else len = SIZE_MAX;
      } else if (len == 127) {
        if (buf_.size() >= 10) {
          len = 0;
          for (int i = 0; i < 8; ++i)
            len = (len << 8) | (unsigned char)buf_[2 + i];
          off = 10;
        } else len = SIZE_MAX;
      }
      if (len != SIZE_MAX) {
        // RFC 6455: clients MUST mask; also cap the declared payload
        // before computing `need` so an attacker-controlled 64-bit
        // length can neither wrap size_t nor balloon buf_
        static const size_t kMaxFrame = 1 << 20;  // 1 MiB
        if (!masked || len > kMaxFrame) {
          send_close();
          return std::nullopt;
        }
        const size_t need = off + (masked ? 4 : 0) + len;
        if (buf_.size() >= need) {
          std::string payload = buf_.substr(off + (masked ? 4 : 0), len);
          if (masked) {
            const unsigned char* mk =
                (const unsigned char*)buf_.data() + off;
            for (size_t i = 0; i < payload.size(); ++i)
              payload[i] = (char)(payload[i] ^ mk[i & 3]);
          }
          buf_.erase(0, need);
          if (opcode == 0x8) return std::nullopt;          // close
          if (opcode == 0x9) {                             // ping -> pong
            std::string pong;
            pong.push_back((char)0x8a);
            pong.push_back((char)payload.size());
            pong += payload;
            send(fd_, pong.data(), pong.size(), MSG_NOSIGNAL);
            continue;
          }
          if (opcode == 0x1 || opcode == 0x2) return payload;
          continue;                                        // pong etc.
        }
      }
    }
    struct pollfd pf{fd_, POLLIN, 0};
    if (poll(&pf, 1, timeout_ms) <= 0) return std::nullopt;
    char tmp[8192];
    ssize_t r = recv(fd_, tmp, sizeof tmp, 0);
    if (r <= 0) return std::nullopt;
    buf_.append(tmp, (size_t)r);
  }
}

std::optional<WsConn> websocket_upgrade(const HttpRequest& req,
                                        ResponseWriter& w) {
  if (req.header("upgrade").find("websocket") == std::string::npos)
    return std::nullopt;
  const std::string key = req.header("sec-websocket-key");
  if (key.empty()) return std::nullopt;
  const std::string magic = key + "258EAFA5-E914-47DA-95CA-C5AB0DC85B11";
  unsigned char sha[20];
  SHA1((const unsigned char*)magic.data(), magic.size(), sha);
  static const char* tbl =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  std::string acc;
  for (int i = 0; i < 20; i += 3) {
    unsigned v = sha[i] << 16;
    if (i + 1 < 20) v |= sha[i + 1] << 8;
    if (i + 2 < 20) v |= sha[i + 2];
    acc.push_back(tbl[(v >> 18) & 63]);
    acc.push_back(tbl[(v >> 12) & 63]);
    acc.push_back(i + 1 < 20 ? tbl[(v >> 6) & 63] : '=');
    acc.push_back(i + 2 < 20 ? tbl[v & 63] : '=');
  }
  std::string resp =
      "HTTP/1.1 101 Switching Protocols\r\n"
      "upgrade: websocket\r\nconnection: Upgrade\r\n"
      "sec-websocket-accept: " + acc + "\r\n\r\n";
  ssize_t sr = send(w.raw_fd(), resp.data(), resp.size(), MSG_NOSIGNAL);
  if (sr != (ssize_t)resp.size()) return std::nullopt;
  w.mark_started();
  w.keep_alive = false;
  return WsConn(w.raw_fd());
}

}  // namespace hs
