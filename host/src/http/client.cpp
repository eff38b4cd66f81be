#include "client.h"

#include <openssl/err.h>
#include <openssl/ssl.h>

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <mutex>
#include <sstream>

namespace hs {

namespace {

int connect_to(const std::string& host, int port, int timeout_ms) {
  addrinfo hints{}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  if (getaddrinfo(host.c_str(), std::to_string(port).c_str(), &hints,
                  &res) != 0 || !res)
    return -1;
  int fd = socket(res->ai_family, SOCK_STREAM, 0);
  if (fd < 0) { freeaddrinfo(res); return -1; }
  // non-blocking connect with timeout
  int fl = fcntl(fd, F_GETFL);
  fcntl(fd, F_SETFL, fl | O_NONBLOCK);
  int rc = connect(fd, res->ai_addr, res->ai_addrlen);
  freeaddrinfo(res);
  if (rc != 0 && errno == EINPROGRESS) {
    pollfd pf{fd, POLLOUT, 0};
    if (poll(&pf, 1, timeout_ms) <= 0) { close(fd); return -1; }
    int err = 0;
    socklen_t len = sizeof err;
    getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &len);
    if (err != 0) { close(fd); return -1; }
  } else if (rc != 0) {
    close(fd);
    return -1;
  }
  fcntl(fd, F_SETFL, fl);
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
  return fd;
}

// plain-or-TLS connection IO
struct Io {
  int fd = -1;
  SSL* ssl = nullptr;
  SSL_CTX* ctx = nullptr;

  ssize_t read(char* p, size_t n) {
    if (ssl) return (ssize_t)SSL_read(ssl, p, (int)n);
    return recv(fd, p, n, 0);
  }
  bool write_all(const char* p, size_t n) {
    while (n) {
      ssize_t w = ssl ? (ssize_t)SSL_write(ssl, p, (int)n)
                      : ::send(fd, p, n, MSG_NOSIGNAL);
      if (w <= 0) return false;
      p += w;
      n -= size_t(w);
    }
    return true;
  }
  void close_all() {
    if (ssl) {
      SSL_shutdown(ssl);
      SSL_free(ssl);
      ssl = nullptr;
    }
    if (ctx) {
      SSL_CTX_free(ctx);
      ctx = nullptr;
    }
    if (fd >= 0) {
      close(fd);
      fd = -1;
    }
  }
};

bool tls_connect(Io& io, const std::string& host, const TlsOpts& opts) {
  static std::once_flag once;
  std::call_once(once, [] {
    SSL_library_init();
    SSL_load_error_strings();
  });
  io.ctx = SSL_CTX_new(TLS_client_method());
  if (!io.ctx) return false;
  if (opts.verify) {
    SSL_CTX_set_verify(io.ctx, SSL_VERIFY_PEER, nullptr);
    if (!opts.ca_file.empty()) {
      if (SSL_CTX_load_verify_locations(io.ctx, opts.ca_file.c_str(),
                                        nullptr) != 1)
        return false;
    } else {
      SSL_CTX_set_default_verify_paths(io.ctx);
    }
  }
  io.ssl = SSL_new(io.ctx);
  if (!io.ssl) return false;
  SSL_set_fd(io.ssl, io.fd);
  SSL_set_tlsext_host_name(io.ssl, host.c_str());   // SNI
  if (opts.verify) SSL_set1_host(io.ssl, host.c_str());
  if (SSL_connect(io.ssl) != 1) return false;
  return true;
}

}  // namespace

std::optional<ClientResponse> http_request(
    const std::string& host, int port, const std::string& method,
    const std::string& target,
    const std::map<std::string, std::string>& headers,
    const std::string& body, int connect_timeout_ms,
    const std::function<bool(const char*, size_t)>& on_chunk,
    const std::function<void(const ClientResponse&)>& on_headers,
    const TlsOpts* tls) {
  Io io;
  io.fd = connect_to(host, port, connect_timeout_ms);
  if (io.fd < 0) return std::nullopt;
  if (tls && tls->enable && !tls_connect(io, host, *tls)) {
    io.close_all();
    return std::nullopt;
  }
  std::ostringstream req;
  req << method << " " << target << " HTTP/1.1\r\n"
      << "host: " << host << ":" << port << "\r\n"
      << "connection: close\r\n";
  bool has_cl = false;
  for (auto& [k, v] : headers) {
    if (k == "host" || k == "connection" || k == "content-length") continue;
    req << k << ": " << v << "\r\n";
  }
  (void)has_cl;
  if (!body.empty() || method == "POST" || method == "PUT" ||
      method == "PATCH")
    req << "content-length: " << body.size() << "\r\n";
  req << "\r\n";
  std::string head = req.str();
  if (!io.write_all(head.data(), head.size()) ||
      !io.write_all(body.data(), body.size())) {
    io.close_all();
    return std::nullopt;
  }
  // read response head
  std::string buf;
  char tmp[16384];
  size_t hdr_end;
  while ((hdr_end = buf.find("\r\n\r\n")) == std::string::npos) {
    ssize_t r = io.read(tmp, sizeof tmp);
    if (r <= 0) { io.close_all(); return std::nullopt; }
    buf.append(tmp, size_t(r));
    if (buf.size() > 256 * 1024) { io.close_all(); return std::nullopt; }
  }
  ClientResponse resp;
  {
    std::string headpart = buf.substr(0, hdr_end);
    size_t le = headpart.find("\r\n");
    std::string status_line = headpart.substr(0, le);
    size_t sp = status_line.find(' ');
    resp.status = sp == std::string::npos
        ? 0 : atoi(status_line.c_str() + sp + 1);
    size_t pos = le + 2;
    while (pos < headpart.size()) {
      size_t e = headpart.find("\r\n", pos);
      if (e == std::string::npos) e = headpart.size();
      std::string line = headpart.substr(pos, e - pos);
      size_t c = line.find(':');
      if (c != std::string::npos) {
        std::string k = line.substr(0, c);
        for (auto& ch : k) ch = char(tolower(ch));
        size_t vs = line.find_first_not_of(' ', c + 1);
        resp.headers[k] = vs == std::string::npos ? "" : line.substr(vs);
      }
      pos = e + 2;
    }
  }
  if (on_headers) on_headers(resp);
  std::string rest = buf.substr(hdr_end + 4);
  const bool chunked =
      resp.headers.count("transfer-encoding") &&
      resp.headers["transfer-encoding"].find("chunked") != std::string::npos;
  long content_len = resp.headers.count("content-length")
      ? atol(resp.headers["content-length"].c_str()) : -1;

  auto emit = [&](const char* p, size_t n) -> bool {
    if (on_chunk) return on_chunk(p, n);
    resp.body.append(p, n);
    return true;
  };

  if (chunked) {
    // de-chunk
    std::string acc = rest;
    while (true) {
      size_t nl = acc.find("\r\n");
      while (nl == std::string::npos) {
        ssize_t r = io.read(tmp, sizeof tmp);
        if (r <= 0) { io.close_all(); return resp; }
        acc.append(tmp, size_t(r));
        nl = acc.find("\r\n");
      }
      long sz = strtol(acc.c_str(), nullptr, 16);
      if (sz == 0) break;
      size_t need = nl + 2 + size_t(sz) + 2;
      while (acc.size() < need) {
        ssize_t r = io.read(tmp, sizeof tmp);
        if (r <= 0) { io.close_all(); return resp; }
        acc.append(tmp, size_t(r));
      }
      if (!emit(acc.data() + nl + 2, size_t(sz))) break;
      acc.erase(0, need);
    }
  } else {
    if (!rest.empty()) emit(rest.data(), rest.size());
    long got = (long)rest.size();
    while (content_len < 0 || got < content_len) {
      ssize_t r = io.read(tmp, sizeof tmp);
      if (r <= 0) break;
      got += r;
      if (!emit(tmp, size_t(r))) break;
    }
  }
  io.close_all();
  return resp;
}

}  // namespace hs
