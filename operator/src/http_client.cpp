#include "http_client.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/err.h>
#include <openssl/ssl.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdlib>
#include <cstring>
#include <sstream>
#include <stdexcept>

namespace pshttp {

Url parse_url(const std::string& url) {
  Url u;
  std::string rest;
  if (url.rfind("https://", 0) == 0) {
    u.tls = true;
    u.port = 443;
    rest = url.substr(8);
  } else if (url.rfind("http://", 0) == 0) {
    u.tls = false;
    u.port = 80;
    rest = url.substr(7);
  } else {
    throw std::runtime_error("unsupported url scheme: " + url);
  }
  auto slash = rest.find('/');
  std::string hostport = slash == std::string::npos ? rest
                                                    : rest.substr(0, slash);
  u.path = slash == std::string::npos ? "/" : rest.substr(slash);
  auto colon = hostport.rfind(':');
  if (colon != std::string::npos) {
    u.host = hostport.substr(0, colon);
    u.port = std::stoi(hostport.substr(colon + 1));
  } else {
    u.host = hostport;
  }
  return u;
}

namespace {

int tcp_connect(const std::string& host, int port, int timeout_sec) {
  struct addrinfo hints;
  memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  std::string port_s = std::to_string(port);
  if (getaddrinfo(host.c_str(), port_s.c_str(), &hints, &res) != 0 || !res)
    throw std::runtime_error("getaddrinfo failed for " + host);
  int fd = -1;
  for (auto* p = res; p; p = p->ai_next) {
    fd = socket(p->ai_family, p->ai_socktype, p->ai_protocol);
    if (fd < 0) continue;
    struct timeval tv = {timeout_sec, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    if (connect(fd, p->ai_addr, p->ai_addrlen) == 0) break;
    close(fd);
    fd = -1;
  }
  freeaddrinfo(res);
  if (fd < 0) throw std::runtime_error("connect failed: " + host);
  return fd;
}

struct Conn {
  int fd = -1;
  SSL_CTX* ctx = nullptr;
  SSL* ssl = nullptr;

  ~Conn() {
    if (ssl) {
      SSL_shutdown(ssl);
      SSL_free(ssl);
    }
    if (ctx) SSL_CTX_free(ctx);
    if (fd >= 0) close(fd);
  }

  ssize_t write_all(const char* buf, size_t n) {
    size_t sent = 0;
    while (sent < n) {
      ssize_t r = ssl ? SSL_write(ssl, buf + sent, (int)(n - sent))
                      : ::send(fd, buf + sent, n - sent, 0);
      if (r <= 0) return -1;
      sent += (size_t)r;
    }
    return (ssize_t)sent;
  }

  ssize_t read_some(char* buf, size_t n) {
    return ssl ? SSL_read(ssl, buf, (int)n) : ::recv(fd, buf, n, 0);
  }
};

}  // namespace

Response request(const std::string& method, const std::string& url,
                 const std::string& body, const std::string& token,
                 const std::string& content_type, int timeout_sec) {
  Url u = parse_url(url);
  Conn c;
  c.fd = tcp_connect(u.host, u.port, timeout_sec);
  if (u.tls) {
    static bool init = [] {
      SSL_library_init();
      SSL_load_error_strings();
      return true;
    }();
    (void)init;
    c.ctx = SSL_CTX_new(TLS_client_method());
    // Verify the peer against the mounted service-account CA by default —
    // a Bearer token travels on this connection, so SSL_VERIFY_NONE would
    // hand credentials to any MITM. Opt-out only via the explicit
    // PS_OPERATOR_TLS_INSECURE=1 escape hatch.
    const char* insecure_env = getenv("PS_OPERATOR_TLS_INSECURE");
    bool insecure = insecure_env && std::string(insecure_env) == "1";
    if (!insecure) {
      const char* ca = getenv("PS_OPERATOR_CA_FILE");
      std::string ca_file = ca && *ca
          ? ca
          : "/var/run/secrets/kubernetes.io/serviceaccount/ca.crt";
      int loaded = SSL_CTX_load_verify_locations(c.ctx, ca_file.c_str(),
                                                 nullptr);
      if (!loaded) loaded = SSL_CTX_set_default_verify_paths(c.ctx);
      if (!loaded)
        throw std::runtime_error(
            "no CA bundle loadable (" + ca_file +
            "); refusing unverified TLS with a Bearer token. "
            "Set PS_OPERATOR_TLS_INSECURE=1 to override.");
      SSL_CTX_set_verify(c.ctx, SSL_VERIFY_PEER, nullptr);
    } else {
      SSL_CTX_set_verify(c.ctx, SSL_VERIFY_NONE, nullptr);
    }
    c.ssl = SSL_new(c.ctx);
    SSL_set_fd(c.ssl, c.fd);
    SSL_set_tlsext_host_name(c.ssl, u.host.c_str());
    if (!insecure) SSL_set1_host(c.ssl, u.host.c_str());
    if (SSL_connect(c.ssl) != 1)
      throw std::runtime_error("TLS handshake failed: " + u.host);
  }

  std::ostringstream req;
  req << method << " " << u.path << " HTTP/1.1\r\n"
      << "Host: " << u.host << "\r\n"
      << "Connection: close\r\n"
      << "Accept: application/json\r\n";
  if (!token.empty()) req << "Authorization: Bearer " << token << "\r\n";
  if (!body.empty())
    req << "Content-Type: " << content_type << "\r\n"
        << "Content-Length: " << body.size() << "\r\n";
  req << "\r\n" << body;
  std::string payload = req.str();
  if (c.write_all(payload.data(), payload.size()) < 0)
    throw std::runtime_error("send failed");

  std::string raw;
  char buf[8192];
  while (true) {
    ssize_t r = c.read_some(buf, sizeof(buf));
    if (r <= 0) break;
    raw.append(buf, (size_t)r);
  }
  Response resp;
  auto hdr_end = raw.find("\r\n\r\n");
  if (hdr_end == std::string::npos)
    throw std::runtime_error("malformed HTTP response");
  std::string headers = raw.substr(0, hdr_end);
  std::string rest = raw.substr(hdr_end + 4);
  {
    auto sp = headers.find(' ');
    resp.status = std::stoi(headers.substr(sp + 1, 3));
  }
  // chunked transfer decoding
  bool chunked = headers.find("chunked") != std::string::npos;
  if (chunked) {
    size_t pos = 0;
    while (pos < rest.size()) {
      auto nl = rest.find("\r\n", pos);
      if (nl == std::string::npos) break;
      long len = strtol(rest.substr(pos, nl - pos).c_str(), nullptr, 16);
      if (len <= 0) break;
      resp.body.append(rest, nl + 2, (size_t)len);
      pos = nl + 2 + (size_t)len + 2;
    }
  } else {
    resp.body = rest;
  }
  return resp;
}

}  // namespace pshttp
