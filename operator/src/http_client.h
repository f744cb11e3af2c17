// Minimal blocking HTTP/1.1 client over POSIX sockets, with optional TLS
// (OpenSSL) for in-cluster https API servers. Supports the verbs the
// operator needs (GET/POST/PUT/PATCH/DELETE) with bearer-token auth.
#pragma once

#include <string>

namespace pshttp {

struct Response {
  int status = 0;
  std::string body;
  bool ok() const { return status >= 200 && status < 300; }
};

struct Url {
  bool tls = false;
  std::string host;
  int port = 80;
  std::string path;
};

Url parse_url(const std::string& url);

// Performs one request. `token` (if non-empty) is sent as a Bearer header.
// TLS peers are verified against the mounted service-account CA
// (PS_OPERATOR_CA_FILE overrides the path); set PS_OPERATOR_TLS_INSECURE=1
// to explicitly skip verification (test clusters only).
Response request(const std::string& method, const std::string& url,
                 const std::string& body, const std::string& token,
                 const std::string& content_type = "application/json",
                 int timeout_sec = 30);

}  // namespace pshttp
