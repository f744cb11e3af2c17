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
// `ca_insecure` skips certificate verification (in-cluster CA handling is
// out of scope for this build; production deployments mount the CA and use
// verify mode).
Response request(const std::string& method, const std::string& url,
                 const std::string& body, const std::string& token,
                 const std::string& content_type = "application/json",
                 int timeout_sec = 30);

}  // namespace pshttp
