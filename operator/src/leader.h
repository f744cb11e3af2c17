// Lease-based leader election + health/metrics endpoints for psoperator
// (parity with the reference manager: controller-runtime leader election
// and :8081 healthz / :8080 metrics, operator/cmd/main.go).
#pragma once

#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cstdio>
#include <cstring>
#include <ctime>
#include <string>
#include <thread>

#include "http_client.h"
#include "psjson.h"
#include "reconciler.h"

namespace psop {

inline std::string now_rfc3339() {
  char buf[64];
  time_t t = time(nullptr);
  struct tm tmv;
  gmtime_r(&t, &tmv);
  strftime(buf, sizeof(buf), "%Y-%m-%dT%H:%M:%S.000000Z", &tmv);
  return buf;
}

// Acquire or renew the coordination.k8s.io Lease. Returns true when this
// process holds the lease (and may reconcile).
inline bool acquire_lease(const Ctx& ctx, const std::string& identity,
                          int duration_sec) {
  using psjson::Value;
  std::string url = ctx.api_server +
                    "/apis/coordination.k8s.io/v1/namespaces/" + ctx.ns +
                    "/leases/production-stack-amd-operator";
  auto mk = [&](const std::string& rv) {
    auto lease = Value::object();
    lease->set("apiVersion", "coordination.k8s.io/v1");
    lease->set("kind", "Lease");
    auto meta = Value::object();
    meta->set("name", "production-stack-amd-operator");
    meta->set("namespace", ctx.ns);
    if (!rv.empty()) meta->set("resourceVersion", rv);
    lease->set("metadata", meta);
    auto spec = Value::object();
    spec->set("holderIdentity", identity);
    spec->set("leaseDurationSeconds", duration_sec);
    spec->set("renewTime", now_rfc3339());
    lease->set("spec", spec);
    return lease;
  };
  try {
    auto r = pshttp::request("GET", url, "", ctx.token);
    if (r.status == 404) {
      std::string base = ctx.api_server +
                         "/apis/coordination.k8s.io/v1/namespaces/" +
                         ctx.ns + "/leases";
      auto c = pshttp::request("POST", base, psjson::dump(mk("")),
                               ctx.token);
      return c.ok();
    }
    if (!r.ok()) return false;
    auto lease = psjson::parse(r.body);
    auto spec = lease ? lease->get("spec") : nullptr;
    std::string holder = spec ? spec->get_str("holderIdentity") : "";
    std::string renew = spec ? spec->get_str("renewTime") : "";
    if (holder != identity && !holder.empty()) {
      // another holder: take over only when its renewTime went stale
      struct tm tmv;
      memset(&tmv, 0, sizeof(tmv));
      if (strptime(renew.c_str(), "%Y-%m-%dT%H:%M:%S", &tmv) != nullptr) {
        time_t then = timegm(&tmv);
        if (time(nullptr) - then <
            (time_t)spec->get_num("leaseDurationSeconds", duration_sec))
          return false;  // still fresh — stand by
      }
    }
    std::string rv =
        lease->get("metadata")
            ? lease->get("metadata")->get_str("resourceVersion")
            : "";
    auto u = pshttp::request("PUT", url, psjson::dump(mk(rv)), ctx.token);
    return u.ok();
  } catch (const std::exception&) {
    return false;
  }
}

// Minimal /healthz + /readyz + /metrics HTTP server (ref main.go's
// health/metrics endpoints). Serves prometheus counters.
struct HealthServer {
  std::atomic<long> reconcile_total{0};
  std::atomic<long> reconcile_actions_total{0};
  std::atomic<bool> is_leader{false};
  std::atomic<bool> ready{false};
  int fd = -1;
  std::thread thr;

  bool start(int port) {
    fd = socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) return false;
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons((uint16_t)port);
    if (bind(fd, (sockaddr*)&addr, sizeof(addr)) != 0 ||
        listen(fd, 16) != 0) {
      close(fd);
      fd = -1;
      return false;
    }
    ready = true;
    thr = std::thread([this] { serve(); });
    thr.detach();
    return true;
  }

  void serve() {
    char buf[2048];
    while (true) {
      int c = accept(fd, nullptr, nullptr);
      if (c < 0) break;
      ssize_t n = recv(c, buf, sizeof(buf) - 1, 0);
      std::string req(buf, n > 0 ? (size_t)n : 0);
      std::string body, ctype = "text/plain";
      int status = 200;
      if (req.rfind("GET /healthz", 0) == 0 ||
          req.rfind("GET /readyz", 0) == 0) {
        body = "ok";
      } else if (req.rfind("GET /metrics", 0) == 0) {
        char m[512];
        snprintf(m, sizeof(m),
                 "# TYPE psoperator_reconcile_total counter\n"
                 "psoperator_reconcile_total %ld\n"
                 "# TYPE psoperator_reconcile_actions_total counter\n"
                 "psoperator_reconcile_actions_total %ld\n"
                 "# TYPE psoperator_leader gauge\n"
                 "psoperator_leader %d\n",
                 reconcile_total.load(), reconcile_actions_total.load(),
                 is_leader.load() ? 1 : 0);
        body = m;
      } else {
        status = 404;
        body = "not found";
      }
      char hdr[256];
      snprintf(hdr, sizeof(hdr),
               "HTTP/1.1 %d %s\r\nContent-Type: %s\r\n"
               "Content-Length: %zu\r\nConnection: close\r\n\r\n",
               status, status == 200 ? "OK" : "Not Found", ctype.c_str(),
               body.size());
      std::string resp = std::string(hdr) + body;
      send(c, resp.data(), resp.size(), 0);
      close(c);
    }
  }
};

}  // namespace psop
