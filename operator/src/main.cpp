// psoperator: native reconciler for the production-stack-amd CRDs
// (VLLMRuntime, VLLMRouter, CacheServer, LoraAdapter).
//
// Usage (in cluster):
//   psoperator --namespace default
// (reads the service-account token + API server from the pod environment)
// Usage (tests / out of cluster):
//   psoperator --api-server http://127.0.0.1:9443 --namespace default --once

#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <fstream>
#include <string>

#include "leader.h"
#include "reconciler.h"

static std::string read_file(const std::string& path) {
  std::ifstream f(path);
  if (!f) return "";
  std::string s((std::istreambuf_iterator<char>(f)),
                std::istreambuf_iterator<char>());
  while (!s.empty() && (s.back() == '\n' || s.back() == '\r')) s.pop_back();
  return s;
}

int main(int argc, char** argv) {
  psop::Ctx ctx;
  ctx.ns = "default";
  int interval = 10;
  bool once = false;
  std::string token_file =
      "/var/run/secrets/kubernetes.io/serviceaccount/token";

  bool leader_elect = false;
  int health_port = 0;
  for (int i = 1; i < argc; i++) {
    std::string a = argv[i];
    auto next = [&]() -> std::string {
      return (i + 1 < argc) ? argv[++i] : "";
    };
    if (a == "--api-server") ctx.api_server = next();
    else if (a == "--namespace") ctx.ns = next();
    else if (a == "--token-file") token_file = next();
    else if (a == "--interval") interval = std::stoi(next());
    else if (a == "--once") once = true;
    else if (a == "--leader-elect") leader_elect = true;
    else if (a == "--health-port") health_port = std::stoi(next());
    else if (a == "--help") {
      printf(
          "psoperator [--api-server URL] [--namespace NS] [--token-file F]\n"
          "           [--interval SEC] [--once] [--leader-elect]\n"
          "           [--health-port P]\n");
      return 0;
    }
  }

  if (ctx.api_server.empty()) {
    const char* host = getenv("KUBERNETES_SERVICE_HOST");
    const char* port = getenv("KUBERNETES_SERVICE_PORT");
    if (host && port)
      ctx.api_server =
          "https://" + std::string(host) + ":" + std::string(port);
    else
      ctx.api_server = "https://kubernetes.default.svc";
  }
  ctx.token = read_file(token_file);

  fprintf(stderr,
          "[psoperator] api=%s ns=%s interval=%ds once=%d leader-elect=%d\n",
          ctx.api_server.c_str(), ctx.ns.c_str(), interval, (int)once,
          (int)leader_elect);

  psop::HealthServer health;
  if (health_port > 0) {
    if (health.start(health_port))
      fprintf(stderr, "[psoperator] health/metrics on :%d\n", health_port);
    else
      fprintf(stderr, "[psoperator] health server failed on :%d\n",
              health_port);
  }

  char host[256] = "psoperator";
  gethostname(host, sizeof(host));
  std::string identity =
      std::string(host) + "_" + std::to_string((long)getpid());

  while (true) {
    bool lead = true;
    if (leader_elect) {
      lead = psop::acquire_lease(ctx, identity, 2 * interval + 10);
      health.is_leader = lead;
      if (!lead)
        fprintf(stderr, "[psoperator] standing by (lease held)\n");
    } else {
      health.is_leader = true;
    }
    int actions = 0;
    if (lead) {
      try {
        actions = psop::reconcile_all(ctx);
      } catch (const std::exception& e) {
        fprintf(stderr, "[psoperator] reconcile error: %s\n", e.what());
      }
      health.reconcile_total++;
      health.reconcile_actions_total += actions;
    }
    if (actions)
      fprintf(stderr, "[psoperator] applied %d change(s)\n", actions);
    if (once) break;
    if (lead)
      psop::watch_or_sleep(ctx, interval);  // event-driven when possible
    else
      sleep(interval);
  }
  return 0;
}
