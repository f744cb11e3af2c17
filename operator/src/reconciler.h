#pragma once

#include <string>

#include "psjson.h"

namespace psop {

struct Ctx {
  std::string api_server;  // e.g. https://kubernetes.default.svc
  std::string ns;
  std::string token;
  std::string group = "production-stack.amd.com";
  std::string version = "v1alpha1";
};

// One reconcile pass over every CR kind. Returns the number of actions
// (creates/updates) performed.
int reconcile_all(const Ctx& ctx);

// Event-driven pacing: block on a watch of the primary CR group for up
// to `interval` seconds; returns immediately when a change event
// arrives (next reconcile picks it up), otherwise sleeps out the
// remainder (spin-safe against apiservers without watch support).
void watch_or_sleep(const Ctx& ctx, int interval);

// Exposed for tests: builders from CR -> desired child objects.
psjson::ValuePtr build_engine_deployment(const Ctx& ctx,
                                         const psjson::ValuePtr& cr);
psjson::ValuePtr build_engine_service(const Ctx& ctx,
                                      const psjson::ValuePtr& cr);
psjson::ValuePtr build_router_deployment(const Ctx& ctx,
                                         const psjson::ValuePtr& cr);
psjson::ValuePtr build_cacheserver_deployment(const Ctx& ctx,
                                              const psjson::ValuePtr& cr);

uint64_t spec_hash(const psjson::ValuePtr& spec);

}  // namespace psop
