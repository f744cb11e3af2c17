// Reconcilers for the production-stack-amd CRDs.
//
// Capability parity with the reference Go operator
// (reference operator/internal/controller/*.go): VLLMRuntime -> engine
// Deployment + Service with argv assembled from the CR spec + drift
// detection via a spec-hash annotation; VLLMRouter -> router Deployment +
// Service; CacheServer -> KV-controller Deployment + Service; LoraAdapter ->
// POST /v1/load_lora_adapter against the base model's pods with
// default/ordered placement. Implemented natively in C++ (no Go toolchain in
// the build image) against the raw Kubernetes REST API.

#include "reconciler.h"

#include <cstdio>
#include <functional>

#include "http_client.h"

namespace psop {

using psjson::Value;
using psjson::ValuePtr;

uint64_t spec_hash(const ValuePtr& spec) {
  // FNV-1a over the canonical (sorted-key) dump
  std::string s = psjson::dump(spec);
  uint64_t h = 1469598103934665603ull;
  for (unsigned char c : s) {
    h ^= c;
    h *= 1099511628211ull;
  }
  return h;
}

static std::string hash_str(const ValuePtr& spec) {
  char buf[32];
  snprintf(buf, sizeof(buf), "%016llx", (unsigned long long)spec_hash(spec));
  return buf;
}

static ValuePtr metadata(const std::string& name, const std::string& ns,
                         const std::string& app_label,
                         const std::string& hash) {
  auto meta = Value::object();
  meta->set("name", name);
  meta->set("namespace", ns);
  auto labels = Value::object();
  labels->set("app", app_label);
  labels->set("app.kubernetes.io/managed-by", "production-stack-amd-operator");
  meta->set("labels", labels);
  auto ann = Value::object();
  ann->set("production-stack.amd.com/spec-hash", hash);
  meta->set("annotations", ann);
  return meta;
}

static ValuePtr container_base(const std::string& name,
                               const std::string& image, int port) {
  auto c = Value::object();
  c->set("name", name);
  c->set("image", image);
  auto ports = Value::array();
  auto p = Value::object();
  p->set("containerPort", port);
  p->set("name", "http");
  ports->push(p);
  c->set("ports", ports);
  auto probe = Value::object();
  auto httpGet = Value::object();
  httpGet->set("path", "/health");
  httpGet->set("port", port);
  probe->set("httpGet", httpGet);
  probe->set("initialDelaySeconds", 10);
  probe->set("periodSeconds", 10);
  c->set("livenessProbe", probe);
  return c;
}

static ValuePtr deployment_skeleton(const ValuePtr& meta,
                                    const std::string& app_label,
                                    int replicas, ValuePtr container) {
  auto d = Value::object();
  d->set("apiVersion", "apps/v1");
  d->set("kind", "Deployment");
  d->set("metadata", meta);
  auto spec = Value::object();
  spec->set("replicas", replicas);
  auto sel = Value::object();
  auto ml = Value::object();
  ml->set("app", app_label);
  sel->set("matchLabels", ml);
  spec->set("selector", sel);
  auto tmpl = Value::object();
  auto tmeta = Value::object();
  auto tlabels = Value::object();
  tlabels->set("app", app_label);
  tlabels->set("environment", "engine");
  tlabels->set("release", "engine");
  tmeta->set("labels", tlabels);
  tmpl->set("metadata", tmeta);
  auto pspec = Value::object();
  auto containers = Value::array();
  containers->push(container);
  pspec->set("containers", containers);
  tmpl->set("spec", pspec);
  spec->set("template", tmpl);
  d->set("spec", spec);
  return d;
}

static void push_arg(ValuePtr args, const std::string& a) {
  args->push(Value::of(a));
}

// ---------------------------------------------------------------------------
ValuePtr build_engine_deployment(const Ctx& ctx, const ValuePtr& cr) {
  auto crmeta = cr->get("metadata");
  auto spec = cr->get("spec");
  std::string name = crmeta->get_str("name");
  auto model = spec ? spec->get("model") : nullptr;
  auto vconf = spec ? spec->get("vllmConfig") : nullptr;
  auto lmc = spec ? spec->get("lmCacheConfig") : nullptr;
  auto dconf = spec ? spec->get("deploymentConfig") : nullptr;

  std::string image = "production-stack-amd/engine:latest";
  int replicas = 1;
  int port = 8000;
  if (dconf) {
    if (!dconf->get_str("image").empty()) image = dconf->get_str("image");
    replicas = (int)dconf->get_num("replicas", 1);
    port = (int)dconf->get_num("port", 8000);
  }
  std::string model_url =
      model ? model->get_str("modelURL", "llama-3-8b") : "llama-3-8b";

  auto c = container_base("engine", image, port);
  auto cmd = Value::array();
  push_arg(cmd, "python3");
  push_arg(cmd, "-m");
  push_arg(cmd, "production_stack_amd.engine.server");
  c->set("command", cmd);
  auto args = Value::array();
  push_arg(args, model_url);
  push_arg(args, "--host");
  push_arg(args, "0.0.0.0");
  push_arg(args, "--port");
  push_arg(args, std::to_string(port));
  push_arg(args, "--served-model-name");
  push_arg(args, name);
  if (vconf) {
    if (vconf->get("maxModelLen")) {
      push_arg(args, "--max-model-len");
      push_arg(args, std::to_string((int)vconf->get_num("maxModelLen")));
    }
    if (vconf->get("tensorParallelSize")) {
      push_arg(args, "--tensor-parallel-size");
      push_arg(args,
               std::to_string((int)vconf->get_num("tensorParallelSize")));
    }
    if (vconf->get("maxNumSeqs")) {
      push_arg(args, "--max-num-seqs");
      push_arg(args, std::to_string((int)vconf->get_num("maxNumSeqs")));
    }
    if (vconf->get("gpuMemoryUtilization")) {
      push_arg(args, "--gpu-memory-utilization");
      char buf[32];
      snprintf(buf, sizeof(buf), "%.2f",
               vconf->get_num("gpuMemoryUtilization", 0.85));
      push_arg(args, buf);
    }
  }
  if (lmc && lmc->get_bool("enabled")) {
    push_arg(args, "--cpu-offload-gb");
    push_arg(args, std::to_string((int)lmc->get_num(
                       "cpuOffloadingBufferSize", 30)));
    if (!lmc->get_str("controllerURL").empty()) {
      push_arg(args, "--kv-controller-url");
      push_arg(args, lmc->get_str("controllerURL"));
    }
  }
  c->set("args", args);

  // GPU resources (amd.com/gpu)
  auto res = Value::object();
  auto lim = Value::object();
  int gpus = dconf ? (int)dconf->get_num("gpus", 1) : 1;
  lim->set("amd.com/gpu", gpus);
  res->set("limits", lim);
  auto reqs = Value::object();
  reqs->set("amd.com/gpu", gpus);
  res->set("requests", reqs);
  c->set("resources", res);

  std::string app = name + "-engine";
  auto meta = metadata(name + "-engine", ctx.ns, app, hash_str(spec));
  return deployment_skeleton(meta, app, replicas, c);
}

ValuePtr build_engine_service(const Ctx& ctx, const ValuePtr& cr) {
  auto crmeta = cr->get("metadata");
  auto spec = cr->get("spec");
  std::string name = crmeta->get_str("name");
  auto dconf = spec ? spec->get("deploymentConfig") : nullptr;
  int port = dconf ? (int)dconf->get_num("port", 8000) : 8000;
  auto svc = Value::object();
  svc->set("apiVersion", "v1");
  svc->set("kind", "Service");
  svc->set("metadata",
           metadata(name + "-engine-service", ctx.ns, name + "-engine",
                    hash_str(spec)));
  auto sspec = Value::object();
  auto sel = Value::object();
  sel->set("app", name + "-engine");
  sspec->set("selector", sel);
  auto ports = Value::array();
  auto p = Value::object();
  p->set("name", "http");
  p->set("port", port);
  p->set("targetPort", port);
  ports->push(p);
  sspec->set("ports", ports);
  svc->set("spec", sspec);
  return svc;
}

ValuePtr build_router_deployment(const Ctx& ctx, const ValuePtr& cr) {
  auto crmeta = cr->get("metadata");
  auto spec = cr->get("spec");
  std::string name = crmeta->get_str("name");
  std::string image = spec ? spec->get_str("image",
                                           "production-stack-amd/router:"
                                           "latest")
                           : "production-stack-amd/router:latest";
  int port = spec ? (int)spec->get_num("port", 8000) : 8000;
  int replicas = spec ? (int)spec->get_num("replicas", 1) : 1;
  auto c = container_base("router", image, port);
  auto cmd = Value::array();
  push_arg(cmd, "python3");
  push_arg(cmd, "-m");
  push_arg(cmd, "production_stack_amd.router.app");
  c->set("command", cmd);
  auto args = Value::array();
  push_arg(args, "--host");
  push_arg(args, "0.0.0.0");
  push_arg(args, "--port");
  push_arg(args, std::to_string(port));
  push_arg(args, "--service-discovery");
  push_arg(args, spec ? spec->get_str("serviceDiscovery", "k8s") : "k8s");
  push_arg(args, "--routing-logic");
  push_arg(args, spec ? spec->get_str("routingLogic", "roundrobin")
                      : "roundrobin");
  if (spec && !spec->get_str("sessionKey").empty()) {
    push_arg(args, "--session-key");
    push_arg(args, spec->get_str("sessionKey"));
  }
  c->set("args", args);
  std::string app = name + "-router";
  auto meta = metadata(name + "-router", ctx.ns, app, hash_str(spec));
  auto d = deployment_skeleton(meta, app, replicas, c);
  return d;
}

ValuePtr build_cacheserver_deployment(const Ctx& ctx, const ValuePtr& cr) {
  auto crmeta = cr->get("metadata");
  auto spec = cr->get("spec");
  std::string name = crmeta->get_str("name");
  std::string image = spec ? spec->get_str("image",
                                           "production-stack-amd/router:"
                                           "latest")
                           : "production-stack-amd/router:latest";
  int port = spec ? (int)spec->get_num("port", 9000) : 9000;
  auto c = container_base("cacheserver", image, port);
  // the KV controller has no /health HTTP endpoint; drop the probe
  c->obj.erase("livenessProbe");
  auto cmd = Value::array();
  push_arg(cmd, "python3");
  push_arg(cmd, "-m");
  push_arg(cmd, "production_stack_amd.kvpool.controller");
  c->set("command", cmd);
  auto args = Value::array();
  push_arg(args, "--host");
  push_arg(args, "0.0.0.0");
  push_arg(args, "--port");
  push_arg(args, std::to_string(port));
  c->set("args", args);
  std::string app = name + "-cacheserver";
  auto meta = metadata(name + "-cacheserver", ctx.ns, app, hash_str(spec));
  return deployment_skeleton(meta, app, 1, c);
}

// ---------------------------------------------------------------------------
namespace {

std::string crd_path(const Ctx& ctx, const std::string& plural) {
  return ctx.api_server + "/apis/" + ctx.group + "/" + ctx.version +
         "/namespaces/" + ctx.ns + "/" + plural;
}

ValuePtr api_get(const Ctx& ctx, const std::string& url) {
  auto r = pshttp::request("GET", url, "", ctx.token);
  if (r.status == 404) return nullptr;
  if (!r.ok()) return nullptr;
  return psjson::parse(r.body);
}

// create-or-replace keyed on the spec-hash annotation
bool apply(const Ctx& ctx, const std::string& base_path,
           const ValuePtr& obj) {
  std::string name = obj->get("metadata")->get_str("name");
  std::string url = base_path + "/" + name;
  auto existing = api_get(ctx, url);
  std::string want_hash = obj->get("metadata")
                              ->get("annotations")
                              ->get_str("production-stack.amd.com/spec-hash");
  if (existing) {
    auto meta = existing->get("metadata");
    auto ann = meta ? meta->get("annotations") : nullptr;
    std::string have_hash =
        ann ? ann->get_str("production-stack.amd.com/spec-hash") : "";
    if (have_hash == want_hash) return false;  // no drift
    // carry resourceVersion for the replace
    if (meta && meta->get("resourceVersion"))
      obj->get("metadata")->set("resourceVersion",
                                meta->get_str("resourceVersion"));
    auto r = pshttp::request("PUT", url, psjson::dump(obj), ctx.token);
    if (!r.ok())
      fprintf(stderr, "[psoperator] PUT %s -> %d\n", url.c_str(), r.status);
    return r.ok();
  }
  auto r = pshttp::request("POST", base_path, psjson::dump(obj), ctx.token);
  if (!r.ok())
    fprintf(stderr, "[psoperator] POST %s -> %d\n", base_path.c_str(),
            r.status);
  return r.ok();
}

int reconcile_vllmruntimes(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "vllmruntimes"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  std::string dep_path =
      ctx.api_server + "/apis/apps/v1/namespaces/" + ctx.ns + "/deployments";
  std::string svc_path =
      ctx.api_server + "/api/v1/namespaces/" + ctx.ns + "/services";
  for (auto& cr : items->arr) {
    if (apply(ctx, dep_path, build_engine_deployment(ctx, cr))) actions++;
    if (apply(ctx, svc_path, build_engine_service(ctx, cr))) actions++;
  }
  return actions;
}

int reconcile_routers(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "vllmrouters"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  std::string dep_path =
      ctx.api_server + "/apis/apps/v1/namespaces/" + ctx.ns + "/deployments";
  for (auto& cr : items->arr)
    if (apply(ctx, dep_path, build_router_deployment(ctx, cr))) actions++;
  return actions;
}

int reconcile_cacheservers(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "cacheservers"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  std::string dep_path =
      ctx.api_server + "/apis/apps/v1/namespaces/" + ctx.ns + "/deployments";
  for (auto& cr : items->arr)
    if (apply(ctx, dep_path, build_cacheserver_deployment(ctx, cr)))
      actions++;
  return actions;
}

// LoraAdapter: discover the base model's ready pods and register the
// adapter on each via the engine's /v1/load_lora_adapter endpoint
// (reference loraadapter_controller.go:553-592 behaviour, default
// placement = all pods).
int reconcile_loraadapters(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "loraadapters"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  for (auto& cr : items->arr) {
    auto spec = cr->get("spec");
    if (!spec) continue;
    std::string base = spec->get_str("baseModel");
    auto src = spec->get("adapterSource");
    std::string adapter_name =
        src ? src->get_str("adapterName",
                           cr->get("metadata")->get_str("name"))
            : cr->get("metadata")->get_str("name");
    std::string adapter_path = src ? src->get_str("adapterPath") : "";
    std::string pods_url = ctx.api_server + "/api/v1/namespaces/" + ctx.ns +
                           "/pods?labelSelector=app%3D" + base + "-engine";
    auto pods = api_get(ctx, pods_url);
    if (!pods) continue;
    auto pitems = pods->get("items");
    if (!pitems) continue;
    for (auto& pod : pitems->arr) {
      auto status = pod->get("status");
      std::string ip = status ? status->get_str("podIP") : "";
      if (ip.empty()) continue;
      auto body = Value::object();
      body->set("lora_name", adapter_name);
      body->set("lora_path", adapter_path);
      auto r = pshttp::request(
          "POST", "http://" + ip + ":8000/v1/load_lora_adapter",
          psjson::dump(body), "");
      if (r.ok()) actions++;
    }
  }
  return actions;
}

}  // namespace

int reconcile_all(const Ctx& ctx) {
  int n = 0;
  n += reconcile_vllmruntimes(ctx);
  n += reconcile_routers(ctx);
  n += reconcile_cacheservers(ctx);
  n += reconcile_loraadapters(ctx);
  return n;
}

}  // namespace psop
