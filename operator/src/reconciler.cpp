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

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <ctime>
#include <unistd.h>
#include <functional>
#include <map>
#include <vector>

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

  // storage PVC mount (/data, HF_HOME) + configData mount (/config)
  auto storage = spec ? spec->get("storage") : nullptr;
  auto cdata = spec ? spec->get("configData") : nullptr;
  auto vmounts = Value::array();
  auto vols = Value::array();
  if (storage && storage->get_bool("enabled")) {
    auto vm = Value::object();
    vm->set("name", "model-storage");
    vm->set("mountPath", "/data");
    vmounts->push(vm);
    auto v = Value::object();
    v->set("name", "model-storage");
    auto pvc = Value::object();
    pvc->set("claimName", name + "-storage");
    v->set("persistentVolumeClaim", pvc);
    vols->push(v);
    auto env = Value::array();
    auto e = Value::object();
    e->set("name", "HF_HOME");
    e->set("value", "/data");
    env->push(e);
    c->set("env", env);
  }
  if (cdata) {
    auto vm = Value::object();
    vm->set("name", "engine-config");
    vm->set("mountPath", "/config");
    vmounts->push(vm);
    auto v = Value::object();
    v->set("name", "engine-config");
    auto cm = Value::object();
    cm->set("name", name + "-config");
    v->set("configMap", cm);
    vols->push(v);
  }
  if (!vmounts->arr.empty()) c->set("volumeMounts", vmounts);

  std::string app = name + "-engine";
  auto meta = metadata(name + "-engine", ctx.ns, app, hash_str(spec));
  auto d = deployment_skeleton(meta, app, replicas, c);
  if (!vols->arr.empty())
    d->get("spec")->get("template")->get("spec")->set("volumes", vols);
  return d;
}

// PVC for model weights (reference vllmruntime_controller.go:148-200):
// spec.storage { enabled, size, storageClassName } -> <name>-storage PVC,
// mounted at /data with HF_HOME pointing at it.
ValuePtr build_engine_pvc(const Ctx& ctx, const ValuePtr& cr) {
  auto spec = cr->get("spec");
  auto storage = spec ? spec->get("storage") : nullptr;
  if (!storage || !storage->get_bool("enabled")) return nullptr;
  std::string name = cr->get("metadata")->get_str("name");
  auto pvc = Value::object();
  pvc->set("apiVersion", "v1");
  pvc->set("kind", "PersistentVolumeClaim");
  pvc->set("metadata", metadata(name + "-storage", ctx.ns, name + "-engine",
                                hash_str(spec)));
  auto pspec = Value::object();
  auto modes = Value::array();
  modes->push(Value::of("ReadWriteOnce"));
  pspec->set("accessModes", modes);
  auto res = Value::object();
  auto req = Value::object();
  req->set("storage", storage->get_str("size", "50Gi"));
  res->set("requests", req);
  pspec->set("resources", res);
  if (!storage->get_str("storageClassName").empty())
    pspec->set("storageClassName", storage->get_str("storageClassName"));
  pvc->set("spec", pspec);
  return pvc;
}

// ConfigMap carrying spec.configData (reference :1435); mounted at /config.
ValuePtr build_engine_configmap(const Ctx& ctx, const ValuePtr& cr) {
  auto spec = cr->get("spec");
  auto data = spec ? spec->get("configData") : nullptr;
  if (!data) return nullptr;
  std::string name = cr->get("metadata")->get_str("name");
  auto cm = Value::object();
  cm->set("apiVersion", "v1");
  cm->set("kind", "ConfigMap");
  cm->set("metadata", metadata(name + "-config", ctx.ns, name + "-engine",
                               hash_str(spec)));
  cm->set("data", data);
  return cm;
}

// KEDA ScaledObject on the engine deployment (reference :1201-1326):
// spec.autoscaling { enabled, minReplicas, maxReplicas, idleReplicaCount,
// prometheusAddress, threshold } -> prometheus trigger on
// vllm:num_requests_waiting, matching the Helm chart's KEDA block.
ValuePtr build_keda_scaledobject(const Ctx& ctx, const ValuePtr& cr) {
  auto spec = cr->get("spec");
  auto as = spec ? spec->get("autoscaling") : nullptr;
  if (!as || !as->get_bool("enabled")) return nullptr;
  std::string name = cr->get("metadata")->get_str("name");
  auto so = Value::object();
  so->set("apiVersion", "keda.sh/v1alpha1");
  so->set("kind", "ScaledObject");
  so->set("metadata", metadata(name + "-scaler", ctx.ns, name + "-engine",
                               hash_str(spec)));
  auto sspec = Value::object();
  auto target = Value::object();
  target->set("apiVersion", "apps/v1");
  target->set("kind", "Deployment");
  target->set("name", name + "-engine");
  sspec->set("scaleTargetRef", target);
  sspec->set("minReplicaCount", (int)as->get_num("minReplicas", 1));
  sspec->set("maxReplicaCount", (int)as->get_num("maxReplicas", 8));
  if (as->get("idleReplicaCount"))
    sspec->set("idleReplicaCount", (int)as->get_num("idleReplicaCount"));
  sspec->set("pollingInterval", (int)as->get_num("pollingInterval", 15));
  sspec->set("cooldownPeriod", (int)as->get_num("cooldownPeriod", 300));
  auto triggers = Value::array();
  auto trig = Value::object();
  trig->set("type", "prometheus");
  auto tm = Value::object();
  tm->set("serverAddress",
          as->get_str("prometheusAddress",
                      "http://prometheus-operated.monitoring.svc:9090"));
  std::string metric = as->get_str("metric", "vllm:num_requests_waiting");
  tm->set("metricName", metric);
  tm->set("query", "sum(" + metric + "{model_name=\"" + name + "\"})");
  char thr[32];
  snprintf(thr, sizeof(thr), "%g", as->get_num("threshold", 10));
  tm->set("threshold", thr);
  trig->set("metadata", tm);
  triggers->push(trig);
  sspec->set("triggers", triggers);
  so->set("spec", sspec);
  return so;
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

ValuePtr build_router_service(const Ctx& ctx, const ValuePtr& cr) {
  auto crmeta = cr->get("metadata");
  auto spec = cr->get("spec");
  std::string name = crmeta->get_str("name");
  int port = spec ? (int)spec->get_num("port", 8000) : 8000;
  int svc_port = spec ? (int)spec->get_num("servicePort", 80) : 80;
  auto svc = Value::object();
  svc->set("apiVersion", "v1");
  svc->set("kind", "Service");
  svc->set("metadata",
           metadata(name + "-router-service", ctx.ns, name + "-router",
                    hash_str(spec)));
  auto sspec = Value::object();
  auto sel = Value::object();
  sel->set("app", name + "-router");
  sspec->set("selector", sel);
  auto ports = Value::array();
  auto p = Value::object();
  p->set("name", "http");
  p->set("port", svc_port);
  p->set("targetPort", port);
  ports->push(p);
  sspec->set("ports", ports);
  svc->set("spec", sspec);
  return svc;
}

// router SA + namespaced Role (pod read for k8s discovery) + binding —
// reference vllmrouter_controller.go:196-539 creates the same trio
ValuePtr build_router_sa(const Ctx& ctx, const ValuePtr& cr) {
  std::string name = cr->get("metadata")->get_str("name");
  auto sa = Value::object();
  sa->set("apiVersion", "v1");
  sa->set("kind", "ServiceAccount");
  sa->set("metadata", metadata(name + "-router-sa", ctx.ns,
                               name + "-router", ""));
  return sa;
}

ValuePtr build_router_role(const Ctx& ctx, const ValuePtr& cr) {
  std::string name = cr->get("metadata")->get_str("name");
  auto role = Value::object();
  role->set("apiVersion", "rbac.authorization.k8s.io/v1");
  role->set("kind", "Role");
  role->set("metadata", metadata(name + "-router-role", ctx.ns,
                                 name + "-router", ""));
  auto rules = Value::array();
  auto r = Value::object();
  auto groups = Value::array();
  push_arg(groups, "");
  r->set("apiGroups", groups);
  auto res = Value::array();
  push_arg(res, "pods");
  auto spec = cr->get("spec");
  if (spec &&
      spec->get_str("serviceDiscovery") == "k8s_service_name") {
    push_arg(res, "services");
    push_arg(res, "endpoints");
  }
  r->set("resources", res);
  auto verbs = Value::array();
  push_arg(verbs, "get");
  push_arg(verbs, "list");
  push_arg(verbs, "watch");
  r->set("verbs", verbs);
  rules->push(r);
  role->set("rules", rules);
  return role;
}

ValuePtr build_router_rolebinding(const Ctx& ctx, const ValuePtr& cr) {
  std::string name = cr->get("metadata")->get_str("name");
  auto rb = Value::object();
  rb->set("apiVersion", "rbac.authorization.k8s.io/v1");
  rb->set("kind", "RoleBinding");
  rb->set("metadata", metadata(name + "-router-rb", ctx.ns,
                               name + "-router", ""));
  auto subjects = Value::array();
  auto subj = Value::object();
  subj->set("kind", "ServiceAccount");
  subj->set("name", name + "-router-sa");
  subj->set("namespace", ctx.ns);
  subjects->push(subj);
  rb->set("subjects", subjects);
  auto ref = Value::object();
  ref->set("apiGroup", "rbac.authorization.k8s.io");
  ref->set("kind", "Role");
  ref->set("name", name + "-router-role");
  rb->set("roleRef", ref);
  return rb;
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
  d->get("spec")->get("template")->get("spec")
      ->set("serviceAccountName", name + "-router-sa");
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

// engine-pod / status calls must never abort a reconcile pass: a single
// unreachable pod would otherwise starve every later CR in the list
pshttp::Response try_request(const std::string& method,
                             const std::string& url,
                             const std::string& body,
                             const std::string& token,
                             const std::string& ctype =
                                 "application/json") {
  try {
    return pshttp::request(method, url, body, token, ctype);
  } catch (const std::exception& e) {
    fprintf(stderr, "[psoperator] %s %s failed: %s\n", method.c_str(),
            url.c_str(), e.what());
    return pshttp::Response{};
  }
}

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

// scale-subresource status (reference :1159-1199): PUT /status with
// replicas + the label selector HPA needs, mirroring the child
// Deployment's observed state.
void update_runtime_status(const Ctx& ctx, const ValuePtr& cr) {
  std::string name = cr->get("metadata")->get_str("name");
  std::string dep_url = ctx.api_server + "/apis/apps/v1/namespaces/" +
                        ctx.ns + "/deployments/" + name + "-engine";
  auto dep = api_get(ctx, dep_url);
  int replicas = 0, ready = 0;
  if (dep) {
    auto dspec = dep->get("spec");
    auto dstat = dep->get("status");
    if (dspec) replicas = (int)dspec->get_num("replicas", 0);
    if (dstat) ready = (int)dstat->get_num("readyReplicas", 0);
  }
  auto status = Value::object();
  status->set("replicas", replicas);
  status->set("readyReplicas", ready);
  status->set("selector", "app=" + name + "-engine");
  status->set("phase", ready >= replicas && replicas > 0 ? "Ready"
                                                         : "Progressing");
  auto patch = Value::object();
  patch->set("status", status);
  std::string st_url = crd_path(ctx, "vllmruntimes") + "/" + name + "/status";
  try_request("PATCH", st_url, psjson::dump(patch), ctx.token,
              "application/merge-patch+json");
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
  std::string pvc_path = ctx.api_server + "/api/v1/namespaces/" + ctx.ns +
                         "/persistentvolumeclaims";
  std::string cm_path =
      ctx.api_server + "/api/v1/namespaces/" + ctx.ns + "/configmaps";
  std::string so_path = ctx.api_server +
                        "/apis/keda.sh/v1alpha1/namespaces/" + ctx.ns +
                        "/scaledobjects";
  for (auto& cr : items->arr) {
    // reference reconcile order: Service -> PVC -> ConfigMap ->
    // Deployment -> ScaledObject -> status (vllmruntime_controller.go:62)
    if (apply(ctx, svc_path, build_engine_service(ctx, cr))) actions++;
    if (auto pvc = build_engine_pvc(ctx, cr))
      if (apply(ctx, pvc_path, pvc)) actions++;
    if (auto cm = build_engine_configmap(ctx, cr))
      if (apply(ctx, cm_path, cm)) actions++;
    if (apply(ctx, dep_path, build_engine_deployment(ctx, cr))) actions++;
    if (auto so = build_keda_scaledobject(ctx, cr))
      if (apply(ctx, so_path, so)) actions++;
    update_runtime_status(ctx, cr);
  }
  return actions;
}

int reconcile_routers(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "vllmrouters"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  std::string base = ctx.api_server;
  std::string dep_path =
      base + "/apis/apps/v1/namespaces/" + ctx.ns + "/deployments";
  std::string svc_path =
      base + "/api/v1/namespaces/" + ctx.ns + "/services";
  std::string sa_path =
      base + "/api/v1/namespaces/" + ctx.ns + "/serviceaccounts";
  std::string role_path = base +
      "/apis/rbac.authorization.k8s.io/v1/namespaces/" + ctx.ns + "/roles";
  std::string rb_path = base +
      "/apis/rbac.authorization.k8s.io/v1/namespaces/" + ctx.ns +
      "/rolebindings";
  for (auto& cr : items->arr) {
    if (apply(ctx, sa_path, build_router_sa(ctx, cr))) actions++;
    if (apply(ctx, role_path, build_router_role(ctx, cr))) actions++;
    if (apply(ctx, rb_path, build_router_rolebinding(ctx, cr))) actions++;
    if (apply(ctx, svc_path, build_router_service(ctx, cr))) actions++;
    if (apply(ctx, dep_path, build_router_deployment(ctx, cr))) actions++;
  }
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

// LoraAdapter: discover the base model's ready pods, place the adapter per
// spec.loraAdapterDeploymentConfig.algorithm (default | ordered |
// equalized — reference loraadapter_types.go:70-79), register via the
// engine's /v1/load_lora_adapter, track status.loadedAdapters, and clean
// up through a finalizer on deletion (loraadapter_controller.go:889-927).
constexpr const char* kLoraFinalizer = "production-stack.amd.com/lora-cleanup";

bool cr_has_finalizer(const ValuePtr& cr) {
  auto meta = cr->get("metadata");
  auto fins = meta ? meta->get("finalizers") : nullptr;
  if (!fins) return false;
  for (auto& f : fins->arr)
    if (f->str == kLoraFinalizer) return true;
  return false;
}

void patch_finalizers(const Ctx& ctx, const std::string& name, bool add) {
  auto fins = Value::array();
  if (add) fins->push(Value::of(kLoraFinalizer));
  auto meta = Value::object();
  meta->set("finalizers", fins);
  auto patch = Value::object();
  patch->set("metadata", meta);
  try_request("PATCH", crd_path(ctx, "loraadapters") + "/" + name,
              psjson::dump(patch), ctx.token,
              "application/merge-patch+json");
}

int reconcile_loraadapters(const Ctx& ctx) {
  auto list = api_get(ctx, crd_path(ctx, "loraadapters"));
  if (!list) return 0;
  auto items = list->get("items");
  if (!items) return 0;
  int actions = 0;
  // adapters per base model in name order (for ordered/equalized)
  std::map<std::string, std::vector<std::string>> by_base;
  for (auto& cr : items->arr) {
    auto spec = cr->get("spec");
    if (spec)
      by_base[spec->get_str("baseModel")].push_back(
          cr->get("metadata")->get_str("name"));
  }
  for (auto& [b, v] : by_base) std::sort(v.begin(), v.end());

  for (auto& cr : items->arr) {
    auto meta = cr->get("metadata");
    auto spec = cr->get("spec");
    if (!spec) continue;
    std::string cr_name = meta->get_str("name");
    std::string base = spec->get_str("baseModel");
    auto src = spec->get("adapterSource");
    std::string adapter_name =
        src ? src->get_str("adapterName", cr_name) : cr_name;
    std::string adapter_path = src ? src->get_str("adapterPath") : "";
    std::string src_type = src ? src->get_str("type", "local") : "local";
    // non-local sources (s3 | http | huggingface — reference
    // loraadapter_types.go:55-58): delegate the fetch to the downloader
    // sidecar (docker/Dockerfile.sidecar), which writes onto the shared
    // volume the engine pods mount and returns the local path.
    if (src_type != "local" && src) {
      const char* dl_env = std::getenv("PS_OPERATOR_DOWNLOADER_URL");
      std::string dl = dl_env ? dl_env : "http://downloader:8000";
      auto req = Value::object();
      req->set("model_id", src->get_str("repository",
                                        src->get_str("adapterPath")));
      if (!adapter_path.empty()) req->set("target_dir", adapter_path);
      auto r = try_request("POST", dl + "/download", psjson::dump(req),
                           "");
      if (r.ok() && !r.body.empty()) {
        try {
          auto resp = psjson::parse(r.body);
          std::string got = resp ? resp->get_str("path") : "";
          if (!got.empty()) adapter_path = got;
        } catch (const std::exception&) {
          // malformed sidecar reply: keep the declared adapterPath
        }
      } else if (adapter_path.empty()) {
        // nothing to load yet; leave status Pending this pass
        continue;
      }
    }
    auto dcfg = spec->get("loraAdapterDeploymentConfig");
    std::string algo = dcfg ? dcfg->get_str("algorithm", "default")
                            : "default";

    std::string pods_url = ctx.api_server + "/api/v1/namespaces/" + ctx.ns +
                           "/pods?labelSelector=app%3D" + base + "-engine";
    auto pods = api_get(ctx, pods_url);
    auto pitems = pods ? pods->get("items") : nullptr;
    std::vector<std::string> ips;
    if (pitems)
      for (auto& pod : pitems->arr) {
        auto status = pod->get("status");
        std::string ip = status ? status->get_str("podIP") : "";
        if (!ip.empty()) ips.push_back(ip);
      }
    std::sort(ips.begin(), ips.end());

    // deletion: unload everywhere, then drop the finalizer
    if (!meta->get_str("deletionTimestamp").empty()) {
      for (auto& ip : ips) {
        auto body = Value::object();
        body->set("lora_name", adapter_name);
        try_request("POST",
                    "http://" + ip + ":8000/v1/unload_lora_adapter",
                    psjson::dump(body), "");
      }
      patch_finalizers(ctx, cr_name, false);
      actions++;
      continue;
    }
    if (!cr_has_finalizer(cr)) {
      patch_finalizers(ctx, cr_name, true);
      actions++;
    }

    // placement: which pods get this adapter
    std::vector<std::string> targets;
    if (algo == "ordered" && dcfg) {
      // explicit replica index list, e.g. {"replicas": [0, 2]}
      auto reps = dcfg->get("replicas");
      if (reps)
        for (auto& r : reps->arr) {
          size_t idx = (size_t)r->num;
          if (idx < ips.size()) targets.push_back(ips[idx]);
        }
      if (targets.empty() && !ips.empty()) targets.push_back(ips[0]);
    } else if (algo == "equalized" && !ips.empty()) {
      // spread this base model's adapters round-robin over pods: adapter
      // rank r (name order) lands on pod r % n (reference :70-79)
      auto& sibs = by_base[base];
      size_t rank = 0;
      for (size_t i = 0; i < sibs.size(); i++)
        if (sibs[i] == cr_name) rank = i;
      targets.push_back(ips[rank % ips.size()]);
    } else {
      targets = ips;  // default: every pod serves the adapter
    }

    auto loaded = Value::array();
    for (auto& ip : targets) {
      auto body = Value::object();
      body->set("lora_name", adapter_name);
      body->set("lora_path", adapter_path);
      auto r = try_request(
          "POST", "http://" + ip + ":8000/v1/load_lora_adapter",
          psjson::dump(body), "");
      if (r.ok()) {
        actions++;
        loaded->push(Value::of(ip));
      }
    }
    // status.loadedAdapters mirror (reference tracks per-pod registrations)
    auto status = Value::object();
    status->set("loadedAdapters", loaded);
    status->set("phase", loaded->arr.empty() ? "Pending" : "Loaded");
    auto patch = Value::object();
    patch->set("status", status);
    try_request("PATCH",
                crd_path(ctx, "loraadapters") + "/" + cr_name + "/status",
                psjson::dump(patch), ctx.token,
                "application/merge-patch+json");
  }
  return actions;
}

}  // namespace

void watch_or_sleep(const Ctx& ctx, int interval) {
  time_t t0 = time(nullptr);
  std::string url = crd_path(ctx, "vllmruntimes") +
      "?watch=true&timeoutSeconds=" + std::to_string(interval);
  auto r = try_request("GET", url, "", ctx.token);
  int elapsed = (int)(time(nullptr) - t0);
  // a real watch stream delivers {"type":"ADDED"|...} event frames; a
  // server without watch support echoes a plain list immediately —
  // treat only genuine events as a wake-up so we never hot-spin
  bool event = r.status == 200 && !r.body.empty() &&
               r.body.find("\"type\"") != std::string::npos;
  if (event) return;  // change event: reconcile now
  if (elapsed < interval) sleep(interval - elapsed);
}

int reconcile_all(const Ctx& ctx) {
  int n = 0;
  n += reconcile_vllmruntimes(ctx);
  n += reconcile_routers(ctx);
  n += reconcile_cacheservers(ctx);
  n += reconcile_loraadapters(ctx);
  return n;
}

}  // namespace psop
