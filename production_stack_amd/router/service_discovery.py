"""Service discovery: which serving-engine endpoints exist and what they run.

Behavioural parity: reference src/vllm_router/service_discovery.py —
EndpointInfo/ModelInfo (:53-174), StaticServiceDiscovery with an optional
active health-check thread (:221-408), K8s pod-IP watcher (:411-889, gated
here on the `kubernetes` package being importable), ExternalOnly (:205-218),
initialize/get singletons (:1343-1368).
"""

from __future__ import annotations

import abc
import enum
import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import requests

logger = logging.getLogger("router.discovery")


@dataclass
class ModelInfo:
    id: str
    object: str = "model"
    created: Optional[int] = None
    owned_by: Optional[str] = None
    parent: Optional[str] = None
    is_adapter: bool = False
    # backend fields outside the OpenAI card schema (max_model_len,
    # permissions, ...) survive the round trip — reference
    # ModelInfo preserves extras (test_main_router_models.py)
    extra: Dict[str, Any] = field(default_factory=dict)

    _KNOWN = ("id", "object", "created", "owned_by", "parent")

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "ModelInfo":
        return ModelInfo(
            id=d.get("id", ""),
            object=d.get("object", "model"),
            created=d.get("created"),
            owned_by=d.get("owned_by"),
            parent=d.get("parent"),
            is_adapter=d.get("parent") is not None,
            extra={k: v for k, v in d.items()
                   if k not in ModelInfo._KNOWN},
        )

    def to_dict(self) -> Dict[str, Any]:
        out = {"id": self.id, "object": self.object}
        if self.created is not None:
            out["created"] = self.created
        if self.owned_by is not None:
            out["owned_by"] = self.owned_by
        if self.parent is not None:
            out["parent"] = self.parent
        out.update(self.extra)
        return out


@dataclass
class EndpointInfo:
    url: str
    model_names: List[str] = field(default_factory=list)
    added_timestamp: float = field(default_factory=time.time)
    model_label: Optional[str] = None  # e.g. "prefill" / "decode" pools
    model_info: Dict[str, ModelInfo] = field(default_factory=dict)
    sleep: bool = False
    pod_name: Optional[str] = None

    def __hash__(self) -> int:
        return hash(self.url)


class ServiceDiscoveryType(str, enum.Enum):
    static = "static"
    k8s_pod_ip = "k8s"
    k8s_service_name = "k8s_service_name"
    external = "external"


class ServiceDiscovery(abc.ABC):
    @abc.abstractmethod
    def get_endpoint_info(self) -> List[EndpointInfo]:
        ...

    def get_health(self) -> bool:
        return True

    def get_unhealthy_endpoint_hashes(self) -> List[str]:
        return []

    def close(self) -> None:
        pass


class ExternalOnlyServiceDiscovery(ServiceDiscovery):
    """No local engines; all models served by external providers."""

    def get_endpoint_info(self) -> List[EndpointInfo]:
        return []


class StaticServiceDiscovery(ServiceDiscovery):
    def __init__(
        self,
        urls: List[str],
        models: List[str],
        aliases: Optional[Dict[str, str]] = None,
        model_labels: Optional[List[str]] = None,
        health_check: bool = False,
        health_check_interval: float = 60.0,
        model_types: Optional[List[str]] = None,
        api_key: Optional[str] = None,
        prefill_model_labels: Optional[List[str]] = None,
        decode_model_labels: Optional[List[str]] = None,
    ) -> None:
        assert len(urls) == len(models) or len(models) in (0, 1), (
            "urls and models must align (or one shared model)"
        )
        self.urls = [u.rstrip("/") for u in urls]
        if len(models) == 1 and len(urls) > 1:
            models = models * len(urls)
        self.models = models
        self.aliases = aliases or {}
        self.model_labels = model_labels or [None] * len(urls)
        self.model_types = model_types or ["chat"] * len(urls)
        self.api_key = api_key
        self.prefill_model_labels = prefill_model_labels or []
        self.decode_model_labels = decode_model_labels or []
        self._added = time.time()
        self._unhealthy: set = set()
        self._sleeping: set = set()
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        if health_check:
            self.health_check_interval = health_check_interval
            self._thread = threading.Thread(
                target=self._health_worker, daemon=True
            )
            self._thread.start()

    # -- health checking ----------------------------------------------
    def _probe(self, url: str, model: str, model_type: str) -> bool:
        from production_stack_amd.router.utils import ModelType

        try:
            headers = {}
            if self.api_key:
                headers["Authorization"] = f"Bearer {self.api_key}"
            endpoint = getattr(
                ModelType, model_type, ModelType.chat
            ).value if hasattr(ModelType, model_type) else "/v1/chat/completions"
            r = requests.post(
                url + endpoint,
                json=ModelType.get_test_payload(model_type, model),
                headers=headers,
                timeout=10,
            )
            return r.status_code == 200
        except requests.RequestException:
            return False

    def _health_worker(self) -> None:
        while not self._stop.wait(self.health_check_interval):
            bad = set()
            for url, model, mtype in zip(
                self.urls, self.models, self.model_types
            ):
                if not self._probe(url, model, mtype):
                    bad.add((url, model))
            with self._lock:
                self._unhealthy = bad

    def check_health_once(self) -> None:
        bad = set()
        for url, model, mtype in zip(self.urls, self.models, self.model_types):
            if not self._probe(url, model, mtype):
                bad.add((url, model))
        with self._lock:
            self._unhealthy = bad

    def set_sleep(self, url: str, sleeping: bool) -> None:
        with self._lock:
            if sleeping:
                self._sleeping.add(url)
            else:
                self._sleeping.discard(url)

    # ------------------------------------------------------------------
    def get_endpoint_info(self) -> List[EndpointInfo]:
        with self._lock:
            unhealthy = set(self._unhealthy)
            sleeping = set(self._sleeping)
        out = []
        for url, model, label in zip(
            self.urls, self.models, self.model_labels
        ):
            if (url, model) in unhealthy:
                continue
            names = [model]
            if self.aliases:
                names += [a for a, m in self.aliases.items() if m == model]
            out.append(
                EndpointInfo(
                    url=url,
                    model_names=names,
                    added_timestamp=self._added,
                    model_label=label,
                    sleep=url in sleeping,
                )
            )
        return out

    def get_unhealthy_endpoint_hashes(self) -> List[str]:
        with self._lock:
            return [f"{u}:{m}" for u, m in self._unhealthy]

    def close(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)


def _resolve_tls_verify(
    sa_dir: str, token: str, insecure: bool
):
    """TLS verify policy for K8s API calls. Never silently downgrade to
    verify=False while a Bearer token is being sent (credential exposure
    to a MITM): use the mounted SA CA when present, else the system trust
    store (fails closed on unknown CAs). verify=False only on the explicit
    insecure flag, with a loud warning."""
    import os

    ca = f"{sa_dir}/ca.crt"
    if insecure:
        logger.warning(
            "TLS verification DISABLED for the K8s API "
            "(insecure_skip_tls_verify=true); Bearer credentials are "
            "exposed to any on-path attacker"
        )
        return False
    if os.path.exists(ca):
        return ca
    if token:
        logger.warning(
            "service-account CA bundle %s missing; falling back to the "
            "system trust store (set insecure_skip_tls_verify=true to "
            "skip verification explicitly)", ca,
        )
    return True


class K8sPodIpServiceDiscovery(ServiceDiscovery):
    """Watches pods matching a label selector; endpoint per ready pod IP.

    Implemented over the raw Kubernetes REST API (streaming watch with
    resourceVersion resume) rather than the `kubernetes` client package —
    same approach as the C++ operator — so it runs (and is testable)
    without the client library. In-cluster credentials default to the
    mounted service-account token/CA; `api_base`/`sa_token` override for
    tests or out-of-cluster use. Parity: reference
    service_discovery.py:411-889 (K8sPodIpServiceDiscovery).
    """

    SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

    def __init__(
        self,
        namespace: str = "default",
        port: int = 8000,
        label_selector: Optional[str] = None,
        api_key: Optional[str] = None,
        api_base: Optional[str] = None,
        sa_token: Optional[str] = None,
        probe_models: bool = True,
        insecure_skip_tls_verify: bool = False,
        watcher_timeout_seconds: int = 30,
    ) -> None:
        self.watcher_timeout_seconds = max(int(watcher_timeout_seconds), 1)
        self.namespace = namespace
        self.port = port
        self.label_selector = label_selector
        self.api_key = api_key
        self.probe_models = probe_models
        self.api_base = api_base or "https://kubernetes.default.svc"
        self._token = sa_token
        if self._token is None:
            try:
                with open(f"{self.SA_DIR}/token") as f:
                    self._token = f.read().strip()
            except OSError:
                self._token = ""
        self._verify = _resolve_tls_verify(
            self.SA_DIR, self._token, insecure_skip_tls_verify
        )
        self._endpoints: Dict[str, EndpointInfo] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._resource_version = ""
        self._thread = threading.Thread(target=self._watch_pods, daemon=True)
        self._thread.start()

    # ---- REST helpers -------------------------------------------------
    def _headers(self) -> Dict[str, str]:
        h = {"Accept": "application/json"}
        if self._token:
            h["Authorization"] = f"Bearer {self._token}"
        return h

    def _pods_url(self) -> str:
        url = f"{self.api_base}/api/v1/namespaces/{self.namespace}/pods"
        if self.label_selector:
            from urllib.parse import quote

            url += f"?labelSelector={quote(self.label_selector)}"
        return url

    def _get_model_names(self, ip: str) -> List[str]:
        if not self.probe_models:
            return []
        try:
            headers = {}
            if self.api_key:
                headers["Authorization"] = f"Bearer {self.api_key}"
            r = requests.get(
                f"http://{ip}:{self.port}/v1/models", headers=headers,
                timeout=5
            )
            return [m["id"] for m in r.json().get("data", [])]
        except requests.RequestException:
            return []

    def _build_endpoint(self, pod: dict):
        """Build (name, EndpointInfo|None) for a pod — including the
        blocking /v1/models probe — WITHOUT holding self._lock, so the
        routing hot path (get_endpoint_info) is never stalled behind a
        5 s HTTP probe."""
        meta = pod.get("metadata", {})
        status = pod.get("status", {})
        name = meta.get("name", "")
        ip = status.get("podIP")
        statuses = status.get("containerStatuses") or []
        ready = (
            bool(statuses)
            and all(c.get("ready") for c in statuses)
            and meta.get("deletionTimestamp") is None
        )
        if not ready or not ip:
            return name, None
        labels = meta.get("labels") or {}
        return name, EndpointInfo(
            url=f"http://{ip}:{self.port}",
            model_names=self._get_model_names(ip),
            model_label=labels.get("model"),
            pod_name=name,
            sleep=labels.get("sleeping") == "true",
        )

    def _apply_pod(self, ev_type: str, pod: dict) -> None:
        name, info = self._build_endpoint(pod)  # probe outside the lock
        with self._lock:
            if ev_type == "DELETED" or info is None:
                self._endpoints.pop(name, None)
            else:
                self._endpoints[name] = info

    def _list_once(self) -> None:
        r = requests.get(self._pods_url(), headers=self._headers(),
                         verify=self._verify, timeout=15)
        r.raise_for_status()
        body = r.json()
        self._resource_version = body.get("metadata", {}).get(
            "resourceVersion", ""
        )
        # build the refreshed map off to the side (probes included) and
        # swap atomically — the router must never observe a cleared or
        # half-rebuilt endpoint set mid-refresh
        new: Dict[str, EndpointInfo] = {}
        for pod in body.get("items", []):
            name, info = self._build_endpoint(pod)
            if info is not None:
                new[name] = info
        with self._lock:
            self._endpoints = new

    def _watch_pods(self) -> None:
        import json as _json

        need_list = True
        while not self._stop.is_set():
            try:
                if need_list:
                    self._list_once()
                    need_list = False
                sep = "&" if "?" in self._pods_url() else "?"
                wt = self.watcher_timeout_seconds
                url = (
                    f"{self._pods_url()}{sep}watch=true"
                    f"&resourceVersion={self._resource_version}"
                    f"&timeoutSeconds={wt}"
                )
                with requests.get(
                    url, headers=self._headers(), verify=self._verify,
                    stream=True, timeout=wt + 10,
                ) as r:
                    if r.status_code == 410:  # resourceVersion expired
                        need_list = True
                        continue
                    r.raise_for_status()
                    # chunk_size=1: watch events are sparse; the default
                    # 512-byte buffering would sit on an event until more
                    # bytes arrive or the stream closes
                    for line in r.iter_lines(chunk_size=1):
                        if self._stop.is_set():
                            return
                        if not line:
                            continue
                        ev = _json.loads(line)
                        if ev.get("type") == "ERROR":
                            # typically 410 Gone inside the stream
                            need_list = True
                            break
                        obj = ev.get("object", {})
                        rv = obj.get("metadata", {}).get("resourceVersion")
                        if rv:
                            # continue the watch from here on reconnect
                            # instead of re-listing every ~30 s cycle
                            self._resource_version = rv
                        self._apply_pod(ev.get("type", ""), obj)
            except Exception as e:
                if self._stop.is_set():
                    return
                logger.warning("k8s watch error: %s", e)
                need_list = True
                time.sleep(1)

    def get_endpoint_info(self) -> List[EndpointInfo]:
        with self._lock:
            return list(self._endpoints.values())

    def close(self) -> None:
        self._stop.set()


class K8sServiceNameServiceDiscovery(ServiceDiscovery):
    """Discovers engine Services (not pods) by label selector; endpoints are
    the stable in-cluster service DNS names. Parity: reference
    service_discovery.py:892-1306 (K8sServiceNameServiceDiscovery)."""

    SA_DIR = K8sPodIpServiceDiscovery.SA_DIR

    def __init__(
        self,
        namespace: str = "default",
        port: int = 80,
        label_selector: Optional[str] = None,
        api_key: Optional[str] = None,
        refresh_interval: float = 30.0,
        api_base: Optional[str] = None,
        sa_token: Optional[str] = None,
        probe_models: bool = True,
        insecure_skip_tls_verify: bool = False,
        watcher_timeout_seconds: int = 30,
    ) -> None:
        self.watcher_timeout_seconds = max(int(watcher_timeout_seconds), 1)
        self.namespace = namespace
        self.port = port
        self.label_selector = label_selector
        self.api_key = api_key
        self.refresh_interval = refresh_interval
        self.probe_models = probe_models
        self.api_base = api_base or "https://kubernetes.default.svc"
        self._token = sa_token
        if self._token is None:
            try:
                with open(f"{self.SA_DIR}/token") as f:
                    self._token = f.read().strip()
            except OSError:
                self._token = ""
        self._verify = _resolve_tls_verify(
            self.SA_DIR, self._token, insecure_skip_tls_verify
        )
        self._endpoints: Dict[str, EndpointInfo] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._refresh_loop,
                                        daemon=True)
        self._thread.start()

    def _refresh_once(self) -> None:
        url = (
            f"{self.api_base}/api/v1/namespaces/{self.namespace}/services"
        )
        if self.label_selector:
            from urllib.parse import quote

            url += f"?labelSelector={quote(self.label_selector)}"
        headers = {"Accept": "application/json"}
        if self._token:
            headers["Authorization"] = f"Bearer {self._token}"
        r = requests.get(url, headers=headers, verify=self._verify,
                         timeout=15)
        r.raise_for_status()
        new: Dict[str, EndpointInfo] = {}
        for svc in r.json().get("items", []):
            meta = svc.get("metadata", {})
            spec = svc.get("spec", {})
            name = meta.get("name", "")
            port = self.port
            if spec.get("ports"):
                port = spec["ports"][0].get("port", port)
            svc_url = f"http://{name}.{self.namespace}.svc:{port}"
            models: List[str] = []
            if self.probe_models:
                try:
                    h = {}
                    if self.api_key:
                        h["Authorization"] = f"Bearer {self.api_key}"
                    rr = requests.get(svc_url + "/v1/models", headers=h,
                                      timeout=5)
                    models = [m["id"] for m in rr.json().get("data", [])]
                except requests.RequestException:
                    pass
            labels = meta.get("labels") or {}
            new[name] = EndpointInfo(
                url=svc_url,
                model_names=models,
                model_label=labels.get("model"),
            )
        with self._lock:
            self._endpoints = new

    def _refresh_loop(self) -> None:
        while not self._stop.is_set():
            try:
                self._refresh_once()
            except Exception as e:
                logger.warning("k8s service refresh failed: %s", e)
            self._stop.wait(self.refresh_interval)

    def get_endpoint_info(self) -> List[EndpointInfo]:
        with self._lock:
            return list(self._endpoints.values())

    def close(self) -> None:
        self._stop.set()


_instance: Optional[ServiceDiscovery] = None


def initialize_service_discovery(
    kind: str, **kwargs: Any
) -> ServiceDiscovery:
    global _instance
    if _instance is not None:
        _instance.close()
    if kind in ("static", ServiceDiscoveryType.static):
        _instance = StaticServiceDiscovery(**kwargs)
    elif kind in ("k8s", "k8s_pod_ip", ServiceDiscoveryType.k8s_pod_ip):
        _instance = K8sPodIpServiceDiscovery(**kwargs)
    elif kind in ("k8s_service_name", ServiceDiscoveryType.k8s_service_name):
        _instance = K8sServiceNameServiceDiscovery(**kwargs)
    elif kind in ("external", ServiceDiscoveryType.external):
        _instance = ExternalOnlyServiceDiscovery()
    else:
        raise ValueError(f"unknown service discovery type {kind!r}")
    return _instance


def get_service_discovery() -> ServiceDiscovery:
    if _instance is None:
        raise RuntimeError("service discovery not initialized")
    return _instance


def reset_service_discovery() -> None:
    global _instance
    if _instance is not None:
        _instance.close()
    _instance = None
