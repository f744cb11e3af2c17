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

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "ModelInfo":
        return ModelInfo(
            id=d.get("id", ""),
            object=d.get("object", "model"),
            created=d.get("created"),
            owned_by=d.get("owned_by"),
            parent=d.get("parent"),
            is_adapter=d.get("parent") is not None,
        )


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


class K8sPodIpServiceDiscovery(ServiceDiscovery):
    """Watches pods matching a label selector; endpoint per ready pod IP.

    Requires the `kubernetes` package (present in cluster images; absent in
    the offline build env, in which case construction raises).
    """

    def __init__(
        self,
        namespace: str = "default",
        port: int = 8000,
        label_selector: Optional[str] = None,
        api_key: Optional[str] = None,
    ) -> None:
        try:
            from kubernetes import client, config, watch  # noqa: F401
        except ImportError as e:  # pragma: no cover
            raise RuntimeError(
                "k8s service discovery requires the `kubernetes` package"
            ) from e
        self.namespace = namespace
        self.port = port
        self.label_selector = label_selector
        self.api_key = api_key
        self._endpoints: Dict[str, EndpointInfo] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        config.load_incluster_config()
        self._core = client.CoreV1Api()
        self._watch = watch.Watch()
        self._thread = threading.Thread(target=self._watch_pods, daemon=True)
        self._thread.start()

    def _get_model_names(self, ip: str) -> List[str]:
        try:
            headers = {}
            if self.api_key:
                headers["Authorization"] = f"Bearer {self.api_key}"
            r = requests.get(
                f"http://{ip}:{self.port}/v1/models", headers=headers, timeout=5
            )
            return [m["id"] for m in r.json().get("data", [])]
        except requests.RequestException:
            return []

    def _watch_pods(self) -> None:  # pragma: no cover - needs a cluster
        while not self._stop.is_set():
            try:
                for event in self._watch.stream(
                    self._core.list_namespaced_pod,
                    namespace=self.namespace,
                    label_selector=self.label_selector,
                    timeout_seconds=30,
                ):
                    pod = event["object"]
                    name = pod.metadata.name
                    ip = pod.status.pod_ip
                    ready = (
                        pod.status.container_statuses is not None
                        and all(c.ready for c in pod.status.container_statuses)
                        and pod.metadata.deletion_timestamp is None
                    )
                    url = f"http://{ip}:{self.port}" if ip else None
                    with self._lock:
                        if event["type"] == "DELETED" or not ready or not url:
                            self._endpoints.pop(name, None)
                        else:
                            labels = pod.metadata.labels or {}
                            self._endpoints[name] = EndpointInfo(
                                url=url,
                                model_names=self._get_model_names(ip),
                                model_label=labels.get("model"),
                                pod_name=name,
                            )
            except Exception as e:
                logger.warning("k8s watch error: %s", e)
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

    def __init__(
        self,
        namespace: str = "default",
        port: int = 80,
        label_selector: Optional[str] = None,
        api_key: Optional[str] = None,
        refresh_interval: float = 30.0,
    ) -> None:
        try:
            from kubernetes import client, config  # noqa: F401
        except ImportError as e:  # pragma: no cover
            raise RuntimeError(
                "k8s service discovery requires the `kubernetes` package"
            ) from e
        self.namespace = namespace
        self.port = port
        self.label_selector = label_selector
        self.api_key = api_key
        self.refresh_interval = refresh_interval
        self._endpoints: Dict[str, EndpointInfo] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        config.load_incluster_config()
        self._core = client.CoreV1Api()
        self._thread = threading.Thread(target=self._refresh_loop,
                                        daemon=True)
        self._thread.start()

    def _refresh_once(self) -> None:  # pragma: no cover - needs a cluster
        svcs = self._core.list_namespaced_service(
            namespace=self.namespace, label_selector=self.label_selector
        )
        new: Dict[str, EndpointInfo] = {}
        for svc in svcs.items:
            name = svc.metadata.name
            port = self.port
            if svc.spec.ports:
                port = svc.spec.ports[0].port
            url = f"http://{name}.{self.namespace}.svc:{port}"
            models: List[str] = []
            try:
                headers = {}
                if self.api_key:
                    headers["Authorization"] = f"Bearer {self.api_key}"
                r = requests.get(url + "/v1/models", headers=headers,
                                 timeout=5)
                models = [m["id"] for m in r.json().get("data", [])]
            except requests.RequestException:
                pass
            labels = svc.metadata.labels or {}
            new[name] = EndpointInfo(
                url=url,
                model_names=models,
                model_label=labels.get("model"),
            )
        with self._lock:
            self._endpoints = new

    def _refresh_loop(self) -> None:  # pragma: no cover
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
