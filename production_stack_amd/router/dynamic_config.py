"""Dynamic router config: poll a YAML/JSON file and hot-reconfigure.

Parity: reference dynamic_config.py (DynamicRouterConfig :43-122, watcher
thread :263-287, reconfigure_all :243-251, surfaced in /health).
"""

from __future__ import annotations

import json
import logging
import threading
from dataclasses import dataclass
from typing import Any, Dict, Optional

import yaml

logger = logging.getLogger("router.dynamic_config")


@dataclass
class DynamicRouterConfig:
    service_discovery: Optional[str] = None
    static_backends: Optional[str] = None
    static_models: Optional[str] = None
    static_aliases: Optional[str] = None
    routing_logic: Optional[str] = None
    session_key: Optional[str] = None
    prefix_min_match_length: Optional[int] = None

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "DynamicRouterConfig":
        known = {
            k: d[k]
            for k in DynamicRouterConfig.__dataclass_fields__
            if k in d
        }
        return DynamicRouterConfig(**known)


class DynamicConfigWatcher:
    def __init__(
        self,
        path: str,
        interval: float = 10.0,
        app=None,
        start: bool = True,
    ) -> None:
        self.path = path
        self.interval = interval
        self.app = app
        self._last: Optional[str] = None
        self.current: Optional[DynamicRouterConfig] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.poll_once()
        if start:
            self._thread = threading.Thread(target=self._worker, daemon=True)
            self._thread.start()

    def _load(self) -> Optional[Dict[str, Any]]:
        try:
            with open(self.path) as f:
                raw = f.read()
        except OSError:
            return None
        if raw == self._last:
            return None
        self._last = raw
        try:
            if self.path.endswith((".yaml", ".yml")):
                d = yaml.safe_load(raw)
            else:
                d = json.loads(raw)
            return _flatten_structured(d)
        except (ValueError, yaml.YAMLError) as e:
            logger.error("invalid dynamic config: %s", e)
            return None

    def poll_once(self) -> bool:
        d = self._load()
        if d is None:
            return False
        cfg = DynamicRouterConfig.from_dict(d)
        self.reconfigure_all(cfg)
        self.current = cfg
        return True

    def reconfigure_all(self, cfg: DynamicRouterConfig) -> None:
        from production_stack_amd.router import routing_logic as rl
        from production_stack_amd.router import service_discovery as sd
        from production_stack_amd.router.utils import (
            parse_static_aliases,
            parse_static_model_names,
            parse_static_urls,
        )

        if cfg.service_discovery == "static" and cfg.static_backends:
            sd.initialize_service_discovery(
                "static",
                urls=parse_static_urls(cfg.static_backends),
                models=parse_static_model_names(cfg.static_models or ""),
                aliases=parse_static_aliases(cfg.static_aliases),
            )
            logger.info("dynamic config: reinitialized static discovery")
        if cfg.routing_logic:
            kwargs: Dict[str, Any] = {}
            if cfg.session_key:
                kwargs["session_key"] = cfg.session_key
            if cfg.prefix_min_match_length:
                kwargs["prefix_min_match_length"] = cfg.prefix_min_match_length
            router = rl.reconfigure_routing_logic(cfg.routing_logic, **kwargs)
            if self.app is not None:
                self.app.state.router = router
            logger.info(
                "dynamic config: reinitialized routing logic to %s",
                cfg.routing_logic,
            )

    def _worker(self) -> None:
        while not self._stop.wait(self.interval):
            try:
                self.poll_once()
            except Exception as e:  # pragma: no cover
                logger.error("dynamic config poll failed: %s", e)

    def get_health(self) -> bool:
        return self._thread is None or self._thread.is_alive()

    def close(self) -> None:
        self._stop.set()


_watcher: Optional[DynamicConfigWatcher] = None


def _flatten_structured(d):
    """Accept the reference's structured YAML schema
    (`models: {name: {static_backends: [...], static_model_type: t}}`,
    `aliases: {alias: model}`) by flattening it to the comma-separated
    static_* strings the router config uses
    (reference parsers/yaml_utils.py:10-38)."""
    if not isinstance(d, dict) or "models" not in d:
        return d
    out = dict(d)
    models = out.pop("models") or {}
    backends, names, types = [], [], []
    for name, det in models.items():
        for b in det.get("static_backends", []):
            backends.append(b)
            names.append(name)
            types.append(det.get("static_model_type", "chat"))
    out.setdefault("static_backends", ",".join(backends))
    out.setdefault("static_models", ",".join(names))
    if any(t != "chat" for t in types):
        out.setdefault("static_model_types", ",".join(types))
    aliases = out.pop("aliases", None)
    if aliases:
        out.setdefault(
            "static_aliases",
            ",".join(f"{a}:{m}" for a, m in aliases.items()),
        )
    return out


def initialize_dynamic_config_watcher(
    path: str, interval: float = 10.0, app=None, start: bool = True
) -> DynamicConfigWatcher:
    global _watcher
    if _watcher is not None:
        _watcher.close()
    _watcher = DynamicConfigWatcher(path, interval, app, start=start)
    return _watcher


def get_dynamic_config_watcher() -> Optional[DynamicConfigWatcher]:
    return _watcher
