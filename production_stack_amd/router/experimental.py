"""Feature gates, request rewriter, custom callbacks.

Parity: reference experimental/feature_gates.py (Alpha/Beta/GA staged gates),
services/request_service/rewriter.py (pluggable pre-proxy body rewrite),
services/callbacks_service/callbacks.py (dynamic import of a user module
with pre_request/post_request hooks).
"""

from __future__ import annotations

import importlib
import logging
from typing import Any, Dict, Optional

logger = logging.getLogger("router.experimental")


# ---------------------------------------------------------------------------
# Feature gates
# ---------------------------------------------------------------------------
class FeatureStage:
    ALPHA = "Alpha"
    BETA = "Beta"
    GA = "GA"


KNOWN_FEATURES: Dict[str, str] = {
    "SemanticCache": FeatureStage.ALPHA,
    "PIIDetection": FeatureStage.ALPHA,
    "KVOffload": FeatureStage.BETA,
}


class FeatureGates:
    def __init__(self, spec: Optional[str] = None) -> None:
        self.enabled: Dict[str, bool] = {
            # Beta+ features default on
            k: v != FeatureStage.ALPHA
            for k, v in KNOWN_FEATURES.items()
        }
        if spec:
            for pair in spec.split(","):
                if "=" not in pair:
                    continue
                name, val = pair.split("=", 1)
                name = name.strip()
                if name not in KNOWN_FEATURES:
                    raise ValueError(f"unknown feature gate {name!r}")
                self.enabled[name] = val.strip().lower() == "true"

    def is_enabled(self, name: str) -> bool:
        return self.enabled.get(name, False)


_gates: Optional[FeatureGates] = None


def initialize_feature_gates(spec: Optional[str] = None) -> FeatureGates:
    global _gates
    _gates = FeatureGates(spec)
    return _gates


def get_feature_gates() -> FeatureGates:
    global _gates
    if _gates is None:
        _gates = FeatureGates()
    return _gates


# ---------------------------------------------------------------------------
# Request rewriter
# ---------------------------------------------------------------------------
class RequestRewriter:
    def rewrite(
        self, endpoint: str, request_json: Dict[str, Any]
    ) -> Optional[Dict[str, Any]]:
        """Return a replacement body, or None to leave it untouched."""
        return None


class NoopRequestRewriter(RequestRewriter):
    pass


def get_request_rewriter(kind: str = "noop") -> RequestRewriter:
    return NoopRequestRewriter()


# ---------------------------------------------------------------------------
# Custom callbacks
# ---------------------------------------------------------------------------
class Callbacks:
    def __init__(self, module) -> None:
        self._module = module

    def pre_request(self, request, request_json, model):
        fn = getattr(self._module, "pre_request", None)
        if fn is None:
            return None
        return fn(request, request_json, model)

    def post_request(self, request, response):
        fn = getattr(self._module, "post_request", None)
        if fn is None:
            return None
        return fn(request, response)


def configure_custom_callbacks(module_path: str) -> Callbacks:
    module = importlib.import_module(module_path)
    logger.info("loaded custom callbacks from %s", module_path)
    return Callbacks(module)
