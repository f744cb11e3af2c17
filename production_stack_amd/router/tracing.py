"""Request tracing.

Parity: reference experimental/otel/tracing.py — OTLP exporter + W3C
traceparent propagation. OpenTelemetry is an optional import (absent in the
offline image); without it this module still implements functional W3C
traceparent extraction/injection and span timing (logged), so distributed
trace context flows router -> engine either way.
"""

from __future__ import annotations

import logging
import random
import re
import time
from contextlib import contextmanager
from typing import Dict, Optional

logger = logging.getLogger("router.tracing")

_TRACEPARENT = re.compile(
    r"^(?P<version>[0-9a-f]{2})-(?P<trace_id>[0-9a-f]{32})-"
    r"(?P<span_id>[0-9a-f]{16})-(?P<flags>[0-9a-f]{2})$"
)

try:  # pragma: no cover - otel not installed in the offline env
    from opentelemetry import trace as otel_trace
    from opentelemetry.exporter.otlp.proto.grpc.trace_exporter import (
        OTLPSpanExporter,
    )
    from opentelemetry.sdk.resources import Resource
    from opentelemetry.sdk.trace import TracerProvider
    from opentelemetry.sdk.trace.export import BatchSpanProcessor

    HAVE_OTEL = True
except ImportError:
    HAVE_OTEL = False

_enabled = False
_service_name = "vllm-router"


def initialize_tracing(
    endpoint: Optional[str],
    service_name: str = "vllm-router",
    secure: bool = False,
) -> None:
    global _enabled, _service_name
    _service_name = service_name
    if not endpoint:
        return
    _enabled = True
    if HAVE_OTEL:  # pragma: no cover
        provider = TracerProvider(
            resource=Resource.create({"service.name": service_name})
        )
        provider.add_span_processor(
            BatchSpanProcessor(
                OTLPSpanExporter(endpoint=endpoint, insecure=not secure)
            )
        )
        otel_trace.set_tracer_provider(provider)
    logger.info(
        "tracing initialized (otel=%s, endpoint=%s)", HAVE_OTEL, endpoint
    )


def is_enabled() -> bool:
    return _enabled


def extract_context(headers) -> Optional[Dict[str, str]]:
    """Parse a W3C traceparent header into its fields."""
    tp = None
    try:
        tp = headers.get("traceparent")
    except AttributeError:
        pass
    if not tp:
        return None
    m = _TRACEPARENT.match(tp.strip())
    if not m:
        return None
    return m.groupdict()


def new_span_id() -> str:
    return f"{random.getrandbits(64):016x}"


def new_trace_id() -> str:
    return f"{random.getrandbits(128):032x}"


def inject_context(
    headers: Dict[str, str], parent: Optional[Dict[str, str]]
) -> Dict[str, str]:
    """Inject a child traceparent into outgoing backend headers."""
    trace_id = parent["trace_id"] if parent else new_trace_id()
    flags = parent["flags"] if parent else "01"
    headers = dict(headers)
    headers["traceparent"] = f"00-{trace_id}-{new_span_id()}-{flags}"
    return headers


@contextmanager
def span(name: str, parent: Optional[Dict[str, str]] = None, **attrs):
    """SERVER/CLIENT span: otel when available, timed log record otherwise."""
    if HAVE_OTEL and _enabled:  # pragma: no cover
        tracer = otel_trace.get_tracer(_service_name)
        with tracer.start_as_current_span(name) as s:
            for k, v in attrs.items():
                s.set_attribute(k, v)
            yield s
        return
    t0 = time.time()
    try:
        yield None
    finally:
        if _enabled:
            logger.debug(
                "span %s took %.3fs attrs=%s", name, time.time() - t0, attrs
            )
