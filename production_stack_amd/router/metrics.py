"""Router Prometheus metrics.

Parity: reference services/metrics_service/__init__.py:1-71 metric names and
routers/metrics_router.py:81-138 fill-on-scrape behaviour (plus
router_{cpu,memory,disk}_usage_percent process gauges).
"""

from __future__ import annotations

import psutil
from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    generate_latest,
)

REGISTRY = CollectorRegistry()

num_requests_running = Gauge(
    "vllm:num_requests_running",
    "Number of running requests per server",
    ["server"],
    registry=REGISTRY,
)
num_requests_waiting = Gauge(
    "vllm:num_requests_waiting",
    "Number of waiting requests per server",
    ["server"],
    registry=REGISTRY,
)
current_qps = Gauge(
    "vllm:current_qps",
    "Sliding-window QPS per server",
    ["server"],
    registry=REGISTRY,
)
avg_ttft = Gauge(
    "vllm:avg_ttft",
    "Average time-to-first-token per server (s)",
    ["server"],
    registry=REGISTRY,
)
avg_latency = Gauge(
    "vllm:avg_latency",
    "Average request latency per server (s)",
    ["server"],
    registry=REGISTRY,
)
in_prefill_requests = Gauge(
    "vllm:in_prefill_requests",
    "Requests in prefill per server",
    ["server"],
    registry=REGISTRY,
)
in_decoding_requests = Gauge(
    "vllm:in_decoding_requests",
    "Requests in decode per server",
    ["server"],
    registry=REGISTRY,
)
finished_requests = Gauge(
    "vllm:finished_requests",
    "Finished requests per server",
    ["server"],
    registry=REGISTRY,
)
gpu_cache_usage = Gauge(
    "vllm:gpu_cache_usage_perc",
    "Engine KV cache usage",
    ["server"],
    registry=REGISTRY,
)
healthy_pods_total = Gauge(
    "vllm:healthy_pods_total",
    "Number of healthy endpoints",
    ["model"],
    registry=REGISTRY,
)
model_input_tokens = Counter(
    "vllm:model_input_tokens_total",
    "Input tokens proxied per model",
    ["model"],
    registry=REGISTRY,
)
model_output_tokens = Counter(
    "vllm:model_output_tokens_total",
    "Output tokens proxied per model",
    ["model"],
    registry=REGISTRY,
)
request_errors = Counter(
    "vllm:request_errors_total",
    "Errored requests",
    ["model", "reason"],
    registry=REGISTRY,
)
request_latency_hist = Histogram(
    "vllm:request_latency_seconds",
    "Request latency histogram",
    ["model"],
    registry=REGISTRY,
)
router_cpu_usage = Gauge(
    "router_cpu_usage_percent", "Router CPU usage", registry=REGISTRY
)
router_memory_usage = Gauge(
    "router_memory_usage_percent", "Router memory usage", registry=REGISTRY
)
router_disk_usage = Gauge(
    "router_disk_usage_percent", "Router disk usage", registry=REGISTRY
)
semantic_cache_hits = Gauge(
    "vllm:semantic_cache_hits", "Semantic cache hits", registry=REGISTRY
)
semantic_cache_misses = Gauge(
    "vllm:semantic_cache_misses", "Semantic cache misses", registry=REGISTRY
)
semantic_cache_hit_rate = Gauge(
    "vllm:semantic_cache_hit_ratio", "Semantic cache hit ratio",
    registry=REGISTRY,
)


def _clear_label_gauges() -> None:
    """Drop every per-label child so endpoints removed from service
    discovery stop being reported (stale-metric fix; reference
    test_stale_metrics.py behavior: labels vanish after clearing and
    only re-populated endpoints reappear)."""
    for g in (healthy_pods_total, num_requests_running,
              num_requests_waiting, gpu_cache_usage, current_qps,
              avg_ttft, avg_latency, in_prefill_requests,
              in_decoding_requests, finished_requests):
        g.clear()


def fill_and_render() -> bytes:
    """Fill the gauges from the stats singletons and render the exposition."""
    _clear_label_gauges()
    from production_stack_amd.router.service_discovery import (
        get_service_discovery,
    )
    from production_stack_amd.router.stats import (
        get_engine_stats_scraper,
        get_request_stats_monitor,
    )

    try:
        endpoints = get_service_discovery().get_endpoint_info()
    except RuntimeError:
        endpoints = []
    scraper = get_engine_stats_scraper()
    engine_stats = scraper.get_engine_stats() if scraper else {}
    request_stats = get_request_stats_monitor().get_request_stats()

    model_health: dict = {}
    for ep in endpoints:
        for m in ep.model_names:
            model_health[m] = model_health.get(m, 0) + 1
    for m, n in model_health.items():
        healthy_pods_total.labels(model=m).set(n)

    for url, es in engine_stats.items():
        num_requests_running.labels(server=url).set(es.num_running_requests)
        num_requests_waiting.labels(server=url).set(es.num_queuing_requests)
        gpu_cache_usage.labels(server=url).set(es.gpu_cache_usage_perc)
    for url, rs in request_stats.items():
        current_qps.labels(server=url).set(rs.qps)
        avg_ttft.labels(server=url).set(rs.ttft)
        avg_latency.labels(server=url).set(rs.avg_latency)
        in_prefill_requests.labels(server=url).set(rs.in_prefill_requests)
        in_decoding_requests.labels(server=url).set(rs.in_decoding_requests)
        finished_requests.labels(server=url).set(rs.finished_requests)

    try:
        import production_stack_amd.router.app as _app_mod  # noqa
    except ImportError:
        _app_mod = None
    sem = getattr(getattr(_app_mod, "_last_app", None), "state", None)
    sem = getattr(sem, "semantic_cache", None) if sem else None
    if sem is not None:
        m = sem.metrics()
        semantic_cache_hits.set(m["semantic_cache_hits"])
        semantic_cache_misses.set(m["semantic_cache_misses"])
        semantic_cache_hit_rate.set(m["semantic_cache_hit_rate"])

    router_cpu_usage.set(psutil.cpu_percent())
    router_memory_usage.set(psutil.virtual_memory().percent)
    try:
        router_disk_usage.set(psutil.disk_usage("/").percent)
    except OSError:
        pass
    return generate_latest(REGISTRY)
