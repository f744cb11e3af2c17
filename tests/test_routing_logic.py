"""Unit tests for the routing algorithms (no HTTP)."""

import asyncio

from production_stack_amd.router.routing_logic import (
    DisaggregatedPrefillOrchestratedRouter,
    DisaggregatedPrefillRouter,
    KvAwareRouter,
    PrefixAwareRouter,
    RoundRobinRouter,
    SessionRouter,
    initialize_routing_logic,
)
from production_stack_amd.router.service_discovery import EndpointInfo
from production_stack_amd.router.stats import RequestStats


class FakeRequest:
    def __init__(self, headers=None):
        self.headers = headers or {}


def eps(*urls, labels=None):
    labels = labels or [None] * len(urls)
    return [EndpointInfo(url=u, model_label=l) for u, l in zip(urls, labels)]


def run(coro):
    return asyncio.run(coro)


def test_round_robin_rotates():
    r = RoundRobinRouter()
    endpoints = eps("http://a", "http://b", "http://c")
    got = [
        run(r.route_request(endpoints, {}, {}, FakeRequest()))
        for _ in range(6)
    ]
    assert got == ["http://a", "http://b", "http://c"] * 2


def test_round_robin_separate_endpoint_sets():
    r = RoundRobinRouter()
    s1 = eps("http://a", "http://b")
    s2 = eps("http://c")
    assert run(r.route_request(s1, {}, {}, FakeRequest())) == "http://a"
    assert run(r.route_request(s2, {}, {}, FakeRequest())) == "http://c"
    assert run(r.route_request(s1, {}, {}, FakeRequest())) == "http://b"


def test_session_stickiness():
    r = SessionRouter(session_key="x-user-id")
    endpoints = eps("http://a", "http://b", "http://c")
    req = FakeRequest({"x-user-id": "alice"})
    first = run(r.route_request(endpoints, {}, {}, req))
    for _ in range(5):
        assert run(r.route_request(endpoints, {}, {}, req)) == first


def test_session_survives_unrelated_node_removal():
    r = SessionRouter(session_key="x-user-id")
    endpoints = eps("http://a", "http://b", "http://c")
    req = FakeRequest({"x-user-id": "alice"})
    first = run(r.route_request(endpoints, {}, {}, req))
    remaining = [e for e in endpoints if e.url != first]
    other = run(r.route_request(remaining, {}, {}, req))
    assert other != first
    # adding the node back restores the original mapping
    assert run(r.route_request(endpoints, {}, {}, req)) == first


def test_session_fallback_to_qps_without_session_id():
    r = SessionRouter(session_key="x-user-id")
    endpoints = eps("http://a", "http://b")
    stats = {"http://a": RequestStats(qps=5.0), "http://b": RequestStats(qps=1.0)}
    assert (
        run(r.route_request(endpoints, {}, stats, FakeRequest()))
        == "http://b"
    )


def test_prefixaware_affinity():
    r = PrefixAwareRouter(prefix_min_match_length=8)
    endpoints = eps("http://a", "http://b")
    long_prompt = {"prompt": "the quick brown fox jumps over the lazy dog " * 20}
    first = run(
        r.route_request(endpoints, {}, {}, FakeRequest(), long_prompt)
    )
    for _ in range(5):
        assert (
            run(r.route_request(endpoints, {}, {}, FakeRequest(), long_prompt))
            == first
        )


def test_prefixaware_falls_back_below_min_match():
    r = PrefixAwareRouter(prefix_min_match_length=1000)
    endpoints = eps("http://a", "http://b")
    stats = {"http://a": RequestStats(qps=9.0), "http://b": RequestStats(qps=0.0)}
    body = {"prompt": "short"}
    assert (
        run(r.route_request(endpoints, {}, stats, FakeRequest(), body))
        == "http://b"
    )


def test_prefixaware_chat_messages():
    r = PrefixAwareRouter(prefix_min_match_length=8)
    endpoints = eps("http://a", "http://b")
    body = {
        "messages": [
            {"role": "system", "content": "You are helpful. " * 30},
            {"role": "user", "content": "hi"},
        ]
    }
    first = run(r.route_request(endpoints, {}, {}, FakeRequest(), body))
    body2 = {
        "messages": [
            {"role": "system", "content": "You are helpful. " * 30},
            {"role": "user", "content": "another question"},
        ]
    }
    assert (
        run(r.route_request(endpoints, {}, {}, FakeRequest(), body2)) == first
    )


def test_disaggregated_prefill_label_split():
    r = DisaggregatedPrefillRouter(
        prefill_model_labels=["prefill"], decode_model_labels=["decode"]
    )
    endpoints = eps(
        "http://p1", "http://d1", labels=["prefill", "decode"]
    )
    assert (
        run(r.route_request(endpoints, {}, {}, FakeRequest(), {"max_tokens": 1}))
        == "http://p1"
    )
    assert (
        run(
            r.route_request(
                endpoints, {}, {}, FakeRequest(), {"max_tokens": 100}
            )
        )
        == "http://d1"
    )


def test_orchestrated_round_robins_pools():
    r = DisaggregatedPrefillOrchestratedRouter(
        prefill_model_labels=["prefill"], decode_model_labels=["decode"]
    )
    endpoints = eps(
        "http://p1", "http://p2", "http://d1", "http://d2",
        labels=["prefill", "prefill", "decode", "decode"],
    )
    assert r.select_prefill_endpoint(endpoints) == "http://p1"
    assert r.select_prefill_endpoint(endpoints) == "http://p2"
    assert r.select_prefill_endpoint(endpoints) == "http://p1"
    assert r.select_decode_endpoint(endpoints) == "http://d1"
    assert r.select_decode_endpoint(endpoints) == "http://d2"


def test_kvaware_falls_back_without_controller():
    r = KvAwareRouter(kv_controller_port=1)  # nothing listening
    endpoints = eps("http://a", "http://b")
    url = run(
        r.route_request(
            endpoints, {}, {}, FakeRequest({"x-user-id": "u1"}),
            {"prompt": "hello"},
        )
    )
    assert url in ("http://a", "http://b")


def test_initialize_by_name():
    for name, cls in [
        ("roundrobin", RoundRobinRouter),
        ("session", SessionRouter),
        ("prefixaware", PrefixAwareRouter),
        ("kvaware", KvAwareRouter),
    ]:
        r = initialize_routing_logic(name)
        assert isinstance(r, cls)
