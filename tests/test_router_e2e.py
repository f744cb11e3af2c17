"""End-to-end router tests: real fake-engine backends over HTTP, router app
exercised through ASGI (reference tests/e2e/test-routing.py pattern)."""

import asyncio
import json

import httpx
import pytest

from production_stack_amd.router import app as app_mod
from production_stack_amd.router.parser import parse_args
from tests.fake_engine import FakeEngineServer

BASE_PORT = 18100


@pytest.fixture(scope="module")
def engines():
    servers = [
        FakeEngineServer(BASE_PORT, model="m1"),
        FakeEngineServer(BASE_PORT + 1, model="m1"),
    ]
    for s in servers:
        s.start()
    yield servers
    for s in servers:
        s.stop()


def make_app(engines, extra_args=None):
    argv = [
        "--service-discovery", "static",
        "--static-backends", ",".join(s.url for s in engines),
        "--static-models", "m1",
        "--routing-logic", "roundrobin",
    ] + (extra_args or [])
    args = parse_args(argv)
    application = app_mod.build_app()
    app_mod.initialize_all(application, args)
    return application


def with_client(application, fn):
    async def go():
        app_mod._http_session = None
        transport = httpx.ASGITransport(app=application)
        async with httpx.AsyncClient(
            transport=transport, base_url="http://router"
        ) as client:
            await fn(client)
        if app_mod._http_session is not None:
            await app_mod._http_session.close()
            app_mod._http_session = None

    asyncio.run(go())


def test_round_robin_distribution(engines):
    application = make_app(engines)
    for s in engines:
        s.seen["requests"].clear()

    async def go(client):
        for _ in range(10):
            r = await client.post(
                "/v1/chat/completions",
                json={
                    "model": "m1",
                    "messages": [{"role": "user", "content": "hi"}],
                    "max_tokens": 2,
                },
            )
            assert r.status_code == 200

    with_client(application, go)
    counts = [len(s.seen["requests"]) for s in engines]
    assert counts == [5, 5], counts


def test_streaming_sse_with_usage(engines):
    application = make_app(engines)

    async def go(client):
        async with client.stream(
            "POST",
            "/v1/chat/completions",
            json={
                "model": "m1",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 3,
                "stream": True,
            },
        ) as r:
            assert r.status_code == 200
            assert "text/event-stream" in r.headers["content-type"]
            body = ""
            async for chunk in r.aiter_text():
                body += chunk
        lines = [
            line[6:] for line in body.splitlines() if line.startswith("data: ")
        ]
        assert lines[-1] == "[DONE]"
        final = json.loads(lines[-2])
        assert final["usage"]["completion_tokens"] == 3

    with_client(application, go)


def test_unknown_model_404(engines):
    application = make_app(engines)

    async def go(client):
        r = await client.post(
            "/v1/chat/completions",
            json={"model": "nope", "messages": [], "max_tokens": 1},
        )
        assert r.status_code == 404

    with_client(application, go)


def test_models_aggregation(engines):
    application = make_app(engines)

    async def go(client):
        r = await client.get("/v1/models")
        data = r.json()
        assert {m["id"] for m in data["data"]} == {"m1"}

    with_client(application, go)


def test_health_and_version(engines):
    application = make_app(engines)

    async def go(client):
        r = await client.get("/health")
        assert r.status_code == 200
        r = await client.get("/version")
        assert "version" in r.json()

    with_client(application, go)


def test_metrics_exposition(engines):
    application = make_app(engines)

    async def go(client):
        await client.post(
            "/v1/completions",
            json={"model": "m1", "prompt": "x", "max_tokens": 2},
        )
        r = await client.get("/metrics")
        text = r.text
        assert "vllm:current_qps" in text
        assert "router_cpu_usage_percent" in text

    with_client(application, go)


def test_failover_reroutes_to_live_backend(engines):
    """Backend list includes a dead URL; failover must recover."""
    class Dead:
        url = "http://127.0.0.1:1"

    argv = [
        "--service-discovery", "static",
        "--static-backends", f"{Dead.url},{engines[0].url}",
        "--static-models", "m1",
        "--routing-logic", "roundrobin",
        "--max-instance-failover-reroute-attempts", "2",
    ]
    args = parse_args(argv)
    application = app_mod.build_app()
    app_mod.initialize_all(application, args)

    async def go(client):
        ok = 0
        for _ in range(4):
            r = await client.post(
                "/v1/completions",
                json={"model": "m1", "prompt": "x", "max_tokens": 1},
            )
            if r.status_code == 200:
                ok += 1
        assert ok == 4

    with_client(application, go)


def test_sleep_wake_filters_endpoint(engines):
    application = make_app(engines)

    async def go(client):
        r = await client.post("/sleep", params={"url": engines[0].url})
        assert r.status_code == 200
        r = await client.get("/is_sleeping")
        assert engines[0].url in r.json()
        # sleeping endpoint is filtered from routing
        for s in engines:
            s.seen["requests"].clear()
        for _ in range(4):
            r = await client.post(
                "/v1/completions",
                json={"model": "m1", "prompt": "x", "max_tokens": 1},
            )
            assert r.status_code == 200
        assert len(engines[0].seen["requests"]) == 0
        assert len(engines[1].seen["requests"]) == 4
        r = await client.post("/wake_up", params={"url": engines[0].url})
        assert r.status_code == 200

    with_client(application, go)


def test_engine_stats_scrape(engines):
    from production_stack_amd.router.stats import get_engine_stats_scraper

    application = make_app(engines)
    scraper = get_engine_stats_scraper()
    scraper.scrape_once()
    stats = scraper.get_engine_stats()
    assert engines[0].url in stats
    assert stats[engines[0].url].gpu_cache_usage_perc == 0.25
    # avoid unused warning
    assert application is not None


def test_files_roundtrip(engines):
    application = make_app(engines)

    async def go(client):
        boundary = "----testboundary"
        body = (
            f"--{boundary}\r\n"
            'Content-Disposition: form-data; name="purpose"\r\n\r\n'
            "batch\r\n"
            f"--{boundary}\r\n"
            'Content-Disposition: form-data; name="file"; filename="in.jsonl"\r\n'
            "Content-Type: application/jsonl\r\n\r\n"
            '{"custom_id": "1"}\r\n'
            f"--{boundary}--\r\n"
        ).encode()
        r = await client.post(
            "/v1/files",
            content=body,
            headers={
                "content-type": f"multipart/form-data; boundary={boundary}"
            },
        )
        assert r.status_code == 200, r.text
        meta = r.json()
        fid = meta["id"]
        r = await client.get(f"/v1/files/{fid}")
        assert r.json()["filename"] == "in.jsonl"
        r = await client.get(f"/v1/files/{fid}/content")
        assert b"custom_id" in r.content

    with_client(application, go)


def test_disaggregated_orchestrated_flow(engines):
    """P->D chaining: prefill gets max_tokens=1 + kv_transfer_params, decode
    gets the kv params back (reference request.py:733-935 flow)."""
    argv = [
        "--service-discovery", "static",
        "--static-backends", ",".join(s.url for s in engines),
        "--static-models", "m1",
        "--static-model-labels", "prefill,decode",
        "--routing-logic", "disaggregated_prefill_orchestrated",
        "--prefill-model-labels", "prefill",
        "--decode-model-labels", "decode",
    ]
    args = parse_args(argv)
    application = app_mod.build_app()
    app_mod.initialize_all(application, args)
    for s in engines:
        s.seen["requests"].clear()

    async def go(client):
        r = await client.post(
            "/v1/completions",
            json={"model": "m1", "prompt": "hello", "max_tokens": 4},
        )
        assert r.status_code == 200

    with_client(application, go)
    p_reqs = engines[0].seen["requests"]
    d_reqs = engines[1].seen["requests"]
    assert len(p_reqs) == 1 and len(d_reqs) == 1
    assert p_reqs[0]["body"]["max_tokens"] == 1
    assert p_reqs[0]["body"]["kv_transfer_params"]["do_remote_decode"] is True
    assert d_reqs[0]["body"]["max_tokens"] == 4
    assert d_reqs[0]["body"]["kv_transfer_params"]["do_remote_prefill"] is True
    assert (
        d_reqs[0]["body"]["kv_transfer_params"]["remote_engine_id"]
        == "fake-engine-1"
    )


def test_dynamic_config_hot_reload(engines, tmp_path):
    """Editing the dynamic-config file swaps routing logic without restart
    (reference dynamic_config.py behaviour)."""
    import json as _json

    from production_stack_amd.router.dynamic_config import (
        initialize_dynamic_config_watcher,
        get_dynamic_config_watcher,
    )
    from production_stack_amd.router.routing_logic import (
        PrefixAwareRouter,
        RoundRobinRouter,
        get_routing_logic,
    )

    cfg_file = tmp_path / "dyn.json"
    cfg_file.write_text(_json.dumps({"routing_logic": "roundrobin"}))
    application = make_app(engines)
    watcher = initialize_dynamic_config_watcher(
        str(cfg_file), interval=3600, app=application, start=False
    )
    assert isinstance(get_routing_logic(), RoundRobinRouter)
    cfg_file.write_text(
        _json.dumps(
            {"routing_logic": "prefixaware", "prefix_min_match_length": 64}
        )
    )
    assert watcher.poll_once()
    assert isinstance(get_routing_logic(), PrefixAwareRouter)
    assert isinstance(application.state.router, PrefixAwareRouter)
    watcher.close()


def test_stress_even_distribution(engines):
    """Stress-test parity (reference tests/e2e/stress-test.sh): many
    concurrent requests through the round-robin router distribute evenly."""
    application = make_app(engines)
    for s in engines:
        s.seen["requests"].clear()

    async def go(client):
        async def one(i):
            r = await client.post(
                "/v1/completions",
                json={"model": "m1", "prompt": f"q{i}", "max_tokens": 1},
            )
            return r.status_code

        results = await asyncio.gather(*(one(i) for i in range(100)))
        assert all(c == 200 for c in results)

    with_client(application, go)
    counts = [len(s.seen["requests"]) for s in engines]
    assert sum(counts) == 100
    assert abs(counts[0] - counts[1]) <= 2, counts


def test_active_health_check_hashes_out_dead_backend(engines):
    """Static discovery health checker (reference service_discovery.py:
    254-289 behaviour): a dead (url, model) pair leaves the rotation."""
    from production_stack_amd.router.service_discovery import (
        StaticServiceDiscovery,
        reset_service_discovery,
    )

    sd = StaticServiceDiscovery(
        urls=[engines[0].url, "http://127.0.0.1:1"],
        models=["m1"],
        health_check=False,
    )
    assert len(sd.get_endpoint_info()) == 2
    sd.check_health_once()
    eps = sd.get_endpoint_info()
    assert [e.url for e in eps] == [engines[0].url]
    assert sd.get_unhealthy_endpoint_hashes() == ["http://127.0.0.1:1:m1"]
    sd.close()


def test_multipart_transcription_proxied_verbatim(engines):
    """A real multipart/form-data transcription request must pass
    through the router byte-for-byte (boundary preserved in the
    forwarded content-type; no JSON-parse 400), with routing driven by
    the form's model field."""
    app = make_app(engines)

    async def go():
        boundary = "testbnd123"
        body = (
            f"--{boundary}\r\n"
            'Content-Disposition: form-data; name="model"\r\n\r\n'
            "m1\r\n"
            f"--{boundary}\r\n"
            'Content-Disposition: form-data; name="file"; '
            'filename="clip.wav"\r\n'
            "Content-Type: audio/wav\r\n\r\n"
            "RIFFxxxxWAVE\r\n"
            f"--{boundary}--\r\n"
        ).encode()
        async with httpx.ASGITransport(app=app) as transport:
            async with app.router.lifespan_context(app):
                async with httpx.AsyncClient(
                    transport=transport, base_url="http://router"
                ) as client:
                    r = await client.post(
                        "/v1/audio/transcriptions",
                        content=body,
                        headers={"content-type":
                                 f"multipart/form-data; boundary={boundary}"},
                        timeout=30,
                    )
        assert r.status_code == 200, r.text
        assert r.json()["text"] == "transcribed:clip.wav"
        seen = [
            req for s in engines for req in s.seen["requests"]
            if isinstance(req, dict)
            and req.get("endpoint") == "/v1/audio/transcriptions"
        ]
        assert seen and "testbnd123" in seen[0]["content_type"]
        assert seen[0]["fields"] == ["file", "model"]

    asyncio.run(go())


def test_speech_binary_response_content_type_preserved(engines):
    """Binary audio responses stream through the router with the
    upstream content-type intact (reference
    test_audio_speech_routing.py behaviors)."""
    app = make_app(engines)

    async def go():
        async with httpx.ASGITransport(app=app) as transport:
            async with app.router.lifespan_context(app):
                async with httpx.AsyncClient(
                    transport=transport, base_url="http://router"
                ) as client:
                    r = await client.post(
                        "/v1/audio/speech",
                        json={"model": "m1", "input": "hello",
                              "voice": "alloy"},
                        timeout=30,
                    )
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("audio/wav")
        assert r.content.startswith(b"RIFF")

    asyncio.run(go())


def test_alias_rewrites_model_in_forwarded_body(engines):
    """Requests for an alias route to the target model's endpoints AND
    the forwarded body carries the resolved model name (reference
    utils.replace_model_in_request_body behavior)."""
    app = make_app(engines, extra_args=["--static-aliases", "ali:m1"])

    async def go():
        async with httpx.ASGITransport(app=app) as transport:
            async with app.router.lifespan_context(app):
                async with httpx.AsyncClient(
                    transport=transport, base_url="http://router"
                ) as client:
                    r = await client.post(
                        "/v1/chat/completions",
                        json={"model": "ali",
                              "messages": [{"role": "user",
                                            "content": "hi"}],
                              "max_tokens": 2},
                        timeout=30,
                    )
        assert r.status_code == 200, r.text
        bodies = [
            req["body"] for s in engines for req in s.seen["requests"]
            if isinstance(req, dict)
            and isinstance(req.get("body"), dict)
            and req["body"].get("model") == "m1"
        ]
        assert bodies, "no backend saw the resolved model name"

    asyncio.run(go())


def test_audio_voices_and_translations_proxied(engines):
    app = make_app(engines)

    async def go():
        async with httpx.ASGITransport(app=app) as transport:
            async with app.router.lifespan_context(app):
                async with httpx.AsyncClient(
                    transport=transport, base_url="http://router"
                ) as client:
                    r = await client.get("/v1/audio/voices", timeout=30)
                    assert r.status_code == 200
                    assert "alloy" in r.json()["voices"]

    asyncio.run(go())
