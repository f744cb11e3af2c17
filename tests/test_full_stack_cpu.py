"""Full-stack CPU integration: router -> two REAL tiny engines over HTTP.

This is BASELINE.json config #1's shape ("router round-robin over 2 CPU
backends") with our own MI355X-native engine (CPU reference ops) as the
backend instead of an external image.
"""

import asyncio
import threading
import time

import httpx
import pytest
import uvicorn

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.server import build_server
from production_stack_amd.router import app as router_app_mod
from production_stack_amd.router.parser import parse_args

BASE_PORT = 18300


class RealEngineServer:
    def __init__(self, port: int) -> None:
        cfg = EngineConfig(
            model="tiny-llama",
            max_model_len=512,
            cache=CacheConfig(num_gpu_blocks=256, block_size=16),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=256
            ),
        )
        self.engine = LLMEngine(cfg, device="cpu")
        self.app = build_server(self.engine, served_model="tiny-llama")
        self.port = port
        config = uvicorn.Config(
            self.app, host="127.0.0.1", port=port, log_level="error"
        )
        self.server = uvicorn.Server(config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    @property
    def url(self):
        return f"http://127.0.0.1:{self.port}"

    def start(self):
        self.thread.start()
        import requests

        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                if requests.get(self.url + "/health", timeout=0.5).ok:
                    return
            except Exception:
                time.sleep(0.1)
        raise RuntimeError("engine server failed to start")

    def stop(self):
        self.server.should_exit = True
        self.thread.join(timeout=5)


@pytest.fixture(scope="module")
def engines():
    servers = [RealEngineServer(BASE_PORT), RealEngineServer(BASE_PORT + 1)]
    for s in servers:
        s.start()
    yield servers
    for s in servers:
        s.stop()


def test_router_over_two_real_engines(engines):
    argv = [
        "--service-discovery", "static",
        "--static-backends", ",".join(s.url for s in engines),
        "--static-models", "tiny-llama",
        "--routing-logic", "roundrobin",
    ]
    args = parse_args(argv)
    application = router_app_mod.build_app()
    router_app_mod.initialize_all(application, args)

    async def go():
        router_app_mod._http_session = None
        transport = httpx.ASGITransport(app=application)
        async with httpx.AsyncClient(
            transport=transport, base_url="http://router", timeout=60
        ) as client:
            # round robin across real engines, completions + chat + stream
            for i in range(4):
                r = await client.post(
                    "/v1/completions",
                    json={
                        "model": "tiny-llama",
                        "prompt": f"hello number {i}",
                        "max_tokens": 4,
                        "temperature": 0,
                        "ignore_eos": True,
                    },
                )
                assert r.status_code == 200, r.text
                assert r.json()["usage"]["completion_tokens"] == 4
            async with client.stream(
                "POST",
                "/v1/chat/completions",
                json={
                    "model": "tiny-llama",
                    "messages": [{"role": "user", "content": "stream me"}],
                    "max_tokens": 3,
                    "temperature": 0,
                    "ignore_eos": True,
                    "stream": True,
                },
            ) as r:
                assert r.status_code == 200
                body = ""
                async for t in r.aiter_text():
                    body += t
                assert "[DONE]" in body
            # scrape engine metrics through the real /metrics endpoints
            from production_stack_amd.router.stats import (
                get_engine_stats_scraper,
            )

            scraper = get_engine_stats_scraper()
            scraper.scrape_once()
            stats = scraper.get_engine_stats()
            assert len(stats) == 2
        if router_app_mod._http_session is not None:
            await router_app_mod._http_session.close()
            router_app_mod._http_session = None

    asyncio.run(go())
    # both engines actually served traffic
    served = [s.engine.stats.num_requests for s in engines]
    assert all(n > 0 for n in served), served


def test_new_apis_through_router(engines):
    """Round-2-late APIs end to end through the router over real
    engines: Anthropic messages, tool calling, structured outputs,
    TTS."""
    import io
    import wave

    argv = [
        "--service-discovery", "static",
        "--static-backends", ",".join(s.url for s in engines),
        "--static-models", "tiny-llama",
        "--routing-logic", "roundrobin",
    ]
    args = parse_args(argv)
    application = router_app_mod.build_app()
    router_app_mod.initialize_all(application, args)

    async def go():
        router_app_mod._http_session = None
        transport = httpx.ASGITransport(app=application)
        async with httpx.AsyncClient(
            transport=transport, base_url="http://router", timeout=60
        ) as client:
            r = await client.post("/v1/messages", json={
                "model": "tiny-llama", "max_tokens": 4,
                "messages": [{"role": "user", "content": "hi"}],
                "temperature": 0, "ignore_eos": True,
            })
            assert r.status_code == 200, r.text
            assert r.json()["type"] == "message"

            r = await client.post("/v1/chat/completions", json={
                "model": "tiny-llama", "max_tokens": 3,
                "temperature": 0,
                "messages": [{"role": "user", "content": "hi"}],
                "tools": [{"type": "function",
                           "function": {"name": "noop",
                                        "parameters": {}}}],
                "ignore_eos": True,
            })
            assert r.status_code == 200, r.text
            msg = r.json()["choices"][0]["message"]
            assert msg["role"] == "assistant"

            r = await client.post("/v1/chat/completions", json={
                "model": "tiny-llama", "max_tokens": 6,
                "temperature": 0,
                "messages": [{"role": "user", "content": "pick"}],
                "response_format": {"type": "json_object"},
            })
            assert r.status_code == 200, r.text
            assert r.json()["choices"][0]["finish_reason"] == "stop"

            r = await client.post("/v1/audio/speech", json={
                "model": "tiny-llama", "input": "w1 w2 w3"})
            assert r.status_code == 200
            with wave.open(io.BytesIO(r.content)) as w:
                assert w.getnframes() > 0

    asyncio.run(go())
