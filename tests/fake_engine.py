"""Fake OpenAI-compatible engine for CPU router tests.

Pattern parity: reference src/tests/perftest/fake-openai-server.py — a
FastAPI mock with SSE chunking, /v1/models, /metrics (vllm:* names),
/tokenize, sleep/wake, configurable latency. Runs under uvicorn in a thread.
"""

from __future__ import annotations

import asyncio
import json
import threading
import time
import uuid
from typing import Optional

import uvicorn
from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse


def build_fake_engine(
    model: str = "fake-model",
    ttft: float = 0.0,
    tokens_per_sec: float = 10000.0,
    label: Optional[str] = None,
) -> FastAPI:
    app = FastAPI()
    state = {
        "requests": [],
        "sleeping": False,
        "running": 0,
    }
    app.state.seen = state

    @app.post("/v1/audio/transcriptions")
    async def transcriptions(request: Request):
        body = await request.body()
        ct = request.headers.get("content-type", "")
        from production_stack_amd.router.app import parse_multipart

        parts = parse_multipart(body, ct)
        state["requests"].append({"endpoint": "/v1/audio/transcriptions",
                                  "content_type": ct,
                                  "fields": sorted(parts)})
        fn = parts.get("file", ("?", b""))[0]
        return {"text": f"transcribed:{fn}", "task": "transcribe"}

    @app.get("/v1/audio/voices")
    async def voices():
        return {"voices": ["alloy", "echo"]}

    @app.post("/v1/audio/speech")
    async def speech(request: Request):
        body = await request.json()
        state["requests"].append({"endpoint": "/v1/audio/speech",
                                  "input": body.get("input")})
        return Response(content=b"RIFF\x00\x01\x02WAVEbinary",
                        media_type="audio/wav")

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [{"id": model, "object": "model"}]}

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def metrics():
        text = "\n".join(
            [
                f"vllm:num_requests_running {state['running']}",
                "vllm:num_requests_waiting 0",
                "vllm:gpu_cache_usage_perc 0.25",
                "vllm:gpu_prefix_cache_hits_total 10",
                "vllm:gpu_prefix_cache_queries_total 20",
            ]
        )
        return PlainTextResponse(text)

    @app.post("/tokenize")
    async def tokenize(request: Request):
        body = await request.json()
        prompt = body.get("prompt", "")
        tokens = [hash(w) % 50000 for w in str(prompt).split()]
        return {"tokens": tokens, "count": len(tokens)}

    @app.get("/is_sleeping")
    async def is_sleeping():
        return {"is_sleeping": state["sleeping"]}

    @app.post("/sleep")
    async def sleep():
        state["sleeping"] = True
        return {"status": "ok"}

    @app.post("/wake_up")
    async def wake_up():
        state["sleeping"] = False
        return {"status": "ok"}

    async def _chat_impl(request: Request, kind: str):
        body = await request.json()
        state["requests"].append(
            {"endpoint": kind, "body": body, "ts": time.time()}
        )
        n_tokens = int(body.get("max_tokens") or 8)
        rid = f"cmpl-{uuid.uuid4().hex[:12]}"
        created = int(time.time())
        delay = 1.0 / tokens_per_sec

        kv_params = None
        if body.get("kv_transfer_params"):
            kv_params = {
                "do_remote_decode": False,
                "do_remote_prefill": True,
                "remote_engine_id": "fake-engine-1",
                "remote_block_ids": [1, 2, 3],
                "remote_host": "127.0.0.1",
                "remote_port": 14001,
            }

        if body.get("stream"):

            async def gen():
                state["running"] += 1
                if ttft:
                    await asyncio.sleep(ttft)
                for i in range(n_tokens):
                    if kind == "chat":
                        chunk = {
                            "id": rid,
                            "object": "chat.completion.chunk",
                            "created": created,
                            "model": body.get("model", model),
                            "choices": [
                                {
                                    "index": 0,
                                    "delta": {"content": f"tok{i} "},
                                    "finish_reason": None,
                                }
                            ],
                        }
                    else:
                        chunk = {
                            "id": rid,
                            "object": "text_completion",
                            "created": created,
                            "model": body.get("model", model),
                            "choices": [
                                {
                                    "index": 0,
                                    "text": f"tok{i} ",
                                    "finish_reason": None,
                                }
                            ],
                        }
                    yield f"data: {json.dumps(chunk)}\n\n".encode()
                    if delay:
                        await asyncio.sleep(delay)
                final = {
                    "id": rid,
                    "object": "chat.completion.chunk",
                    "created": created,
                    "choices": [
                        {"index": 0, "delta": {}, "finish_reason": "stop"}
                    ],
                    "usage": {
                        "prompt_tokens": 5,
                        "completion_tokens": n_tokens,
                        "total_tokens": 5 + n_tokens,
                    },
                }
                yield f"data: {json.dumps(final)}\n\n".encode()
                yield b"data: [DONE]\n\n"
                state["running"] -= 1

            return StreamingResponse(gen(), media_type="text/event-stream")

        if ttft:
            await asyncio.sleep(ttft)
        text = " ".join(f"tok{i}" for i in range(n_tokens))
        resp = {
            "id": rid,
            "object": "chat.completion" if kind == "chat" else "text_completion",
            "created": created,
            "model": body.get("model", model),
            "choices": [
                {
                    "index": 0,
                    "finish_reason": "stop",
                    **(
                        {"message": {"role": "assistant", "content": text}}
                        if kind == "chat"
                        else {"text": text}
                    ),
                }
            ],
            "usage": {
                "prompt_tokens": 5,
                "completion_tokens": n_tokens,
                "total_tokens": 5 + n_tokens,
            },
        }
        if kv_params:
            resp["kv_transfer_params"] = kv_params
        return JSONResponse(resp)

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        return await _chat_impl(request, "chat")

    @app.post("/v1/completions")
    async def completions(request: Request):
        return await _chat_impl(request, "completion")

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        body = await request.json()
        state["requests"].append({"endpoint": "embeddings", "body": body})
        return {
            "object": "list",
            "data": [{"object": "embedding", "index": 0, "embedding": [0.0]}],
            "model": body.get("model", model),
        }

    return app


class FakeEngineServer:
    """Runs a fake engine under uvicorn in a daemon thread."""

    def __init__(self, port: int, **kwargs) -> None:
        self.port = port
        self.app = build_fake_engine(**kwargs)
        config = uvicorn.Config(
            self.app,
            host="127.0.0.1",
            port=port,
            log_level="error",
            lifespan="off",
        )
        self.server = uvicorn.Server(config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    @property
    def seen(self):
        return self.app.state.seen

    def start(self) -> None:
        self.thread.start()
        deadline = time.time() + 10
        import requests as _requests

        while time.time() < deadline:
            try:
                if (
                    _requests.get(self.url + "/health", timeout=0.5).status_code
                    == 200
                ):
                    return
            except Exception:
                time.sleep(0.05)
        raise RuntimeError(f"fake engine on :{self.port} failed to start")

    def stop(self) -> None:
        self.server.should_exit = True
        self.thread.join(timeout=5)
