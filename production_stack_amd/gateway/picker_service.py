"""Endpoint-picker service for Gateway API Inference Extension integration.

Capability parity with the reference's Go EPP plugins
(reference src/gateway_inference_extension/{prefix_aware_picker,
kv_aware_picker,roundrobin_picker}.go): the same three picking algorithms
exposed as a sidecar HTTP service. The gateway's ext-proc shim calls
POST /pick with the candidate pods and the request body; the response names
the chosen endpoint. (The reference links these as in-process Go plugins;
without a Go toolchain in this build the logic is served out-of-process —
same algorithms, same chunk size, same tie-breaks.)
"""

from __future__ import annotations

import argparse
import asyncio
import logging
from typing import List, Optional

from fastapi import FastAPI, Request

from production_stack_amd.router.hashtrie import HashTrie
from production_stack_amd.router.routing_logic import extract_prompt_text

try:  # compiled C++ picker core (csrc/gateway_pickers.cpp)
    from production_stack_amd import _gwpick
except ImportError:  # pragma: no cover - source-only checkout
    _gwpick = None

logger = logging.getLogger("gateway.picker")

CHUNK_SIZE = 128  # matches reference prefix_aware_picker.go:25


def build_picker_app(
    algorithm: str = "prefixaware",
    kv_controller_host: str = "127.0.0.1",
    kv_controller_port: int = 9000,
    min_match: int = CHUNK_SIZE,
) -> FastAPI:
    app = FastAPI(title="production-stack-amd endpoint picker")
    trie = HashTrie(chunk_size=CHUNK_SIZE)
    native = (_gwpick.NativePicker(CHUNK_SIZE, min_match)
              if _gwpick is not None else None)
    rr_state = {"idx": 0}
    kv_client = {"c": None}

    async def pick_roundrobin(endpoints: List[str], body) -> str:
        url = endpoints[rr_state["idx"] % len(endpoints)]
        rr_state["idx"] += 1
        return url

    async def pick_prefixaware(endpoints: List[str], body) -> str:
        text = extract_prompt_text(body or {})
        if native is not None:  # GIL-free C++ trie walk
            return native.pick_prefixaware(text, endpoints)
        matched, cands = await trie.longest_prefix_match(
            text, set(endpoints)
        )
        if matched >= min_match and cands:
            url = sorted(cands)[0]
        else:
            url = await pick_roundrobin(endpoints, body)
        await trie.insert(text, url)
        return url

    async def pick_kvaware(endpoints: List[str], body) -> str:
        from production_stack_amd.kvpool.client import ControllerClient

        text = extract_prompt_text(body or {})
        # tokenize with the ENGINE's tokenizer (its /tokenize endpoint)
        # so the controller lookup compares real token ids; a local
        # stand-in hash would never match the registered prefixes
        tokens = None
        try:
            import httpx

            async with httpx.AsyncClient(timeout=5) as hc:
                tr = await hc.post(endpoints[0] + "/tokenize",
                                   json={"prompt": text})
                if tr.status_code == 200:
                    tokens = tr.json().get("tokens")
        except Exception:
            tokens = None
        if not tokens:
            return await pick_roundrobin(endpoints, body)
        try:
            if kv_client["c"] is None:
                kv_client["c"] = ControllerClient(
                    kv_controller_host, kv_controller_port
                )
            matches = await kv_client["c"].lookup(tokens)
            live = {u: n for u, n in matches.items() if u in set(endpoints)}
            if live:
                best = max(live, key=live.get)
                if live[best] > 0:
                    return best
        except (ConnectionError, OSError, asyncio.TimeoutError):
            pass
        return await pick_roundrobin(endpoints, body)

    pickers = {
        "roundrobin": pick_roundrobin,
        "prefixaware": pick_prefixaware,
        "kvaware": pick_kvaware,
    }
    picker = pickers[algorithm]

    @app.post("/pick")
    async def pick(request: Request):
        payload = await request.json()
        endpoints = payload.get("endpoints") or []
        if not endpoints:
            return {"error": "no endpoints"}
        url = await picker(endpoints, payload.get("body"))
        return {"endpoint": url, "algorithm": algorithm}

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    return app


def main() -> None:
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9002)
    ap.add_argument(
        "--algorithm",
        default="prefixaware",
        choices=["roundrobin", "prefixaware", "kvaware"],
    )
    ap.add_argument("--kv-controller-host", default="127.0.0.1")
    ap.add_argument("--kv-controller-port", type=int, default=9000)
    args = ap.parse_args()
    app = build_picker_app(
        args.algorithm, args.kv_controller_host, args.kv_controller_port
    )
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
