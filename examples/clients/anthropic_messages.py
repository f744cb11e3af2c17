#!/usr/bin/env python3
"""Anthropic Messages API example (SSE streaming)."""

import json
import sys

import requests

BASE = sys.argv[1] if len(sys.argv) > 1 else "http://localhost:8001"

with requests.post(f"{BASE}/v1/messages", json={
    "model": "llama-3-8b", "max_tokens": 64,
    "system": "be terse",
    "messages": [{"role": "user", "content": "hello!"}],
    "stream": True,
}, stream=True) as r:
    for line in r.iter_lines():
        if line.startswith(b"data: "):
            ev = json.loads(line[6:])
            if ev.get("type") == "content_block_delta":
                print(ev["delta"]["text"], end="", flush=True)
    print()
