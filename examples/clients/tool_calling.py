#!/usr/bin/env python3
"""Tool-calling client example (works against the router or an engine)."""

import json
import sys

import requests

BASE = sys.argv[1] if len(sys.argv) > 1 else "http://localhost:8001"

tools = [{
    "type": "function",
    "function": {
        "name": "get_weather",
        "description": "Current weather for a city",
        "parameters": {
            "type": "object",
            "properties": {"city": {"type": "string"}},
            "required": ["city"],
        },
    },
}]

messages = [{"role": "user", "content": "What's the weather in SF?"}]
r = requests.post(f"{BASE}/v1/chat/completions", json={
    "model": "llama-3-8b", "messages": messages, "tools": tools,
}).json()
msg = r["choices"][0]["message"]
print("assistant:", json.dumps(msg, indent=2))

for call in msg.get("tool_calls", []):
    args = json.loads(call["function"]["arguments"])
    messages.append(msg)
    messages.append({
        "role": "tool",
        "tool_call_id": call["id"],
        "content": json.dumps({"city": args.get("city"), "temp_c": 18}),
    })
    r = requests.post(f"{BASE}/v1/chat/completions", json={
        "model": "llama-3-8b", "messages": messages, "tools": tools,
    }).json()
    print("after tool result:", r["choices"][0]["message"]["content"])
