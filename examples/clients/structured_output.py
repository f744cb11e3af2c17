#!/usr/bin/env python3
"""Structured-output client example: enforced JSON schema + choice."""

import json
import sys

import requests

BASE = sys.argv[1] if len(sys.argv) > 1 else "http://localhost:8001"

schema = {
    "type": "object",
    "additionalProperties": False,
    "properties": {"name": {"type": "string"},
                   "age": {"type": "number"}},
    "required": ["name", "age"],
}
r = requests.post(f"{BASE}/v1/chat/completions", json={
    "model": "llama-3-8b",
    "messages": [{"role": "user", "content": "Invent a user profile."}],
    "response_format": {"type": "json_schema",
                        "json_schema": {"schema": schema}},
}).json()
print(json.loads(r["choices"][0]["message"]["content"]))

r = requests.post(f"{BASE}/v1/chat/completions", json={
    "model": "llama-3-8b",
    "messages": [{"role": "user", "content": "Safe to deploy?"}],
    "guided_choice": ["yes", "no"],
}).json()
print("verdict:", r["choices"][0]["message"]["content"])
