"""OpenAI-compatible L7 request router (reference vllm_router parity,
MI355X-native engines behind it)."""

from production_stack_amd.router.app import build_app, initialize_all, main

__all__ = ["build_app", "initialize_all", "main"]
