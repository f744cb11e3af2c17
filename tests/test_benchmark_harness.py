"""The multi-round-QA harness runs against our real engine server (CPU)."""

import asyncio
import importlib.util
import os
import sys

import pytest

spec = importlib.util.spec_from_file_location(
    "multi_round_qa",
    os.path.join(
        os.path.dirname(__file__), "..", "benchmarks", "multi-round-qa",
        "multi_round_qa.py",
    ),
)
mrq = importlib.util.module_from_spec(spec)
spec.loader.exec_module(mrq)


@pytest.mark.timeout(180)
def test_harness_against_real_engine():
    from tests.test_full_stack_cpu import RealEngineServer

    server = RealEngineServer(18500)
    server.start()
    try:
        args = mrq.parse_args(
            [
                "--base-url", server.url,
                "--model", "tiny-llama",
                "--num-users", "3",
                "--num-rounds", "2",
                "--qps", "50",
                "--shared-system-prompt", "40",
                "--user-history-prompt", "30",
                "--question-len", "10",
                "--answer-len", "8",
                "--time-limit", "60",
                "--init-user-interval", "0.1",
            ]
        )
        summary = asyncio.run(mrq.main_async(args))
    finally:
        server.stop()
    assert summary["errors"] == 0, summary
    assert summary["finished_requests"] == 6
    assert summary["output_tokens_per_s"] > 0
    assert summary["ttft_p50_s"] is not None
