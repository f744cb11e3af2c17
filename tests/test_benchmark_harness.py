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


def test_sharegpt_loader_and_question_sequence(tmp_path):
    """ShareGPT conversations drive the per-user question sequence;
    exhausted conversations fall back to generated text."""
    import json as _json
    import random

    data = [
        {"conversations": [
            {"from": "human", "value": "q-a-1"},
            {"from": "gpt", "value": "ans"},
            {"from": "human", "value": "q-a-2"},
        ]},
        {"conversations": [
            {"from": "user", "value": "q-b-1"},
            {"from": "gpt", "value": "ans"},
        ]},
        {"conversations": [{"from": "gpt", "value": "no humans"}]},
    ]
    p = tmp_path / "sharegpt.json"
    p.write_text(_json.dumps(data))
    rng = random.Random(0)
    per_user = mrq.load_sharegpt(str(p), 4, rng)
    assert len(per_user) == 4
    flat = {tuple(q) for q in per_user}
    assert flat <= {("q-a-1", "q-a-2"), ("q-b-1",)}

    u = mrq.UserSession(0, "sys", questions=["q-a-1", "q-a-2"])
    u.rounds_done = 1
    assert u.questions[u.rounds_done] == "q-a-2"
