"""bench.py driver-contract tests: single-process and torchrun multi-rank
(gloo, CPU) must emit exactly one valid JSON line from rank 0."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.join(os.path.dirname(__file__), "..")

COMMON = [
    "--device", "cpu", "--model", "tiny-llama", "--users", "2",
    "--steps", "8", "--warmup", "2", "--max-model-len", "1600",
    "--max-num-batched-tokens", "256",
]


def test_bench_single_process():
    r = subprocess.run(
        [sys.executable, "bench.py"] + COMMON,
        cwd=ROOT, capture_output=True, timeout=420, text=True,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    out = json.loads(lines[0])
    assert out["metric"] == "output_tokens_per_sec"
    assert out["n_gpus"] == 1
    assert out["value"] > 0
    assert out["steps"] == 8 and out["warmup"] == 2
    assert out["dtype"] == "bfloat16" and out["data"] == "synthetic"
    assert out["higher_is_better"] is True and out["scaling"] == "weak"
    assert "ms_per_step" in out and "config" in out


@pytest.mark.timeout(400)
def test_bench_torchrun_two_ranks():
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29617",
            "bench.py",
        ] + COMMON + ["--gpus", "2"],
        cwd=ROOT, capture_output=True, timeout=380, text=True,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0
