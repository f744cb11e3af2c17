"""Sanitizer tier (the reference has no race/memory
sanitizers anywhere; this framework adds them):

  * psoperator-tsan: ThreadSanitizer over the operator's concurrent
    health-server + reconcile-loop paths (the -race equivalent for the
    C++ HTTP/JSON layer).
  * psoperator-asan: AddressSanitizer+UBSan over the JSON parser, HTTP
    client and reconcile builders against the fake API server.

GPU-kernel sanitizing is covered separately: every HIP kernel has
numerics tests against fp32 references (tests/test_kernels_gpu.py) and
the v5/gemm8p kernels carry explicit multi-run race screens
(csrc/tools/debug_v5c.py, gemm8p_bench.py); ROCm in this image ships no
compute-sanitizer equivalent, so device-code ASan is out of scope.
"""

import os
import subprocess
import time

import pytest
import requests

from tests.test_operator import PORT, fake_k8s  # noqa: F401 (fixture)

OP_DIR = os.path.join(os.path.dirname(__file__), "..", "operator")


def _build(target: str) -> str:
    path = os.path.join(OP_DIR, target)
    subprocess.run(["make", target], cwd=OP_DIR, check=True,
                   capture_output=True)
    return path


@pytest.mark.parametrize("target,env", [
    ("psoperator-asan",
     {"ASAN_OPTIONS": "detect_leaks=0,abort_on_error=1"}),
    ("psoperator-tsan",
     {"TSAN_OPTIONS": "halt_on_error=1"}),
])
def test_sanitized_operator_reconcile(fake_k8s, target, env):  # noqa: F811
    """One full reconcile pass of every CR kind under the sanitizer: any
    data race / heap error aborts with a non-zero exit."""
    binary = _build(target)
    fake_k8s.crs["vllmruntimes"] = [
        {
            "metadata": {"name": "san"},
            "spec": {
                "model": {"modelURL": "llama-3-8b"},
                "deploymentConfig": {"replicas": 1},
                "storage": {"enabled": True, "size": "10Gi"},
                "configData": {"a": "b"},
                "autoscaling": {"enabled": True},
            },
        }
    ]
    r = subprocess.run(
        [binary, "--api-server", f"http://127.0.0.1:{PORT}",
         "--namespace", "default", "--token-file", "/dev/null",
         "--leader-elect", "--once"],
        capture_output=True, timeout=120, env={**os.environ, **env},
    )
    assert r.returncode == 0, r.stderr.decode()[-3000:]
    assert b"WARNING: ThreadSanitizer" not in r.stderr
    assert b"AddressSanitizer" not in r.stderr


def test_tsan_concurrent_health_and_reconcile(fake_k8s):  # noqa: F811
    """TSan race check across the health-server thread and the reconcile
    loop while /metrics is hammered (the shared-counter paths)."""
    binary = _build("psoperator-tsan")
    proc = subprocess.Popen(
        [binary, "--api-server", f"http://127.0.0.1:{PORT}",
         "--namespace", "default", "--token-file", "/dev/null",
         "--health-port", "18093", "--interval", "1"],
        stderr=subprocess.PIPE,
        env={**os.environ, "TSAN_OPTIONS": "halt_on_error=1"},
    )
    try:
        deadline = time.time() + 15
        hits = 0
        while time.time() < deadline and hits < 30:
            try:
                requests.get("http://127.0.0.1:18093/metrics", timeout=0.5)
                requests.get("http://127.0.0.1:18093/healthz", timeout=0.5)
                hits += 1
            except Exception:
                time.sleep(0.2)
            if proc.poll() is not None:
                break
        assert hits >= 5, "health server never came up"
        assert proc.poll() is None, (
            "operator died under TSan: "
            + proc.stderr.read().decode()[-3000:]
        )
    finally:
        proc.terminate()
        try:
            _, err = proc.communicate(timeout=5)
        except subprocess.TimeoutExpired:
            proc.kill()
            _, err = proc.communicate()
        assert b"WARNING: ThreadSanitizer" not in (err or b"")
