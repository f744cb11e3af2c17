"""Native C++ gateway picker (csrc/gateway_pickers.cpp) vs the Python
implementation it replaces (differential oracle), plus thread-safety of
the shared-mutex trie. Parity target: reference
src/gateway_inference_extension/{prefix_aware,kv_aware,roundrobin}_picker.go.
"""

import threading

import pytest

_gwpick = pytest.importorskip("production_stack_amd._gwpick")

PODS = ["http://pod-a:8000", "http://pod-b:8000", "http://pod-c:8000"]


def test_roundrobin_cycles():
    p = _gwpick.NativePicker(128, 128)
    picks = [p.pick_roundrobin(PODS) for _ in range(6)]
    assert picks[:3] == picks[3:]
    assert set(picks) == set(PODS)


def test_prefixaware_matches_python_picker():
    from production_stack_amd.gateway.extproc import Picker

    native = Picker("prefixaware")
    assert native._native is not None, "native picker not built"
    python = Picker("prefixaware", use_native=False)
    prompts = []
    for i in range(50):
        if i % 7 == 0:
            prompts.append("short")  # below chunk -> rr fallback + seed
        else:
            prompts.append("sys" * 80 + f"user{i % 5}" + "x" * (i % 3) * 64)
    for t in prompts:
        assert native.pick(t, PODS) == python.pick(t, PODS)


def test_prefixaware_sticky_after_seed():
    p = _gwpick.NativePicker(128, 128)
    prompt = "a" * 400
    first = p.pick_prefixaware(prompt, PODS)
    for _ in range(10):
        assert p.pick_prefixaware(prompt, PODS) == first
    # longer prompt sharing the prefix routes to the same pod
    assert p.pick_prefixaware(prompt + "b" * 50, PODS) == first


def test_kvaware_best_score_and_fallback():
    p = _gwpick.NativePicker(128, 128)
    assert p.pick_kvaware({PODS[1]: 900, PODS[2]: 100}, PODS) == PODS[1]
    # dead pod's score ignored
    assert p.pick_kvaware({"http://gone:1": 999, PODS[2]: 5}, PODS) == PODS[2]
    # no positive score -> round robin over live pods
    assert p.pick_kvaware({}, PODS) in PODS


def test_remove_endpoint_invalidates_routing():
    p = _gwpick.NativePicker(128, 128)
    prompt = "c" * 512
    first = p.pick_prefixaware(prompt, PODS)
    p.remove_endpoint(first)
    remaining = [x for x in PODS if x != first]
    assert p.pick_prefixaware(prompt, remaining) in remaining


def test_trie_thread_safety():
    trie = _gwpick.PrefixTrie(128)
    errs = []

    def worker(tid):
        try:
            for i in range(300):
                text = f"t{tid}" * 40 + "z" * (i % 4) * 64
                trie.insert(text, PODS[tid % len(PODS)])
                trie.longest_prefix_match(text, PODS)
                if i % 97 == 0:
                    trie.remove_endpoint(PODS[(tid + 1) % len(PODS)])
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
