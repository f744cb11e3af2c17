"""LoRA execution: adapters change outputs, multi-adapter batches stay
isolated, unload restores base behaviour."""

import pytest
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.lora import (
    LoRAAdapter,
    save_synthetic_adapter,
)
from production_stack_amd.engine.sampling import SamplingParams

PROMPT = list(range(50, 90))


def make_engine():
    cfg = EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        cache=CacheConfig(num_gpu_blocks=128, block_size=16,
                          enable_prefix_caching=False),
        scheduler=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=512),
    )
    return LLMEngine(cfg, device="cpu")


def _gen(eng, lora=None, rid="r"):
    p = SamplingParams(max_tokens=6, temperature=0.0, ignore_eos=True)
    eng.add_request(rid, PROMPT, p, lora_name=lora)
    out = []
    while eng.has_unfinished():
        for r in eng.step():
            out.extend(r.new_token_ids)
    return out


def test_adapter_loads_and_changes_output(tmp_path):
    eng = make_engine()
    base = _gen(eng, rid="base")
    adir = str(tmp_path / "a1")
    save_synthetic_adapter(adir, hidden=128, q_size=128, kv_size=64,
                           num_layers=2, seed=3)
    eng.load_lora("a1", adir)
    ad = eng.lora_adapters["a1"]
    assert isinstance(ad, LoRAAdapter)
    assert set(ad.layers) == {0, 1}
    with_lora = _gen(eng, lora="a1", rid="lora")
    assert with_lora != base  # the adapter must actually act
    base2 = _gen(eng, rid="base2")
    assert base2 == base  # base model untouched


def test_unknown_adapter_rejected():
    eng = make_engine()
    with pytest.raises(ValueError):
        eng.add_request(
            "x", PROMPT, SamplingParams(max_tokens=2), lora_name="nope"
        )


def test_mixed_adapter_batch_isolated(tmp_path):
    """base + adapter requests in ONE batch: each matches its solo run."""
    a1 = str(tmp_path / "a1")
    a2 = str(tmp_path / "a2")
    save_synthetic_adapter(a1, hidden=128, q_size=128, kv_size=64,
                           num_layers=2, seed=11)
    save_synthetic_adapter(a2, hidden=128, q_size=128, kv_size=64,
                           num_layers=2, seed=22)

    solo = {}
    for name, path in (("base", None), ("a1", a1), ("a2", a2)):
        eng = make_engine()
        if path:
            eng.load_lora(name, path)
        solo[name] = _gen(eng, lora=None if path is None else name)

    eng = make_engine()
    eng.load_lora("a1", a1)
    eng.load_lora("a2", a2)
    p = SamplingParams(max_tokens=6, temperature=0.0, ignore_eos=True)
    eng.add_request("rb", PROMPT, p, lora_name=None)
    eng.add_request("r1", PROMPT, p, lora_name="a1")
    eng.add_request("r2", PROMPT, p, lora_name="a2")
    outs = {"rb": [], "r1": [], "r2": []}
    while eng.has_unfinished():
        for r in eng.step():
            outs[r.request_id].extend(r.new_token_ids)
    assert outs["rb"] == solo["base"]
    assert outs["r1"] == solo["a1"]
    assert outs["r2"] == solo["a2"]
    assert outs["r1"] != outs["rb"]
    assert outs["r1"] != outs["r2"]
