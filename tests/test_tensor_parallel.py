"""Tensor parallelism over torch.distributed (gloo, world_size 2, CPU).

Verifies the TP-sharded model (q/kv heads + MLP split, all-reduced o/down
projections) produces the same greedy tokens as the TP=1 model with the
same weights — the multi-GPU path is correct by construction before it ever
touches RCCL.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
    ParallelConfig,
)

PROMPT = list(range(20, 84))  # 64 tokens


def _config(weights_path, tp, async_sched=False):
    return EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        weights_path=weights_path,
        async_scheduling=async_sched,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=256),
        parallel=ParallelConfig(tensor_parallel_size=tp),
    )


def _tp_worker(rank, world, weights_path, port, q, async_sched=False):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.parallel import state as pstate

    try:
        eng = LLMEngine(_config(weights_path, world, async_sched),
                        device="cpu")
        out = eng.generate(
            [PROMPT],
            SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True),
        )["offline-0"]
        if rank == 0:
            q.put(("ok", out))
    except Exception as e:  # pragma: no cover
        if rank == 0:
            q.put(("err", repr(e)))
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()
        pstate.destroy()


@pytest.mark.timeout(180)
def test_tp2_matches_tp1(tmp_path):
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.weights import save_hf_safetensors

    # build reference TP=1 engine with random weights; export them
    eng1 = LLMEngine(_config(None, 1), device="cpu")
    wdir = str(tmp_path / "w")
    save_hf_safetensors(eng1.runner.model, wdir)
    want = eng1.generate(
        [PROMPT], SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    )["offline-0"]

    # reload via the safetensors loader (TP=1) to validate the loader alone
    eng1b = LLMEngine(_config(wdir, 1), device="cpu")
    got1 = eng1b.generate(
        [PROMPT], SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    )["offline-0"]
    assert got1 == want

    # TP=2 over gloo
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = []
    port = 29541
    for rank in range(2):
        p = ctx.Process(target=_tp_worker, args=(rank, 2, wdir, port, q))
        p.start()
        procs.append(p)
    status, out = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", out
    assert out == want, f"TP2 {out} != TP1 {want}"


@pytest.mark.timeout(180)
def test_tp2_lockstep_with_async_scheduling(tmp_path):
    """TP=2 lockstep generate with async scheduling: every rank runs the
    same one-step-lagged pipeline (device sampling is deterministic, so
    the ranks stay in lockstep) and must match the TP=1 sync output."""
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.weights import save_hf_safetensors
    import torch.multiprocessing as mp

    ref = LLMEngine(_config(None, 1), device="cpu")
    wdir = str(tmp_path / "w")
    save_hf_safetensors(ref.runner.model, wdir)
    want = ref.generate(
        [PROMPT],
        SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True),
    )["offline-0"]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, wdir, 29791, q, True))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, out = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", out
    assert out == want, f"TP2-async {out} != TP1 {want}"
