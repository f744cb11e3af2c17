"""Pipeline parallelism: 2 stages over gloo (CPU) must match the 1-process
model with the same weights."""

import os

import pytest
import torch.multiprocessing as mp

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ParallelConfig,
    SchedulerConfig,
)

PROMPT = list(range(40, 100))  # 60 tokens
PORT = 29771


def _config(weights_path, pp, mb=1):
    return EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        weights_path=weights_path,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=256),
        parallel=ParallelConfig(pipeline_parallel_size=pp,
                                pp_microbatches=mb),
    )


def _pp_worker(rank, world, weights_path, q, mb=1, prompts=None, port=PORT):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    import torch.distributed as dist

    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams

    try:
        eng = LLMEngine(_config(weights_path, world, mb), device="cpu")
        if eng.is_pp_worker:
            eng.run_pp_worker()
            return
        outs = eng.generate(
            prompts or [PROMPT],
            SamplingParams(max_tokens=6, temperature=0.0, ignore_eos=True),
        )
        eng.stop_pp_workers()
        q.put(("ok", outs if prompts else outs["offline-0"]))
    except Exception as e:
        q.put(("err", repr(e)))
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_pp2_matches_single_process(tmp_path):
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.weights import save_hf_safetensors

    ref = LLMEngine(_config(None, 1), device="cpu")
    wdir = str(tmp_path / "w")
    save_hf_safetensors(ref.runner.model, wdir)
    want = ref.generate(
        [PROMPT], SamplingParams(max_tokens=6, temperature=0.0,
                                 ignore_eos=True)
    )["offline-0"]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_pp_worker, args=(r, 2, wdir, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    status, out = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", out
    assert out == want, f"PP2 {out} != single {want}"


@pytest.mark.timeout(240)
def test_pp2_microbatched_matches_single_process(tmp_path):
    """pp_microbatches=2: batched multi-seq steps split at sequence
    boundaries and pipelined in flight must match the 1-process engine."""
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.weights import save_hf_safetensors

    prompts = [PROMPT, list(range(5, 45)), list(range(150, 171))]
    ref = LLMEngine(_config(None, 1), device="cpu")
    wdir = str(tmp_path / "w")
    save_hf_safetensors(ref.runner.model, wdir)
    want = ref.generate(
        prompts, SamplingParams(max_tokens=6, temperature=0.0,
                                ignore_eos=True)
    )

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_pp_worker,
                    args=(r, 2, wdir, q, 2, prompts, PORT + 3))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, out = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", out
    assert out == want, f"PP2-mb2 {out} != single {want}"
