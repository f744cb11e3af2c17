"""TP serving mode: rank 0 schedules + broadcasts, rank 1 replays forwards
(the HTTP-driven topology, here exercised directly over gloo)."""

import os

import pytest
import torch.multiprocessing as mp

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ParallelConfig,
    SchedulerConfig,
)

PROMPT = list(range(60, 124))
PORT = 29881


def _config(weights_path, tp):
    return EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        weights_path=weights_path,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=256),
        parallel=ParallelConfig(tensor_parallel_size=tp),
    )


def _worker(rank, world, weights_path, q):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(PORT),
        RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    import torch.distributed as dist

    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams

    try:
        eng = LLMEngine(_config(weights_path, world), device="cpu")
        if eng.is_tp_worker:
            eng.run_tp_worker()
            return
        eng.runner.tp_serving = True
        out = eng.generate(
            [PROMPT],
            SamplingParams(max_tokens=6, temperature=0.0, ignore_eos=True),
        )["offline-0"]
        eng.runner.tp_coord.stop_workers()
        q.put(("ok", out))
    except Exception as e:
        q.put(("err", repr(e)))
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_tp_serving_mode(tmp_path):
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
        ctx.Process(target=_worker, args=(r, 2, wdir, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    status, out = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", out
    assert out == want
