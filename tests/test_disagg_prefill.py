"""Disaggregated prefill: KV blocks move prefill-engine -> decode-engine
over torch.distributed p2p (gloo here, RCCL/xGMI on the GPU box) with the
TCP side channel, and the decode engine adopts them as prefix-cache hits.

2-process CPU test (BASELINE config #4's mechanism at small scale).
"""

import asyncio
import os

import pytest
import torch.multiprocessing as mp

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)

PROMPT = list(range(300, 380))  # 80 tokens: 5 full blocks, 4 transferable
MASTER_PORT = 29661
SIDE_PORT = 29662


def _config(weights_path):
    return EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        weights_path=weights_path,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=256),
    )


def _prefill_proc(weights_path, q_out, done_evt):
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.parallel.kv_transfer import KVTransferService

    eng = LLMEngine(_config(weights_path), device="cpu")
    svc = KVTransferService(
        eng, kv_rank=0, kv_world=2, master_port=MASTER_PORT,
        side_port=SIDE_PORT, backend="gloo",
    )
    # prefill request: 1 token, like the router's orchestrated P-phase
    eng.generate(
        [PROMPT], SamplingParams(max_tokens=1, temperature=0.0,
                                 ignore_eos=True)
    )
    params = svc.register_prefilled("req-1", PROMPT)
    assert len(params["remote_block_ids"]) == 5
    q_out.put(params)

    async def serve():
        await svc.start_side_channel()
        await asyncio.to_thread(done_evt.wait, 60)
        await svc.stop()

    asyncio.run(serve())


def _decode_proc(weights_path, q_in, q_out, done_evt):
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.parallel.kv_transfer import KVTransferService

    eng = LLMEngine(_config(weights_path), device="cpu")
    svc = KVTransferService(
        eng, kv_rank=1, kv_world=2, master_port=MASTER_PORT,
        side_port=SIDE_PORT + 1, backend="gloo",
    )
    params = q_in.get(timeout=60)

    async def pull():
        return await svc.pull_into_prefix_cache(
            params["remote_request_id"],
            PROMPT,
            params["remote_host"],
            params["remote_port"],
            params["remote_engine_id"],
        )

    adopted = asyncio.run(pull())
    out = eng.generate(
        [PROMPT], SamplingParams(max_tokens=6, temperature=0.0,
                                 ignore_eos=True)
    )["offline-0"]
    q_out.put(
        {
            "adopted": adopted,
            "tokens": out,
            "prefix_hits": eng.block_manager.prefix_hits,
        }
    )
    done_evt.set()


@pytest.mark.timeout(240)
def test_disaggregated_prefill_kv_transfer(tmp_path):
    from production_stack_amd.engine.engine import LLMEngine
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.weights import save_hf_safetensors

    ref = LLMEngine(_config(None), device="cpu")
    wdir = str(tmp_path / "w")
    save_hf_safetensors(ref.runner.model, wdir)
    want = ref.generate(
        [PROMPT], SamplingParams(max_tokens=6, temperature=0.0,
                                 ignore_eos=True)
    )["offline-0"]

    ctx = mp.get_context("spawn")
    params_q = ctx.Queue()
    result_q = ctx.Queue()
    done = ctx.Event()
    p0 = ctx.Process(target=_prefill_proc, args=(wdir, params_q, done))
    p1 = ctx.Process(
        target=_decode_proc, args=(wdir, params_q, result_q, done)
    )
    p0.start()
    p1.start()
    res = result_q.get(timeout=200)
    p1.join(timeout=60)
    p0.join(timeout=60)
    # 4 of the 5 full prompt blocks are adoptable (last token recomputed)
    assert res["adopted"] == 64, res
    assert res["prefix_hits"] >= 4, res
    assert res["tokens"] == want, f"{res['tokens']} != {want}"
