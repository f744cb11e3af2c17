"""Pipeline parallelism: stage-partitioned engine over torch.distributed.

Capability parity: the reference exposes PP through KubeRay +
`--pipeline-parallel-size` (reference helm/templates/ray-cluster.yaml:716);
here it is native: rank 0 runs the scheduler and the first stage, per-step
metadata is broadcast to all stages, activations (hidden + residual) move
stage-to-stage with send/recv (RCCL p2p over xGMI on GPU, gloo on CPU), and
the last stage samples and returns tokens to rank 0.

With `ParallelConfig.pp_microbatches > 1` each step's batch is split at
sequence boundaries into M microbatches that are issued back-to-back:
rank 0 computes and sends microbatch i's stage-0 activations before
waiting on anything, so stage k runs microbatch i while stage k-1 runs
i+1 (classic in-flight pipelining — GPU work overlaps because RCCL
send/recv enqueue on streams without host-blocking on completion).
Sampled tokens are collected per microbatch after the last send.
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch
import torch.distributed as dist

from production_stack_amd.engine.models.llama import BatchMeta
from production_stack_amd.engine.sampling import SamplingParams

logger = logging.getLogger("engine.pipeline")


def _meta_to_payload(meta: BatchMeta) -> dict:
    def cpu(t):
        return t.cpu() if t is not None else None

    return {
        "positions": cpu(meta.positions),
        "slot_mapping": cpu(meta.slot_mapping),
        "num_prefill_tokens": meta.num_prefill_tokens,
        "prefill_token_seq": cpu(meta.prefill_token_seq),
        "prefill_token_pos": cpu(meta.prefill_token_pos),
        "prefill_block_tables": cpu(meta.prefill_block_tables),
        "num_decode_seqs": meta.num_decode_seqs,
        "decode_seq_lens": cpu(meta.decode_seq_lens),
        "decode_block_tables": cpu(meta.decode_block_tables),
        "prefill_tiles": cpu(meta.prefill_tiles),
    }


def _payload_to_meta(p: dict, device: torch.device) -> BatchMeta:
    def dev(t):
        return t.to(device) if t is not None else None

    return BatchMeta(
        positions=dev(p["positions"]),
        slot_mapping=dev(p["slot_mapping"]),
        num_prefill_tokens=p["num_prefill_tokens"],
        prefill_token_seq=dev(p["prefill_token_seq"]),
        prefill_token_pos=dev(p["prefill_token_pos"]),
        prefill_block_tables=dev(p["prefill_block_tables"]),
        num_decode_seqs=p["num_decode_seqs"],
        decode_seq_lens=dev(p["decode_seq_lens"]),
        decode_block_tables=dev(p["decode_block_tables"]),
        prefill_tiles=dev(p["prefill_tiles"]),
    )


class PipelineCoordinator:
    def __init__(self, runner, pp_rank: int, pp_size: int) -> None:
        self.runner = runner
        self.rank = pp_rank
        self.size = pp_size
        self.device = runner.device
        self.hidden = runner.model_cfg.hidden_size

    # ---- rank 0 (driver) ---------------------------------------------
    @torch.no_grad()
    def drive(
        self,
        token_t: torch.Tensor,
        meta: BatchMeta,
        sample_rows: torch.Tensor,
        params: List[SamplingParams],
    ) -> torch.Tensor:
        """Run one step through the pipeline; returns sampled token ids."""
        return self.drive_many([(token_t, meta, sample_rows, params)])

    @torch.no_grad()
    def drive_many(self, microbatches) -> torch.Tensor:
        """Issue all microbatches through the pipeline before collecting
        any sampled tokens; returns token ids concatenated in mb order."""
        mb_payloads = []
        for token_t, meta, sample_rows, params in microbatches:
            mb_payloads.append({
                "meta": _meta_to_payload(meta),
                "tokens": token_t.cpu(),
                "sample_rows": sample_rows.cpu(),
                "params": [
                    (p.greedy, p.temperature, p.top_p, p.top_k)
                    for p in params
                ],
            })
        dist.broadcast_object_list([{"op": "step", "mbs": mb_payloads}],
                                   src=0)
        # isend: a blocking send of mb i+1 would deadlock against the last
        # stage blocking on its sampled-token send for mb i (gloo
        # rendezvous); isend also lets RCCL enqueue all stage-0 work so the
        # stages genuinely overlap.
        works, acts = [], []
        for token_t, meta, _, _ in microbatches:
            act = self.runner.model(
                token_t, meta, self.runner.kv_caches
            ).contiguous()
            works.append(dist.isend(act, dst=1))
            acts.append(act)  # keep alive until matched
        outs = []
        for _, _, _, params in microbatches:
            if not params:
                continue
            sampled = torch.empty(len(params), dtype=torch.long)
            dist.recv(sampled, src=self.size - 1)
            outs.append(sampled)
        for w in works:
            w.wait()
        del acts
        if not outs:
            return torch.empty(0, dtype=torch.long)
        return torch.cat(outs)

    def stop_workers(self) -> None:
        try:
            dist.broadcast_object_list([{"op": "stop"}], src=0)
        except RuntimeError:
            pass

    # ---- ranks 1..size-1 ----------------------------------------------
    @torch.no_grad()
    def worker_loop(self) -> None:
        model = self.runner.model
        while True:
            box = [None]
            dist.broadcast_object_list(box, src=0)
            payload = box[0]
            if payload is None or payload.get("op") == "stop":
                logger.info("pp worker rank %d stopping", self.rank)
                return
            for mb in payload["mbs"]:
                meta = _payload_to_meta(mb["meta"], self.device)
                T = mb["tokens"].shape[0]
                act = torch.empty(
                    (2, T, self.hidden), dtype=torch.bfloat16,
                    device=self.device
                )
                dist.recv(act, src=self.rank - 1)
                hidden, residual = act[0], act[1]
                out = model(hidden, meta, self.runner.kv_caches, residual)
                if not model.is_last:
                    dist.send(out.contiguous(), dst=self.rank + 1)
                    continue
                params_raw = mb["params"]
                if not params_raw:
                    continue
                rows = mb["sample_rows"].to(self.device)
                logits = model.compute_logits(out[rows])
                params = [
                    SamplingParams(
                        temperature=t, top_p=tp, top_k=tk, max_tokens=1
                    )
                    for (_, t, tp, tk) in params_raw
                ]
                sampled = self.runner.sample_params(logits, params)
                dist.send(sampled.cpu(), dst=0)
