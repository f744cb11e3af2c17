"""Tensor-parallel serving coordination.

With TP > 1 every rank must execute the same forward (the per-layer
all-reduces are collective); only rank 0 runs the scheduler, HTTP server and
sampler. Rank 0 broadcasts the per-step batch (tokens + attention metadata)
to the TP group before executing; worker ranks replay the forward and
discard the output. Mirrors engine/pipeline.py's payload format.
"""

from __future__ import annotations

import logging

import torch
import torch.distributed as dist

from production_stack_amd.engine.pipeline import (
    _meta_to_payload,
    _payload_to_meta,
)

logger = logging.getLogger("engine.tp")


class TPCoordinator:
    def __init__(self, runner, rank: int, size: int) -> None:
        self.runner = runner
        self.rank = rank
        self.size = size
        self.device = runner.device

    # ---- rank 0 -------------------------------------------------------
    @torch.no_grad()
    def broadcast_step(self, token_t: torch.Tensor, meta) -> None:
        payload = {
            "op": "step",
            "meta": _meta_to_payload(meta),
            "tokens": token_t.cpu(),
        }
        dist.broadcast_object_list([payload], src=0)

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
                logger.info("tp worker rank %d stopping", self.rank)
                return
            meta = _payload_to_meta(payload["meta"], self.device)
            tokens = payload["tokens"].to(self.device)
            model(tokens, meta, self.runner.kv_caches)
