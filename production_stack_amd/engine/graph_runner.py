"""hipGraph capture for decode-only steps.

A decode step of Llama-3-8B is ~350 kernel launches of mostly <100 us
kernels; eager launch overhead was ~25% of step time (profiles/ run1). The
whole decode forward (embedding -> layers -> final norm -> logits) is
captured once per batch-size bucket into a hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) against static input buffers and replayed each step.

Padding rows are made harmless by: slot_mapping = -1 (KV append skips),
seq_len = 1 with block_table row 0 (reads one garbage block, output row is
discarded), token id 0, position 0.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from production_stack_amd.engine.models.llama import BatchMeta

logger = logging.getLogger("engine.graphs")

BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 384, 512]


class DecodeGraphRunner:
    def __init__(
        self,
        model,
        kv_caches,
        max_batch: int,
        max_blocks_per_seq: int,
        device: torch.device,
    ) -> None:
        self.model = model
        self.kv_caches = kv_caches
        self.device = device
        self.max_blocks = max_blocks_per_seq
        self.buckets = [b for b in BUCKETS if b <= max_batch]
        if not self.buckets or self.buckets[-1] < max_batch:
            self.buckets.append(max_batch)
        B = self.buckets[-1]
        self.max_batch = B

        dev = device
        self.tokens = torch.zeros(B, dtype=torch.long, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.slots = torch.full((B,), -1, dtype=torch.long, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.block_tables = torch.zeros(
            (B, max_blocks_per_seq), dtype=torch.int32, device=dev
        )
        # per-row LoRA slot (-1 = none); captured into the graphs when the
        # model carries BGMV slot stacks
        self.lora_idx = torch.full((B,), -1, dtype=torch.int32, device=dev)
        # pinned host staging
        self.h_tokens = torch.zeros(B, dtype=torch.long, pin_memory=True)
        self.h_positions = torch.zeros(B, dtype=torch.int32, pin_memory=True)
        self.h_slots = torch.full((B,), -1, dtype=torch.long, pin_memory=True)
        self.h_seq_lens = torch.ones(B, dtype=torch.int32, pin_memory=True)
        self.h_block_tables = torch.zeros(
            (B, max_blocks_per_seq), dtype=torch.int32, pin_memory=True
        )
        self.h_lora_idx = torch.full((B,), -1, dtype=torch.int32,
                                     pin_memory=True)
        self.use_lora = getattr(model, "lora_slots", None) is not None

        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.logits: Dict[int, torch.Tensor] = {}
        self.pool = None

    def _meta(self, b: int) -> BatchMeta:
        return BatchMeta(
            positions=self.positions[:b],
            slot_mapping=self.slots[:b],
            num_prefill_tokens=0,
            prefill_token_seq=None,
            prefill_token_pos=None,
            prefill_block_tables=None,
            num_decode_seqs=b,
            decode_seq_lens=self.seq_lens[:b],
            decode_block_tables=self.block_tables[:b],
            lora_idx=self.lora_idx[:b] if self.use_lora else None,
        )

    @torch.no_grad()
    def capture_all(self) -> None:
        torch.cuda.synchronize()
        for b in reversed(self.buckets):  # big first: pool sized once
            self._capture(b)
        torch.cuda.synchronize()
        logger.info(
            "captured %d decode graphs (buckets %s)",
            len(self.graphs),
            self.buckets,
        )

    @torch.no_grad()
    def _capture(self, b: int) -> None:
        meta = self._meta(b)
        # warmup (allocates GEMM workspaces outside capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                hidden = self.model(self.tokens[:b], meta, self.kv_caches)
                self.model.compute_logits(hidden)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self.pool):
            hidden = self.model(self.tokens[:b], meta, self.kv_caches)
            logits = self.model.compute_logits(hidden)
        if self.pool is None:
            self.pool = g.pool()
        self.graphs[b] = g
        self.logits[b] = logits

    def bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if b >= n:
                return b
        return None

    @torch.no_grad()
    def run(
        self,
        tokens: np.ndarray,
        positions: np.ndarray,
        slots: np.ndarray,
        seq_lens: np.ndarray,
        block_tables: List[List[int]],
        lora_idx: Optional[np.ndarray] = None,
        prev_fill=None,  # (rows, src_rows, prev_sampled_dev) async gather
    ) -> torch.Tensor:
        n = len(tokens)
        b = self.bucket_for(n)
        assert b is not None
        # fill pinned staging (pad region reset)
        self.h_tokens[:n] = torch.from_numpy(tokens)
        self.h_tokens[n:b] = 0
        self.h_positions[:n] = torch.from_numpy(positions)
        self.h_positions[n:b] = 0
        self.h_slots[:n] = torch.from_numpy(slots)
        self.h_slots[n:b] = -1
        self.h_seq_lens[:n] = torch.from_numpy(seq_lens)
        self.h_seq_lens[n:b] = 1
        hbt = self.h_block_tables.numpy()
        for i, bt in enumerate(block_tables):
            hbt[i, : len(bt)] = bt
        hbt[n:b, 0] = 0
        if self.use_lora:
            if lora_idx is None:
                self.h_lora_idx[:b] = -1
            else:
                self.h_lora_idx[:n] = torch.from_numpy(lora_idx)
                self.h_lora_idx[n:b] = -1
            self.lora_idx[:b].copy_(self.h_lora_idx[:b], non_blocking=True)
        # H2D into the static buffers
        self.tokens[:b].copy_(self.h_tokens[:b], non_blocking=True)
        self.positions[:b].copy_(self.h_positions[:b], non_blocking=True)
        self.slots[:b].copy_(self.h_slots[:b], non_blocking=True)
        self.seq_lens[:b].copy_(self.h_seq_lens[:b], non_blocking=True)
        self.block_tables[:b].copy_(
            self.h_block_tables[:b], non_blocking=True
        )
        if prev_fill is not None:
            rows, src, prev_dev = prev_fill
            ridx = torch.tensor(rows, dtype=torch.long, device=self.device)
            sidx = torch.tensor(src, dtype=torch.long, device=self.device)
            self.tokens[ridx] = prev_dev[sidx]
        self.graphs[b].replay()
        return self.logits[b][:n]
