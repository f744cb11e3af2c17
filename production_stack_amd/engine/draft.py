"""Draft-model speculative decoding (vLLM `--speculative-model`
analogue; the reference stack exposes it through the engine flags it
renders, deployment-vllm-multi.yaml argv surface).

A smaller model of the same family proposes K tokens autoregressively;
the target verifies them in one multi-token MFMA chunk (scheduler's
speculative verification path). Acceptance in model_runner:

- greedy target: exact match against the target's own argmax (lossless);
- stochastic target: full Leviathan rejection sampling using the draft's
  per-position proposal distribution q — accept d w.p. min(1, p(d)/q(d)),
  else emit a sample from normalize(max(p - q, 0)); the emitted token is
  marginally ~ p either way.

KV layout trick: the draft runner allocates its own KV tensors with the
SAME block count and block size as the target and indexes them through
the target's block tables, so the draft needs no block manager of its
own. Draft-side KV written at speculative positions that get rejected is
stale — harmless, because `Sequence._draft_progress` only advances over
tokens that became real, and the next catch-up pass recomputes exactly
those positions with the real tokens (overwriting the stale entries).
Preemption resets `_draft_progress` together with `num_computed`
(sequence.reset_for_recompute), which also invalidates the draft KV with
the old block table.
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional, Tuple

import torch

from production_stack_amd.engine.config import EngineConfig, ParallelConfig


class DraftModelProposer:
    def __init__(self, config: EngineConfig, device: torch.device,
                 num_blocks: int) -> None:
        from production_stack_amd.engine.model_runner import ModelRunner

        draft_cfg = dataclasses.replace(
            config,
            model=config.speculative_model,
            weights_path=config.speculative_weights_path,
            quantization=None,
            enable_lora=False,
            async_scheduling=False,
            parallel=ParallelConfig(),
        )
        self.model_cfg = draft_cfg.model_config()
        self.k = config.scheduler.num_speculative_tokens
        self.max_model_len = config.max_model_len
        self.runner = ModelRunner(draft_cfg, self.model_cfg, device)
        self.runner.allocate_kv_cache(num_blocks)
        self._gen = torch.Generator(device="cpu").manual_seed(
            config.seed + 777
        )

    def load_target_weights(self, target_model) -> None:
        """Testing hook: same-architecture draft copies target weights
        (acceptance becomes ~100% on greedy)."""
        self.runner.model.load_state_dict(target_model.state_dict())

    @torch.no_grad()
    def propose(self, seq, bm) -> Tuple[List[int], Optional[list]]:
        """Returns (draft_tokens, per-token proposal distributions or None
        for greedy drafts)."""
        from production_stack_amd.engine.scheduler import (
            ScheduledSeq,
            SchedulerOutput,
        )

        if getattr(seq, "mm_embeds", None):
            return [], None  # draft has no multimodal projector
        k = min(self.k, self.max_model_len - seq.num_tokens - 1)
        if k <= 0:
            return [], None
        # speculative positions need slots before the draft writes KV
        if not bm.ensure_capacity(seq, seq.num_tokens + k):
            return [], None
        progress = getattr(seq, "_draft_progress", 0)
        saved_nc = seq.num_computed
        saved_out = len(seq.output_token_ids)
        greedy = seq.params.greedy
        drafts: List[int] = []
        qs: Optional[list] = None if greedy else []
        try:
            seq.num_computed = progress
            while len(drafts) < k:
                chunk = seq.num_tokens - seq.num_computed
                if chunk <= 0:
                    break
                out1 = SchedulerOutput(
                    scheduled=[ScheduledSeq(seq, chunk)]
                )
                token_t, meta, s_seqs, rows_t = self.runner.prepare(
                    out1, bm
                )
                hidden = self.runner.model(
                    token_t, meta, self.runner.kv_caches
                )
                logits = self.runner.model.compute_logits(
                    hidden[rows_t]
                )[0]
                seq.num_computed += chunk
                if greedy:
                    d = int(torch.argmax(logits))
                else:
                    q = self.runner._filtered_probs(logits, seq.params)
                    d = int(torch.multinomial(q, 1, generator=self._gen))
                    qs.append(q)
                drafts.append(d)
                if len(drafts) < k:
                    seq.output_token_ids.append(d)
        finally:
            seq.num_computed = saved_nc
            del seq.output_token_ids[saved_out:]
        # positions [old_progress, num_tokens) now hold real-token KV
        seq._draft_progress = seq.num_tokens
        return drafts, qs
