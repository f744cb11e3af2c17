"""Model runner: batch preparation, forward execution, sampling.

Batch layout per step: [all prefill-chunk tokens..., all decode tokens...].
Single-token chunks run through the decode paged-attention kernel (its
attention pattern — last token over full context — is exactly a 1-token
chunk), multi-token chunks through the chunked-prefill kernel.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from production_stack_amd import ops
from production_stack_amd.engine.block_manager import BlockManager
from production_stack_amd.engine.config import EngineConfig, ModelConfig
from production_stack_amd.engine.models.llama import BatchMeta, LlamaForCausalLM
from production_stack_amd.engine.scheduler import SchedulerOutput
from production_stack_amd.engine.sequence import Sequence


class ModelRunner:
    def __init__(
        self,
        config: EngineConfig,
        model_cfg: ModelConfig,
        device: torch.device,
    ) -> None:
        self.config = config
        self.model_cfg = model_cfg
        self.device = device
        tp = config.parallel.tensor_parallel_size
        pp = getattr(config.parallel, "pipeline_parallel_size", 1)
        self.pp_size = pp
        self.pp_rank = 0
        if pp > 1:
            import torch.distributed as dist

            self.pp_rank = dist.get_rank() if dist.is_initialized() else 0
        self.model = LlamaForCausalLM(
            model_cfg, tp=tp, pp_rank=self.pp_rank, pp_size=pp
        ).to(device)
        self.pipeline = None
        if pp > 1:
            from production_stack_amd.engine.pipeline import (
                PipelineCoordinator,
            )

            self.pipeline = PipelineCoordinator(self, self.pp_rank, pp)
        self.tp_coord = None
        if tp > 1:
            import torch.distributed as dist

            from production_stack_amd.engine.tp_worker import TPCoordinator
            from production_stack_amd.parallel import state as ps

            # only needed when driven over HTTP (rank 0 schedules); tests
            # that run lockstep generate() on every rank leave this unused
            if dist.is_initialized():
                self.tp_coord = TPCoordinator(self, ps.tp_rank(), tp)
        if config.weights_path:
            from production_stack_amd.engine.weights import load_safetensors

            load_safetensors(self.model, config.weights_path)
        else:
            self.model.random_init(config.seed)
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self.num_blocks = 0
        self.graphs = None
        self.lora_registry = None  # set by the engine (name -> adapter)
        self.lora_slots = None
        if (config.enable_lora
                and config.parallel.tensor_parallel_size == 1):
            from production_stack_amd.engine.lora import LoRASlotManager

            self.lora_slots = LoRASlotManager(
                self.model_cfg, config.max_loras, config.max_lora_rank,
                self.device,
            )
            self.model.lora_slots = self.lora_slots
            for layer in self.model.layers:
                layer.lora_slots = self.lora_slots
        if config.quantization == "fp8":
            self.model.quantize_fp8()
        self._generator = torch.Generator(device="cpu").manual_seed(
            config.seed + 12345
        )
        # request_id -> logprob of the most recent sampled token (only for
        # requests that asked for logprobs; read by engine.step)
        self.last_logprobs: Dict[str, float] = {}
        # request_id -> [(token_id, logprob) x topN] for the latest
        # sampled position (params.logprobs > 0 requests)
        self.last_top_logprobs: Dict[str, list] = {}
        # request_id -> [None, lp1, lp2...] accumulated across prefill
        # chunks for params.prompt_logprobs requests
        self.prompt_logprobs: Dict[str, list] = {}
        self.spec_proposed = 0
        self.spec_accepted = 0
        # async-scheduling state (execute_async/finalize_async)
        self._prev_tokens_dev = None
        self._prev_rows: Dict[str, int] = {}
        self._async_host = [None, None]
        self._async_host_i = 0

    # ------------------------------------------------------------------
    @property
    def kv_elem_bytes(self) -> int:
        return (
            1
            if self.config.cache.kv_cache_dtype in ("fp8", "fp8_e4m3")
            else 2
        )

    def profile_num_blocks(self) -> int:
        """Size the KV cache from free HBM after weights (288 GB per GPU)."""
        cache_cfg = self.config.cache
        if cache_cfg.num_gpu_blocks is not None:
            return cache_cfg.num_gpu_blocks
        if self.device.type != "cuda":
            return 512
        free, total = torch.cuda.mem_get_info(self.device)
        budget = int(
            total * cache_cfg.gpu_memory_utilization
            - (total - free)
        )
        per_block = self.model.kv_bytes_per_block(cache_cfg.block_size)
        if self.kv_elem_bytes == 1:
            per_block //= 2
        n = max(budget // per_block, 16)
        return int(n)

    def allocate_kv_cache(self, num_blocks: int) -> None:
        self.num_blocks = num_blocks
        cfg = self.model_cfg
        bs = self.config.cache.block_size
        kh = self.model.kv_heads
        kv_dt = self.config.cache.kv_cache_dtype
        dtype = (
            torch.float8_e4m3fn
            if kv_dt in ("fp8", "fp8_e4m3")
            else torch.bfloat16
        )
        self.kv_caches = []
        for _ in range(self.model.num_local_layers):
            k = torch.zeros(
                (num_blocks, kh, bs, cfg.head_dim),
                dtype=dtype,
                device=self.device,
            )
            v = torch.zeros_like(k)
            self.kv_caches.append((k, v))

    def capture_decode_graphs(self, max_batch: int) -> None:
        """hipGraph-capture decode-only steps (see graph_runner.py)."""
        if (
            self.device.type != "cuda"
            or self.config.enforce_eager
            or self.pp_size > 1
            or self.config.parallel.tensor_parallel_size > 1
            # MoE routing is data-dependent (per-expert row gathers):
            # a captured decode graph would freeze one routing pattern
            or self.model_cfg.num_experts > 0
        ):
            return
        from production_stack_amd.engine.graph_runner import DecodeGraphRunner
        from production_stack_amd.ops import gemm_policy

        # fp8-quantized GEMMs run through scaled_mm: nothing to autotune
        if self.config.quantization != "fp8":
            try:
                layer0 = self.model.layers[0]
                weights = [
                    layer0.qkv_proj, layer0.o_proj, layer0.gate_up_proj,
                    layer0.down_proj,
                ]
                from production_stack_amd.engine.graph_runner import BUCKETS

                buckets = [b for b in BUCKETS if b <= max_batch]
                gemm_policy.tune(weights, buckets, self.device)
            except Exception:
                import logging

                logging.getLogger("engine.runner").exception(
                    "gemm autotune failed; using hipBLASLt everywhere"
                )

        max_blocks = (
            self.config.max_model_len + self.config.cache.block_size - 1
        ) // self.config.cache.block_size
        try:
            runner = DecodeGraphRunner(
                self.model, self.kv_caches, max_batch, max_blocks, self.device
            )
            runner.capture_all()
            self.graphs = runner
        except Exception:
            import logging

            logging.getLogger("engine.runner").exception(
                "hipGraph capture failed; continuing in eager mode"
            )
            self.graphs = None

    # ------------------------------------------------------------------
    def prepare(
        self, out: SchedulerOutput, bm: BlockManager
    ) -> Tuple[torch.Tensor, BatchMeta, List[Sequence], torch.Tensor]:
        """Returns (token_ids, meta, seqs_to_sample, sample_row_indices)."""
        bs = bm.block_size
        prefills = [s for s in out.scheduled if not s.is_decode]
        decodes = [s for s in out.scheduled if s.is_decode]

        tokens: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        mm_rows: List[int] = []
        mm_parts: List = []
        p_token_seq: List[int] = []
        p_token_pos: List[int] = []
        p_tables: List[List[int]] = []
        p_tiles: List[List[int]] = []  # (seq_row, q_tok0, q_pos0, n_rows)
        sample_rows: List[int] = []
        sample_seqs: List[Sequence] = []

        import numpy as np

        TILE = ops.prefill_tile_rows(
            self.model_cfg.num_q_heads, self.model_cfg.num_kv_heads
        )
        for row, ss in enumerate(prefills):
            seq = ss.seq
            all_ids = seq.token_ids()
            start = seq.num_computed
            end = start + ss.num_tokens
            p_tables.append(seq.block_table)
            q_flat0 = len(tokens)
            for t0 in range(0, ss.num_tokens, TILE):
                p_tiles.append(
                    [row, q_flat0 + t0, start + t0,
                     min(TILE, ss.num_tokens - t0)]
                )
            pos_arr = np.arange(start, end, dtype=np.int64)
            bt_arr = np.asarray(seq.block_table, dtype=np.int64)
            slot_arr = bt_arr[pos_arr // bs] * bs + pos_arr % bs
            if ss.draft_tokens:
                # speculative verification chunk: 1 real token + drafts,
                # logits needed at every position
                tokens.extend([all_ids[start]] + ss.draft_tokens)
                base = len(tokens) - ss.num_tokens
                sample_rows.extend(range(base, len(tokens)))
                sample_seqs.extend([seq] * ss.num_tokens)
            else:
                tokens.extend(all_ids[start:end])
                if end == seq.num_tokens:
                    sample_rows.append(len(tokens) - 1)
                    sample_seqs.append(seq)
            positions.extend(pos_arr.tolist())
            slots.extend(slot_arr.tolist())
            p_token_seq.extend([row] * ss.num_tokens)
            p_token_pos.extend(pos_arr.tolist())
            # multimodal: slice each media embed span into this chunk
            for off, emb in getattr(seq, "mm_embeds", []):
                lo = max(off, start)
                hi = min(off + emb.shape[0], end)
                if lo < hi:
                    mm_rows.extend(range(q_flat0 + (lo - start),
                                         q_flat0 + (hi - start)))
                    mm_parts.append(emb[lo - off:hi - off])

        d_seq_lens: List[int] = []
        d_tables: List[List[int]] = []
        for ss in decodes:
            seq = ss.seq
            pos = seq.num_computed
            # read the one input token directly: token_ids() would build
            # a prompt+output list copy per seq per step (~5 ms/step at
            # 256 long sequences, measured on the CPU profile)
            npr = seq.num_prompt
            tokens.append(
                seq.prompt_token_ids[pos] if pos < npr
                else seq.output_token_ids[pos - npr]
            )
            positions.append(pos)
            slots.append(seq.block_table[pos // bs] * bs + pos % bs)
            d_seq_lens.append(pos + 1)
            d_tables.append(seq.block_table)
            sample_rows.append(len(tokens) - 1)
            sample_seqs.append(seq)

        max_bt = max((len(t) for t in p_tables + d_tables), default=1)

        def pad_tables(tabs: List[List[int]]) -> Optional[torch.Tensor]:
            if not tabs:
                return None
            # numpy staging: a torch.tensor per row measured ~2.5 ms/step
            # at 256 seqs on the host profile
            arr = np.zeros((len(tabs), max_bt), dtype=np.int32)
            for i, t in enumerate(tabs):
                if t:
                    arr[i, : len(t)] = t
            return torch.from_numpy(arr).to(dev, non_blocking=True)

        dev = self.device
        num_prefill_tokens = len(p_token_seq)

        # per-adapter row groups (LoRA): map flat batch rows to adapters
        lora_rows: Dict[str, List[int]] = {}
        row = 0
        for ss in prefills:
            if ss.seq.lora_name:
                lora_rows.setdefault(ss.seq.lora_name, []).extend(
                    range(row, row + ss.num_tokens)
                )
            row += ss.num_tokens
        for ss in decodes:
            if ss.seq.lora_name:
                lora_rows.setdefault(ss.seq.lora_name, []).append(row)
            row += 1
        lora_groups = []
        lora_idx_t = None
        if lora_rows and self.lora_slots is not None:
            import numpy as np

            idx = np.full(row, -1, dtype=np.int32)
            for name, rows in lora_rows.items():
                slot = self.lora_slots.slot_by_name.get(name)
                if slot is not None:
                    idx[rows] = slot
            lora_idx_t = torch.from_numpy(idx).to(dev, non_blocking=True)
        elif lora_rows and self.lora_registry is not None:
            for name, rows in lora_rows.items():
                ad = self.lora_registry.get(name)
                if ad is not None:
                    lora_groups.append(
                        (ad, torch.tensor(rows, dtype=torch.long,
                                          device=dev))
                    )
        elif self.lora_slots is not None:
            lora_idx_t = torch.full((row,), -1, dtype=torch.int32,
                                    device=dev)

        meta = BatchMeta(
            positions=torch.tensor(positions, dtype=torch.int32).to(
                dev, non_blocking=True
            ),
            slot_mapping=torch.tensor(slots, dtype=torch.long).to(
                dev, non_blocking=True
            ),
            num_prefill_tokens=num_prefill_tokens,
            prefill_token_seq=(
                torch.tensor(p_token_seq, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                if p_token_seq
                else None
            ),
            prefill_token_pos=(
                torch.tensor(p_token_pos, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                if p_token_pos
                else None
            ),
            prefill_block_tables=pad_tables(p_tables),
            num_decode_seqs=len(decodes),
            decode_seq_lens=(
                torch.tensor(d_seq_lens, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                if d_seq_lens
                else None
            ),
            decode_block_tables=pad_tables(d_tables),
            prefill_tiles=(
                torch.tensor(p_tiles, dtype=torch.int32).to(
                    dev, non_blocking=True
                )
                if p_tiles
                else None
            ),
            lora_groups=lora_groups,
            lora_idx=lora_idx_t,
            mm_rows=(
                torch.tensor(mm_rows, dtype=torch.long).to(
                    dev, non_blocking=True
                )
                if mm_rows
                else None
            ),
            mm_embeds=(
                torch.cat(mm_parts).to(dev, non_blocking=True)
                if mm_parts
                else None
            ),
        )
        token_t = torch.tensor(tokens, dtype=torch.long).to(
            dev, non_blocking=True
        )
        rows_t = torch.tensor(sample_rows, dtype=torch.long).to(
            dev, non_blocking=True
        )
        return token_t, meta, sample_seqs, rows_t

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _execute_decode_graph(
        self, out: SchedulerOutput, bm: BlockManager, defer_sample=False
    ):
        """Fast path: decode-only step via hipGraph replay. With
        defer_sample the (logits, seqs) pair is returned un-sampled so a
        caller can enqueue more GPU work before the first D2H sync."""
        import numpy as np

        scheduled = out.scheduled
        n = len(scheduled)
        if (
            self.graphs is None
            or n == 0
            or self.graphs.bucket_for(n) is None
            or any(not s.is_decode for s in scheduled)
            or (self.lora_slots is None
                and any(s.seq.lora_name for s in scheduled))
        ):
            return None
        bs = bm.block_size
        tokens = np.empty(n, dtype=np.int64)
        positions = np.empty(n, dtype=np.int32)
        slots = np.empty(n, dtype=np.int64)
        seq_lens = np.empty(n, dtype=np.int32)
        tables: List[List[int]] = []
        seqs: List[Sequence] = []
        lora_idx = None
        if self.lora_slots is not None:
            lora_idx = np.full(n, -1, dtype=np.int32)
        ph_rows: List[int] = []
        ph_src: List[int] = []
        for i, ss in enumerate(scheduled):
            seq = ss.seq
            pos = seq.num_computed
            npr = seq.num_prompt
            # direct read (token_ids() copies prompt+output per seq per
            # step — this is the graph-fill hot loop)
            tok = (seq.prompt_token_ids[pos] if pos < npr
                   else seq.output_token_ids[pos - npr])
            if tok < 0:  # async placeholder: gather from prev device toks
                row = self._prev_rows.get(seq.request_id)
                if row is None or self._prev_tokens_dev is None:
                    return None
                ph_rows.append(i)
                ph_src.append(row)
                tok = 0
            tokens[i] = tok
            positions[i] = pos
            slots[i] = seq.block_table[pos // bs] * bs + pos % bs
            seq_lens[i] = pos + 1
            tables.append(seq.block_table)
            seqs.append(seq)
            if lora_idx is not None and seq.lora_name:
                lora_idx[i] = self.lora_slots.slot_by_name.get(
                    seq.lora_name, -1
                )
        prev_fill = None
        if ph_rows:
            prev_fill = (ph_rows, ph_src, self._prev_tokens_dev)
        logits = self.graphs.run(tokens, positions, slots, seq_lens, tables,
                                 lora_idx, prev_fill)
        if defer_sample:
            return logits, seqs
        sampled = self.sample(logits, seqs)
        return {
            seq.request_id: int(tok) for seq, tok in zip(seqs, sampled)
        }

    @torch.no_grad()
    def execute(
        self, out: SchedulerOutput, bm: BlockManager
    ) -> Dict[str, int]:
        """Run one step; returns request_id -> sampled token."""
        fast = self._execute_decode_graph(out, bm)
        if fast is not None:
            return fast
        n_mb = self.config.parallel.pp_microbatches
        if self.pipeline is not None and n_mb > 1 and len(out.scheduled) > 1:
            return self._execute_pp_microbatched(out, bm, n_mb)
        # mixed step: replay the decode rows through their captured graph
        # and run only the prefill chunks eagerly (per-row outputs are
        # independent; both enqueue on the same stream so KV ordering
        # holds). Without this, one prefill chunk forced the entire
        # 256-row step into eager mode.
        if (
            self.graphs is not None
            and self.pipeline is None
            and not getattr(self, "tp_serving", False)
            and not self.config.unified_mixed_steps
        ):
            dec = [s for s in out.scheduled if s.is_decode]
            pre = [s for s in out.scheduled if not s.is_decode]
            if dec and pre:
                got = self._execute_decode_graph(
                    SchedulerOutput(scheduled=dec), bm, defer_sample=True
                )
                if got is not None:
                    # decode replay is in flight; enqueue the prefill
                    # forward BEFORE the first D2H sync so the two halves
                    # pipeline on the stream (host profile showed two
                    # serialized GPU waits per mixed step without this)
                    logits_d, seqs_d = got
                    token_t, meta, sample_seqs, rows_t = self.prepare(
                        SchedulerOutput(scheduled=pre), bm
                    )
                    logits_p = None
                    if token_t.numel():
                        hidden = self.model(token_t, meta, self.kv_caches)
                        self._collect_prompt_logprobs(
                            SchedulerOutput(scheduled=pre), hidden
                        )
                        if sample_seqs:
                            logits_p = self.model.compute_logits(
                                hidden[rows_t]
                            )
                    result = {
                        seq.request_id: int(tok)
                        for seq, tok in zip(
                            seqs_d, self.sample(logits_d, seqs_d)
                        )
                    }
                    if logits_p is not None:
                        out_p = SchedulerOutput(scheduled=pre)
                        toks_p = self._spec_stochastic_fix(
                            out_p, sample_seqs, logits_p,
                            self.sample(logits_p, sample_seqs),
                        )
                        result.update(self._collect_sampled(
                            out_p, sample_seqs, toks_p, logits_p,
                        ))
                    return result
        token_t, meta, sample_seqs, rows_t = self.prepare(out, bm)
        if token_t.numel() == 0:
            return {}
        if self.pipeline is not None:
            tokens = self.pipeline.drive(
                token_t, meta, rows_t, [s.params for s in sample_seqs]
            )
            return {
                seq.request_id: int(tok)
                for seq, tok in zip(sample_seqs, tokens)
            }
        if self.tp_coord is not None and self.tp_coord.rank == 0 and                 getattr(self, "tp_serving", False):
            self.tp_coord.broadcast_step(token_t, meta)
        hidden = self.model(token_t, meta, self.kv_caches)
        self._collect_prompt_logprobs(out, hidden)
        if not sample_seqs:
            return {}
        sel = hidden[rows_t]
        logits = self.model.compute_logits(sel)
        tokens = self._spec_stochastic_fix(
            out, sample_seqs, logits, self.sample(logits, sample_seqs)
        )
        return self._collect_sampled(out, sample_seqs, tokens, logits)

    @torch.no_grad()
    def _execute_pp_microbatched(
        self, out: SchedulerOutput, bm: BlockManager, n_mb: int
    ) -> Dict[str, int]:
        """Split the step at sequence boundaries into up to n_mb
        microbatches (token-count balanced, greedy) and pipeline them
        through the stages in flight (pipeline.drive_many)."""
        scheduled = out.scheduled
        n_mb = min(n_mb, len(scheduled))
        # greedy balance by token count: big (prefill) seqs first
        order = sorted(range(len(scheduled)),
                       key=lambda i: -scheduled[i].num_tokens)
        groups: List[List[int]] = [[] for _ in range(n_mb)]
        loads = [0] * n_mb
        for i in order:
            g = loads.index(min(loads))
            groups[g].append(i)
            loads[g] += scheduled[i].num_tokens
        mbs = []
        all_seqs: List[Sequence] = []
        for g in groups:
            if not g:
                continue
            sub = SchedulerOutput(scheduled=[scheduled[i] for i in sorted(g)])
            token_t, meta, sample_seqs, rows_t = self.prepare(sub, bm)
            if token_t.numel() == 0:
                continue
            mbs.append((token_t, meta, rows_t,
                        [s.params for s in sample_seqs]))
            all_seqs.extend(sample_seqs)
        if not mbs:
            return {}
        tokens = self.pipeline.drive_many(mbs)
        return {
            seq.request_id: int(tok)
            for seq, tok in zip(all_seqs, tokens)
        }

    @torch.no_grad()
    def execute_async(self, out: SchedulerOutput, bm: BlockManager):
        """Async-scheduling fast path: launch this step's GPU work and
        device-side greedy sampling WITHOUT any host sync; returns a
        handle the engine finalizes on the next step (one-step-lagged
        pipeline), or None when the step needs the sync path (non-greedy
        sampling, speculative chunks, logprobs, TP/PP, no graphs).

        Decode rows whose input token is still the -1 placeholder gather
        it on-device from the previous step's sampled tensor."""
        if (
            self.pipeline is not None
            or getattr(self, "tp_serving", False)
            or self.lora_slots is not None and any(
                s.seq.lora_name for s in out.scheduled)
        ):
            return None
        for ss in out.scheduled:
            p = ss.seq.params
            if not p.greedy or p.logprobs is not None or p.needs_penalties:
                return None
            if getattr(p, "response_format", None):
                return None  # guided masks need last step's token on host
            if getattr(p, "prompt_logprobs", None) is not None:
                return None  # collected on the sync prefill path
            if ss.draft_tokens:
                return None
        dec = [ss for ss in out.scheduled if ss.is_decode]
        pre = [ss for ss in out.scheduled if not ss.is_decode]
        logits_d, seqs_d = None, []
        use_graph = (
            self.graphs is not None and dec
            and self.graphs.bucket_for(len(dec)) is not None
            and not (pre and self.config.unified_mixed_steps)
        )
        if use_graph:
            got = self._execute_decode_graph(
                SchedulerOutput(scheduled=dec), bm, defer_sample=True
            )
            if got is None:
                use_graph = False
            else:
                logits_d, seqs_d = got
        eager = pre if use_graph else out.scheduled
        logits_p, seqs_p = None, []
        if eager:
            sub = SchedulerOutput(scheduled=list(eager))
            token_t, meta, seqs_p, rows_t = self.prepare(sub, bm)
            if token_t.numel():
                # resolve -1 placeholders (decode rows sit after the
                # prefill-chunk rows) from the prev step's device tokens
                prefill_rows = sum(
                    ss.num_tokens for ss in eager if not ss.is_decode
                )
                decode_rids = [
                    ss.seq.request_id for ss in eager if ss.is_decode
                ]
                ph = (token_t < 0).nonzero().flatten()
                if ph.numel():
                    if self._prev_tokens_dev is None:
                        return None
                    src = []
                    for i in ph.tolist():
                        if i < prefill_rows:
                            # a preempted seq re-admitted as recompute
                            # prefill still carries -1 outputs: the sync
                            # path must resolve them first
                            return None
                        row = self._prev_rows.get(
                            decode_rids[i - prefill_rows]
                        )
                        if row is None:
                            return None
                        src.append(row)
                    gather = torch.tensor(src, dtype=torch.long,
                                          device=token_t.device)
                    token_t[ph] = self._prev_tokens_dev[gather]
                hidden = self.model(token_t, meta, self.kv_caches)
                if seqs_p:
                    logits_p = self.model.compute_logits(hidden[rows_t])
        sample_seqs = list(seqs_d) + list(seqs_p)
        if not sample_seqs:
            return None
        parts = []
        if logits_d is not None and len(seqs_d):
            parts.append(ops.greedy_sample(logits_d))
        if logits_p is not None and len(seqs_p):
            parts.append(ops.greedy_sample(logits_p))
        sampled_dev = parts[0] if len(parts) == 1 else torch.cat(parts)
        # async D2H into pinned staging; two buffers alternate so the
        # in-flight previous step's values survive until its finalize
        self._async_host_i ^= 1
        host = self._async_host[self._async_host_i]
        if host is None or host.numel() < sampled_dev.numel():
            host = torch.empty(
                max(sampled_dev.numel(), 512), dtype=torch.long,
                pin_memory=(self.device.type == "cuda"),
            )
            self._async_host[self._async_host_i] = host
        host[: sampled_dev.numel()].copy_(sampled_dev, non_blocking=True)
        ev = None
        if self.device.type == "cuda":
            ev = torch.cuda.Event()
            ev.record()
        self._prev_tokens_dev = sampled_dev
        self._prev_rows = {
            seq.request_id: i for i, seq in enumerate(sample_seqs)
        }
        return {
            "out": out,
            "sample_seqs": sample_seqs,
            "host": host,
            "n": sampled_dev.numel(),
            "event": ev,
        }

    def finalize_async(self, handle) -> Dict[str, int]:
        """Wait for the handle's D2H and return request_id -> token."""
        if handle["event"] is not None:
            handle["event"].synchronize()
        vals = handle["host"][: handle["n"]].tolist()
        return {
            seq.request_id: int(t)
            for seq, t in zip(handle["sample_seqs"], vals)
        }

    def _collect_prompt_logprobs(self, out, hidden) -> None:
        """vLLM prompt_logprobs: for requesting sequences, log-softmax
        the prefill-chunk rows and gather each next prompt token.
        Row j of a chunk starting at `start` predicts position
        start+j+1; only prompt positions are collected. No-op (and no
        extra compute) unless a scheduled seq asked for it."""
        row = 0
        for ss in out.scheduled:
            if ss.is_decode:
                continue
            seq, n = ss.seq, ss.num_tokens
            p = seq.params
            if (getattr(p, "prompt_logprobs", None) is not None
                    and not ss.draft_tokens):
                start = seq.num_computed
                npr = seq.num_prompt
                count = max(0, min(n, npr - start - 1))
                if count > 0:
                    idx_rows = torch.arange(
                        row, row + count, device=hidden.device
                    )
                    lg = torch.log_softmax(
                        self.model.compute_logits(
                            hidden.index_select(0, idx_rows)
                        ).float(),
                        dim=-1,
                    )
                    ids = torch.tensor(
                        seq.prompt_token_ids[start + 1:start + 1 + count],
                        dtype=torch.long, device=lg.device,
                    )
                    vals = lg.gather(1, ids.unsqueeze(1)).squeeze(1)
                    acc = self.prompt_logprobs.setdefault(
                        seq.request_id, [None]
                    )
                    acc.extend(float(v) for v in vals.cpu())
            row += n

    def _collect_sampled(self, out, sample_seqs, tokens,
                         logits=None) -> Dict[str, object]:
        """Map sampled rows back to requests; speculative chunks contribute
        their accepted-prefix token list (draft j is accepted when it
        equals the model's own prediction at the previous position)."""
        draft_map = {
            ss.seq.request_id: ss.draft_tokens
            for ss in out.scheduled
            if getattr(ss, "draft_tokens", None)
        }
        result: Dict[str, object] = {}
        i = 0
        n = len(sample_seqs)
        while i < n:
            seq = sample_seqs[i]
            rid = seq.request_id
            drafts = draft_map.get(rid)
            if not drafts:
                result[rid] = int(tokens[i])
                i += 1
                continue
            k1 = len(drafts) + 1
            row_toks = [int(t) for t in tokens[i : i + k1]]
            accepted = [row_toks[0]]
            for j, d in enumerate(drafts):
                if row_toks[j] != d:
                    break
                accepted.append(row_toks[j + 1])
            self.spec_proposed += len(drafts)
            self.spec_accepted += len(accepted) - 1
            if seq.params.logprobs is not None and logits is not None:
                # per-accepted-token logprobs (row j's raw log-softmax at
                # the token emitted there) — spec chunks emit several
                # tokens per step, so last_logprobs carries a list
                rows = logits[i:i + len(accepted)].float()
                lg = torch.log_softmax(rows, dim=-1)
                idx = torch.tensor(accepted, dtype=torch.long,
                                   device=lg.device)
                vals = lg.gather(1, idx.unsqueeze(1)).squeeze(1)
                self.last_logprobs[rid] = [float(v) for v in vals.cpu()]
            result[rid] = accepted
            i += k1
        return result

    def sample(
        self, logits: torch.Tensor, seqs: List[Sequence]
    ) -> torch.Tensor:
        params = [s.params for s in seqs]
        if any(p.needs_penalties for p in params):
            logits = self._apply_penalties(logits, seqs)
        if any(getattr(p, "response_format", None) for p in params):
            logits = self._apply_guided(logits, seqs)
        sampled = self.sample_params(logits, params)
        want_lp = [i for i, p in enumerate(params)
                   if p.logprobs is not None]
        if want_lp:
            idx = torch.tensor(want_lp, dtype=torch.long,
                               device=logits.device)
            lf = logits.index_select(0, idx).float()
            lp = torch.log_softmax(lf, dim=-1)
            chosen = sampled[idx.cpu()].to(lp.device)
            vals = lp.gather(1, chosen.unsqueeze(1)).squeeze(1).cpu()
            for j, i in enumerate(want_lp):
                rid = seqs[i].request_id
                self.last_logprobs[rid] = float(vals[j])
                n_top = params[i].logprobs or 0
                if n_top > 0:
                    k = min(int(n_top), 20, lp.shape[-1])
                    tv, ti = lp[j].topk(k)
                    self.last_top_logprobs[rid] = list(zip(
                        (int(t) for t in ti.cpu()),
                        (float(v) for v in tv.cpu()),
                    ))
        return sampled

    def _apply_guided(
        self, logits: torch.Tensor, seqs: List[Sequence]
    ) -> torch.Tensor:
        """Structured outputs (`response_format` JSON mode): mask each
        guided row to the tokens whose text keeps the output a valid JSON
        prefix; once the document completes only EOS stays legal
        (engine/guided.py)."""
        from production_stack_amd.engine.guided import (
            guided_state_from_response_format,
        )

        tok = getattr(self, "tokenizer", None)
        if tok is None:
            return logits
        logits = logits.float().clone()
        eos = self.model_cfg.eos_token_id
        for i, seq in enumerate(seqs):
            p = seq.params
            rf = getattr(p, "response_format", None)
            if not rf:
                continue
            gs = getattr(p, "_guided_state", None)
            if gs is None:
                gs = guided_state_from_response_format(rf)
                if gs is None:
                    continue
                p._guided_state = gs
            gs.advance(tok, seq.output_token_ids)
            allowed, _ = gs.allowed_mask(tok, logits[i], eos)
            idx = torch.tensor(allowed, dtype=torch.long,
                               device=logits.device)
            row = torch.full_like(logits[i], float("-inf"))
            row[idx] = logits[i][idx]
            if not torch.isfinite(row[idx]).any():
                # e.g. min_tokens suppressed EOS while the grammar
                # completed: the grammar wins (uniform over allowed)
                row[idx] = 0.0
            logits[i] = row
        return logits

    def _apply_penalties(
        self, logits: torch.Tensor, seqs: List[Sequence]
    ) -> torch.Tensor:
        """OpenAI-style presence/frequency penalties, HF-style repetition
        penalty, logit_bias, and min_tokens EOS suppression. Applied only
        to rows whose params ask for it (penalty-free batches skip this
        entirely)."""
        logits = logits.float().clone()
        V = logits.shape[-1]
        eos = self.model_cfg.eos_token_id
        for i, seq in enumerate(seqs):
            p = seq.params
            if not p.needs_penalties:
                continue
            row = logits[i]
            out_ids = [t for t in seq.output_token_ids if t >= 0]
            if out_ids and (p.presence_penalty or p.frequency_penalty
                            or p.repetition_penalty != 1.0):
                ids = torch.tensor(out_ids, dtype=torch.long,
                                   device=row.device)
                counts = torch.bincount(ids, minlength=V).to(row.dtype)
                seen = counts > 0
                if p.frequency_penalty:
                    row -= p.frequency_penalty * counts
                if p.presence_penalty:
                    row[seen] -= p.presence_penalty
                if p.repetition_penalty != 1.0:
                    pos = seen & (row > 0)
                    neg = seen & (row <= 0)
                    row[pos] /= p.repetition_penalty
                    row[neg] *= p.repetition_penalty
            if p.logit_bias:
                for tid, b in p.logit_bias.items():
                    t = int(tid)
                    if 0 <= t < V:
                        row[t] += float(b)
            if p.min_tokens and len(out_ids) < p.min_tokens:
                row[eos] = float("-inf")
                for t in p.stop_token_ids:
                    if 0 <= t < V:
                        row[t] = float("-inf")
        return logits

    def sample_params(
        self, logits: torch.Tensor, params: List
    ) -> torch.Tensor:
        """Batched sampling: greedy rows via the HIP argmax kernel, the rest
        through grouped tensor ops (temperature -> per-group top-k ->
        sorted-cumsum top-p -> multinomial), one batch per distinct top_k."""
        greedy_mask = torch.tensor([p.greedy for p in params])
        result = torch.empty(len(params), dtype=torch.long)
        if bool(greedy_mask.all()):
            return ops.greedy_sample(logits).cpu()
        if bool(greedy_mask.any()):
            g_rows = greedy_mask.nonzero().flatten()
            g_idx = g_rows.to(logits.device)
            result[g_rows] = ops.greedy_sample(
                logits.index_select(0, g_idx)
            ).cpu()
        sampled_rows = (~greedy_mask).nonzero().flatten()
        groups: Dict[int, List[int]] = {}
        for i in sampled_rows.tolist():
            groups.setdefault(params[i].top_k, []).append(i)
        for top_k, rows in groups.items():
            idx = torch.tensor(rows, dtype=torch.long, device=logits.device)
            lf = logits.index_select(0, idx).float()
            temps = torch.tensor(
                [max(params[i].temperature, 1e-5) for i in rows],
                device=lf.device,
            ).unsqueeze(1)
            lf = lf / temps
            if 0 < top_k < lf.shape[-1]:
                kth = torch.topk(lf, top_k, dim=-1).values[:, -1:]
                lf = lf.masked_fill(lf < kth, float("-inf"))
            probs = torch.softmax(lf, dim=-1)
            min_ps = torch.tensor(
                [getattr(params[i], "min_p", 0.0) for i in rows],
                device=lf.device,
            ).unsqueeze(1)
            if bool((min_ps > 0).any()):
                cutoff = probs.max(dim=-1, keepdim=True).values * min_ps
                probs = probs.masked_fill(probs < cutoff, 0.0)
                probs = probs / probs.sum(dim=-1, keepdim=True)
            top_ps = torch.tensor(
                [params[i].top_p for i in rows], device=lf.device
            ).unsqueeze(1)
            if bool((top_ps < 1.0).any()):
                sp, si = torch.sort(probs, descending=True, dim=-1)
                cum = torch.cumsum(sp, dim=-1)
                keep = (cum - sp) < top_ps
                keep[:, 0] = True
                sp = sp * keep
                sp = sp / sp.sum(dim=-1, keepdim=True)
                sp_cpu = sp.cpu()
                picks = torch.multinomial(
                    sp_cpu, 1, generator=self._generator
                )
                self._redraw_seeded(sp_cpu, picks, rows, params)
                chosen = torch.gather(si.cpu(), 1, picks).flatten()
            else:
                probs_cpu = probs.cpu()
                picks = torch.multinomial(
                    probs_cpu, 1, generator=self._generator
                )
                self._redraw_seeded(probs_cpu, picks, rows, params)
                chosen = picks.flatten()
            result[torch.tensor(rows)] = chosen
        return result

    def _filtered_probs(self, logit_row: torch.Tensor, p) -> torch.Tensor:
        """CPU fp32 probabilities after exactly the temperature/top-k/top-p
        filtering sample_params applies — the target distribution used by
        rejection-sampled speculative decoding."""
        lf = logit_row.detach().float() / max(p.temperature, 1e-5)
        v = lf.shape[-1]
        if 0 < p.top_k < v:
            kth = torch.topk(lf, p.top_k).values[-1]
            lf = lf.masked_fill(lf < kth, float("-inf"))
        probs = torch.softmax(lf, dim=-1)
        if getattr(p, "min_p", 0.0) > 0:
            cutoff = probs.max() * p.min_p
            probs = probs.masked_fill(probs < cutoff, 0.0)
            probs = probs / probs.sum()
        if p.top_p < 1.0:
            sp, si = torch.sort(probs, descending=True)
            cum = torch.cumsum(sp, dim=-1)
            keep = (cum - sp) < p.top_p
            keep[0] = True
            mask = torch.zeros_like(probs, dtype=torch.bool)
            mask.scatter_(0, si[keep], True)
            probs = probs * mask
            probs = probs / probs.sum()
        return probs.cpu()

    def _spec_generator(self, p) -> torch.Generator:
        if p.seed is None:
            return self._generator
        g = getattr(p, "_seed_generator", None)
        if g is None:
            g = torch.Generator(device="cpu").manual_seed(int(p.seed))
            p._seed_generator = g
        return g

    def _spec_stochastic_fix(self, out, sample_seqs, logits, tokens):
        """Rejection-sampling acceptance for speculative chunks under
        stochastic sampling (Leviathan et al. 2023 with a point-mass
        draft: accept draft d at position j with prob p_j(d); on reject
        emit a sample from p_j with d's mass removed — either way the
        emitted token is marginally distributed exactly as p_j, so
        speculation never changes the output distribution).

        Decisions are written back into the sampled-token rows so
        _collect_sampled's exact-match walk emits the right accepted
        prefix: an accepted position holds the draft token, the first
        rejected position holds the residual resample (always != draft),
        and a fully-accepted chunk falls through to the bonus row's own
        target sample."""
        draft_map = {
            ss.seq.request_id: (ss.draft_tokens,
                                getattr(ss, "draft_probs", None))
            for ss in out.scheduled
            if getattr(ss, "draft_tokens", None)
        }
        if not draft_map:
            return tokens
        i = 0
        n = len(sample_seqs)
        while i < n:
            seq = sample_seqs[i]
            drafts, qs = draft_map.get(seq.request_id, (None, None))
            if not drafts:
                i += 1
                continue
            p = seq.params
            if not p.greedy:
                gen = self._spec_generator(p)
                for j, d in enumerate(drafts):
                    probs = self._filtered_probs(logits[i + j], p)
                    pd = float(probs[d])
                    q = qs[j] if qs is not None else None
                    if q is None:
                        accept_p = pd  # point-mass draft (n-gram)
                    else:
                        accept_p = min(1.0, pd / max(float(q[d]), 1e-20))
                    if float(torch.rand(1, generator=gen)) < accept_p:
                        tokens[i + j] = d
                        continue
                    if q is None:
                        probs[d] = 0.0
                    else:
                        # Leviathan residual: normalize(max(p - q, 0))
                        probs = torch.clamp(probs - q, min=0.0)
                    s = float(probs.sum())
                    if s > 0:
                        tokens[i + j] = int(torch.multinomial(
                            probs / s, 1, generator=gen
                        ))
                    # s == 0: degenerate overlap (p ~= q point mass);
                    # keep the row's own target sample
                    break
            i += len(drafts) + 1
        return tokens

    def _redraw_seeded(
        self, probs_cpu: torch.Tensor, picks: torch.Tensor,
        rows: List[int], params: List,
    ) -> None:
        """OpenAI-parity per-request determinism: rows carrying
        SamplingParams.seed draw from their own persistent generator
        (seeded once per request) instead of the shared engine one."""
        for j, i in enumerate(rows):
            p = params[i]
            if p.seed is None:
                continue
            g = getattr(p, "_seed_generator", None)
            if g is None:
                g = torch.Generator(device="cpu").manual_seed(int(p.seed))
                p._seed_generator = g
            picks[j] = torch.multinomial(probs_cpu[j], 1, generator=g)
