"""LLMEngine: the synchronous core loop (add_request / step / abort).

This is the in-process API used by bench.py and wrapped by the async OpenAI
HTTP server (engine/server.py). One engine instance == one replica == one
GPU (TP>1: one engine per rank, rank 0 drives scheduling — see
parallel/state.py).
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional, Union

import torch

from production_stack_amd.engine.block_manager import BlockManager
from production_stack_amd.engine.config import EngineConfig
from production_stack_amd.engine.model_runner import ModelRunner
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.scheduler import Scheduler
from production_stack_amd.engine.sequence import (
    RequestOutput,
    Sequence,
    SeqStatus,
)
from production_stack_amd.engine.tokenizer import get_tokenizer
from production_stack_amd.parallel import state as pstate


class EngineStats:
    def __init__(self) -> None:
        self.prompt_tokens = 0
        self.generation_tokens = 0
        self.num_requests = 0
        self.num_finished = 0


class LLMEngine:
    def __init__(
        self, config: EngineConfig, device: Optional[str] = None
    ) -> None:
        self.config = config
        self.model_cfg = config.model_config()
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if config.parallel.pipeline_parallel_size > 1:
            pstate.init_distributed(1)  # join the world group; PP uses p2p
        else:
            pstate.init_distributed(config.parallel.tensor_parallel_size)
        torch.manual_seed(config.seed)
        self.tokenizer = get_tokenizer(
            config.tokenizer, self.model_cfg.vocab_size
        )
        self.runner = ModelRunner(config, self.model_cfg, self.device)
        self.runner.tokenizer = self.tokenizer  # guided decoding masks
        num_blocks = self.runner.profile_num_blocks()
        self.runner.allocate_kv_cache(num_blocks)
        self.block_manager = BlockManager(
            num_blocks,
            config.cache.block_size,
            config.cache.enable_prefix_caching,
        )
        self.scheduler = Scheduler(
            config.scheduler, self.block_manager, config.max_model_len
        )
        if config.parallel.pipeline_parallel_size > 1:
            self.scheduler.allow_spec = False
        if (
            config.speculative_model
            and config.scheduler.num_speculative_tokens > 0
            and config.parallel.pipeline_parallel_size == 1
            and config.parallel.tensor_parallel_size == 1
        ):
            from production_stack_amd.engine.draft import (
                DraftModelProposer,
            )

            self.scheduler.draft_proposer = DraftModelProposer(
                config, self.device, num_blocks
            )
        if config.cache.cpu_offload_gb > 0 or config.cache.remote_kv_url:
            from production_stack_amd.kvpool.offload import HostKVPool

            self.host_pool = HostKVPool(
                self.runner.kv_caches,
                config.cache.block_size,
                max(config.cache.cpu_offload_gb, 0.25),
                self.device,
                offload_dtype=config.cache.offload_dtype,
                remote_url=config.cache.remote_kv_url,
                remote_serde=config.cache.remote_kv_serde,
            )
            self.block_manager.offload_pool = self.host_pool
        else:
            self.host_pool = None
        self.runner.capture_decode_graphs(
            min(config.scheduler.max_num_seqs, 512)
        )
        self.stats = EngineStats()
        self._sleeping = False
        self._pending = None  # async-scheduling in-flight step
        self.lora_adapters: Dict[str, object] = {}
        self.runner.lora_registry = self.lora_adapters
        # coarse lock: engine-loop steps vs out-of-band block mutations
        # (KV transfer registration/adoption)
        self.lock = threading.RLock()

    # ------------------------------------------------------------------
    def add_request(
        self,
        request_id: str,
        prompt: Union[str, List[int]],
        params: SamplingParams,
        arrival_time: Optional[float] = None,
        lora_name: Optional[str] = None,
        mm_embeds=None,
    ) -> None:
        params.validate(self.config.max_model_len)
        if lora_name is not None and lora_name not in self.lora_adapters:
            raise ValueError(f"unknown LoRA adapter {lora_name!r}")
        if isinstance(prompt, str):
            token_ids = self.tokenizer.encode(prompt)
        else:
            token_ids = list(prompt)
        if not token_ids:
            token_ids = [self.model_cfg.bos_token_id]
        if mm_embeds and self.config.parallel.tensor_parallel_size > 1:
            raise ValueError(
                "multimodal embeddings are not supported with TP > 1")
        seq = Sequence(request_id, token_ids, params, arrival_time,
                       lora_name=lora_name, mm_embeds=mm_embeds)
        self.scheduler.add(seq)
        self.stats.num_requests += 1
        self.stats.prompt_tokens += len(token_ids)

    def abort_request(self, request_id: str) -> None:
        self.scheduler.abort(request_id)
        self.runner.prompt_logprobs.pop(request_id, None)
        self.runner.last_logprobs.pop(request_id, None)
        self.runner.last_top_logprobs.pop(request_id, None)

    def has_unfinished(self) -> bool:
        return self.scheduler.has_unfinished() or self._pending is not None

    # ------------------------------------------------------------------
    def step(self) -> List[RequestOutput]:
        with self.lock:
            return self._step_locked()

    # ---- async scheduling (one-step-lagged sampling) -----------------
    def _step_locked_async(self) -> List[RequestOutput]:
        out = self.scheduler.schedule()
        if out.is_empty and not out.capacity_stopped:
            return self._finalize_pending()
        if self.host_pool is not None:
            self.host_pool.make_compute_wait()
        handle = self.runner.execute_async(out, self.block_manager)
        if handle is None:
            # resolve in-flight placeholders, then take the sync path for
            # this step (prepare must see real token values)
            prev = self._finalize_pending()
            # finalize may have finished (and freed) sequences this step's
            # schedule still references — executing them would read a
            # dead block table
            out.scheduled = [
                s for s in out.scheduled if not s.seq.finished
            ]
            if not out.scheduled:
                return prev + self._build_outputs(out)
            sampled = self.runner.execute(out, self.block_manager)
            fin = self.scheduler.on_step_done(
                out, sampled, self.model_cfg.eos_token_id,
                detok=self.tokenizer.decode,
            )
            self.stats.num_finished += len(fin)
            return prev + self._build_outputs(out)
        ph_idx = self.scheduler.advance_async(
            out, {s.request_id for s in handle["sample_seqs"]}
        )
        prev = self._finalize_pending()
        self._pending = (handle, out, ph_idx)
        return prev + self._build_outputs(
            out, include_scheduled=False
        )

    def _finalize_pending(self) -> List[RequestOutput]:
        if self._pending is None:
            return []
        handle, p_out, ph_idx = self._pending
        self._pending = None
        sampled = self.runner.finalize_async(handle)
        finished = self.scheduler.finalize_async(
            sampled, ph_idx, self.model_cfg.eos_token_id,
            detok=self.tokenizer.decode,
        )
        self.stats.num_finished += len(finished)
        return self._build_outputs(p_out, count_finished=False)

    def _step_locked(self) -> List[RequestOutput]:
        if self.config.async_scheduling:
            return self._step_locked_async()
        out = self.scheduler.schedule()
        if out.is_empty and not out.capacity_stopped:
            return []
        if self.host_pool is not None:
            self.host_pool.make_compute_wait()
        sampled = self.runner.execute(out, self.block_manager)
        finished = self.scheduler.on_step_done(
            out, sampled, self.model_cfg.eos_token_id,
            detok=self.tokenizer.decode,
        )
        self.stats.num_finished += len(finished)
        return self._build_outputs(out)

    def _build_outputs(
        self, out, include_scheduled: bool = True,
        count_finished: bool = True,
    ) -> List[RequestOutput]:
        now = time.time()
        results: List[RequestOutput] = []
        for seq in out.capacity_stopped:
            results.append(
                RequestOutput(
                    request_id=seq.request_id,
                    new_token_ids=[],
                    text_delta="",
                    finished=True,
                    finish_reason="length",
                    num_prompt_tokens=seq.num_prompt,
                    num_output_tokens=len(seq.output_token_ids),
                    num_cached_tokens=seq.num_cached_prompt_tokens,
                )
            )
            self.stats.num_finished += 1
        if not include_scheduled:
            out.capacity_stopped = []  # don't re-emit on a later build
            return results
        for ss in out.scheduled:
            seq = ss.seq
            if seq.status is SeqStatus.PREEMPTED:
                continue
            new = seq.drain_new_tokens()
            if not new and not seq.finished:
                continue
            lps = None
            tops = None
            if seq.params.logprobs is not None and new:
                top = self.runner.last_top_logprobs.pop(
                    seq.request_id, None)
                if top is not None and len(new) == 1:
                    # alternatives belong to the latest sampled position
                    # only; multi-token (speculative) drains skip them
                    # rather than replicate a stale row
                    tops = [top]
                lp = self.runner.last_logprobs.pop(seq.request_id, None)
                if isinstance(lp, list):
                    # speculative chunk: one value per accepted token
                    lps = lp[-len(new):] if len(lp) >= len(new) else (
                        [lp[0]] * (len(new) - len(lp)) + lp)
                elif lp is not None:
                    # one token sampled per step
                    lps = [lp] * len(new)
            first = seq.first_token_time is None and bool(new)
            if first:
                seq.first_token_time = now
            plp = None
            if first or seq.finished:
                plp = self.runner.prompt_logprobs.pop(
                    seq.request_id, None)
            self.stats.generation_tokens += len(new)
            # decode only up to the streaming cursor: under async
            # scheduling the tail may still hold -1 placeholders
            text_delta = self.tokenizer.stream_decode(
                seq.output_token_ids[: seq._stream_cursor],
                seq.detok_state,
            ) if new else ""
            reason = None
            if seq.status is SeqStatus.FINISHED_STOPPED:
                reason = "stop"
            elif seq.status is SeqStatus.FINISHED_LENGTH:
                reason = "length"
            elif seq.status is SeqStatus.FINISHED_ABORTED:
                reason = "abort"
            if reason is not None:
                gs = getattr(seq.params, "_guided_state", None)
                if gs is not None:
                    # surface json_schema validation failure instead of
                    # silently returning a non-conforming object
                    reason = gs.finish_reason() or reason
            results.append(
                RequestOutput(
                    request_id=seq.request_id,
                    new_token_ids=new,
                    text_delta=text_delta,
                    finished=seq.finished,
                    finish_reason=reason,
                    prompt_logprobs=plp,
                    num_prompt_tokens=seq.num_prompt,
                    num_output_tokens=len(seq.output_token_ids),
                    num_cached_tokens=seq.num_cached_prompt_tokens,
                    first_token=first,
                    new_logprobs=lps,
                    new_top_logprobs=tops,
                )
            )
        return results

    # ------------------------------------------------------------------
    def generate(
        self,
        prompts: List[Union[str, List[int]]],
        params: Union[SamplingParams, List[SamplingParams]],
    ) -> Dict[str, List[int]]:
        """Offline batch API: run all prompts to completion."""
        if isinstance(params, SamplingParams):
            params = [params] * len(prompts)
        for i, (p, sp) in enumerate(zip(prompts, params)):
            self.add_request(f"offline-{i}", p, sp)
        outputs: Dict[str, List[int]] = {}
        while self.has_unfinished():
            for r in self.step():
                outputs.setdefault(r.request_id, []).extend(r.new_token_ids)
        return outputs

    # ---- observability (scraped into vllm:* metric names) -------------
    def engine_metrics(self) -> Dict[str, float]:
        bm = self.block_manager
        return {
            "num_requests_running": float(self.scheduler.num_running),
            "num_requests_waiting": float(self.scheduler.num_waiting),
            "gpu_cache_usage_perc": bm.usage,
            "gpu_prefix_cache_hits_total": float(bm.prefix_hits),
            "gpu_prefix_cache_queries_total": float(bm.prefix_queries),
            "prompt_tokens_total": float(self.stats.prompt_tokens),
            "generation_tokens_total": float(self.stats.generation_tokens),
            "spec_decode_proposed_total": float(self.runner.spec_proposed),
            "spec_decode_accepted_total": float(self.runner.spec_accepted),
            **(self.host_pool.metrics() if self.host_pool else {}),
        }

    # ---- LoRA ----------------------------------------------------------
    def load_lora(self, name: str, path: str) -> None:
        from production_stack_amd.engine.lora import LoRAAdapter

        ad = LoRAAdapter.load(
            name, path, self.device, torch.bfloat16
        )
        self.lora_adapters[name] = ad
        if self.runner.lora_slots is not None:
            self.runner.lora_slots.register(ad)

    def unload_lora(self, name: str) -> None:
        self.lora_adapters.pop(name, None)
        if self.runner.lora_slots is not None:
            self.runner.lora_slots.unregister(name)

    # ---- tensor parallelism (HTTP-served mode) -------------------------
    @property
    def is_tp_worker(self) -> bool:
        from production_stack_amd.parallel import state as ps

        return (
            self.config.parallel.tensor_parallel_size > 1
            and ps.tp_rank() > 0
        )

    def run_tp_worker(self) -> None:
        assert self.is_tp_worker
        self.runner.tp_coord.worker_loop()

    # ---- pipeline parallelism ------------------------------------------
    @property
    def is_pp_worker(self) -> bool:
        return self.runner.pp_size > 1 and self.runner.pp_rank > 0

    def run_pp_worker(self) -> None:
        """Blocks serving pipeline stages (ranks > 0)."""
        assert self.is_pp_worker
        self.runner.pipeline.worker_loop()

    def stop_pp_workers(self) -> None:
        if self.runner.pp_size > 1 and self.runner.pp_rank == 0:
            self.runner.pipeline.stop_workers()

    # ---- multimodal encoders (vision/audio adapters; lazy) -------------
    def get_vision_encoder(self):
        if getattr(self, "_vision_encoder", None) is None:
            from production_stack_amd.engine.models.multimodal import (
                VisionEncoder,
            )

            self._vision_encoder = VisionEncoder(
                self.model_cfg.hidden_size, seed=self.config.seed + 7
            ).to(self.device)
        return self._vision_encoder

    def get_audio_encoder(self):
        if getattr(self, "_audio_encoder", None) is None:
            from production_stack_amd.engine.models.multimodal import (
                AudioEncoder,
            )

            self._audio_encoder = AudioEncoder(
                self.model_cfg.hidden_size, seed=self.config.seed + 9
            ).to(self.device)
        return self._audio_encoder

    # ---- sleep / wake (reference request.py:1041-1128 parity) ----------
    def sleep(self, level: int = 1) -> None:
        self._sleeping = True

    def wake_up(self) -> None:
        self._sleeping = False

    @property
    def is_sleeping(self) -> bool:
        return self._sleeping
