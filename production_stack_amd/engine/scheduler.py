"""Continuous-batching scheduler with chunked prefill and preemption.

Capability target: the engine behaviour the reference stack configures via
--max-num-seqs / --enable-chunked-prefill / --enable-prefix-caching
(reference helm/templates/deployment-vllm-multi.yaml:127-221) and observes
through vllm:num_requests_running / vllm:num_requests_waiting.

Unified token model: a sequence needs KV for all `num_tokens` tokens; each
step schedules a chunk of up to `budget` uncomputed tokens; the step whose
chunk reaches the end samples the next token. Decode is the 1-token case.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Deque, Dict, List, Optional

from production_stack_amd.engine.block_manager import BlockManager
from production_stack_amd.engine.config import SchedulerConfig
from production_stack_amd.engine.sequence import Sequence, SeqStatus


@dataclass
class ScheduledSeq:
    seq: Sequence
    num_tokens: int  # chunk size this step
    # speculative drafts verified within this chunk (chunk = 1 real token
    # + the drafts); empty for normal chunks
    draft_tokens: List[int] = field(default_factory=list)
    # draft-model proposal distributions per draft position (None for
    # greedy/n-gram point-mass drafts); used by rejection sampling
    draft_probs: Optional[list] = None

    @property
    def is_decode(self) -> bool:
        return self.num_tokens == 1


@dataclass
class SchedulerOutput:
    scheduled: List[ScheduledSeq] = field(default_factory=list)
    preempted: List[Sequence] = field(default_factory=list)
    # sequences force-finished because they can no longer fit in the cache
    capacity_stopped: List[Sequence] = field(default_factory=list)

    @property
    def num_tokens(self) -> int:
        return sum(s.num_tokens for s in self.scheduled)

    @property
    def is_empty(self) -> bool:
        return not self.scheduled


class Scheduler:
    def __init__(
        self,
        config: SchedulerConfig,
        block_manager: BlockManager,
        max_model_len: int,
    ) -> None:
        self.config = config
        self.bm = block_manager
        self.max_model_len = max_model_len
        # engine clears this under PP: the PP drive path maps one sampled
        # token per request and cannot carry a k+1-row verification chunk
        self.allow_spec = True
        # set by the engine when --speculative-model is configured
        self.draft_proposer = None
        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []
        self._by_id: Dict[str, Sequence] = {}

    # ------------------------------------------------------------------
    def add(self, seq: Sequence) -> None:
        if seq.request_id in self._by_id:
            raise ValueError(
                f"duplicate request_id {seq.request_id!r}: a request with "
                "this id is still live")
        if seq.num_prompt + 1 > self.max_model_len:
            raise ValueError(
                f"prompt of {seq.num_prompt} tokens exceeds max_model_len "
                f"{self.max_model_len}"
            )
        self._by_id[seq.request_id] = seq
        self.waiting.append(seq)

    def abort(self, request_id: str) -> Optional[Sequence]:
        seq = self._by_id.pop(request_id, None)
        if seq is None or seq.finished:
            return None
        seq.status = SeqStatus.FINISHED_ABORTED
        if seq in self.running:
            self.running.remove(seq)
            self.bm.free_seq(seq)
        else:
            try:
                self.waiting.remove(seq)
            except ValueError:
                pass
        return seq

    @property
    def num_running(self) -> int:
        return len(self.running)

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    def has_unfinished(self) -> bool:
        return bool(self.running or self.waiting)

    # ------------------------------------------------------------------
    def _preempt_last(self, out: SchedulerOutput, keep: Sequence) -> bool:
        """Preempt the most recently admitted running seq (not `keep`)."""
        for victim in reversed(self.running):
            if victim is keep:
                continue
            self.running.remove(victim)
            self.bm.free_seq(victim)
            victim.reset_for_recompute()
            self.waiting.appendleft(victim)
            out.preempted.append(victim)
            # it may already carry a chunk in this step's schedule: drop it
            out.scheduled = [s for s in out.scheduled if s.seq is not victim]
            return True
        return False

    def schedule(self) -> SchedulerOutput:
        out = SchedulerOutput()
        budget = self.config.max_num_batched_tokens

        # 1. running sequences, oldest first: decodes (remaining == 1) and
        #    in-flight chunked prefills.
        for seq in list(self.running):
            if budget <= 0:
                break
            if seq not in self.running:
                # preempted as a victim earlier in this same call
                continue
            remaining = seq.num_tokens - seq.num_computed
            if remaining <= 0:
                continue
            chunk = min(remaining, budget, self.config.max_prefill_chunk)
            drafts: List[int] = []
            dprobs = None
            if (
                remaining == 1
                and self.config.num_speculative_tokens > 0
                and self.allow_spec
                # guided rows mask per-position grammar state; a k+1-row
                # chunk would share one state across positions
                and getattr(seq.params, "response_format", None) is None
                and (
                    seq.params.greedy
                    # stochastic sampling: rejection-sampling acceptance
                    # (model_runner._spec_stochastic_fix) keeps the
                    # output distribution exact; penalty rows stay
                    # non-speculative (their per-position state would
                    # drift across a multi-token chunk). Logprob rows
                    # speculate: _collect_sampled records one value per
                    # accepted token.
                    or not seq.params.needs_penalties
                )
                and seq.output_token_ids
                and seq.output_token_ids[-1] >= 0
            ):
                drafts, dprobs = self._propose_drafts(seq)
                if drafts and not self.bm.ensure_capacity(
                    seq, seq.num_computed + 1 + len(drafts)
                ):
                    drafts, dprobs = [], None  # fall back to plain decode
                chunk = min(1 + len(drafts), budget)
                if chunk <= len(drafts):
                    drafts = drafts[: max(chunk - 1, 0)]
                    if dprobs is not None:
                        dprobs = dprobs[: len(drafts)]
                    chunk = 1 + len(drafts)
            target = seq.num_computed + chunk
            while not self.bm.ensure_capacity(seq, target):
                if not self._preempt_last(out, keep=seq):
                    # no other sequence to evict: this one alone exceeds the
                    # whole KV cache — finish it instead of livelocking on
                    # preempt-recompute.
                    self.running.remove(seq)
                    self.bm.free_seq(seq)
                    seq.status = SeqStatus.FINISHED_LENGTH
                    self._by_id.pop(seq.request_id, None)
                    out.capacity_stopped.append(seq)
                    out.scheduled = [
                        s for s in out.scheduled if s.seq is not seq
                    ]
                    chunk = 0
                    break
                if seq in out.preempted:
                    chunk = 0
                    break
            if chunk > 0:
                out.scheduled.append(
                    ScheduledSeq(seq, chunk, draft_tokens=drafts,
                                 draft_probs=dprobs)
                )
                budget -= chunk

        # 2. admit waiting sequences.
        while (
            self.waiting
            and budget > 0
            and len(self.running) < self.config.max_num_seqs
        ):
            seq = self.waiting[0]
            if not self.bm.can_allocate_prompt(seq):
                break
            remaining_est = seq.num_tokens  # before cache hits
            if (
                not self.config.enable_chunked_prefill
                and remaining_est > budget
            ):
                break
            self.waiting.popleft()
            self.bm.allocate_prompt(seq)  # sets num_computed via cache hits
            seq.status = SeqStatus.RUNNING
            self.running.append(seq)
            remaining = seq.num_tokens - seq.num_computed
            chunk = min(remaining, budget, self.config.max_prefill_chunk)
            out.scheduled.append(ScheduledSeq(seq, chunk))
            budget -= chunk

        return out

    # ------------------------------------------------------------------
    def _propose_drafts(self, seq: Sequence):
        """Returns (draft_tokens, draft_probs|None). With a draft model
        configured, the model proposes (engine/draft.py); otherwise
        prompt-lookup: the longest trailing n-gram that recurs in the
        sequence's own history proposes the tokens that followed it
        (byte-encoded rfind so the scan is C-speed)."""
        if self.draft_proposer is not None:
            try:
                return self.draft_proposer.propose(seq, self.bm)
            except Exception:  # draft failure must never kill serving
                import logging

                logging.getLogger("engine.scheduler").exception(
                    "draft proposer failed; plain decode")
                return [], None
        import numpy as np

        k = self.config.num_speculative_tokens
        k = min(k, self.max_model_len - seq.num_tokens - 1)
        if k <= 0:
            return [], None
        out_ids = seq.output_token_ids
        if len(out_ids) >= 1024:
            win = out_ids[-1024:]
        else:  # avoid materialising the full prompt+output list
            win = seq.prompt_token_ids[
                max(len(seq.prompt_token_ids) - (1024 - len(out_ids)),
                    0):] + out_ids
        arr = np.asarray(win, dtype=np.int32).tobytes()
        for n in range(self.config.ngram_max, self.config.ngram_min - 1, -1):
            if len(win) <= n:
                continue
            pat = np.asarray(win[-n:], dtype=np.int32).tobytes()
            # search excluding the trailing n-gram itself
            hay = arr[: -4]  # allow overlap up to the final token
            idx = hay.rfind(pat)
            while idx >= 0 and idx % 4 != 0:
                idx = hay.rfind(pat, 0, idx + len(pat) - 1)
            if idx < 0:
                continue
            start = idx // 4 + n
            drafts = win[start : start + k]
            if drafts:
                return list(drafts), None
        return [], None

    def on_step_done(
        self,
        output: SchedulerOutput,
        sampled: Dict[str, object],
        eos_token_id: int,
        detok=None,
    ) -> List[Sequence]:
        """Advance state after the model ran. Returns newly finished seqs.
        A sampled value may be a single token or (speculative decoding) a
        list of accepted tokens."""
        finished: List[Sequence] = []
        for ss in output.scheduled:
            seq = ss.seq
            if seq.finished or seq.status is SeqStatus.PREEMPTED:
                continue
            seq.num_computed += ss.num_tokens
            tok = sampled.get(seq.request_id)
            if tok is not None:
                toks = tok if isinstance(tok, list) else [tok]
                p = seq.params
                for t in toks:
                    seq.append_token(t)
                    stop_hit = (
                        (t == eos_token_id and not p.ignore_eos)
                        or t in p.stop_token_ids
                    )
                    if not stop_hit and p.stop and detok is not None:
                        tail = detok(seq.output_token_ids[-16:])
                        stop_hit = any(st in tail for st in p.stop)
                    if stop_hit:
                        seq.status = SeqStatus.FINISHED_STOPPED
                    elif len(seq.output_token_ids) >= p.max_tokens:
                        seq.status = SeqStatus.FINISHED_LENGTH
                    elif seq.num_tokens >= self.max_model_len:
                        seq.status = SeqStatus.FINISHED_LENGTH
                    if seq.finished:
                        break
                seq.num_computed = min(seq.num_computed, seq.num_tokens - 1)
            if seq.finished:
                self.running.remove(seq)
                self.bm.register_computed_blocks(seq)
                self.bm.free_seq(seq)
                self._by_id.pop(seq.request_id, None)
                finished.append(seq)
            else:
                self.bm.register_computed_blocks(seq)
        return finished

    # ---- async scheduling (one-step-lagged sampling) -----------------
    def advance_async(self, output: SchedulerOutput,
                      sampled_rids) -> Dict[str, int]:
        """Bookkeep a launched-but-unsampled step: advance num_computed
        and append a -1 placeholder for every row that will receive a
        token. Returns request_id -> placeholder index (into
        output_token_ids) for finalize_async."""
        ph_idx: Dict[str, int] = {}
        for ss in output.scheduled:
            seq = ss.seq
            if seq.finished or seq.status is SeqStatus.PREEMPTED:
                continue
            seq.num_computed += ss.num_tokens
            if seq.request_id in sampled_rids:
                seq.append_token(-1)
                ph_idx[seq.request_id] = len(seq.output_token_ids) - 1
                seq.num_computed = min(seq.num_computed,
                                       seq.num_tokens - 1)
            self.bm.register_computed_blocks(seq)
        return ph_idx

    def finalize_async(self, sampled: Dict[str, int],
                       ph_idx: Dict[str, int], eos_token_id: int,
                       detok=None) -> List[Sequence]:
        """Resolve placeholders with the now-arrived tokens; run the stop
        checks that were deferred. A stop truncates any newer in-flight
        placeholder (its step's work is simply discarded)."""
        finished: List[Sequence] = []
        for rid, tok in sampled.items():
            seq = self._by_id.get(rid)
            if seq is None:
                continue  # aborted while in flight
            idx = ph_idx.get(rid)
            if idx is None or idx >= len(seq.output_token_ids):
                continue
            seq.output_token_ids[idx] = tok
            if seq.finished:
                continue
            p = seq.params
            stop_hit = (
                (tok == eos_token_id and not p.ignore_eos)
                or tok in p.stop_token_ids
            )
            if not stop_hit and p.stop and detok is not None:
                real = [t for t in seq.output_token_ids[: idx + 1]][-16:]
                stop_hit = any(st in detok(real) for st in p.stop)
            n_real = idx + 1
            if stop_hit:
                seq.status = SeqStatus.FINISHED_STOPPED
            elif n_real >= p.max_tokens:
                seq.status = SeqStatus.FINISHED_LENGTH
            elif seq.num_prompt + n_real >= self.max_model_len:
                seq.status = SeqStatus.FINISHED_LENGTH
            if seq.finished:
                # drop any newer in-flight placeholder tokens
                if len(seq.output_token_ids) > n_real:
                    del seq.output_token_ids[n_real:]
                    seq.num_computed = min(
                        seq.num_computed, seq.num_tokens - 1
                    )
                if seq in self.running:
                    self.running.remove(seq)
                self.bm.register_computed_blocks(seq)
                self.bm.free_seq(seq)
                self._by_id.pop(seq.request_id, None)
                finished.append(seq)
        return finished
