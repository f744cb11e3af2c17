#!/usr/bin/env python3
"""Flagship serving benchmark: multi-round-QA-shaped workload, Llama-3-8B,
one engine replica per GPU (data parallel; BASELINE.json configs).

Workload shape follows the reference's canonical multi-round-qa harness
(reference benchmarks/multi-round-qa/run.sh: shared system prompt, growing
per-user chat history, ~100-token answers), run in steady-state saturation:
each rank keeps `--users` concurrent conversations alive, resubmitting a
user's next round (history grown) the moment their answer completes. Data is
synthetic token ids; weights are random-init bf16 (no network in the
environment).

A "step" is one engine step (one continuous-batching iteration: chunked
prefills + decodes). The headline metric is aggregate output tokens/s across
all ranks; p50 TTFT over rounds submitted inside the timed window is
reported alongside.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
it is launched under torch.distributed.run with one rank per GPU over RCCL.
Rank 0 prints exactly one JSON line.
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams

SYSTEM_PROMPT_TOKENS = 1000
QUESTION_TOKENS = 100
ANSWER_TOKENS = 100
NUM_ROUNDS = 10          # reference run.sh: 10 rounds per conversation
USER_INFO_TOKENS = 2500  # reference CHAT_HISTORY=20000, scaled to fit
# 320 users' KV in 288 GB HBM (20k x 320 users = 880 GB of KV exists on
# no single GPU; the reference's own sweep needs a cluster-scale KV
# tier for it). Histories are append-only within a conversation exactly
# like the reference harness (ChatHistory only ever appends), and a
# fresh conversation (new user info) starts after NUM_ROUNDS.
VOCAB_LOW, VOCAB_HIGH = 16, 128000


class User:
    """One user = a stream of conversations, each NUM_ROUNDS rounds with
    an append-only history (reference multi-round-qa.py ChatHistory
    semantics: system prompt + per-user info on round 1, then question/
    answer pairs appended — never trimmed). Starting users at random
    round phases makes the population stationary from the first step."""

    def __init__(self, uid, rng, system, vocab_high=VOCAB_HIGH,
                 start_round=0, info_tokens=USER_INFO_TOKENS,
                 num_rounds=NUM_ROUNDS):
        self.uid = uid
        self.rng = rng
        self.vocab_high = vocab_high
        self.system = system
        self.info_tokens = info_tokens
        self.num_rounds = num_rounds
        self.round = 0
        self.submit_time = 0.0
        # randomized FIRST answer length desynchronizes the user rounds
        self.first_len = int(rng.integers(10, 2 * ANSWER_TOKENS))
        self._new_conversation()
        for _ in range(start_round):
            self.round += 1
            self.history.extend(self.rng.integers(
                VOCAB_LOW, vocab_high,
                size=QUESTION_TOKENS + ANSWER_TOKENS).tolist())

    def _new_conversation(self) -> None:
        self.round = 0
        self.conv = getattr(self, "conv", -1) + 1
        # per-conversation round count drawn around NUM_ROUNDS (mean 10):
        # with a FIXED length the closed loop bus-bunches — users drift
        # into synchronized reset waves whose 3.6k-token prefills stall
        # everyone (measured: 60-step windows oscillated 4.8-6.6k tok/s).
        # Randomized lengths keep resets de-correlated and the process
        # mixing toward a stationary distribution.
        self.rounds_this_conv = int(self.rng.integers(
            max(1, self.num_rounds - 4), self.num_rounds + 5))
        self.history = list(self.system) + self.rng.integers(
            VOCAB_LOW, self.vocab_high, size=self.info_tokens
        ).tolist()

    def answer_len(self) -> int:
        return self.first_len if self.round <= 1 else ANSWER_TOKENS

    def next_prompt(self) -> list:
        if self.round >= getattr(self, "rounds_this_conv",
                                 self.num_rounds):
            self._new_conversation()
        self.round += 1
        q = self.rng.integers(
            VOCAB_LOW, self.vocab_high, size=QUESTION_TOKENS
        ).tolist()
        self.history.extend(q)
        return list(self.history)

    def complete(self, answer_tokens: list) -> None:
        self.history.extend(answer_tokens)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=48)
    ap.add_argument("--users", type=int, default=320, help="conversations per GPU (reference run.sh canonical: 320)")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--max-model-len", type=int, default=8192)
    ap.add_argument("--max-num-batched-tokens", type=int, default=2048)
    ap.add_argument("--kv-cache-dtype", default="auto",
                    choices=["auto", "bf16", "fp8", "fp8_e4m3"])
    ap.add_argument("--device", default=None)
    ap.add_argument("--profile-host", action="store_true",
                    help="cProfile the measured loop; report to stderr")
    ap.add_argument("--no-async-scheduling", action="store_true",
                    help="disable one-step-lagged sampling")
    ap.add_argument("--quantization", default=None, choices=[None, "fp8"])
    ap.add_argument("--no-unified-mixed-steps", dest="unified_mixed_steps",
                    action="store_false", default=True,
                    help="restore the split graph+eager mixed-step path "
                         "(unified reads weights once per step; default)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available() and args.device != "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")

    cfg = EngineConfig(
        model=args.model,
        max_model_len=args.max_model_len,
        async_scheduling=not args.no_async_scheduling,
        unified_mixed_steps=args.unified_mixed_steps,
        quantization=args.quantization,
        seed=1234 + rank,
        cache=CacheConfig(
            block_size=16,
            gpu_memory_utilization=0.85,
            enable_prefix_caching=True,
            num_gpu_blocks=None if use_cuda else 2048,
            kv_cache_dtype=args.kv_cache_dtype,
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=max(args.users * 2, 64),
            max_num_batched_tokens=args.max_num_batched_tokens,
        ),
    )
    engine = LLMEngine(cfg, device=None if use_cuda else "cpu")

    vocab_high = min(VOCAB_HIGH, engine.model_cfg.vocab_size - 1)
    rng = np.random.default_rng(4242 + rank)
    system = rng.integers(
        VOCAB_LOW, vocab_high, size=SYSTEM_PROMPT_TOKENS
    ).tolist()
    # stationary ensemble: users start at uniformly random rounds of
    # their conversation; size per-user info (and, for tiny test
    # windows, the round count) so a full conversation fits the window
    per_round = QUESTION_TOKENS + ANSWER_TOKENS
    budget = (args.max_model_len - SYSTEM_PROMPT_TOKENS
              - ANSWER_TOKENS - QUESTION_TOKENS - 16)
    rounds_eff = NUM_ROUNDS
    info = min(USER_INFO_TOKENS, budget - (rounds_eff + 4) * per_round)
    if info < 256:
        info = max(0, min(256, budget - per_round))
        rounds_eff = max(1, (budget - info) // per_round)
    users = [
        User(u, rng, system, vocab_high,
             start_round=int(rng.integers(0, rounds_eff)),
             info_tokens=info, num_rounds=rounds_eff)
        for u in range(args.users)
    ]
    answers: dict = {}
    ttfts: list = []
    in_window = False
    in_steady = False
    window_tokens = 0
    done_once: set = set()

    def submit(user: User) -> None:
        prompt = user.next_prompt()
        rid = f"u{user.uid}-c{user.conv}-r{user.round}"
        user.submit_time = time.perf_counter()
        user.steady_submit = in_steady
        answers[rid] = (user, [])
        engine.add_request(
            rid, prompt,
            SamplingParams(max_tokens=user.answer_len(), temperature=0.0,
                           ignore_eos=True),
        )

    for u in users:
        submit(u)

    def run_step() -> None:
        nonlocal window_tokens
        for out in engine.step():
            entry = answers.get(out.request_id)
            if entry is None:
                continue
            user, toks = entry
            toks.extend(out.new_token_ids)
            if in_window:
                window_tokens += len(out.new_token_ids)
            # TTFT is only meaningful in steady state: rounds submitted
            # while the initial 320-user prefill wave is still draining
            # measure queueing of a cold start, not serving latency.
            if out.first_token and in_steady and user.steady_submit:
                ttfts.append(time.perf_counter() - user.submit_time)
            if out.finished:
                user.complete(toks)
                del answers[out.request_id]
                done_once.add(user.uid)
                submit(user)

    # Steady-state guard: the timed window must measure the serving steady
    # state no matter what --steps/--warmup the driver passes.  The initial
    # prefill wave of `users` x ~1.1k-token prompts needs
    # ~users*1100/max_num_batched_tokens steps to drain, far more than a
    # typical --warmup.  So before honoring --warmup we keep stepping until
    # >=90% of users have completed their first round (mixed
    # prefill-over-decode steady state), with a hard time cap as a
    # safety net.  Only then do the driver's warmup steps run.
    steady_target = max(1, int(0.9 * len(users)))
    t_guard = time.perf_counter()
    guard_cap_s = 300.0 if use_cuda else 120.0
    guard_steps = 0
    max_wait = max(2, int(0.05 * len(users)))
    stable = 0
    while True:
        if time.perf_counter() - t_guard >= guard_cap_s:
            break
        ok = (len(done_once) >= steady_target
              and engine.scheduler.num_waiting <= max_wait)
        stable = stable + 1 if ok else 0
        if stable >= 20:  # equilibrium must HOLD, not just flicker
            break
        run_step()
        guard_steps += 1
    # the drain criterion exits before the ensemble's age structure is
    # stationary: conversation resets (the expensive 3.6k-token
    # prefills) only ramp in over ~1 conversation period (~1000 steps),
    # so a 20-step window right after the drain read ~30% above what
    # 60-200-step windows agree on. A fixed mixing period lets short
    # driver windows read the converged value
    for _ in range(400 if use_cuda else 5):
        if time.perf_counter() - t_guard >= guard_cap_s:
            break
        run_step()
        guard_steps += 1
    in_steady = True
    for _ in range(args.warmup):
        run_step()

    if dist:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    in_window = True
    prof = None
    if args.profile_host:
        import cProfile

        prof = cProfile.Profile()
        prof.enable()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    if use_cuda:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    in_window = False
    if prof is not None:
        import io
        import pstats

        prof.disable()
        buf = io.StringIO()
        pstats.Stats(prof, stream=buf).sort_stats("cumulative").print_stats(25)
        print(buf.getvalue(), file=sys.stderr)

    # aggregate: sum tokens over ranks, max elapsed over ranks
    total_tokens = window_tokens
    if dist:
        t = torch.tensor([float(window_tokens)], dtype=torch.float64)
        e = torch.tensor([elapsed], dtype=torch.float64)
        if use_cuda:
            t, e = t.cuda(), e.cuda()
        dist.all_reduce(t)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        total_tokens = t.item()
        elapsed = e.item()

    if rank == 0:
        value = total_tokens / elapsed
        ttft_p50 = (
            float(np.percentile(ttfts, 50) * 1000.0) if ttfts else None
        )
        ttft_p90 = (
            float(np.percentile(ttfts, 90) * 1000.0) if ttfts else None
        )
        print(
            json.dumps(
                {
                    "metric": "output_tokens_per_sec",
                    "value": round(value, 2),
                    "unit": "tokens/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(elapsed * 1000.0 / args.steps, 3),
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bfloat16",
                    "data": "synthetic",
                    "ttft_p50_ms": ttft_p50,
                    "ttft_p90_ms": ttft_p90,
                    "config": {
                        "model": args.model,
                        "workload": "multi-round-qa",
                        "users_per_gpu": args.users,
                        "system_prompt_tokens": SYSTEM_PROMPT_TOKENS,
                        "user_history_tokens": USER_INFO_TOKENS,
                        "rounds_per_conversation": NUM_ROUNDS,
                        "question_tokens": QUESTION_TOKENS,
                        "answer_tokens": ANSWER_TOKENS,
                        "max_model_len": args.max_model_len,
                        "parallelism": f"dp{world}",
                        "prefix_caching": True,
                        "kv_cache_dtype": args.kv_cache_dtype,
                "quantization": args.quantization or "none",
                    },
                }
            )
        )

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
