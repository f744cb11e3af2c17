"""Draft-model speculative decoding (engine/draft.py): greedy losslessness,
full Leviathan rejection sampling with real proposal distributions, and
draft-KV consistency across catch-up passes."""

import numpy as np
import pytest
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.scheduler import (
    ScheduledSeq,
    SchedulerOutput,
)
from production_stack_amd.engine.sequence import Sequence


def mk(spec_model=None, k=4, seed=5, blocks=128):
    cfg = EngineConfig(
        model="tiny-llama",
        max_model_len=512,
        seed=seed,
        speculative_model=spec_model,
        cache=CacheConfig(num_gpu_blocks=blocks, block_size=16),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256,
            num_speculative_tokens=k,
        ),
    )
    return LLMEngine(cfg, device="cpu")


PROMPT = [7, 8, 9, 10] * 8


def test_draft_model_greedy_lossless_perfect_draft():
    """Draft == target: every draft accepted, output identical to the
    plain engine."""
    plain = mk(None, k=0)
    p = SamplingParams(max_tokens=24, temperature=0.0, ignore_eos=True)
    want = plain.generate([PROMPT], p)["offline-0"]

    spec = mk("tiny-llama")
    spec.runner.model.load_state_dict(plain.runner.model.state_dict())
    spec.scheduler.draft_proposer.load_target_weights(spec.runner.model)
    got = spec.generate([PROMPT], p)["offline-0"]
    assert got == want
    assert spec.runner.spec_proposed > 0
    assert spec.runner.spec_accepted == spec.runner.spec_proposed


def test_draft_model_greedy_lossless_bad_draft():
    """A garbage draft model changes nothing about the output — only the
    acceptance rate."""
    plain = mk(None, k=0)
    p = SamplingParams(max_tokens=24, temperature=0.0, ignore_eos=True)
    want = plain.generate([PROMPT], p)["offline-0"]

    spec = mk("tiny-llama")
    spec.runner.model.load_state_dict(plain.runner.model.state_dict())
    spec.scheduler.draft_proposer.runner.model.random_init(991)
    got = spec.generate([PROMPT], p)["offline-0"]
    assert got == want
    assert spec.runner.spec_proposed > 0
    assert spec.runner.spec_accepted < spec.runner.spec_proposed


def test_draft_model_multi_request_and_multi_round():
    """Several concurrent requests across several rounds: catch-up passes
    must keep the shared-block-table draft KV consistent."""
    plain = mk(None, k=0)
    p = SamplingParams(max_tokens=30, temperature=0.0, ignore_eos=True)
    prompts = [PROMPT, list(range(60, 100)), [5, 6] * 10]
    want = plain.generate(prompts, p)

    spec = mk("tiny-llama")
    spec.runner.model.load_state_dict(plain.runner.model.state_dict())
    spec.scheduler.draft_proposer.load_target_weights(spec.runner.model)
    got = spec.generate(prompts, p)
    assert got == want


def test_rejection_sampling_with_draft_distribution():
    """d ~ q then accept w.p. min(1, p(d)/q(d)) else residual: the
    emitted token must be marginally ~ p."""
    eng = mk(None, k=0)
    runner = eng.runner
    V = 8
    logits_row = torch.tensor([2.0, 1.0, 0.5, 0.0, -1., -2., -3., -4.])
    p_target = torch.softmax(logits_row, dim=-1)
    # a deliberately different proposal distribution
    q = torch.softmax(torch.tensor(
        [0.0, 2.0, 1.0, -1., 0.5, -2., -3., -4.]), dim=-1)

    params = SamplingParams(max_tokens=4, temperature=1.0)
    rng = np.random.default_rng(0)
    counts = np.zeros(V)
    trials = 4000
    for _ in range(trials):
        d = int(rng.choice(V, p=q.numpy()))
        seq = Sequence("r0", [1, 2, 3], params)
        out = SchedulerOutput(scheduled=[
            ScheduledSeq(seq, 2, draft_tokens=[d], draft_probs=[q])])
        logits = torch.stack([logits_row, logits_row])
        toks = torch.tensor([0, 0], dtype=torch.long)
        fixed = runner._spec_stochastic_fix(out, [seq, seq], logits, toks)
        counts[int(fixed[0])] += 1
    freq = counts / trials
    for t in range(V):
        assert abs(freq[t] - float(p_target[t])) < 0.03, (t, freq[t])


def test_draft_model_stochastic_end_to_end():
    """Stochastic target + model drafts: runs to max_tokens, proposes,
    and stays deterministic under per-request seeds."""
    def run():
        spec = mk("tiny-llama", seed=5)
        spec.scheduler.draft_proposer.load_target_weights(
            spec.runner.model)
        p = SamplingParams(max_tokens=20, temperature=0.7, top_p=0.9,
                           seed=11, ignore_eos=True)
        return spec, spec.generate([PROMPT], p)["offline-0"]

    a_eng, a = run()
    assert len(a) == 20
    assert a_eng.runner.spec_proposed > 0
    _, b = run()
    assert a == b


def test_draft_progress_resets_on_preemption():
    """Preempt-recompute resets _draft_progress with num_computed so the
    draft re-prefills under the new block table."""
    seq = Sequence("r1", [1, 2, 3], SamplingParams(max_tokens=4))
    seq._draft_progress = 7
    seq.reset_for_recompute()
    assert getattr(seq, "_draft_progress", 0) == 0


def test_spec_with_logprobs_matches_plain():
    """Speculative chunks now carry logprobs: one value per accepted
    token, numerically matching the plain engine's per-step logprobs."""
    def run(spec):
        eng = mk("tiny-llama" if spec else None, k=4 if spec else 0)
        return eng

    plain = run(False)
    p = SamplingParams(max_tokens=16, temperature=0.0, ignore_eos=True,
                       logprobs=1)

    def collect(eng, params):
        eng.add_request("lp", PROMPT, params)
        toks, lps = [], []
        while eng.has_unfinished():
            for out in eng.step():
                toks.extend(out.new_token_ids)
                if out.new_logprobs:
                    lps.extend(out.new_logprobs)
        return toks, lps

    want_toks, want_lps = collect(plain, p)
    spec = run(True)
    spec.runner.model.load_state_dict(plain.runner.model.state_dict())
    spec.scheduler.draft_proposer.load_target_weights(spec.runner.model)
    got_toks, got_lps = collect(
        spec, SamplingParams(max_tokens=16, temperature=0.0,
                             ignore_eos=True, logprobs=1))
    assert got_toks == want_toks
    assert spec.runner.spec_proposed > 0
    assert len(got_lps) == len(want_lps) == 16
    for a, b in zip(got_lps, want_lps):
        assert abs(a - b) < 1e-4, (a, b)


def test_spec_with_top_logprobs_alignment():
    """logprobs=N + speculation: chosen-token logprobs stay one per
    token; top alternatives attach only to single-token positions
    (never replicated across an accepted chunk)."""
    spec = mk("tiny-llama")
    spec.scheduler.draft_proposer.load_target_weights(spec.runner.model)
    p = SamplingParams(max_tokens=12, temperature=0.0, ignore_eos=True,
                       logprobs=3)
    spec.add_request("tl", PROMPT, p)
    n_toks = n_lps = 0
    tops_rows = []
    while spec.has_unfinished():
        for out in spec.step():
            n_toks += len(out.new_token_ids)
            if out.new_logprobs:
                n_lps += len(out.new_logprobs)
            if out.new_top_logprobs:
                assert len(out.new_top_logprobs) <= len(
                    out.new_token_ids)
                tops_rows.extend(
                    t for t in out.new_top_logprobs if t)
    assert n_toks == n_lps == 12
    assert spec.runner.spec_proposed > 0
    for row in tops_rows:
        assert 1 <= len(row) <= 3
