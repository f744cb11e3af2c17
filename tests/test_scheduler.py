from production_stack_amd.engine.block_manager import BlockManager
from production_stack_amd.engine.config import SchedulerConfig
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.scheduler import Scheduler
from production_stack_amd.engine.sequence import Sequence, SeqStatus


def mk(bm_blocks=64, **kw):
    cfg = SchedulerConfig(**kw)
    bm = BlockManager(bm_blocks, 16)
    return Scheduler(cfg, bm, max_model_len=512), bm


def seq(rid, n_prompt, max_tokens=8):
    return Sequence(
        rid, list(range(100, 100 + n_prompt)), SamplingParams(max_tokens=max_tokens)
    )


def run_step(sched, sample_token=7):
    out = sched.schedule()
    sampled = {}
    for ss in out.scheduled:
        s = ss.seq
        if s.num_computed + ss.num_tokens == s.num_tokens:
            sampled[s.request_id] = sample_token
    sched.on_step_done(out, sampled, eos_token_id=2)
    return out, sampled


def test_basic_prefill_then_decode():
    sched, _ = mk()
    s = seq("a", 20, max_tokens=3)
    sched.add(s)
    out, sampled = run_step(sched)
    assert out.scheduled[0].num_tokens == 20
    assert sampled == {"a": 7}
    assert s.output_token_ids == [7]
    out, _ = run_step(sched)
    assert out.scheduled[0].num_tokens == 1  # decode
    run_step(sched)
    assert s.finished and s.status is SeqStatus.FINISHED_LENGTH
    assert not sched.has_unfinished()


def test_chunked_prefill():
    sched, _ = mk(max_num_batched_tokens=16)
    s = seq("a", 40, max_tokens=2)
    sched.add(s)
    out, sampled = run_step(sched)
    assert out.scheduled[0].num_tokens == 16 and not sampled
    out, sampled = run_step(sched)
    assert out.scheduled[0].num_tokens == 16 and not sampled
    out, sampled = run_step(sched)
    assert out.scheduled[0].num_tokens == 8 and sampled == {"a": 7}


def test_decode_priority_over_new_prefill():
    sched, _ = mk(max_num_batched_tokens=16)
    a = seq("a", 16, max_tokens=8)
    sched.add(a)
    run_step(sched)  # a prefilled + first token
    b = seq("b", 100, max_tokens=8)
    sched.add(b)
    out, _ = run_step(sched)
    kinds = [(ss.seq.request_id, ss.num_tokens) for ss in out.scheduled]
    assert kinds[0] == ("a", 1)  # decode first
    assert kinds[1][0] == "b" and kinds[1][1] == 15  # remaining budget


def test_eos_stops():
    sched, _ = mk()
    s = seq("a", 8, max_tokens=100)
    sched.add(s)
    out = sched.schedule()
    sched.on_step_done(out, {"a": 2}, eos_token_id=2)
    assert s.status is SeqStatus.FINISHED_STOPPED


def test_preemption_on_oom():
    # 8 blocks of 16 = 128 tokens capacity
    sched, bm = mk(bm_blocks=8, max_num_batched_tokens=256)
    a = seq("a", 60, max_tokens=30)  # grows to 90 tokens = 6 blocks
    b = seq("b", 60, max_tokens=30)
    sched.add(a)
    sched.add(b)
    run_step(sched)
    assert sched.num_running == 2
    # decode until blocks run out: a is at 60(+1), block 4 boundary at 64
    for _ in range(10):
        run_step(sched)
    # someone must have been preempted, but progress continues
    assert sched.has_unfinished()
    total = sched.num_running + sched.num_waiting
    assert total == 2
    # run to completion: preempted seq recomputes and both finish
    for _ in range(400):
        if not sched.has_unfinished():
            break
        run_step(sched)
    assert not sched.has_unfinished()
    assert a.finished and b.finished
    assert bm.num_free == 8


def test_abort():
    sched, bm = mk()
    s = seq("a", 16, max_tokens=100)
    sched.add(s)
    run_step(sched)
    sched.abort("a")
    assert s.status is SeqStatus.FINISHED_ABORTED
    assert sched.num_running == 0
    assert bm.num_free == bm.num_blocks


def test_max_num_seqs_cap():
    sched, _ = mk(max_num_seqs=2, max_num_batched_tokens=1024)
    for i in range(5):
        sched.add(seq(f"s{i}", 16, max_tokens=4))
    out = sched.schedule()
    assert len(out.scheduled) == 2
    assert sched.num_waiting == 3


def test_capacity_stop_when_single_seq_exceeds_cache():
    """A lone sequence that outgrows the entire cache finishes with
    reason=length instead of livelocking on self-preemption."""
    sched, _ = mk(bm_blocks=4, max_num_batched_tokens=256)  # 64-token cache
    s = seq("a", 32, max_tokens=400)
    sched.add(s)
    for _ in range(200):
        if not sched.has_unfinished():
            break
        run_step(sched)
    assert s.finished and s.status is SeqStatus.FINISHED_LENGTH
    assert len(s.output_token_ids) > 0


def test_speculative_drafts_respect_capacity_and_model_len():
    """Draft proposal must clamp at max_model_len and fall back cleanly
    when KV blocks for the drafts cannot be allocated."""
    from production_stack_amd.engine.block_manager import BlockManager
    from production_stack_amd.engine.config import SchedulerConfig
    from production_stack_amd.engine.scheduler import Scheduler
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.sequence import Sequence, SeqStatus

    bm = BlockManager(num_blocks=8, block_size=16,
                      enable_prefix_caching=False)
    cfg = SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=64,
                          num_speculative_tokens=4)
    # model_len barely above the sequence: k must clamp to fit
    sched = Scheduler(cfg, bm, max_model_len=68)
    seq = Sequence("s1", [5, 6] * 31,  # 62-token periodic prompt
                   SamplingParams(max_tokens=8, temperature=0.0,
                                  ignore_eos=True))
    sched.add(seq)
    out = sched.schedule()  # prefill
    sched.on_step_done(out, {"s1": 5}, eos_token_id=99999)
    out2 = sched.schedule()
    ss = out2.scheduled[0]
    # 63 tokens now; max_model_len 68 -> k <= 68 - 63 - 1 = 4 but the
    # proposal is also bounded by what fits
    assert ss.num_tokens - 1 == len(ss.draft_tokens)
    assert seq.num_computed + ss.num_tokens <= 68
    sched.on_step_done(out2, {"s1": [6] * min(2, ss.num_tokens)},
                       eos_token_id=99999)
    assert seq.num_computed == seq.num_tokens - 1  # decode-ready again


def test_speculative_acceptance_burst_bookkeeping():
    """A fully-accepted draft burst must leave num_computed = num_tokens-1
    and block accounting consistent."""
    from production_stack_amd.engine.block_manager import BlockManager
    from production_stack_amd.engine.config import SchedulerConfig
    from production_stack_amd.engine.scheduler import Scheduler
    from production_stack_amd.engine.sampling import SamplingParams
    from production_stack_amd.engine.sequence import Sequence

    bm = BlockManager(num_blocks=64, block_size=16,
                      enable_prefix_caching=False)
    cfg = SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=256,
                          num_speculative_tokens=3)
    sched = Scheduler(cfg, bm, max_model_len=512)
    seq = Sequence("s1", [1, 2, 3, 4] * 10,
                   SamplingParams(max_tokens=32, temperature=0.0,
                                  ignore_eos=True))
    sched.add(seq)
    out = sched.schedule()
    sched.on_step_done(out, {"s1": 1}, eos_token_id=99999)
    out2 = sched.schedule()
    ss = out2.scheduled[0]
    assert len(ss.draft_tokens) == 3
    accepted = [int(t) for t in (2, 3, 4, 1)]  # all drafts accepted
    sched.on_step_done(out2, {"s1": accepted}, eos_token_id=99999)
    assert len(seq.output_token_ids) == 1 + 4
    assert seq.num_computed == seq.num_tokens - 1
    # drain to completion without accounting drift
    while sched.has_unfinished():
        o = sched.schedule()
        sched.on_step_done(
            o, {"s1": 7}, eos_token_id=99999
        )
    assert bm.num_free == bm.num_blocks
