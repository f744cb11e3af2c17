"""GPU end-to-end engine tests (mini model with kernel-supported head_dim)."""

import pytest
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams

pytestmark = pytest.mark.gpu


def make_engine(**kw):
    cfg = EngineConfig(
        model=kw.pop("model", "mini-llama"),
        max_model_len=kw.pop("max_model_len", 1024),
        cache=CacheConfig(
            num_gpu_blocks=kw.pop("num_gpu_blocks", 256), block_size=16
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=kw.pop("max_num_seqs", 16),
            max_num_batched_tokens=kw.pop("max_num_batched_tokens", 2048),
        ),
        **kw,
    )
    return LLMEngine(cfg, device="cuda")


def test_extension_loaded():
    from production_stack_amd import ops

    assert ops.HAVE_EXT, "HIP extension must load on the GPU box"


def test_generate_deterministic():
    eng = make_engine()
    p = SamplingParams(max_tokens=16, temperature=0.0, ignore_eos=True)
    prompt = list(range(100, 180))
    o1 = eng.generate([prompt], p)["offline-0"]
    o2 = eng.generate([prompt], p)["offline-0"]
    assert o1 == o2 and len(o1) == 16


def test_gpu_matches_cpu_reference_engine():
    """Same tiny model on GPU (HIP kernels) vs CPU (torch reference):
    greedy tokens must agree (same random-init seed)."""
    from production_stack_amd.ops import gemm_policy

    p = SamplingParams(max_tokens=12, temperature=0.0, ignore_eos=True)
    prompt = list(range(7, 87))
    gpu = make_engine()
    # plain hipBLASLt on both sides: the autotuned skinny kernel changes
    # summation order, which flips random-init argmaxes
    gemm_policy.reset()
    out_gpu = gpu.generate([prompt], p)["offline-0"]
    cfg = EngineConfig(
        model="mini-llama",
        max_model_len=1024,
        cache=CacheConfig(num_gpu_blocks=256, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=16, max_num_batched_tokens=2048),
    )
    # CPU random_init uses a CPU generator -> different weights; instead copy
    # the GPU engine's weights down.
    cpu = LLMEngine(cfg, device="cpu")
    sd = {k: v.cpu() for k, v in gpu.runner.model.state_dict().items()}
    cpu.runner.model.load_state_dict(sd)
    out_cpu = cpu.generate([prompt], p)["offline-0"]
    # bf16 kernel-vs-reference drift can flip argmax on random-init logits
    # (uniform-ish random logits make argmax extremely noise-sensitive);
    # require agreement on the early tokens.
    agree = sum(a == b for a, b in zip(out_gpu, out_cpu))
    assert agree >= 4, f"{out_gpu} vs {out_cpu}"


def test_chunked_prefill_matches_single_shot_gpu():
    prompt = list(range(10, 500))
    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    big = make_engine()
    out_big = big.generate([prompt], p)["offline-0"]
    small = make_engine(max_num_batched_tokens=128)
    out_small = small.generate([prompt], p)["offline-0"]
    assert out_big == out_small


def test_prefix_cache_reuse_gpu():
    eng = make_engine()
    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    prompt = list(range(3, 300))
    cold = eng.generate([prompt], p)["offline-0"]
    warm = eng.generate([prompt], p)["offline-0"]
    assert eng.block_manager.prefix_hits > 0
    assert cold == warm


def test_batched_mixed_phase_gpu():
    """Batch where prefills and decodes run in one step."""
    eng = make_engine()
    p = SamplingParams(max_tokens=6, temperature=0.0, ignore_eos=True)
    prompts = [list(range(50, 120)), list(range(200, 230)), [9, 8, 7]]
    batched = eng.generate(prompts, p)
    for i, prompt in enumerate(prompts):
        solo = make_engine().generate([prompt], p)["offline-0"]
        assert batched[f"offline-{i}"] == solo


def test_flagship_8b_one_decode():
    free, _ = torch.cuda.mem_get_info()
    if free < 30e9:
        pytest.skip("needs ~30 GB free HBM")
    eng = make_engine(
        model="llama-3-8b", num_gpu_blocks=512, max_model_len=2048
    )
    p = SamplingParams(max_tokens=2, temperature=0.0, ignore_eos=True)
    out = eng.generate([list(range(1000, 1128))], p)["offline-0"]
    assert len(out) == 2


def test_lora_served_through_decode_graphs(tmp_path):
    """enable_lora: adapter requests must run through the captured decode
    graphs (BGMV slots), produce adapter-specific outputs, and leave base
    requests untouched."""
    from production_stack_amd.engine.lora import save_synthetic_adapter
    from production_stack_amd.engine.sampling import SamplingParams

    adir = str(tmp_path / "ad")
    save_synthetic_adapter(adir, hidden=512, q_size=512, kv_size=256,
                           num_layers=4, seed=9)
    eng = make_engine(enable_lora=True, max_loras=2, max_lora_rank=8)
    assert eng.runner.graphs is not None
    assert eng.runner.graphs.use_lora
    eng.load_lora("ad", adir)

    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    prompt = list(range(300, 340))
    eng.add_request("base", prompt, p)
    eng.add_request("lora", prompt, p, lora_name="ad")
    outs = {"base": [], "lora": []}
    while eng.has_unfinished():
        for r in eng.step():
            outs[r.request_id].extend(r.new_token_ids)
    assert len(outs["base"]) == 8 and len(outs["lora"]) == 8
    assert outs["base"] != outs["lora"], "adapter must change the tokens"

    # same request on a fresh eager-LoRA engine must agree with the
    # graph+BGMV path
    ref = make_engine()
    ref.runner.model.load_state_dict(eng.runner.model.state_dict())
    ref.load_lora("ad", adir)
    ref.add_request("lora", prompt, p, lora_name="ad")
    ref_out = []
    while ref.has_unfinished():
        for r in ref.step():
            ref_out.extend(r.new_token_ids)
    agree = sum(a == b for a, b in zip(outs["lora"], ref_out))
    assert agree >= 4, f"{outs['lora']} vs {ref_out}"


def test_mistral_sliding_window_gpu_matches_cpu():
    """mini-mistral (window=64): windowed HIP kernels (decode + MFMA
    prefill) must agree with the CPU reference implementation."""
    from production_stack_amd.engine.config import (
        CacheConfig as CC,
        EngineConfig as EC,
        SchedulerConfig as SC,
    )
    from production_stack_amd.ops import gemm_policy

    def cfg():
        return EC(
            model="mini-mistral",
            max_model_len=1024,
            cache=CC(num_gpu_blocks=256, block_size=16),
            scheduler=SC(max_num_seqs=8, max_num_batched_tokens=2048),
        )

    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    prompt = list(range(9, 209))  # 200 tokens >> window 64
    gemm_policy.reset()
    gpu = LLMEngine(cfg(), device="cuda")
    out_gpu = gpu.generate([prompt], p)["offline-0"]
    cpu = LLMEngine(cfg(), device="cpu")
    cpu.runner.model.load_state_dict(
        {k: v.cpu() for k, v in gpu.runner.model.state_dict().items()}
    )
    out_cpu = cpu.generate([prompt], p)["offline-0"]
    agree = sum(a == b for a, b in zip(out_gpu, out_cpu))
    assert agree >= 4, f"{out_gpu} vs {out_cpu}"


def test_speculative_ngram_gpu_matches_plain():
    """Speculative chunks run through the MFMA prefill path on GPU and
    must reproduce the plain greedy output exactly."""
    from production_stack_amd.engine.config import SchedulerConfig as SC

    def mk(spec):
        cfg = EngineConfig(
            model="mini-llama",
            max_model_len=1024,
            seed=2,
            cache=CacheConfig(num_gpu_blocks=256, block_size=16),
            scheduler=SC(max_num_seqs=8, max_num_batched_tokens=2048,
                         num_speculative_tokens=4 if spec else 0),
        )
        return LLMEngine(cfg, device="cuda")

    p = SamplingParams(max_tokens=24, temperature=0.0, ignore_eos=True)
    prompt = [11, 12, 13, 14, 15] * 10
    plain = mk(False)
    want = plain.generate([prompt], p)["offline-0"]
    spec = mk(True)
    spec.runner.model.load_state_dict(plain.runner.model.state_dict())
    got = spec.generate([prompt], p)["offline-0"]
    assert got == want, f"{got} != {want}"
    assert spec.runner.spec_proposed > 0

    # stochastic sampling through the same chunks: rejection-sampling
    # acceptance (near-greedy temperature so drafts fire), seeded
    # determinism across two engines
    def fresh():
        return SamplingParams(max_tokens=20, temperature=0.05, seed=7,
                              ignore_eos=True)

    got1 = spec.generate([prompt], fresh())["offline-0"]
    assert len(got1) == 20
    spec2 = mk(True)
    spec2.runner.model.load_state_dict(plain.runner.model.state_dict())
    got2 = spec2.generate([prompt], fresh())["offline-0"]
    assert got2 == got1, f"{got2} != {got1}"


def test_async_scheduling_gpu_matches_sync():
    """Async (one-step-lagged) scheduling on GPU: decode graphs + device
    token gather must reproduce the synchronous outputs exactly."""
    p = SamplingParams(max_tokens=16, temperature=0.0, ignore_eos=True)
    prompts = [list(range(100, 180)), list(range(7, 40)), [5, 6, 7]]
    sync = make_engine()
    want = sync.generate(prompts, p)
    a = make_engine(async_scheduling=True)
    a.runner.model.load_state_dict(sync.runner.model.state_dict())
    got = a.generate(prompts, p)
    assert got == want, f"{got} != {want}"


def test_fp8_weight_quantization_gpu():
    """fp8 weights run through torch._scaled_mm inside the captured
    decode graphs; output matches the CPU fp8 reference engine's early
    tokens."""
    eng = make_engine(quantization="fp8")
    assert eng.runner.graphs is not None
    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    prompt = list(range(50, 120))
    out = eng.generate([prompt], p)["offline-0"]
    assert len(out) == 8
    # weights on the GPU engine were quantized from ITS random init; build
    # the CPU reference from the same quantized tensors
    cfg = EngineConfig(
        model="mini-llama",
        max_model_len=1024,
        cache=CacheConfig(num_gpu_blocks=256, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=16,
                                  max_num_batched_tokens=2048),
        quantization="fp8",
    )
    cpu = LLMEngine(cfg, device="cpu")
    for lg, lc in zip(eng.runner.model.layers, cpu.runner.model.layers):
        lc.fp8_w = {
            k: (wq.cpu(), sc.cpu()) for k, (wq, sc) in lg.fp8_w.items()
        }
        lc.input_norm.data = lg.input_norm.data.cpu()
        lc.post_attn_norm.data = lg.post_attn_norm.data.cpu()
    cpu.runner.model.embed.data = eng.runner.model.embed.data.cpu()
    cpu.runner.model.final_norm.data = (
        eng.runner.model.final_norm.data.cpu()
    )
    cpu.runner.model.lm_head.data = eng.runner.model.lm_head.data.cpu()
    want = cpu.generate([prompt], p)["offline-0"]
    agree = sum(a == b for a, b in zip(out[:4], want[:4]))
    assert agree >= 2, (out, want)


def test_multimodal_vision_injection_gpu():
    """Vision embeddings injected over placeholder tokens on the GPU path
    (eager prefill + decode graphs): deterministic per image, different
    across images."""
    import io

    import numpy as np
    from PIL import Image

    from production_stack_amd.engine.models.multimodal import (
        decode_image,
        media_placeholder_tokens,
    )

    eng = make_engine()
    enc = eng.get_vision_encoder()

    def png(color):
        img = Image.new("RGB", (64, 64), color)
        buf = io.BytesIO()
        img.save(buf, format="PNG")
        return buf.getvalue()

    def run(data, tag):
        emb = enc(decode_image(data).to(eng.device))
        toks = [5, 6, 7] + media_placeholder_tokens(
            data, emb.shape[0], eng.model_cfg.vocab_size) + [9, 10]
        eng.add_request(tag, toks, SamplingParams(
            max_tokens=8, temperature=0.0, ignore_eos=True),
            mm_embeds=[(3, emb)])
        out = []
        while eng.has_unfinished():
            for o in eng.step():
                if o.request_id == tag:
                    out.extend(o.new_token_ids)
        return out

    a1 = run(png((255, 0, 0)), "a1")
    a2 = run(png((255, 0, 0)), "a2")
    b = run(png((0, 0, 255)), "b1")
    assert a1 == a2
    assert a1 != b


def test_unified_mixed_steps_matches_split_path():
    """The unified eager mixed step (default) must be token-identical to
    the split graph+eager path on the same greedy requests."""
    outs = {}
    for unified in (True, False):
        eng = make_engine(unified_mixed_steps=unified)
        prompts = [list(range(20 + i, 140 + i)) for i in range(6)]
        for i, p in enumerate(prompts):
            eng.add_request(f"r{i}", p, SamplingParams(
                max_tokens=24, temperature=0.0, ignore_eos=True))
        # stagger two more so steps stay mixed (prefill + decode)
        got = {f"r{i}": [] for i in range(8)}
        step = 0
        while eng.has_unfinished():
            if step == 2:
                for i in (6, 7):
                    eng.add_request(f"r{i}", list(range(40 + i, 200 + i)),
                                    SamplingParams(max_tokens=24,
                                                   temperature=0.0,
                                                   ignore_eos=True))
            for o in eng.step():
                got[o.request_id].extend(o.new_token_ids)
            step += 1
        outs[unified] = got
        del eng
        torch.cuda.empty_cache()
    assert outs[True] == outs[False]


def test_guided_json_gpu():
    """Guided JSON masking on a cuda engine: every emitted token keeps
    the output a valid JSON prefix (mask indexes device logits)."""
    import json as _json

    from production_stack_amd.engine.guided import JsonPrefixValidator
    from test_guided import JsonToyTokenizer

    eng = make_engine()
    tok = JsonToyTokenizer(eng.model_cfg.vocab_size)
    eng.tokenizer = tok
    eng.runner.tokenizer = tok
    p = SamplingParams(max_tokens=32, temperature=1.0, seed=9,
                       response_format={"type": "json_object"})
    eng.add_request("gj", [33, 34, 35], p)
    toks, reason = [], None
    for _ in range(60):
        for out in eng.step():
            if out.request_id == "gj":
                toks.extend(out.new_token_ids)
                if out.finished:
                    reason = out.finish_reason
        if reason:
            break
    text = "".join(tok.decode_token(t) for t in toks
                   if t != tok.eos_token_id)
    assert JsonPrefixValidator().feed_text(text), text
    assert reason in ("stop", "length")
    if reason == "stop":
        _json.loads(text)


def test_draft_model_spec_gpu():
    """Draft-model speculation on GPU. Exact equality with the plain
    engine is asserted on CPU (tests/test_draft_spec.py) where both
    paths run identical fp32 kernels; on GPU the verification chunk
    (MFMA prefill) and graph decode can flip bf16 argmax near-ties, so
    the invariants here are: deterministic across runs, near-total
    acceptance with a perfect draft, and ≤1 near-tie divergence from
    the plain engine's output."""
    from production_stack_amd.engine.config import SchedulerConfig as SC

    p = SamplingParams(max_tokens=20, temperature=0.0, ignore_eos=True)
    prompt = [11, 12, 13, 14, 15] * 10
    plain = make_engine()
    want = plain.generate([prompt], p)["offline-0"]

    def run():
        cfg = EngineConfig(
            model="mini-llama",
            max_model_len=1024,
            speculative_model="mini-llama",
            cache=CacheConfig(num_gpu_blocks=256, block_size=16),
            scheduler=SC(max_num_seqs=8, max_num_batched_tokens=2048,
                         num_speculative_tokens=4),
        )
        spec = LLMEngine(cfg, device="cuda")
        spec.runner.model.load_state_dict(plain.runner.model.state_dict())
        spec.scheduler.draft_proposer.load_target_weights(
            spec.runner.model)
        return spec, spec.generate([prompt], p)["offline-0"]

    eng1, got1 = run()
    assert len(got1) == 20
    assert eng1.runner.spec_proposed > 0
    assert (eng1.runner.spec_accepted
            >= 0.5 * eng1.runner.spec_proposed)
    diffs = sum(a != b for a, b in zip(got1, want))
    assert diffs <= 1, f"{got1} != {want}"
    _, got2 = run()
    assert got2 == got1


def test_mixtral_moe_gpu_matches_cpu():
    """mini-mixtral: MoE routing + per-expert GEMMs on GPU (hipBLASLt +
    fused SiLU-mul; decode graphs disabled for MoE) agree with the CPU
    reference on early tokens, and GPU runs are deterministic."""
    from production_stack_amd.engine.config import (
        CacheConfig as CC,
        EngineConfig as EC,
        SchedulerConfig as SC,
    )

    def cfg():
        return EC(
            model="mini-mixtral",
            max_model_len=1024,
            cache=CC(num_gpu_blocks=256, block_size=16),
            scheduler=SC(max_num_seqs=8, max_num_batched_tokens=2048),
        )

    p = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True)
    prompt = list(range(9, 129))
    gpu = LLMEngine(cfg(), device="cuda")
    assert gpu.runner.graphs is None  # data-dependent routing: no capture
    out_gpu = gpu.generate([prompt], p)["offline-0"]
    gpu2 = LLMEngine(cfg(), device="cuda")
    gpu2.runner.model.load_state_dict(gpu.runner.model.state_dict())
    assert gpu2.generate([prompt], p)["offline-0"] == out_gpu
    cpu = LLMEngine(cfg(), device="cpu")
    cpu.runner.model.load_state_dict(
        {k: v.cpu() for k, v in gpu.runner.model.state_dict().items()}
    )
    out_cpu = cpu.generate([prompt], p)["offline-0"]
    # MoE routing amplifies bf16 cross-device noise (a near-tie router
    # flip reroutes a token through a different expert), so only the
    # early tokens are expected to agree
    agree = sum(a == b for a, b in zip(out_gpu[:4], out_cpu[:4]))
    assert agree >= 2, f"{out_gpu} vs {out_cpu}"


def test_logprobs_and_prompt_logprobs_gpu():
    """Sampled-token logprobs, top alternatives and prompt_logprobs on a
    cuda engine: finite, correctly shaped, greedy-consistent (the chosen
    token's logprob equals the max alternative)."""
    eng = make_engine()
    p = SamplingParams(max_tokens=4, temperature=0.0, ignore_eos=True,
                       logprobs=3, prompt_logprobs=1)
    eng.add_request("lp", list(range(50, 90)), p)
    lps, tops, plp = [], [], None
    while eng.has_unfinished():
        for out in eng.step():
            if out.new_logprobs:
                lps.extend(out.new_logprobs)
            if out.new_top_logprobs:
                tops.extend(t for t in out.new_top_logprobs if t)
            if out.prompt_logprobs is not None:
                plp = out.prompt_logprobs
    assert len(lps) == 4 and all(v <= 0 for v in lps)
    assert plp is not None and plp[0] is None and len(plp) == 40
    assert all(v <= 0 for v in plp[1:])
    for row, chosen in zip(tops, lps):
        assert abs(max(v for _, v in row) - chosen) < 1e-4
