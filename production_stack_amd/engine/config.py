"""Engine configuration.

Mirrors the engine-flag surface the reference deploys through helm
(`vllm serve` argv assembled at reference helm/templates/
deployment-vllm-multi.yaml:127-221 — SURVEY.md section 2.9): model, dtype,
max-model-len, max-num-seqs, gpu-memory-utilization, enable-prefix-caching,
enable-chunked-prefill, kv-transfer role. Architecture definitions are local
(random-init weights by default: no network in the build environment).
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    name: str
    hidden_size: int
    num_layers: int
    num_q_heads: int
    num_kv_heads: int
    head_dim: int
    intermediate_size: int
    vocab_size: int
    rope_theta: float = 500000.0
    max_position: int = 8192
    rms_norm_eps: float = 1e-5
    bos_token_id: int = 1
    eos_token_id: int = 2
    # family knobs: Qwen2 uses qkv biases; Mistral attends over a sliding
    # window; small models often tie lm_head to the embedding
    qkv_bias: bool = False
    tie_word_embeddings: bool = False
    sliding_window: Optional[int] = None
    # Llama-3.1-style RoPE frequency scaling (None = off):
    # (factor, low_freq_factor, high_freq_factor, original_max_position)
    rope_scaling: Optional[tuple] = None
    # Mixtral-style sparse MoE MLP: num_experts > 0 replaces the dense
    # gate/up/down with per-expert projections + a top-k router
    num_experts: int = 0
    num_experts_per_tok: int = 2

    @property
    def q_size(self) -> int:
        return self.num_q_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


# Named architectures (shapes follow the public HF configs for the Llama
# family; weights are random-init — see BASELINE.md: synthetic data /
# random-init weights).
ARCHITECTURES = {
    "llama-3-8b": ModelConfig(
        name="llama-3-8b",
        hidden_size=4096,
        num_layers=32,
        num_q_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=8192,
    ),
    "llama-3-70b": ModelConfig(
        name="llama-3-70b",
        hidden_size=8192,
        num_layers=80,
        num_q_heads=64,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=28672,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=8192,
    ),
    "llama-3.1-8b": ModelConfig(
        name="llama-3.1-8b",
        hidden_size=4096,
        num_layers=32,
        num_q_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=131072,
        rope_scaling=(8.0, 1.0, 4.0, 8192),
    ),
    "llama-3.1-70b": ModelConfig(
        name="llama-3.1-70b",
        hidden_size=8192,
        num_layers=80,
        num_q_heads=64,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=28672,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=131072,
        rope_scaling=(8.0, 1.0, 4.0, 8192),
    ),
    "llama-2-7b": ModelConfig(
        name="llama-2-7b",
        hidden_size=4096,
        num_layers=32,
        num_q_heads=32,
        num_kv_heads=32,
        head_dim=128,
        intermediate_size=11008,
        vocab_size=32000,
        rope_theta=10000.0,
        max_position=4096,
    ),
    "llama-2-13b": ModelConfig(
        name="llama-2-13b",
        hidden_size=5120,
        num_layers=40,
        num_q_heads=40,
        num_kv_heads=40,
        head_dim=128,
        intermediate_size=13824,
        vocab_size=32000,
        rope_theta=10000.0,
        max_position=4096,
    ),
    "llama-3.2-1b": ModelConfig(
        name="llama-3.2-1b",
        hidden_size=2048,
        num_layers=16,
        num_q_heads=32,
        num_kv_heads=8,
        head_dim=64,
        intermediate_size=8192,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=8192,
        tie_word_embeddings=True,
    ),
    "llama-3.2-3b": ModelConfig(
        name="llama-3.2-3b",
        hidden_size=3072,
        num_layers=28,
        num_q_heads=24,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=8192,
        vocab_size=128256,
        rope_theta=500000.0,
        max_position=8192,
        tie_word_embeddings=True,
    ),
    "qwen2-72b": ModelConfig(
        name="qwen2-72b",
        hidden_size=8192,
        num_layers=80,
        num_q_heads=64,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=29568,
        vocab_size=152064,
        rope_theta=1000000.0,
        max_position=32768,
        qkv_bias=True,
    ),
    "mistral-7b": ModelConfig(
        name="mistral-7b",
        hidden_size=4096,
        num_layers=32,
        num_q_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=32000,
        rope_theta=10000.0,
        max_position=32768,
        sliding_window=4096,
    ),

    "mixtral-8x7b": ModelConfig(
        name="mixtral-8x7b",
        hidden_size=4096,
        num_layers=32,
        num_q_heads=32,
        num_kv_heads=8,
        head_dim=128,
        intermediate_size=14336,
        vocab_size=32000,
        rope_theta=1e6,
        max_position=32768,
        num_experts=8,
        num_experts_per_tok=2,
    ),
    "tiny-mixtral": ModelConfig(
        name="tiny-mixtral",
        hidden_size=64,
        num_layers=2,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=16,
        intermediate_size=128,
        vocab_size=512,
        max_position=512,
        num_experts=4,
        num_experts_per_tok=2,
    ),
    "qwen2-7b": ModelConfig(
        name="qwen2-7b",
        hidden_size=3584,
        num_layers=28,
        num_q_heads=28,
        num_kv_heads=4,
        head_dim=128,
        intermediate_size=18944,
        vocab_size=152064,
        rope_theta=1000000.0,
        max_position=32768,
        qkv_bias=True,
    ),
    # Small config for tests (CPU-runnable).
    "tiny-llama": ModelConfig(
        name="tiny-llama",
        hidden_size=128,
        num_layers=2,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=32,
        intermediate_size=256,
        vocab_size=1024,
        rope_theta=10000.0,
        max_position=2048,
    ),
    "tiny-mistral": ModelConfig(
        name="tiny-mistral",
        hidden_size=128,
        num_layers=2,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=32,
        intermediate_size=256,
        vocab_size=1024,
        rope_theta=10000.0,
        max_position=2048,
        sliding_window=48,
    ),
    "tiny-qwen2": ModelConfig(
        name="tiny-qwen2",
        hidden_size=128,
        num_layers=2,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=32,
        intermediate_size=256,
        vocab_size=1024,
        rope_theta=10000.0,
        max_position=2048,
        qkv_bias=True,
        tie_word_embeddings=True,
    ),
    "mini-mistral": ModelConfig(
        name="mini-mistral",
        hidden_size=512,
        num_layers=4,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=1024,
        vocab_size=2048,
        rope_theta=10000.0,
        max_position=4096,
        sliding_window=64,
    ),
    # GPU-runnable small config with kernel-supported head_dim.
    "mini-mixtral": ModelConfig(
        name="mini-mixtral",
        hidden_size=512,
        num_layers=4,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=1024,
        vocab_size=2048,
        rope_theta=10000.0,
        max_position=4096,
        num_experts=4,
        num_experts_per_tok=2,
    ),
    "mini-llama": ModelConfig(
        name="mini-llama",
        hidden_size=512,
        num_layers=4,
        num_q_heads=4,
        num_kv_heads=2,
        head_dim=128,
        intermediate_size=1024,
        vocab_size=2048,
        rope_theta=10000.0,
        max_position=4096,
    ),
}


@dataclass
class CacheConfig:
    block_size: int = 16
    # Either an explicit block count (tests) or a fraction of free GPU memory.
    num_gpu_blocks: Optional[int] = None
    gpu_memory_utilization: float = 0.85
    enable_prefix_caching: bool = True
    # Host-DRAM KV offload pool size in GiB (0 disables). LMCache-equivalent
    # surface: reference LMCACHE_MAX_LOCAL_CPU_SIZE
    # (deployment-vllm-multi.yaml:336-343).
    cpu_offload_gb: float = 0.0
    # remote cacheserver data plane ("host:port"), shared across instances
    # (reference cacheserverSpec / LMCACHE_REMOTE_URL)
    remote_kv_url: Optional[str] = None
    remote_kv_serde: str = "cachegen"  # raw | cachegen | cachegen4
    # "bf16" (raw) or "int8" (CacheGen-style row-quantized serde, halves
    # host-pool bytes; reference LMCACHE remote serde surface)
    offload_dtype: str = "bf16"
    # KV cache storage dtype: "auto"/"bf16" or "fp8"/"fp8_e4m3" (OCP e4m3;
    # halves decode KV bandwidth; vLLM --kv-cache-dtype parity)
    kv_cache_dtype: str = "auto"


@dataclass
class SchedulerConfig:
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 8192
    enable_chunked_prefill: bool = True
    # Cap on how many prompt tokens a single prefill chunk may carry.
    max_prefill_chunk: int = 8192
    # n-gram (prompt-lookup) speculative decoding: propose up to K draft
    # tokens per greedy decode by matching the sequence's trailing n-gram
    # against its own history; the drafts are verified in one multi-token
    # step through the prefill path. 0 = off.
    num_speculative_tokens: int = 0
    ngram_min: int = 2
    ngram_max: int = 4


@dataclass
class ParallelConfig:
    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    # >1 splits each step's batch into M microbatches and issues them
    # back-to-back through the PP stages so stage k computes microbatch i
    # while stage k-1 computes i+1 (in-flight pipelining; see pipeline.py)
    pp_microbatches: int = 1
    rank: int = 0
    # disaggregated prefill role: None | "prefill" | "decode"
    kv_role: Optional[str] = None


@dataclass
class EngineConfig:
    model: str = "llama-3-8b"
    dtype: str = "bfloat16"
    max_model_len: int = 4096
    seed: int = 0
    weights_path: Optional[str] = None  # safetensors dir (optional)
    tokenizer: str = "synthetic"  # or a path to a tokenizer.json dir
    enforce_eager: bool = False  # True disables hipGraph decode capture
    # Mixed prefill+decode steps: True (default) runs one unified eager
    # batch so each layer's weights are read ONCE per step instead of
    # once for the decode graph replay and once for the prefill GEMMs —
    # measured +15-25% output tok/s AND lower TTFT at every load from 32
    # to 320 users on MI355X (graph replay remains for pure-decode
    # steps). False restores the split graph+eager mixed path.
    unified_mixed_steps: bool = True
    # graph-safe batched LoRA (BGMV slots); when False, adapters still run
    # through the eager grouped path
    # one-step-lagged sampling: the host schedules/launches step N+1 while
    # step N's sampled tokens are still in flight (decode inputs gather
    # from the previous step's device tensor). Greedy-exact; steps with
    # non-greedy sampling, penalties, logprobs or speculative chunks fall
    # back to the synchronous path transparently.
    async_scheduling: bool = False
    # weight quantization: None (bf16) or "fp8" (OCP e4m3 weights +
    # dynamic per-tensor activation quant through the fp8 MFMA pipe)
    quantization: Optional[str] = None
    # draft-model speculative decoding (engine/draft.py): architecture
    # name of the (smaller, same-vocab) proposer; None = n-gram drafts
    speculative_model: Optional[str] = None
    speculative_weights_path: Optional[str] = None
    enable_lora: bool = False
    max_loras: int = 4
    max_lora_rank: int = 16
    cache: CacheConfig = field(default_factory=CacheConfig)
    scheduler: SchedulerConfig = field(default_factory=SchedulerConfig)
    parallel: ParallelConfig = field(default_factory=ParallelConfig)

    def model_config(self) -> ModelConfig:
        if self.model not in ARCHITECTURES:
            raise ValueError(
                f"unknown model architecture {self.model!r}; "
                f"known: {sorted(ARCHITECTURES)}"
            )
        cfg = ARCHITECTURES[self.model]
        if self.max_model_len > cfg.max_position:
            cfg = dataclasses.replace(cfg, max_position=self.max_model_len)
        return cfg
