"""production_stack_amd: a brand-new MI355X-native multi-replica LLM serving
stack with the capabilities of vllm-project/production-stack.

Layers (see SURVEY.md for the reference blueprint):
  router/   - OpenAI-compatible L7 request router (round-robin / session /
              prefix-aware / KV-aware / disaggregated-prefill routing,
              service discovery, stats, failover, dynamic config).
  engine/   - the serving engine: continuous-batching scheduler, paged KV
              cache with prefix caching, Llama-family models, OpenAI HTTP
              front with SSE streaming and vllm:* Prometheus metrics.
  ops/      - hand-written CDNA4 HIP kernels (paged attention, RMSNorm,
              RoPE, SiLU-mul, KV append, sampling) + CPU references.
  kvpool/   - KV offload/tiering (HBM <-> pinned host DRAM) and the KV
              controller powering KV-aware routing.
  parallel/ - tensor parallelism and disaggregated-prefill KV transfer over
              RCCL / xGMI.
"""

__version__ = "0.2.0"
