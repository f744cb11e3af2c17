"""Llama-family causal LM built on the MI355X op set.

Compute mapping (MI355X-first):
  * plain GEMMs (qkv / o / gate-up / down / lm_head) -> torch F.linear, which
    lowers to hipBLASLt on ROCm;
  * everything else on the hot path is a hand-written HIP kernel via
    production_stack_amd.ops: fused residual-add RMSNorm, fused rope over
    q+k, paged-KV append, paged attention (prefill + decode), SiLU-mul.
  * weights are plain bf16 Parameters; qkv and gate/up are pre-fused so each
    layer issues exactly 4 GEMMs.

Supports tensor parallelism: q/kv heads and the MLP intermediate dim are
sharded across the TP group; o_proj and down_proj outputs are all-reduced
(RCCL over xGMI on the GPU box, gloo in CPU tests).
"""

from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from production_stack_amd import ops
from production_stack_amd.ops import gemm_policy
from production_stack_amd.engine.config import ModelConfig
from production_stack_amd.parallel import state as pstate


class BatchMeta:
    """Per-step metadata consumed by the attention layers."""

    def __init__(
        self,
        positions: torch.Tensor,  # [T] int32
        slot_mapping: torch.Tensor,  # [T] int64
        num_prefill_tokens: int,
        prefill_token_seq: Optional[torch.Tensor],  # [Tp] int32
        prefill_token_pos: Optional[torch.Tensor],  # [Tp] int32
        prefill_block_tables: Optional[torch.Tensor],  # [Sp, MB] int32
        num_decode_seqs: int,
        decode_seq_lens: Optional[torch.Tensor],  # [Sd] int32
        decode_block_tables: Optional[torch.Tensor],  # [Sd, MB] int32
        prefill_tiles: Optional[torch.Tensor] = None,  # [NT, 4] int32
        lora_groups=None,  # [(LoRAAdapter, row_idx_tensor)]
        lora_idx: Optional[torch.Tensor] = None,  # [T] int32 slot per row
        mm_rows: Optional[torch.Tensor] = None,   # [M] long batch rows
        mm_embeds: Optional[torch.Tensor] = None,  # [M, H] bf16
    ) -> None:
        self.positions = positions
        self.slot_mapping = slot_mapping
        self.num_prefill_tokens = num_prefill_tokens
        self.prefill_token_seq = prefill_token_seq
        self.prefill_token_pos = prefill_token_pos
        self.prefill_block_tables = prefill_block_tables
        self.num_decode_seqs = num_decode_seqs
        self.decode_seq_lens = decode_seq_lens
        self.decode_block_tables = decode_block_tables
        self.prefill_tiles = prefill_tiles
        self.lora_groups = lora_groups or []
        self.lora_idx = lora_idx
        self.mm_rows = mm_rows
        self.mm_embeds = mm_embeds


def build_cos_sin_cache(
    head_dim: int, max_position: int, theta: float,
    rope_scaling=None,
) -> torch.Tensor:
    """[max_position, head_dim] fp32: cos[half] || sin[half] (host-side).

    rope_scaling = (factor, low_freq_factor, high_freq_factor,
    original_max_position) applies the Llama-3.1 frequency remap: long
    wavelengths divide by `factor`, short ones pass through, the band in
    between interpolates smoothly (HF modeling_rope_utils llama3 rule)."""
    half = head_dim // 2
    inv = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
    )
    if rope_scaling is not None:
        factor, low_f, high_f, orig_max = rope_scaling
        low_wl = orig_max / low_f
        high_wl = orig_max / high_f
        wavelen = 2 * math.pi / inv
        scaled = torch.where(wavelen > low_wl, inv / factor, inv)
        smooth = (orig_max / wavelen - low_f) / (high_f - low_f)
        mid = (1 - smooth) * (inv / factor) + smooth * inv
        in_band = (wavelen <= low_wl) & (wavelen >= high_wl)
        inv = torch.where(in_band, mid, scaled)
    t = torch.arange(max_position, dtype=torch.float64)
    freqs = torch.outer(t, inv)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float().contiguous()


class LlamaLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, tp: int, layer_idx: int = 0) -> None:
        super().__init__()
        self.cfg = cfg
        self.layer_idx = layer_idx
        assert cfg.num_q_heads % tp == 0, "q heads must divide TP"
        assert cfg.num_kv_heads % tp == 0 or tp % cfg.num_kv_heads == 0
        self.q_heads = cfg.num_q_heads // tp
        self.kv_heads = max(cfg.num_kv_heads // tp, 1)
        self.head_dim = cfg.head_dim
        self.inter = cfg.intermediate_size // tp
        h = cfg.hidden_size
        qs = self.q_heads * self.head_dim
        kvs = self.kv_heads * self.head_dim
        self.qkv_proj = nn.Parameter(
            torch.empty(qs + 2 * kvs, h, dtype=torch.bfloat16)
        )
        self.o_proj = nn.Parameter(torch.empty(h, qs, dtype=torch.bfloat16))
        self.n_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        if self.n_experts:
            # Mixtral-style sparse MoE: stacked per-expert projections +
            # a top-k softmax router; each expert reuses the dense
            # hipBLASLt GEMM + fused SiLU-mul kernels, so no new device
            # code is involved (EP across GPUs is a round-3 item)
            self.moe_gate = nn.Parameter(
                torch.empty(self.n_experts, h, dtype=torch.bfloat16)
            )
            self.experts_gate_up = nn.Parameter(
                torch.empty(self.n_experts, 2 * self.inter, h,
                            dtype=torch.bfloat16)
            )
            self.experts_down = nn.Parameter(
                torch.empty(self.n_experts, h, self.inter,
                            dtype=torch.bfloat16)
            )
        else:
            self.gate_up_proj = nn.Parameter(
                torch.empty(2 * self.inter, h, dtype=torch.bfloat16)
            )
            self.down_proj = nn.Parameter(
                torch.empty(h, self.inter, dtype=torch.bfloat16)
            )
        self.input_norm = nn.Parameter(torch.empty(h, dtype=torch.bfloat16))
        self.post_attn_norm = nn.Parameter(
            torch.empty(h, dtype=torch.bfloat16)
        )
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.window = cfg.sliding_window or 0
        self.qkv_bias = (
            nn.Parameter(torch.empty(qs + 2 * kvs, dtype=torch.bfloat16))
            if cfg.qkv_bias else None
        )

    def forward(
        self,
        hidden: torch.Tensor,  # [T, H]
        residual: Optional[torch.Tensor],
        meta: BatchMeta,
        kv_cache: tuple,  # (k_cache, v_cache) [NB, KH, BS, HD]
        cos_sin: torch.Tensor,
    ) -> tuple:
        cfg = self.cfg
        fp8 = getattr(self, "fp8_w", None)
        # fused act-quant path: (add-)RMSNorm emits fp8 + row scales
        # directly, skipping the bf16 round trip;
        # LoRA needs the bf16 normed activations, so it keeps the plain
        # norm + per-tensor fp8 GEMM path.
        fuse_q = (fp8 is not None and hidden.is_cuda
                  and meta.lora_idx is None and not meta.lora_groups
                  and ops.fp8_rowwise_supported(hidden.device))
        if fuse_q:
            if residual is None:
                residual = hidden.clone()
                xq, sx = ops.rms_norm_fp8(
                    hidden, self.input_norm, cfg.rms_norm_eps)
            else:
                xq, sx = ops.rms_norm_fp8(
                    hidden, self.input_norm, cfg.rms_norm_eps,
                    residual=residual)
            qkv = ops.fp8_linear_rowwise(xq, sx, *fp8["qkv"])
        else:
            if residual is None:
                residual = hidden
                hidden = ops.rms_norm(hidden, self.input_norm,
                                      cfg.rms_norm_eps)
            else:
                hidden, residual = ops.fused_add_rms_norm(
                    hidden, residual, self.input_norm, cfg.rms_norm_eps
                )
            if fp8 is not None:
                qkv = ops.fp8_linear(hidden, *fp8["qkv"])
            else:
                qkv = gemm_policy.linear(hidden, self.qkv_proj)
        if self.qkv_bias is not None:
            qkv = qkv + self.qkv_bias
        qs = self.q_heads * self.head_dim
        kvs = self.kv_heads * self.head_dim
        slots = getattr(self, "lora_slots", None)
        if slots is not None and meta.lora_idx is not None:
            li = self.layer_idx
            slots.apply("q", li, qkv, hidden, meta.lora_idx)
            slots.apply("k", li, qkv, hidden, meta.lora_idx)
            slots.apply("v", li, qkv, hidden, meta.lora_idx)
        elif meta.lora_groups:
            from production_stack_amd.engine.lora import apply_lora_slice

            li = self.layer_idx
            apply_lora_slice(qkv, hidden, meta.lora_groups, li, "q", 0, qs)
            apply_lora_slice(qkv, hidden, meta.lora_groups, li, "k", qs, kvs)
            apply_lora_slice(
                qkv, hidden, meta.lora_groups, li, "v", qs + kvs, kvs
            )
        T = qkv.shape[0]
        k_cache, v_cache = kv_cache
        if qkv.is_cuda:
            # fused in-place RoPE + KV append straight on the packed qkv;
            # attention consumes the q heads as a strided view (no copies)
            ops.fused_rope_cache(
                qkv, meta.positions, cos_sin, meta.slot_mapping,
                k_cache, v_cache, self.q_heads, self.head_dim,
            )
            qh = qkv.view(T, -1, self.head_dim)[:, : self.q_heads]
        else:
            q = qkv[:, :qs].contiguous()
            k = qkv[:, qs : qs + kvs].contiguous()
            v = qkv[:, qs + kvs :].contiguous()
            q, k = ops.rotary_embedding(
                meta.positions, q, k, cos_sin, self.head_dim
            )
            ops.reshape_and_cache(
                k.view(T, self.kv_heads, self.head_dim),
                v.view(T, self.kv_heads, self.head_dim),
                k_cache,
                v_cache,
                meta.slot_mapping,
            )
            qh = q.view(T, self.q_heads, self.head_dim)
        outs: List[torch.Tensor] = []
        tp = meta.num_prefill_tokens
        if tp > 0:
            if (
                qh.is_cuda
                and self.head_dim == 128
                and meta.prefill_tiles is not None
            ):
                outs.append(
                    ops.paged_attn_prefill_mfma(
                        qh[:tp],
                        k_cache,
                        v_cache,
                        meta.prefill_block_tables,
                        meta.prefill_tiles,
                        self.scale,
                        self.window,
                    )
                )
            else:
                outs.append(
                    ops.paged_attn_prefill(
                        qh[:tp] if qh.is_cuda else qh[:tp].contiguous(),
                        k_cache,
                        v_cache,
                        meta.prefill_block_tables,
                        meta.prefill_token_seq,
                        meta.prefill_token_pos,
                        self.scale,
                        self.window,
                    )
                )
        if meta.num_decode_seqs > 0:
            outs.append(
                ops.paged_attn_decode(
                    qh[tp:] if qh.is_cuda else qh[tp:].contiguous(),
                    k_cache,
                    v_cache,
                    meta.decode_block_tables,
                    meta.decode_seq_lens,
                    self.scale,
                    self.window,
                )
            )
        attn = torch.cat(outs, dim=0) if len(outs) > 1 else outs[0]
        if getattr(self, "fp8_w", None) is not None:
            attn_out = ops.fp8_linear(attn.reshape(T, qs),
                                      *self.fp8_w["o"])
        else:
            attn_out = gemm_policy.linear(attn.view(T, qs), self.o_proj)
        if slots is not None and meta.lora_idx is not None:
            slots.apply("o", self.layer_idx, attn_out,
                        attn.reshape(T, qs), meta.lora_idx)
        elif meta.lora_groups:
            apply_lora_slice(
                attn_out, attn.view(T, qs), meta.lora_groups,
                self.layer_idx, "o",
            )
        attn_out = pstate.tp_all_reduce(attn_out)

        if self.n_experts:
            hidden, residual = ops.fused_add_rms_norm(
                attn_out, residual, self.post_attn_norm, cfg.rms_norm_eps
            )
            mlp_out = self._moe_forward(hidden)
            mlp_out = pstate.tp_all_reduce(mlp_out)
            return mlp_out, residual
        if fuse_q:
            xq, sx = ops.rms_norm_fp8(
                attn_out, self.post_attn_norm, cfg.rms_norm_eps,
                residual=residual)
            hidden = attn_out  # bf16 normed not materialized on this path
            gate_up = ops.fp8_linear_rowwise(xq, sx, *fp8["gate_up"])
        else:
            hidden, residual = ops.fused_add_rms_norm(
                attn_out, residual, self.post_attn_norm, cfg.rms_norm_eps
            )
            if fp8 is not None:
                gate_up = ops.fp8_linear(hidden, *fp8["gate_up"])
            else:
                gate_up = gemm_policy.linear(hidden, self.gate_up_proj)
        if slots is not None and meta.lora_idx is not None:
            slots.apply("gate", self.layer_idx, gate_up, hidden,
                        meta.lora_idx)
            slots.apply("up", self.layer_idx, gate_up, hidden,
                        meta.lora_idx)
        elif meta.lora_groups:
            apply_lora_slice(
                gate_up, hidden, meta.lora_groups, self.layer_idx,
                "gate", 0, self.inter,
            )
            apply_lora_slice(
                gate_up, hidden, meta.lora_groups, self.layer_idx,
                "up", self.inter, self.inter,
            )
        if fuse_q:
            aq, sa = ops.silu_and_mul_fp8(gate_up)
            act = gate_up[:, : self.inter]  # only LoRA reads act; unused
            mlp_out = ops.fp8_linear_rowwise(aq, sa, *fp8["down"])
        else:
            act = ops.silu_and_mul(gate_up)
            if fp8 is not None:
                mlp_out = ops.fp8_linear(act, *fp8["down"])
            else:
                mlp_out = gemm_policy.linear(act, self.down_proj)
        if slots is not None and meta.lora_idx is not None:
            slots.apply("down", self.layer_idx, mlp_out, act, meta.lora_idx)
        elif meta.lora_groups:
            apply_lora_slice(
                mlp_out, act, meta.lora_groups, self.layer_idx, "down"
            )
        mlp_out = pstate.tp_all_reduce(mlp_out)
        return mlp_out, residual

    def _moe_forward(self, x: torch.Tensor) -> torch.Tensor:
        """HF Mixtral routing rule: softmax over all experts, take the
        top-k, renormalize the selected weights; each expert runs the
        dense GEMM + fused SiLU-mul path over its assigned rows."""
        logits = F.linear(x.float(), self.moe_gate.float())
        probs = torch.softmax(logits, dim=-1)
        weights, sel = torch.topk(probs, self.top_k, dim=-1)
        weights = weights / weights.sum(dim=-1, keepdim=True)
        weights = weights.to(x.dtype)
        out = torch.zeros_like(x)
        for e in range(self.n_experts):
            mask = sel == e  # [T, K]
            rows = mask.any(dim=-1).nonzero().flatten()
            if rows.numel() == 0:
                continue
            xe = x.index_select(0, rows)
            act = ops.silu_and_mul(
                gemm_policy.linear(xe, self.experts_gate_up[e])
            )
            ye = gemm_policy.linear(act, self.experts_down[e])
            w = (weights * mask.to(weights.dtype)).sum(-1)
            out.index_add_(
                0, rows, ye * w.index_select(0, rows).unsqueeze(-1)
            )
        return out


class LlamaForCausalLM(nn.Module):
    """Optionally a pipeline stage: with pp_size > 1 this rank holds only
    layers [layer_start, layer_end); embedding lives on the first stage,
    final norm + lm_head on the last."""

    def __init__(
        self, cfg: ModelConfig, tp: int = 1, pp_rank: int = 0,
        pp_size: int = 1,
    ) -> None:
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        per = (cfg.num_layers + pp_size - 1) // pp_size
        self.layer_start = pp_rank * per
        self.layer_end = min(cfg.num_layers, self.layer_start + per)
        self.is_first = pp_rank == 0
        self.is_last = pp_rank == pp_size - 1
        if self.is_first:
            self.embed = nn.Parameter(
                torch.empty(
                    cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16
                )
            )
        self.layers = nn.ModuleList(
            LlamaLayer(cfg, tp, layer_idx=self.layer_start + i)
            for i in range(self.layer_end - self.layer_start)
        )
        if self.is_last:
            self.final_norm = nn.Parameter(
                torch.empty(cfg.hidden_size, dtype=torch.bfloat16)
            )
            if cfg.tie_word_embeddings and self.is_first:
                self.lm_head = self.embed  # tied (Qwen2-small style)
            else:
                self.lm_head = nn.Parameter(
                    torch.empty(
                        cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16
                    )
                )
        self.register_buffer(
            "cos_sin",
            build_cos_sin_cache(cfg.head_dim, cfg.max_position,
                                cfg.rope_theta, cfg.rope_scaling),
            persistent=False,
        )

    @torch.no_grad()
    def random_init(self, seed: int = 0) -> None:
        dev = next(self.parameters()).device
        gen = torch.Generator(device=dev).manual_seed(seed)
        for name, p in self.named_parameters():
            if "norm" in name:
                p.fill_(1.0)
            elif "bias" in name:
                p.normal_(0.0, 0.002, generator=gen)
            else:
                p.normal_(0.0, 0.02, generator=gen)

    @torch.no_grad()
    def quantize_fp8(self) -> None:
        """Convert the four per-layer projections to OCP fp8-e4m3 with
        per-output-channel scales (vLLM --quantization fp8 analogue);
        halves
        weight HBM and runs GEMMs through the fp8 MFMA pipe. Embedding,
        lm_head and norms stay bf16."""
        for layer in self.layers:
            fp8_w = {}
            proj_names = [("qkv", "qkv_proj"), ("o", "o_proj")]
            if not layer.n_experts:  # MoE experts stay bf16 for now
                proj_names += [("gate_up", "gate_up_proj"),
                               ("down", "down_proj")]
            for key, pname in proj_names:
                w = getattr(layer, pname)
                if (w.is_cuda
                        and ops.fp8_rowwise_supported(w.device)):
                    w_q, scale = ops.fp8_quantize_weight_rowwise(w.data)
                else:
                    w_q, scale = ops.fp8_quantize_weight(w.data)
                fp8_w[key] = (w_q, scale.to(w.device))
                # free the bf16 copy (replace with a tiny stub so
                # state_dict/save paths still see the attribute)
                w.data = torch.empty(0, dtype=torch.bfloat16,
                                     device=w.device)
            layer.fp8_w = fp8_w

    @property
    def kv_heads(self) -> int:
        return max(self.cfg.num_kv_heads // self.tp, 1)

    @property
    def num_local_layers(self) -> int:
        return self.layer_end - self.layer_start

    def kv_bytes_per_block(self, block_size: int) -> int:
        return (
            2  # k and v
            * self.kv_heads
            * block_size
            * self.cfg.head_dim
            * 2  # bf16
            * self.num_local_layers
        )

    @torch.no_grad()
    def forward(
        self,
        token_ids: torch.Tensor,  # [T] long (first stage) | hidden [T, H]
        meta: BatchMeta,
        kv_caches: List[tuple],
        residual: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if self.is_first:
            hidden = F.embedding(token_ids, self.embed)
            if meta.mm_rows is not None:
                # multimodal injection: encoder embeddings overwrite the
                # placeholder positions (vision/audio adapter pattern)
                hidden = hidden.index_copy(
                    0, meta.mm_rows, meta.mm_embeds.to(hidden.dtype)
                )
            residual = None
        else:
            hidden = token_ids  # mid-pipeline: activations from prev stage
        for layer, cache in zip(self.layers, kv_caches):
            hidden, residual = layer(
                hidden, residual, meta, cache, self.cos_sin
            )
        if not self.is_last:
            # caller ships (hidden, residual) to the next stage
            return torch.stack([hidden, residual])
        # final residual add + norm
        hidden, _ = ops.fused_add_rms_norm(
            hidden, residual, self.final_norm, self.cfg.rms_norm_eps
        )
        return hidden

    @torch.no_grad()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return F.linear(hidden, self.lm_head)
