"""Load HF-format Llama safetensors into the fused-weight model.

Handles the name mapping from HF checkpoints
(model.layers.N.self_attn.{q,k,v}_proj.weight etc.) onto the pre-fused
qkv_proj / gate_up_proj parameters, with TP sharding.
"""

from __future__ import annotations

import glob
import os

import torch

from production_stack_amd.parallel import state as pstate


def load_safetensors(model, path: str) -> None:
    from safetensors.torch import load_file

    files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no safetensors under {path}")
    state = {}
    for f in files:
        state.update(load_file(f))

    tp = pstate.tp_size()
    rank = pstate.tp_rank()

    def shard(t: torch.Tensor, dim: int) -> torch.Tensor:
        if tp == 1:
            return t
        n = t.shape[dim] // tp
        return t.narrow(dim, rank * n, n)

    cfg = model.cfg

    def get(name):
        return state[name].to(torch.bfloat16)

    layer_start = getattr(model, "layer_start", 0)
    with torch.no_grad():
        if hasattr(model, "embed"):
            model.embed.copy_(get("model.embed_tokens.weight"))
        if hasattr(model, "final_norm"):
            model.final_norm.copy_(get("model.norm.weight"))
        if hasattr(model, "lm_head"):
            if "lm_head.weight" in state:
                model.lm_head.copy_(get("lm_head.weight"))
            else:  # tied embeddings
                model.lm_head.copy_(get("model.embed_tokens.weight"))
        for i, layer in enumerate(model.layers):
            pre = f"model.layers.{layer_start + i}."
            q = shard(get(pre + "self_attn.q_proj.weight"), 0)
            k = shard(get(pre + "self_attn.k_proj.weight"), 0)
            v = shard(get(pre + "self_attn.v_proj.weight"), 0)
            layer.qkv_proj.copy_(torch.cat([q, k, v], dim=0))
            if layer.qkv_bias is not None:
                qb = shard(get(pre + "self_attn.q_proj.bias"), 0)
                kb = shard(get(pre + "self_attn.k_proj.bias"), 0)
                vb = shard(get(pre + "self_attn.v_proj.bias"), 0)
                layer.qkv_bias.copy_(torch.cat([qb, kb, vb], dim=0))
            layer.o_proj.copy_(shard(get(pre + "self_attn.o_proj.weight"), 1))
            if layer.n_experts:
                # HF Mixtral layout: block_sparse_moe.gate +
                # experts.E.{w1=gate, w3=up, w2=down}
                layer.moe_gate.copy_(
                    get(pre + "block_sparse_moe.gate.weight"))
                for e in range(layer.n_experts):
                    ep = pre + f"block_sparse_moe.experts.{e}."
                    g = shard(get(ep + "w1.weight"), 0)
                    u = shard(get(ep + "w3.weight"), 0)
                    layer.experts_gate_up[e].copy_(
                        torch.cat([g, u], dim=0))
                    layer.experts_down[e].copy_(
                        shard(get(ep + "w2.weight"), 1))
            else:
                g = shard(get(pre + "mlp.gate_proj.weight"), 0)
                u = shard(get(pre + "mlp.up_proj.weight"), 0)
                layer.gate_up_proj.copy_(torch.cat([g, u], dim=0))
                layer.down_proj.copy_(
                    shard(get(pre + "mlp.down_proj.weight"), 1))
            layer.input_norm.copy_(get(pre + "input_layernorm.weight"))
            layer.post_attn_norm.copy_(
                get(pre + "post_attention_layernorm.weight")
            )


def save_hf_safetensors(model, path: str) -> None:
    """Export the fused-weight model to HF-format safetensors (inverse of
    load_safetensors; used by tests and for interchange)."""
    from safetensors.torch import save_file

    os.makedirs(path, exist_ok=True)
    cfg = model.cfg
    state = {
        "model.embed_tokens.weight": model.embed.detach().cpu(),
        "model.norm.weight": model.final_norm.detach().cpu(),
    }
    if model.lm_head is not model.embed:  # tied weights saved once
        state["lm_head.weight"] = model.lm_head.detach().cpu()
    layer_start = getattr(model, "layer_start", 0)
    for i, layer in enumerate(model.layers):
        pre = f"model.layers.{layer_start + i}."
        qs = layer.q_heads * layer.head_dim
        kvs = layer.kv_heads * layer.head_dim
        qkv = layer.qkv_proj.detach().cpu()
        state[pre + "self_attn.q_proj.weight"] = qkv[:qs].clone()
        state[pre + "self_attn.k_proj.weight"] = qkv[qs : qs + kvs].clone()
        state[pre + "self_attn.v_proj.weight"] = qkv[qs + kvs :].clone()
        if layer.qkv_bias is not None:
            qb = layer.qkv_bias.detach().cpu()
            state[pre + "self_attn.q_proj.bias"] = qb[:qs].clone()
            state[pre + "self_attn.k_proj.bias"] = qb[qs : qs + kvs].clone()
            state[pre + "self_attn.v_proj.bias"] = (
                qb[qs + kvs :].clone()
            )
        state[pre + "self_attn.o_proj.weight"] = layer.o_proj.detach().cpu()
        if layer.n_experts:
            state[pre + "block_sparse_moe.gate.weight"] = (
                layer.moe_gate.detach().cpu())
            for e in range(layer.n_experts):
                ep = pre + f"block_sparse_moe.experts.{e}."
                gu = layer.experts_gate_up[e].detach().cpu()
                state[ep + "w1.weight"] = gu[: layer.inter].clone()
                state[ep + "w3.weight"] = gu[layer.inter :].clone()
                state[ep + "w2.weight"] = (
                    layer.experts_down[e].detach().cpu().clone())
        else:
            gu = layer.gate_up_proj.detach().cpu()
            state[pre + "mlp.gate_proj.weight"] = gu[: layer.inter].clone()
            state[pre + "mlp.up_proj.weight"] = gu[layer.inter :].clone()
            state[pre + "mlp.down_proj.weight"] = (
                layer.down_proj.detach().cpu())
        state[pre + "input_layernorm.weight"] = (
            layer.input_norm.detach().cpu()
        )
        state[pre + "post_attention_layernorm.weight"] = (
            layer.post_attn_norm.detach().cpu()
        )
    save_file(state, os.path.join(path, "model.safetensors"))
