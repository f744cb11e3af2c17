"""LoRA adapters: loading (HF PEFT safetensors format) and batched
application.

Capability parity: the reference drives engine LoRA through
/v1/load_lora_adapter + the LoraAdapter CRD (reference
loraadapter_controller.go:553-592); here the adapters are actually executed:
requests whose model name is a loaded adapter run with low-rank deltas
applied around the base GEMMs. Multi-adapter batches are supported by
grouping batch rows per adapter (dense per-group GEMMs; ranks are small so
this stays cheap relative to the base projections).
"""

from __future__ import annotations

import json
import os
import re
from typing import Dict, List, Optional, Tuple

import torch

# projections we support, keyed by HF PEFT module name
TARGETS = {
    "q_proj": "q",
    "k_proj": "k",
    "v_proj": "v",
    "o_proj": "o",
    "gate_proj": "gate",
    "up_proj": "up",
    "down_proj": "down",
}

_LAYER_RE = re.compile(r"layers\.(\d+)\.(?:self_attn|mlp)\.(\w+)\.lora_(A|B)")


class LoRAAdapter:
    def __init__(self, name: str, rank: int, scaling: float) -> None:
        self.name = name
        self.rank = rank
        self.scaling = scaling
        # layer -> proj key -> (A [r, in], B [out, r])
        self.layers: Dict[int, Dict[str, Tuple[torch.Tensor, torch.Tensor]]] = {}

    @staticmethod
    def load(name: str, path: str, device, dtype=torch.bfloat16
             ) -> "LoRAAdapter":
        from safetensors.torch import load_file

        cfg_path = os.path.join(path, "adapter_config.json")
        alpha, r = 16.0, 8
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                cfg = json.load(f)
            alpha = float(cfg.get("lora_alpha", 16))
            r = int(cfg.get("r", 8))
        weights_file = None
        for cand in ("adapter_model.safetensors", "model.safetensors"):
            p = os.path.join(path, cand)
            if os.path.exists(p):
                weights_file = p
                break
        if weights_file is None:
            raise FileNotFoundError(f"no adapter safetensors under {path}")
        state = load_file(weights_file)
        ad = LoRAAdapter(name, r, alpha / r)
        halves: Dict[Tuple[int, str], Dict[str, torch.Tensor]] = {}
        for key, tensor in state.items():
            m = _LAYER_RE.search(key)
            if not m:
                continue
            layer, proj, ab = int(m.group(1)), m.group(2), m.group(3)
            if proj not in TARGETS:
                continue
            halves.setdefault((layer, TARGETS[proj]), {})[ab] = tensor
        for (layer, proj), d in halves.items():
            if "A" in d and "B" in d:
                ad.layers.setdefault(layer, {})[proj] = (
                    d["A"].to(device=device, dtype=dtype),
                    d["B"].to(device=device, dtype=dtype),
                )
        if not ad.layers:
            raise ValueError(f"adapter at {path} has no supported weights")
        return ad

    def delta(
        self, x: torch.Tensor, layer: int, proj: str
    ) -> Optional[torch.Tensor]:
        w = self.layers.get(layer, {}).get(proj)
        if w is None:
            return None
        A, B = w
        return ((x @ A.t()) @ B.t()) * self.scaling


def apply_lora_slice(
    out: torch.Tensor,
    x: torch.Tensor,
    groups: List[Tuple[LoRAAdapter, torch.Tensor]],
    layer: int,
    proj: str,
    col_offset: int = 0,
    width: Optional[int] = None,
) -> None:
    """out[rows, col_offset:+width] += adapter delta for each group."""
    for ad, rows in groups:
        xr = x.index_select(0, rows)
        d = ad.delta(xr, layer, proj)
        if d is None:
            continue
        w = d.shape[1] if width is None else width
        sub = out.index_select(0, rows)
        sub[:, col_offset : col_offset + w] += d.to(sub.dtype)
        out.index_copy_(0, rows, sub)


def save_synthetic_adapter(
    path: str,
    hidden: int,
    q_size: int,
    kv_size: int,
    num_layers: int,
    rank: int = 8,
    seed: int = 0,
    targets=("q_proj", "v_proj"),
) -> None:
    """Write a random PEFT-format adapter (tests / offline demos)."""
    from safetensors.torch import save_file

    gen = torch.Generator().manual_seed(seed)
    state = {}
    out_dims = {"q_proj": q_size, "k_proj": kv_size, "v_proj": kv_size,
                "o_proj": hidden}
    for layer in range(num_layers):
        for t in targets:
            pre = (
                f"base_model.model.model.layers.{layer}.self_attn.{t}"
            )
            state[pre + ".lora_A.weight"] = torch.randn(
                rank, hidden, generator=gen
            ) * 0.05
            state[pre + ".lora_B.weight"] = torch.randn(
                out_dims[t], rank, generator=gen
            ) * 0.05
    os.makedirs(path, exist_ok=True)
    save_file(state, os.path.join(path, "adapter_model.safetensors"))
    with open(os.path.join(path, "adapter_config.json"), "w") as f:
        json.dump({"r": rank, "lora_alpha": 16,
                   "target_modules": list(targets)}, f)


class LoRASlotManager:
    """Stacked per-slot adapter weights for graph-safe batched LoRA.

    One pair of stacks per logical projection:
      A[proj]: [L, S, R, IN]  bf16   B[proj]: [L, S, W, R]  bf16
    plus a shared per-slot scale vector. The BGMV kernel (csrc/
    lora_bgmv.hip) indexes slots per token, so the decode hipGraph can stay
    captured while adapters load/unload (tensor storage is fixed; only
    contents and the per-step idx buffer change) — the same design point as
    vLLM's punica stacks, built MI355X-native.
    """

    def __init__(self, model_cfg, max_loras: int, max_rank: int,
                 device, dtype=torch.bfloat16) -> None:
        h = model_cfg.hidden_size
        q = model_cfg.num_q_heads * model_cfg.head_dim
        kv = model_cfg.num_kv_heads * model_cfg.head_dim
        inter = model_cfg.intermediate_size
        L = model_cfg.num_layers
        self.S = max_loras
        self.R = max_rank
        self.device = device
        # proj -> (IN, W, target tensor name, column offset in target)
        self.spec = {
            "q": (h, q, "qkv", 0),
            "k": (h, kv, "qkv", q),
            "v": (h, kv, "qkv", q + kv),
            "o": (q, h, "o", 0),
            "gate": (h, inter, "gate_up", 0),
            "up": (h, inter, "gate_up", inter),
            "down": (inter, h, "down", 0),
        }
        self.A: Dict[str, torch.Tensor] = {}
        self.B: Dict[str, torch.Tensor] = {}
        for p, (IN, W, _, _) in self.spec.items():
            self.A[p] = torch.zeros(
                (L, max_loras, max_rank, IN), dtype=dtype, device=device
            )
            self.B[p] = torch.zeros(
                (L, max_loras, W, max_rank), dtype=dtype, device=device
            )
        self.scale = torch.zeros(max_loras, dtype=torch.float32,
                                 device=device)
        self.slot_by_name: Dict[str, int] = {}
        self._free = list(range(max_loras))

    def register(self, adapter: "LoRAAdapter") -> int:
        if adapter.name in self.slot_by_name:
            return self.slot_by_name[adapter.name]
        if not self._free:
            raise RuntimeError(
                f"all {self.S} LoRA slots in use (max_loras)"
            )
        if adapter.rank > self.R:
            raise ValueError(
                f"adapter rank {adapter.rank} > max_lora_rank {self.R}"
            )
        slot = self._free.pop(0)
        for li, projs in adapter.layers.items():
            for p, (A, B) in projs.items():
                r = A.shape[0]
                self.A[p][li, slot, :r].copy_(A)
                self.B[p][li, slot, :, :r].copy_(B)
        self.scale[slot] = adapter.scaling
        self.slot_by_name[adapter.name] = slot
        return slot

    def unregister(self, name: str) -> None:
        slot = self.slot_by_name.pop(name, None)
        if slot is None:
            return
        for p in self.spec:
            self.A[p][:, slot].zero_()
            self.B[p][:, slot].zero_()
        self.scale[slot] = 0.0
        self._free.append(slot)

    def apply(self, proj: str, layer: int, out: torch.Tensor,
              x: torch.Tensor, idx: torch.Tensor) -> None:
        """out[:, off:+W] += per-token adapter delta (BGMV kernel)."""
        from production_stack_amd import ops

        _, _, _, off = self.spec[proj]
        ops.lora_bgmv(out, x, self.A[proj][layer], self.B[proj][layer],
                      self.scale, idx, off)
