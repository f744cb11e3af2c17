"""Host-DRAM KV offload pool (LMCache-capability equivalent).

Full KV blocks that get hash-registered in the GPU prefix cache are copied
to a pinned host pool on a dedicated side HIP stream (D2H overlapped with
compute). When a prompt's prefix misses in HBM but hits the host pool, the
block is restored H2D instead of recomputed. Config surface mirrors the
reference's LMCACHE_LOCAL_CPU / LMCACHE_MAX_LOCAL_CPU_SIZE intent
(reference deployment-vllm-multi.yaml:336-343).

MI355X notes: one host record = all layers' K+V for one 16-token block
(layers x 2 x KH x 16 x HD bf16, 2 MiB for Llama-3-8B), transferred as
2 x layers contiguous 64 KiB copies on the side stream; compute-stream
ordering is enforced with events, never a device-wide sync.
"""

from __future__ import annotations

import logging
from collections import OrderedDict, deque
from typing import Deque, Dict, List, Optional, Tuple

import torch

logger = logging.getLogger("kvpool.offload")


class HostKVPool:
    def __init__(
        self,
        kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
        block_size: int,
        capacity_gb: float,
        device: torch.device,
    ) -> None:
        self.kv_caches = kv_caches
        self.device = device
        self.layers = len(kv_caches)
        k0 = kv_caches[0][0]
        self.kh, self.bs, self.hd = k0.shape[1], k0.shape[2], k0.shape[3]
        assert self.bs == block_size
        self.block_elems = self.kh * self.bs * self.hd
        block_bytes = self.layers * 2 * self.block_elems * 2
        self.capacity = max(int(capacity_gb * (1 << 30)) // block_bytes, 1)
        pin = device.type == "cuda"
        self.store = torch.empty(
            (self.capacity, self.layers, 2, self.block_elems),
            dtype=torch.bfloat16,
            pin_memory=pin,
        )
        self.free_slots: Deque[int] = deque(range(self.capacity))
        self.slot_of: "OrderedDict[int, int]" = OrderedDict()  # hash -> slot
        self.stream = (
            torch.cuda.Stream(device) if device.type == "cuda" else None
        )
        self._restore_events: List[torch.cuda.Event] = []
        self.offloaded = 0
        self.restored = 0
        self.evicted = 0

    # ------------------------------------------------------------------
    def has(self, h: int) -> bool:
        return h in self.slot_of

    def _take_slot(self) -> int:
        if self.free_slots:
            return self.free_slots.popleft()
        _, slot = self.slot_of.popitem(last=False)  # LRU
        self.evicted += 1
        return slot

    def offload(self, h: int, block_id: int) -> None:
        """Async D2H of a (now immutable) full block."""
        if h in self.slot_of:
            self.slot_of.move_to_end(h)
            return
        slot = self._take_slot()
        dst = self.store[slot]
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                for li, (kc, vc) in enumerate(self.kv_caches):
                    dst[li, 0].copy_(
                        kc[block_id].flatten(), non_blocking=True
                    )
                    dst[li, 1].copy_(
                        vc[block_id].flatten(), non_blocking=True
                    )
        else:
            for li, (kc, vc) in enumerate(self.kv_caches):
                dst[li, 0].copy_(kc[block_id].flatten())
                dst[li, 1].copy_(vc[block_id].flatten())
        self.slot_of[h] = slot
        self.offloaded += 1

    def restore(self, h: int, block_id: int) -> bool:
        """Async H2D into a freshly allocated GPU block."""
        slot = self.slot_of.get(h)
        if slot is None:
            return False
        self.slot_of.move_to_end(h)
        src = self.store[slot]
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                for li, (kc, vc) in enumerate(self.kv_caches):
                    kc[block_id].flatten().copy_(
                        src[li, 0], non_blocking=True
                    )
                    vc[block_id].flatten().copy_(
                        src[li, 1], non_blocking=True
                    )
                ev = torch.cuda.Event()
                ev.record(self.stream)
                self._restore_events.append(ev)
        else:
            for li, (kc, vc) in enumerate(self.kv_caches):
                kc[block_id].flatten().copy_(src[li, 0])
                vc[block_id].flatten().copy_(src[li, 1])
        self.restored += 1
        return True

    def make_compute_wait(self) -> None:
        """Compute stream must not read restored blocks before H2D lands."""
        if self.stream is None:
            return
        cur = torch.cuda.current_stream()
        for ev in self._restore_events:
            cur.wait_event(ev)
        self._restore_events.clear()

    def metrics(self) -> Dict[str, float]:
        return {
            "cpu_offload_blocks": float(len(self.slot_of)),
            "cpu_offload_capacity_blocks": float(self.capacity),
            "cpu_offloaded_total": float(self.offloaded),
            "cpu_restored_total": float(self.restored),
            "cpu_evicted_total": float(self.evicted),
        }
