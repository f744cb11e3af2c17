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
import queue
import threading
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
        offload_dtype: str = "bf16",
        remote_url: Optional[str] = None,
        remote_serde: str = "cachegen",  # raw | cachegen | cachegen4
    ) -> None:
        self.remote_serde = remote_serde
        self.kv_caches = kv_caches
        self.device = device
        self.layers = len(kv_caches)
        k0 = kv_caches[0][0]
        self.kh, self.bs, self.hd = k0.shape[1], k0.shape[2], k0.shape[3]
        assert self.bs == block_size
        self.block_elems = self.kh * self.bs * self.hd
        self.quantized = offload_dtype == "int8"
        self.rows = self.layers * 2 * self.kh * self.bs  # scale rows/block
        elem_bytes = 1 if self.quantized else 2
        block_bytes = self.layers * 2 * self.block_elems * elem_bytes
        if self.quantized:
            block_bytes += self.rows * 4
        self.capacity = max(int(capacity_gb * (1 << 30)) // block_bytes, 1)
        pin = device.type == "cuda"
        self.store = torch.empty(
            (self.capacity, self.layers, 2, self.block_elems),
            dtype=torch.int8 if self.quantized else torch.bfloat16,
            pin_memory=pin,
        )
        self.scale_store = (
            torch.empty(
                (self.capacity, self.rows), dtype=torch.float32,
                pin_memory=pin,
            )
            if self.quantized
            else None
        )
        # GPU staging for the quantized path
        if self.quantized:
            self._stage = torch.empty(
                (self.layers, 2, self.block_elems),
                dtype=torch.bfloat16,
                device=device,
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
        # ---- optional remote tier (kvpool.cacheserver data plane) ----
        self.remote = None
        self._push_q: Optional[queue.Queue] = None
        self._push_thread: Optional[threading.Thread] = None
        self.remote_pushed = 0
        self.remote_restored = 0
        if remote_url:
            from production_stack_amd.kvpool.cacheserver import RemoteKVClient

            self.remote = RemoteKVClient(remote_url)
            self._push_q = queue.Queue(maxsize=1024)
            self._push_thread = threading.Thread(
                target=self._push_loop, daemon=True
            )
            self._push_thread.start()

    # ------------------------------------------------------------------
    def has(self, h: int) -> bool:
        if h in self.slot_of:
            return True
        return self.remote is not None and self.remote.exists(h)

    def _take_slot(self) -> int:
        if self.free_slots:
            return self.free_slots.popleft()
        _, slot = self.slot_of.popitem(last=False)  # LRU
        self.evicted += 1
        return slot

    def _gather_quant(self, block_id: int):
        """D2D gather + row-wise int8 quantization on the side stream."""
        from production_stack_amd import ops

        for li, (kc, vc) in enumerate(self.kv_caches):
            self._stage[li, 0].copy_(kc[block_id].flatten(),
                                     non_blocking=True)
            self._stage[li, 1].copy_(vc[block_id].flatten(),
                                     non_blocking=True)
        return ops.kv_quant(self._stage.view(-1, self.hd))

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
                if self.quantized:
                    q, scales = self._gather_quant(block_id)
                    dst.view(-1, self.hd).copy_(q, non_blocking=True)
                    self.scale_store[slot].copy_(scales, non_blocking=True)
                else:
                    for li, (kc, vc) in enumerate(self.kv_caches):
                        dst[li, 0].copy_(
                            kc[block_id].flatten(), non_blocking=True
                        )
                        dst[li, 1].copy_(
                            vc[block_id].flatten(), non_blocking=True
                        )
        else:
            if self.quantized:
                from production_stack_amd.ops import reference

                rows = torch.cat(
                    [
                        torch.stack(
                            [kc[block_id].flatten(), vc[block_id].flatten()]
                        )
                        for kc, vc in self.kv_caches
                    ]
                ).view(-1, self.hd)
                q, scales = reference.kv_quant(rows)
                dst.view(-1, self.hd).copy_(q)
                self.scale_store[slot].copy_(scales)
            else:
                for li, (kc, vc) in enumerate(self.kv_caches):
                    dst[li, 0].copy_(kc[block_id].flatten())
                    dst[li, 1].copy_(vc[block_id].flatten())
        self.slot_of[h] = slot
        self.offloaded += 1
        if self.remote is not None:
            ev = None
            if self.stream is not None:
                ev = torch.cuda.Event()
                ev.record(self.stream)
            try:
                self._push_q.put_nowait((h, slot, ev))
            except queue.Full:
                pass  # lossy best-effort: the local tier still has it

    def restore(self, h: int, block_id: int) -> bool:
        """Async H2D into a freshly allocated GPU block."""
        slot = self.slot_of.get(h)
        if slot is None and self.remote is not None:
            slot = self._fetch_remote(h)
        if slot is None:
            return False
        self.slot_of.move_to_end(h)
        src = self.store[slot]
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                if self.quantized:
                    from production_stack_amd import ops

                    qdev = src.to(self.device, non_blocking=True)
                    sdev = self.scale_store[slot].to(
                        self.device, non_blocking=True
                    )
                    deq = ops.kv_dequant(
                        qdev.view(-1, self.hd), sdev
                    ).view(self.layers, 2, self.block_elems)
                    for li, (kc, vc) in enumerate(self.kv_caches):
                        kc[block_id].flatten().copy_(
                            deq[li, 0], non_blocking=True
                        )
                        vc[block_id].flatten().copy_(
                            deq[li, 1], non_blocking=True
                        )
                else:
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
            if self.quantized:
                from production_stack_amd.ops import reference

                deq = reference.kv_dequant(
                    src.view(-1, self.hd), self.scale_store[slot]
                ).view(self.layers, 2, self.block_elems)
                for li, (kc, vc) in enumerate(self.kv_caches):
                    kc[block_id].flatten().copy_(deq[li, 0])
                    vc[block_id].flatten().copy_(deq[li, 1])
            else:
                for li, (kc, vc) in enumerate(self.kv_caches):
                    kc[block_id].flatten().copy_(src[li, 0])
                    vc[block_id].flatten().copy_(src[li, 1])
        self.restored += 1
        return True

    # ---- remote tier ---------------------------------------------------
    def _record_bytes(self, slot: int):
        scales = None
        if self.quantized:
            scales = (
                self.scale_store[slot].contiguous().view(torch.uint8)
                .numpy().tobytes()
            )
            if self.remote_serde in ("cachegen", "cachegen4"):
                # CacheGen-style serde: entropy-encode the int8 record
                # (optionally re-binned to 4-bit levels) before it goes
                # over the wire; _fetch_remote auto-detects the PSKV blob
                from production_stack_amd import ops

                q = self.store[slot].contiguous().view(torch.int8)
                if self.remote_serde == "cachegen4":
                    q = (
                        (q.float() / 16.0).round().clamp(-8, 7) * 16
                    ).to(torch.int8)
                blob = ops.cachegen_encode(q.view(-1, self.hd))
                return blob.numpy().tobytes(), scales
        data = self.store[slot].contiguous().view(torch.uint8)
        return data.numpy().tobytes(), scales

    def _push_loop(self) -> None:
        while True:
            item = self._push_q.get()
            if item is None:
                return
            h, slot, ev = item
            try:
                if ev is not None:
                    ev.synchronize()  # D2H of this record has landed
                if self.slot_of.get(h) != slot:
                    continue  # slot was LRU-recycled before we shipped it
                data, scales = self._record_bytes(slot)
                if self.remote.put(h, data, scales):
                    self.remote_pushed += 1
            except Exception:  # network best-effort; never kill the engine
                logger.exception("remote KV push failed")

    def _fetch_remote(self, h: int) -> Optional[int]:
        """Blocking fetch into a local slot; returns the slot or None."""
        rec = self.remote.get(h)
        if rec is None:
            return None
        data, scales = rec
        slot = self._take_slot()
        flat = torch.frombuffer(bytearray(data), dtype=torch.uint8)
        dst = self.store[slot].view(torch.uint8)
        if flat.numel() >= 4 and bytes(flat[:4].tolist()) == b"VKSP":
            from production_stack_amd import ops

            dec = ops.cachegen_decode(flat, self.hd)
            dst.copy_(dec.view(torch.uint8).view(dst.shape))
        else:
            dst.copy_(flat.view(dst.shape))
        if self.quantized and scales is not None:
            sflat = torch.frombuffer(bytearray(scales), dtype=torch.uint8)
            self.scale_store[slot].view(torch.uint8).copy_(sflat)
        self.slot_of[h] = slot
        self.remote_restored += 1
        return slot

    def stop(self) -> None:
        if self._push_q is not None:
            self._push_q.put(None)
            self._push_thread.join(timeout=5)
        if self.remote is not None:
            self.remote.close()

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
            "remote_pushed_total": float(self.remote_pushed),
            "remote_restored_total": float(self.remote_restored),
        }
