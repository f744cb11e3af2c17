"""Paged KV block allocator with hash-based prefix caching.

Equivalent capability to the prefix-cache + paged-KV machinery the reference
stack assumes in its engine (it scrapes vllm:gpu_prefix_cache_{hits,queries}
and vllm:gpu_cache_usage_perc — reference stats/engine_stats.py:63-76), built
for a 288 GB HBM3E budget: block reuse is chain-hashed per full block, freed
hashed blocks stay resident in an LRU evictable pool until memory pressure
actually reclaims them.
"""

from __future__ import annotations

from collections import OrderedDict, deque
from typing import Dict, List, Optional

from production_stack_amd.engine.sequence import Sequence


class BlockManager:
    def __init__(
        self,
        num_blocks: int,
        block_size: int,
        enable_prefix_caching: bool = True,
    ) -> None:
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.enable_prefix_caching = enable_prefix_caching
        self.free: deque[int] = deque(range(num_blocks))
        self.ref_count: List[int] = [0] * num_blocks
        self.block_hash: List[Optional[int]] = [None] * num_blocks
        # hash -> block id (block may be in use or evictable)
        self.cached: Dict[int, int] = {}
        # blocks with ref 0 whose contents remain valid, LRU order
        self.evictable: "OrderedDict[int, None]" = OrderedDict()
        # per-seq chain hash of the last registered full block
        self._tail_hash: Dict[str, Optional[int]] = {}
        self.prefix_queries = 0
        self.prefix_hits = 0
        # optional host-DRAM offload tier (kvpool.offload.HostKVPool)
        self.offload_pool = None

    # ------------------------------------------------------------------
    @property
    def num_free(self) -> int:
        return len(self.free) + len(self.evictable)

    @property
    def usage(self) -> float:
        """Fraction of blocks referenced by live sequences."""
        return 1.0 - self.num_free / self.num_blocks

    def _pop_block(self) -> Optional[int]:
        if self.free:
            return self.free.popleft()
        if self.evictable:
            blk, _ = self.evictable.popitem(last=False)
            h = self.block_hash[blk]
            if h is not None and self.cached.get(h) == blk:
                del self.cached[h]
            self.block_hash[blk] = None
            return blk
        return None

    @staticmethod
    def chain_hash(prev: Optional[int], tokens: tuple) -> int:
        # prev=None normalizes to 0: hash(None) is address-derived on
        # CPython < 3.12, so a None seed would make the chain differ
        # ACROSS PROCESSES — breaking kvaware lookup (engine registers,
        # controller matches) and cross-instance prefix exchange.
        # int/tuple-of-int hashing is seed-stable.
        return hash((prev if prev is not None else 0, tokens))

    # ------------------------------------------------------------------
    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate_prompt(self, seq: Sequence) -> bool:
        # worst case (no cache hits)
        return self.blocks_needed(seq.num_tokens) <= self.num_free

    def allocate_prompt(self, seq: Sequence) -> None:
        """Build the sequence's block table, reusing cached prefix blocks.

        Sets seq.num_computed / seq.num_cached_prompt_tokens to the reused
        span (capped so at least one prompt token is still computed).
        """
        if seq.block_table:
            raise AssertionError(
                f"allocate_prompt on seq {seq.request_id} with a live block "
                "table (would leak blocks)"
            )
        bs = self.block_size
        all_tokens = seq.token_ids()  # prompt + generated (recompute case)
        n_prompt = len(all_tokens)
        n_blocks = self.blocks_needed(n_prompt)
        table: List[int] = []
        matched = 0
        prev_hash: Optional[int] = None
        # prompt_logprobs needs logits at every prompt position — a
        # prefix-cache hit would skip those rows (vLLM also recomputes)
        want_prompt_lp = (
            getattr(seq.params, "prompt_logprobs", None) is not None
        )
        if self.enable_prefix_caching and not want_prompt_lp:
            n_full = n_prompt // bs
            # never reuse ALL tokens: the last one must be computed so there
            # are logits to sample from
            if n_full * bs == n_prompt:
                n_full -= 1
            for i in range(n_full):
                tokens = tuple(all_tokens[i * bs : (i + 1) * bs])
                h = self.chain_hash(prev_hash, tokens)
                self.prefix_queries += 1
                blk = self.cached.get(h)
                if blk is None:
                    # HBM miss: try the host-DRAM offload tier
                    pool = self.offload_pool
                    if pool is not None and pool.has(h):
                        nb = self._pop_block()
                        if nb is None:
                            break
                        if pool.restore(h, nb):
                            self.ref_count[nb] += 1
                            self.cached[h] = nb
                            self.block_hash[nb] = h
                            table.append(nb)
                            prev_hash = h
                            matched += 1
                            self.prefix_hits += 1
                            continue
                        # restore raced an eviction: hand the block back
                        self.free.appendleft(nb)
                    break
                self.prefix_hits += 1
                if self.ref_count[blk] == 0:
                    self.evictable.pop(blk, None)
                self.ref_count[blk] += 1
                table.append(blk)
                prev_hash = h
                matched += 1
        for _ in range(n_blocks - matched):
            blk = self._pop_block()
            if blk is None:
                # roll back
                for b in table:
                    self._release_block(b)
                raise RuntimeError("out of KV blocks during prompt allocation")
            self.ref_count[blk] += 1
            table.append(blk)
        seq.block_table = table
        seq.num_cached_prompt_tokens = matched * bs
        seq.num_computed = matched * bs
        self._tail_hash[seq.request_id] = prev_hash

    def ensure_capacity(self, seq: Sequence, num_tokens: int) -> bool:
        """Grow the block table to cover num_tokens. False = out of memory."""
        need = self.blocks_needed(num_tokens)
        while len(seq.block_table) < need:
            blk = self._pop_block()
            if blk is None:
                return False
            self.ref_count[blk] += 1
            seq.block_table.append(blk)
        return True

    def register_computed_blocks(self, seq: Sequence) -> None:
        """Hash-register blocks that became full (called after each step)."""
        if not self.enable_prefix_caching:
            return
        bs = self.block_size
        n_full = seq.num_computed // bs
        # count already registered full blocks for this seq
        done = getattr(seq, "_registered_full", None)
        if done is None:
            done = seq.num_cached_prompt_tokens // bs
        if n_full <= done:
            # no new full block (15 of every 16 decode steps): skip the
            # O(len) token_ids() materialisation entirely
            seq._registered_full = done  # type: ignore[attr-defined]
            return
        tokens = seq.token_ids()
        prev = self._tail_hash.get(seq.request_id)
        for i in range(done, n_full):
            blk = seq.block_table[i]
            chunk = tokens[i * bs : (i + 1) * bs]
            if any(t < 0 for t in chunk):
                # async placeholder not yet resolved: register next step
                n_full = i
                break
            h = self.chain_hash(prev, tuple(chunk))
            if h not in self.cached and self.block_hash[blk] is None:
                self.cached[h] = blk
                self.block_hash[blk] = h
                if self.offload_pool is not None:
                    self.offload_pool.offload(h, blk)
            prev = h
        seq._registered_full = n_full  # type: ignore[attr-defined]
        self._tail_hash[seq.request_id] = prev

    def _release_block(self, blk: int) -> None:
        self.ref_count[blk] -= 1
        if self.ref_count[blk] > 0:
            return
        h = self.block_hash[blk]
        if h is not None and self.cached.get(h) == blk:
            self.evictable[blk] = None  # keep contents for prefix reuse
        else:
            self.block_hash[blk] = None
            self.free.append(blk)

    def free_seq(self, seq: Sequence) -> None:
        for blk in seq.block_table:
            self._release_block(blk)
        seq.block_table = []
        self._tail_hash.pop(seq.request_id, None)
        if hasattr(seq, "_registered_full"):
            del seq._registered_full

    def slot(self, seq: Sequence, position: int) -> int:
        bs = self.block_size
        return seq.block_table[position // bs] * bs + position % bs

    def reset_prefix_cache(self) -> None:
        for blk in list(self.evictable):
            h = self.block_hash[blk]
            if h is not None:
                self.cached.pop(h, None)
            self.block_hash[blk] = None
            self.free.append(blk)
        self.evictable.clear()
