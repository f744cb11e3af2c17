"""Chunked prefix hash trie for prefix-aware routing.

Behavioural parity: reference src/vllm_router/prefix/hashtrie.py (128-char
chunking, xxhash64 node keys, longest-prefix match returning the endpoint
set of the deepest matched node).
"""

from __future__ import annotations

import asyncio
from typing import Dict, Optional, Set, Tuple

import xxhash


class TrieNode:
    __slots__ = ("children", "endpoints", "lock")

    def __init__(self) -> None:
        self.children: Dict[int, "TrieNode"] = {}
        self.endpoints: Set[str] = set()
        self.lock = asyncio.Lock()


def _chunk_hashes(text: str, chunk_size: int):
    for i in range(0, len(text), chunk_size):
        yield xxhash.xxh64_intdigest(text[i : i + chunk_size])


class HashTrie:
    def __init__(self, chunk_size: int = 128) -> None:
        self.chunk_size = chunk_size
        self.root = TrieNode()

    async def insert(self, text: str, endpoint: str) -> None:
        node = self.root
        for h in _chunk_hashes(text, self.chunk_size):
            async with node.lock:
                child = node.children.get(h)
                if child is None:
                    child = TrieNode()
                    node.children[h] = child
            node = child
            async with node.lock:
                node.endpoints.add(endpoint)

    async def longest_prefix_match(
        self, text: str, available: Optional[Set[str]] = None
    ) -> Tuple[int, Set[str]]:
        """Returns (matched_chars, endpoint set at the deepest match that
        intersects `available`)."""
        node = self.root
        matched = 0
        selected: Set[str] = set()
        for i, h in enumerate(_chunk_hashes(text, self.chunk_size)):
            async with node.lock:
                child = node.children.get(h)
            if child is None:
                break
            eps = child.endpoints
            if available is not None:
                eps = eps & available
            if not eps:
                break
            node = child
            selected = set(eps)
            matched = min((i + 1) * self.chunk_size, len(text))
        return matched, selected

    async def remove_endpoint(self, endpoint: str) -> None:
        """Drop a dead endpoint from the whole trie (iterative DFS)."""
        stack = [self.root]
        while stack:
            node = stack.pop()
            async with node.lock:
                node.endpoints.discard(endpoint)
                stack.extend(node.children.values())
