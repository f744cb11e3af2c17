"""Consistent-hash ring for session-sticky routing.

Replaces the reference's uhashring dependency (not available offline) with a
standard ketama-style ring: each node gets `vnodes` virtual points on a
2^64 ring keyed by xxhash64; lookup walks clockwise from the key's point.
"""

from __future__ import annotations

import bisect
from typing import Iterable, List, Optional

import xxhash


class HashRing:
    def __init__(self, nodes: Iterable[str] = (), vnodes: int = 100) -> None:
        self.vnodes = vnodes
        self._points: List[int] = []
        self._owners: List[str] = []
        self._nodes: set = set()
        for n in nodes:
            self.add_node(n)

    @staticmethod
    def _hash(key: str) -> int:
        return xxhash.xxh64_intdigest(key)

    def add_node(self, node: str) -> None:
        if node in self._nodes:
            return
        self._nodes.add(node)
        for i in range(self.vnodes):
            p = self._hash(f"{node}#{i}")
            idx = bisect.bisect(self._points, p)
            self._points.insert(idx, p)
            self._owners.insert(idx, node)

    def remove_node(self, node: str) -> None:
        if node not in self._nodes:
            return
        self._nodes.remove(node)
        keep = [
            (p, o)
            for p, o in zip(self._points, self._owners)
            if o != node
        ]
        self._points = [p for p, _ in keep]
        self._owners = [o for _, o in keep]

    def update_nodes(self, nodes: Iterable[str]) -> None:
        target = set(nodes)
        for n in list(self._nodes - target):
            self.remove_node(n)
        for n in target - self._nodes:
            self.add_node(n)

    def get_node(self, key: str) -> Optional[str]:
        if not self._points:
            return None
        p = self._hash(key)
        idx = bisect.bisect(self._points, p) % len(self._points)
        return self._owners[idx]

    def __len__(self) -> int:
        return len(self._nodes)
