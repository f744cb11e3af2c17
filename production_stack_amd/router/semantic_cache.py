"""Semantic cache (feature-gated, Alpha).

Parity: reference experimental/semantic_cache/ — embed the chat request,
search past responses by cosine similarity, return the cached answer on a
hit, store new answers on completion. The reference uses
sentence-transformers + FAISS; neither ships in the offline image, so the
default embedder is a deterministic hashed n-gram bag-of-words vector and
search is a dense numpy cosine scan (exact, fine for 10^4 entries). A
sentence-transformers embedder is used automatically when importable.
"""

from __future__ import annotations

import hashlib
import logging
import threading
import time
from typing import Any, Dict, List, Optional

import numpy as np

logger = logging.getLogger("router.semantic_cache")


class HashedNGramEmbedder:
    def __init__(self, dim: int = 512) -> None:
        self.dim = dim

    def embed(self, text: str) -> np.ndarray:
        v = np.zeros(self.dim, dtype=np.float32)
        words = text.lower().split()
        grams = words + [
            " ".join(words[i : i + 2]) for i in range(len(words) - 1)
        ]
        for g in grams:
            h = int.from_bytes(
                hashlib.blake2b(g.encode(), digest_size=8).digest(), "little"
            )
            v[h % self.dim] += 1.0 if (h >> 63) else -1.0
        n = np.linalg.norm(v)
        return v / n if n > 0 else v


def make_embedder():
    try:  # pragma: no cover - optional heavy dependency
        from sentence_transformers import SentenceTransformer

        model = SentenceTransformer("all-MiniLM-L6-v2")

        class STEmbedder:
            dim = model.get_sentence_embedding_dimension()

            def embed(self, text: str) -> np.ndarray:
                v = model.encode([text])[0].astype(np.float32)
                n = np.linalg.norm(v)
                return v / n if n > 0 else v

        return STEmbedder()
    except Exception:
        return HashedNGramEmbedder()


class SemanticCache:
    def __init__(
        self,
        threshold: float = 0.95,
        max_entries: int = 10000,
        ttl: float = 3600.0,
        embedder=None,
    ) -> None:
        self.threshold = threshold
        self.max_entries = max_entries
        self.ttl = ttl
        self.embedder = embedder or make_embedder()
        self._vecs: Optional[np.ndarray] = None
        self._entries: List[Dict[str, Any]] = []
        self._lock = threading.Lock()
        self.hits = 0
        self.misses = 0

    @staticmethod
    def request_text(body: Dict[str, Any]) -> str:
        parts = [body.get("model", "")]
        for m in body.get("messages") or []:
            c = m.get("content")
            if isinstance(c, str):
                parts.append(f"{m.get('role')}:{c}")
        if isinstance(body.get("prompt"), str):
            parts.append(body["prompt"])
        return "\n".join(parts)

    def search(self, body: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        text = self.request_text(body)
        q = self.embedder.embed(text)
        now = time.time()
        with self._lock:
            if self._vecs is None or not len(self._entries):
                self.misses += 1
                return None
            sims = self._vecs @ q
            best = int(np.argmax(sims))
            entry = self._entries[best]
            if (
                sims[best] >= self.threshold
                and now - entry["ts"] <= self.ttl
            ):
                self.hits += 1
                return entry["response"]
            self.misses += 1
            return None

    def store(
        self, body: Dict[str, Any], response: Dict[str, Any]
    ) -> None:
        text = self.request_text(body)
        v = self.embedder.embed(text)
        with self._lock:
            self._entries.append({"ts": time.time(), "response": response})
            if self._vecs is None:
                self._vecs = v[None, :]
            else:
                self._vecs = np.vstack([self._vecs, v])
            if len(self._entries) > self.max_entries:
                self._entries.pop(0)
                self._vecs = self._vecs[1:]

    def metrics(self) -> Dict[str, float]:
        total = self.hits + self.misses
        return {
            "semantic_cache_hits": float(self.hits),
            "semantic_cache_misses": float(self.misses),
            "semantic_cache_hit_rate": self.hits / total if total else 0.0,
            "semantic_cache_entries": float(len(self._entries)),
        }
