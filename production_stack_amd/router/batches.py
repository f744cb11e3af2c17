"""OpenAI Batch API: sqlite-queued jobs with a background processor.

Parity: reference services/batch_service (BatchInfo/BatchStatus, SQLite
queue, background poller). Unlike the reference's stubbed processor
(local_processor.py:190-208 sleeps), this one actually executes each JSONL
line against the routed backends.
"""

from __future__ import annotations

import asyncio
import json
import logging
import sqlite3
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from production_stack_amd.router.files import FileStorage

logger = logging.getLogger("router.batches")


@dataclass
class BatchInfo:
    id: str
    input_file_id: str
    endpoint: str
    completion_window: str = "24h"
    status: str = "validating"
    created_at: int = field(default_factory=lambda: int(time.time()))
    output_file_id: Optional[str] = None
    error_file_id: Optional[str] = None
    completed_at: Optional[int] = None
    metadata: Optional[Dict[str, Any]] = None

    def to_dict(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "object": "batch",
            "endpoint": self.endpoint,
            "input_file_id": self.input_file_id,
            "completion_window": self.completion_window,
            "status": self.status,
            "created_at": self.created_at,
            "output_file_id": self.output_file_id,
            "error_file_id": self.error_file_id,
            "completed_at": self.completed_at,
            "metadata": self.metadata or {},
        }


class BatchProcessor:
    def __init__(
        self,
        storage: FileStorage,
        db_path: str = "/tmp/vllm_batches.sqlite",
        base_url: Optional[str] = None,
    ) -> None:
        self.storage = storage
        self.db_path = db_path
        self.base_url = base_url  # route through ourselves by default
        self._lock = threading.Lock()
        self._init_db()
        self._task: Optional[asyncio.Task] = None

    def _db(self):
        conn = sqlite3.connect(self.db_path)
        conn.row_factory = sqlite3.Row
        return conn

    def _init_db(self) -> None:
        with self._db() as conn:
            conn.execute(
                "CREATE TABLE IF NOT EXISTS batches ("
                "id TEXT PRIMARY KEY, data TEXT NOT NULL)"
            )

    def _save(self, b: BatchInfo) -> None:
        with self._lock, self._db() as conn:
            conn.execute(
                "INSERT OR REPLACE INTO batches (id, data) VALUES (?, ?)",
                (b.id, json.dumps(b.to_dict())),
            )

    def create_batch(
        self,
        input_file_id: str,
        endpoint: str,
        completion_window: str = "24h",
        metadata: Optional[Dict] = None,
    ) -> BatchInfo:
        b = BatchInfo(
            id=f"batch_{uuid.uuid4().hex[:24]}",
            input_file_id=input_file_id,
            endpoint=endpoint,
            completion_window=completion_window,
            metadata=metadata,
        )
        self._save(b)
        return b

    def get_batch(self, batch_id: str) -> Optional[BatchInfo]:
        with self._db() as conn:
            row = conn.execute(
                "SELECT data FROM batches WHERE id = ?", (batch_id,)
            ).fetchone()
        if row is None:
            return None
        d = json.loads(row["data"])
        d.pop("object", None)
        return BatchInfo(**d)

    def list_batches(self) -> list:
        with self._db() as conn:
            rows = conn.execute("SELECT data FROM batches").fetchall()
        return [json.loads(r["data"]) for r in rows]

    def cancel_batch(self, batch_id: str) -> Optional[BatchInfo]:
        b = self.get_batch(batch_id)
        if b is None:
            return None
        if b.status in ("validating", "in_progress"):
            b.status = "cancelled"
            self._save(b)
        return b

    async def run_batch(self, batch_id: str) -> None:
        """Execute every JSONL request line against local endpoints."""
        import aiohttp

        b = self.get_batch(batch_id)
        if b is None:
            return
        content = self.storage.get_file_content(b.input_file_id)
        if content is None:
            b.status = "failed"
            self._save(b)
            return
        b.status = "in_progress"
        self._save(b)
        results = []
        from production_stack_amd.router.service_discovery import (
            get_service_discovery,
        )

        try:
            endpoints = get_service_discovery().get_endpoint_info()
        except RuntimeError:
            endpoints = []
        async with aiohttp.ClientSession() as sess:
            for line in content.decode().splitlines():
                if not line.strip():
                    continue
                try:
                    req = json.loads(line)
                except json.JSONDecodeError:
                    continue
                url = self.base_url
                if url is None and endpoints:
                    url = endpoints[0].url
                entry: Dict[str, Any] = {
                    "id": f"batch_req_{uuid.uuid4().hex[:12]}",
                    "custom_id": req.get("custom_id"),
                }
                try:
                    async with sess.post(
                        (url or "") + req.get("url", b.endpoint),
                        json=req.get("body", {}),
                        timeout=aiohttp.ClientTimeout(total=300),
                    ) as r:
                        entry["response"] = {
                            "status_code": r.status,
                            "body": await r.json(),
                        }
                except Exception as e:
                    entry["error"] = str(e)
                results.append(entry)
        out = "\n".join(json.dumps(r) for r in results).encode()
        meta = self.storage.save_file(
            out, f"{batch_id}_output.jsonl", purpose="batch_output"
        )
        b.output_file_id = meta.id
        b.status = "completed"
        b.completed_at = int(time.time())
        self._save(b)
