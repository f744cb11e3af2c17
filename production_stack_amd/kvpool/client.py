"""Clients for the KV controller.

ControllerClient — used by the router's KvAwareRouter for lookups.
EngineReporter — runs inside each engine server, streaming prefix-block
registrations/evictions and heartbeats.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Dict, List, Optional

from production_stack_amd.kvpool.protocol import recv_msg, send_msg

logger = logging.getLogger("kvpool.client")


class ControllerClient:
    def __init__(self, host: str, port: int, timeout: float = 2.0) -> None:
        self.host = host
        self.port = port
        self.timeout = timeout
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._lock = asyncio.Lock()

    async def _connect(self) -> None:
        self._reader, self._writer = await asyncio.wait_for(
            asyncio.open_connection(self.host, self.port), self.timeout
        )

    async def _call(self, msg: dict) -> dict:
        async with self._lock:
            if self._writer is None or self._writer.is_closing():
                await self._connect()
            try:
                await send_msg(self._writer, msg)
                return await asyncio.wait_for(
                    recv_msg(self._reader), self.timeout
                )
            except (ConnectionError, asyncio.IncompleteReadError):
                self._writer = None
                raise ConnectionError("controller connection lost")

    async def lookup(self, token_ids: List[int]) -> Dict[str, int]:
        resp = await self._call({"type": "lookup", "tokens": list(token_ids)})
        return resp.get("matches", {})

    async def stats(self) -> Dict[str, int]:
        resp = await self._call({"type": "stats"})
        return resp.get("instances", {})

    async def close(self) -> None:
        if self._writer is not None:
            self._writer.close()
            self._writer = None


class EngineReporter:
    """Periodically reports an engine's prefix-cache contents.

    Reads newly registered/evicted block hashes off the BlockManager and
    pushes deltas; sends heartbeats even when idle.
    """

    def __init__(
        self,
        engine,
        url: str,
        host: str,
        port: int,
        interval: float = 1.0,
    ) -> None:
        self.engine = engine
        self.url = url
        self.host = host
        self.port = port
        self.interval = interval
        self._known: set = set()
        self._client = ControllerClient(host, port, timeout=5.0)
        self._task: Optional[asyncio.Task] = None
        self._stop = asyncio.Event()

    def snapshot_delta(self):
        bm = self.engine.block_manager
        current = set(bm.cached.keys())
        insert = list(current - self._known)
        evict = list(self._known - current)
        self._known = current
        return insert, evict

    async def run_once(self) -> None:
        insert, evict = self.snapshot_delta()
        if insert or evict:
            await self._client._call(
                {
                    "type": "update",
                    "url": self.url,
                    "insert": insert,
                    "evict": evict,
                }
            )
        else:
            await self._client._call({"type": "heartbeat", "url": self.url})

    async def _loop(self) -> None:
        try:
            await self._client._call({"type": "register", "url": self.url})
        except (ConnectionError, OSError, asyncio.TimeoutError):
            logger.warning("KV controller unreachable; reporter idle")
        while not self._stop.is_set():
            try:
                await self.run_once()
            except (ConnectionError, OSError, asyncio.TimeoutError):
                pass
            try:
                await asyncio.wait_for(self._stop.wait(), self.interval)
            except asyncio.TimeoutError:
                pass

    def start(self) -> None:
        self._task = asyncio.get_running_loop().create_task(self._loop())

    async def stop(self) -> None:
        self._stop.set()
        if self._task:
            await self._task
        await self._client.close()
