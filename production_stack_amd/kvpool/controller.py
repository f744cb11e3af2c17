"""KV controller: tracks which replica holds which KV prefix blocks.

Control-plane parity with the LMCache controller the reference's kvaware
routing talks to (reference routing_logic.py:252-428: worker registration,
heartbeats, LookupMsg -> longest-prefix instance). Engines stream their
block-hash registrations/evictions; the router's KvAwareRouter sends lookup
requests with token ids.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Dict, Optional, Set

from production_stack_amd.kvpool.protocol import (
    chain_hashes,
    recv_msg,
    send_msg,
)

logger = logging.getLogger("kvpool.controller")


class InstanceState:
    def __init__(self, url: str) -> None:
        self.url = url
        self.hashes: Set[int] = set()
        self.last_heartbeat = time.time()


class KVController:
    def __init__(
        self,
        host: str = "127.0.0.1",
        port: int = 9000,
        block_size: int = 16,
        heartbeat_timeout: float = 60.0,
    ) -> None:
        self.host = host
        self.port = port
        self.block_size = block_size
        self.heartbeat_timeout = heartbeat_timeout
        self.instances: Dict[str, InstanceState] = {}
        self._server: Optional[asyncio.AbstractServer] = None

    # ------------------------------------------------------------------
    def _instance(self, url: str) -> InstanceState:
        if url not in self.instances:
            self.instances[url] = InstanceState(url)
        return self.instances[url]

    def handle(self, msg: dict) -> Optional[dict]:
        t = msg.get("type")
        if t == "register":
            inst = self._instance(msg["url"])
            inst.last_heartbeat = time.time()
            inst.hashes.clear()
            return {"ok": True}
        if t == "deregister":
            self.instances.pop(msg["url"], None)
            return {"ok": True}
        if t == "heartbeat":
            self._instance(msg["url"]).last_heartbeat = time.time()
            return {"ok": True}
        if t == "update":
            inst = self._instance(msg["url"])
            inst.last_heartbeat = time.time()
            inst.hashes.update(msg.get("insert", []))
            for h in msg.get("evict", []):
                inst.hashes.discard(h)
            return {"ok": True}
        if t == "lookup":
            return {"ok": True, "matches": self.lookup(msg["tokens"])}
        if t == "stats":
            return {
                "ok": True,
                "instances": {
                    url: len(i.hashes) for url, i in self.instances.items()
                },
            }
        return {"ok": False, "error": f"unknown message type {t!r}"}

    def lookup(self, token_ids) -> Dict[str, int]:
        """Longest chain-hash prefix match per live instance -> token count."""
        hashes = chain_hashes(token_ids, self.block_size)
        now = time.time()
        out: Dict[str, int] = {}
        for url, inst in self.instances.items():
            if now - inst.last_heartbeat > self.heartbeat_timeout:
                continue
            matched = 0
            for h in hashes:
                if h in inst.hashes:
                    matched += 1
                else:
                    break
            out[url] = matched * self.block_size
        return out

    # ------------------------------------------------------------------
    async def _client_loop(
        self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter
    ) -> None:
        try:
            while True:
                msg = await recv_msg(reader)
                resp = self.handle(msg)
                if msg.get("rsvp", True):
                    await send_msg(writer, resp)
        except (asyncio.IncompleteReadError, ConnectionError):
            pass
        finally:
            writer.close()

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._client_loop, self.host, self.port
        )
        logger.info("KV controller listening on %s:%d", self.host, self.port)

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None


async def run_controller(host: str = "0.0.0.0", port: int = 9000) -> None:
    c = KVController(host, port)
    await c.start()
    await asyncio.Event().wait()


def main() -> None:  # console entry for the cacheserver deployment
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9000)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    asyncio.run(run_controller(args.host, args.port))


if __name__ == "__main__":
    main()
