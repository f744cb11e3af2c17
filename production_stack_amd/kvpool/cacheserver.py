"""Remote KV cache server: the data plane of the shared cache tier.

Capability parity: the reference deploys a central cache server that
engines PUT KV blocks into and GET on prefix miss so one instance's
prefill work is reusable by every other instance
(reference: cacheserver in values.yaml `cacheserverSpec` +
tutorials/05-offload-kv-cache.md; LMCache remote backend semantics).

MI355X-native shape: one record = every layer's K+V for one 16-token
block exactly as the host pool lays it out (layers x 2 x KH*16*HD bf16,
or int8 + fp32 row scales when the pool quantizes), so the engine side
is a straight pinned-DRAM <-> socket move with no re-serialization; the
server is a flat LRU byte store with the same 4-byte-length msgpack
framing the KV controller speaks (protocol.py).

Server ops:
  {"op": "put", "key": int, "data": bytes, "scales": bytes|None}
  {"op": "get", "key": int}        -> {"hit": bool, "data", "scales"}
  {"op": "exists", "keys": [int]}  -> {"hits": [0/1, ...]}
  {"op": "stats"}                  -> counters + bytes/records
"""

from __future__ import annotations

import asyncio
import logging
import socket
import struct
import threading
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

import msgpack

from production_stack_amd.kvpool.protocol import recv_msg, send_msg

logger = logging.getLogger("kvpool.cacheserver")


class CacheStore:
    """LRU byte store keyed by chain hash."""

    def __init__(self, capacity_gb: float = 4.0) -> None:
        self.capacity_bytes = int(capacity_gb * (1 << 30))
        self.used = 0
        self.records: "OrderedDict[int, Tuple[bytes, Optional[bytes]]]" = (
            OrderedDict()
        )
        self.puts = 0
        self.gets = 0
        self.hits = 0
        self.evictions = 0

    def put(self, key: int, data: bytes, scales: Optional[bytes]) -> None:
        if key in self.records:
            old = self.records.pop(key)
            self.used -= len(old[0]) + (len(old[1]) if old[1] else 0)
        sz = len(data) + (len(scales) if scales else 0)
        while self.used + sz > self.capacity_bytes and self.records:
            _, (d, s) = self.records.popitem(last=False)
            self.used -= len(d) + (len(s) if s else 0)
            self.evictions += 1
        self.records[key] = (data, scales)
        self.used += sz
        self.puts += 1

    def get(self, key: int):
        self.gets += 1
        rec = self.records.get(key)
        if rec is None:
            return None
        self.records.move_to_end(key)
        self.hits += 1
        return rec

    def exists(self, keys: List[int]) -> List[int]:
        return [1 if k in self.records else 0 for k in keys]

    def stats(self) -> Dict[str, int]:
        return {
            "records": len(self.records),
            "bytes": self.used,
            "capacity_bytes": self.capacity_bytes,
            "puts": self.puts,
            "gets": self.gets,
            "hits": self.hits,
            "evictions": self.evictions,
        }


class CacheServer:
    def __init__(self, host: str = "0.0.0.0", port: int = 9400,
                 capacity_gb: float = 4.0) -> None:
        self.host = host
        self.port = port
        self.store = CacheStore(capacity_gb)
        self._server: Optional[asyncio.AbstractServer] = None
        self._conn_tasks: set = set()

    async def _handle(self, reader: asyncio.StreamReader,
                      writer: asyncio.StreamWriter) -> None:
        task = asyncio.current_task()
        if task is not None:
            self._conn_tasks.add(task)
        try:
            while True:
                msg = await recv_msg(reader)
                op = msg.get("op")
                if op == "put":
                    self.store.put(msg["key"], msg["data"],
                                   msg.get("scales"))
                    await send_msg(writer, {"ok": True})
                elif op == "get":
                    rec = self.store.get(msg["key"])
                    if rec is None:
                        await send_msg(writer, {"hit": False})
                    else:
                        await send_msg(
                            writer,
                            {"hit": True, "data": rec[0], "scales": rec[1]},
                        )
                elif op == "exists":
                    await send_msg(
                        writer, {"hits": self.store.exists(msg["keys"])}
                    )
                elif op == "stats":
                    await send_msg(writer, self.store.stats())
                else:
                    await send_msg(writer, {"error": f"bad op {op!r}"})
        except (asyncio.IncompleteReadError, ConnectionResetError,
                asyncio.CancelledError):
            pass
        finally:
            self._conn_tasks.discard(task)
            writer.close()

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._handle, self.host, self.port
        )
        addr = self._server.sockets[0].getsockname()
        self.port = addr[1]
        logger.info("cacheserver listening on %s:%d", addr[0], addr[1])

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
        # drain live connection handlers so no coroutine is abandoned
        # mid-await when the loop stops (one tick so accepted-but-not-
        # yet-started handlers register in _conn_tasks)
        await asyncio.sleep(0.05)
        for t in list(self._conn_tasks):
            t.cancel()
        if self._conn_tasks:
            await asyncio.gather(*self._conn_tasks,
                                 return_exceptions=True)


async def run_cacheserver(host: str, port: int, capacity_gb: float) -> None:
    srv = CacheServer(host, port, capacity_gb)
    await srv.start()
    await asyncio.Event().wait()


def main() -> None:  # console entry: ps-cacheserver
    import argparse

    ap = argparse.ArgumentParser(description="KV cache server (data plane)")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=9400)
    ap.add_argument("--capacity-gb", type=float, default=4.0)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    asyncio.run(run_cacheserver(args.host, args.port, args.capacity_gb))


# ---------------------------------------------------------------------------
# Engine-side synchronous client (runs on the host-pool's push thread and,
# for gets, inline on the allocate path — both are CPU-side socket I/O that
# never touches a HIP stream).
# ---------------------------------------------------------------------------
class RemoteKVClient:
    def __init__(self, url: str, timeout: float = 5.0) -> None:
        host, port = url.rsplit(":", 1)
        self.addr = (host, int(port))
        self.timeout = timeout
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def _connect(self) -> socket.socket:
        if self._sock is None:
            s = socket.create_connection(self.addr, timeout=self.timeout)
            s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            self._sock = s
        return self._sock

    def _call(self, msg: dict) -> dict:
        with self._lock:
            try:
                s = self._connect()
                payload = msgpack.packb(msg, use_bin_type=True)
                s.sendall(struct.pack(">I", len(payload)) + payload)
                header = self._read_exact(s, 4)
                (n,) = struct.unpack(">I", header)
                return msgpack.unpackb(self._read_exact(s, n), raw=False)
            except OSError:
                self.close()
                raise

    def _read_exact(self, s: socket.socket, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = s.recv(n - len(buf))
            if not chunk:
                raise ConnectionResetError("cacheserver closed")
            buf += chunk
        return buf

    def put(self, key: int, data: bytes,
            scales: Optional[bytes] = None) -> bool:
        try:
            return bool(self._call(
                {"op": "put", "key": key, "data": data, "scales": scales}
            ).get("ok"))
        except OSError:
            return False

    def get(self, key: int):
        try:
            r = self._call({"op": "get", "key": key})
        except OSError:
            return None
        if not r.get("hit"):
            return None
        return r["data"], r.get("scales")

    def exists(self, key: int) -> bool:
        try:
            r = self._call({"op": "exists", "keys": [key]})
            return bool(r["hits"][0])
        except OSError:
            return False

    def stats(self) -> Optional[dict]:
        try:
            return self._call({"op": "stats"})
        except OSError:
            return None

    def close(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            finally:
                self._sock = None


if __name__ == "__main__":
    main()
