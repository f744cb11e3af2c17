"""Wire protocol: 4-byte big-endian length prefix + msgpack payload."""

from __future__ import annotations

import asyncio
import struct
from typing import Any

import msgpack

MAX_MSG = 64 * 1024 * 1024


async def send_msg(writer: asyncio.StreamWriter, obj: Any) -> None:
    payload = msgpack.packb(obj, use_bin_type=True)
    writer.write(struct.pack(">I", len(payload)) + payload)
    await writer.drain()


async def recv_msg(reader: asyncio.StreamReader) -> Any:
    header = await reader.readexactly(4)
    (n,) = struct.unpack(">I", header)
    if n > MAX_MSG:
        raise ValueError(f"message too large: {n}")
    payload = await reader.readexactly(n)
    return msgpack.unpackb(payload, raw=False)


def chain_hashes(token_ids, block_size: int = 16):
    """Chain hash per full block — MUST match BlockManager.chain_hash.
    The seed for the first block is 0, NOT None: hash(None) is
    address-derived on CPython < 3.12 and differs across processes,
    which would silently break every cross-process prefix match
    (kvaware lookups, remote-tier keys). int/tuple-of-int hashing is
    stable regardless of PYTHONHASHSEED."""
    out = []
    prev = 0
    for i in range(len(token_ids) // block_size):
        h = hash((prev, tuple(token_ids[i * block_size : (i + 1) * block_size])))
        out.append(h)
        prev = h
    return out
