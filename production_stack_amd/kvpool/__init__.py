"""KV cache pool layer: host-DRAM offload/tiering and the KV controller
(control plane for KV-aware routing).

Capability parity with the LMCache layer the reference stack drives through
LMCACHE_* env vars and the ZMQ controller ports (SURVEY.md sections 2.7 and
2.9). The control plane here is msgpack-over-TCP (asyncio) instead of ZMQ
(pyzmq is not in the offline image); the wire concepts are the same:
registration, heartbeats, per-instance prefix index, LookupMsg.
"""
