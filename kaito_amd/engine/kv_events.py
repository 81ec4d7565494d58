"""KV-cache event bus on :5557 — the surface the reference exposes over ZMQ
for EPP KVCache-aware routing (pkg/model/interface.go:430-437, KV events
port consts.go:143-145; event types BlockStored / BlockRemoved /
AllBlocksCleared).

pyzmq is not available in the MI355X image, so the publisher speaks the
real ZMTP 3.0 wire protocol directly (engine/zmtp.py): an UNMODIFIED
pyzmq/libzmq SUB socket can connect to tcp://host:5557, subscribe, and
receive the events — matching the reference's contract where external
EPP builds consume vLLM's ZMQ event stream.

Message format (vLLM ZmqEventPublisher framing): multipart
[topic, seq (8-byte big-endian), payload] where payload is msgpack
{"event": str, "block_hashes": [int], "ts": float}.
"""
from __future__ import annotations

import struct
import time
from typing import Callable, List, Optional

import msgpack

from . import zmtp

BLOCK_STORED = "BlockStored"
BLOCK_REMOVED = "BlockRemoved"
ALL_BLOCKS_CLEARED = "AllBlocksCleared"
DEFAULT_PORT = 5557
DEFAULT_TOPIC = b"kv-events"


class KVEventPublisher:
    """ZMTP PUB socket publishing KV-cache events."""

    def __init__(self, host: str = "0.0.0.0", port: int = DEFAULT_PORT,
                 topic: bytes = DEFAULT_TOPIC):
        self._pub = zmtp.PubSocket(host, port)
        self.port = self._pub.port
        self.topic = topic
        self._seq = 0

    def publish(self, event: str, block_hashes: Optional[List[int]] = None):
        payload = msgpack.packb({"event": event,
                                 "block_hashes": block_hashes or [],
                                 "ts": time.time()})
        self._pub.send_multipart([
            self.topic, struct.pack(">Q", self._seq), payload])
        self._seq += 1

    def block_stored(self, hashes: List[int]):
        self.publish(BLOCK_STORED, hashes)

    def block_removed(self, hashes: List[int]):
        self.publish(BLOCK_REMOVED, hashes)

    def all_cleared(self):
        self.publish(ALL_BLOCKS_CLEARED)

    def close(self):
        self._pub.close()


class KVEventSubscriber:
    """ZMTP SUB client for the event stream (used by the routing side and
    tests; any libzmq SUB socket works equally)."""

    def __init__(self, host: str = "127.0.0.1", port: int = DEFAULT_PORT,
                 on_event: Optional[Callable[[dict], None]] = None,
                 topics: tuple = (b"",)):
        self.on_event = on_event
        self.events: List[dict] = []
        self.seqs: List[int] = []
        self._sub = zmtp.SubSocket(host, port, topics=topics,
                                   on_message=self._on_message)

    def _on_message(self, parts: List[bytes]):
        if len(parts) != 3:
            return
        _topic, seq, payload = parts
        evt = msgpack.unpackb(payload)
        self.seqs.append(struct.unpack(">Q", seq)[0])
        self.events.append(evt)
        if self.on_event:
            self.on_event(evt)

    def close(self):
        self._sub.close()
