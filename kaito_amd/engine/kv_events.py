"""KV-cache event bus on :5557 — the surface the reference exposes over ZMQ
for EPP KVCache-aware routing (pkg/model/interface.go:430-437, KV events
port consts.go:143-145; event types BlockStored / BlockRemoved /
AllBlocksCleared).

pyzmq is not available in the MI355X image, so the wire protocol is a
minimal TCP pub/sub: 4-byte big-endian length prefix + msgpack payload
{"event": str, "block_hashes": [int], "ts": float}. KVEventSubscriber is
the matching in-repo client (used by the routing side and tests).
"""
from __future__ import annotations

import socket
import struct
import threading
import time
from typing import Callable, List, Optional

import msgpack

BLOCK_STORED = "BlockStored"
BLOCK_REMOVED = "BlockRemoved"
ALL_BLOCKS_CLEARED = "AllBlocksCleared"
DEFAULT_PORT = 5557


class KVEventPublisher:
    def __init__(self, host: str = "0.0.0.0", port: int = DEFAULT_PORT):
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(16)
        self.port = self._srv.getsockname()[1]
        self._subs: List[socket.socket] = []
        self._lock = threading.Lock()
        self._stop = False
        self._accept_thread = threading.Thread(target=self._accept_loop,
                                               daemon=True)
        self._accept_thread.start()

    def _accept_loop(self):
        while not self._stop:
            try:
                self._srv.settimeout(0.5)
                conn, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            with self._lock:
                self._subs.append(conn)

    def publish(self, event: str, block_hashes: Optional[List[int]] = None):
        frame = msgpack.packb({"event": event,
                               "block_hashes": block_hashes or [],
                               "ts": time.time()})
        data = struct.pack(">I", len(frame)) + frame
        with self._lock:
            dead = []
            for s in self._subs:
                try:
                    s.sendall(data)
                except OSError:
                    dead.append(s)
            for s in dead:
                self._subs.remove(s)
                s.close()

    def block_stored(self, hashes: List[int]):
        self.publish(BLOCK_STORED, hashes)

    def block_removed(self, hashes: List[int]):
        self.publish(BLOCK_REMOVED, hashes)

    def all_cleared(self):
        self.publish(ALL_BLOCKS_CLEARED)

    def close(self):
        self._stop = True
        try:
            self._srv.close()
        except OSError:
            pass
        with self._lock:
            for s in self._subs:
                s.close()
            self._subs.clear()


class KVEventSubscriber:
    def __init__(self, host: str = "127.0.0.1", port: int = DEFAULT_PORT,
                 on_event: Optional[Callable[[dict], None]] = None):
        self._sock = socket.create_connection((host, port), timeout=5)
        self.on_event = on_event
        self.events: List[dict] = []
        self._stop = False
        self._thread = threading.Thread(target=self._recv_loop, daemon=True)
        self._thread.start()

    def _recv_loop(self):
        buf = b""
        while not self._stop:
            try:
                self._sock.settimeout(0.5)
                chunk = self._sock.recv(65536)
            except socket.timeout:
                continue
            except OSError:
                return
            if not chunk:
                return
            buf += chunk
            while len(buf) >= 4:
                n = struct.unpack(">I", buf[:4])[0]
                if len(buf) < 4 + n:
                    break
                evt = msgpack.unpackb(buf[4:4 + n])
                buf = buf[4 + n:]
                self.events.append(evt)
                if self.on_event:
                    self.on_event(evt)

    def close(self):
        self._stop = True
        try:
            self._sock.close()
        except OSError:
            pass
