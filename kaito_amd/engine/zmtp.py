"""Minimal ZMTP 3.0 (ZeroMQ wire protocol, rfc.zeromq.org/spec/23) PUB/SUB.

The reference emits vLLM KV-cache events over a real ZMQ PUB socket on
:5557 (pkg/model/interface.go:430-437, consts.go:143-145) and the EPP /
llm-d ecosystem consumes them with libzmq clients. pyzmq is not in this
image, so this module speaks the ZMTP byte protocol directly over TCP —
an UNMODIFIED pyzmq/libzmq SUB socket can connect, subscribe, and
receive multipart messages; equally, our SUB client can read from a real
ZMQ PUB server.

Scope: TCP transport, NULL security mechanism, PUB and SUB socket types,
ZMTP 3.0 framing (short/long frames, command frames, READY handshake),
3.0-style subscriptions (\x01/\x00 message frames) plus acceptance of
3.1 SUBSCRIBE/CANCEL command frames from newer peers.
"""
from __future__ import annotations

import socket
import struct
import threading
from typing import Callable, List, Optional, Tuple

# frame flag bits
FLAG_MORE = 0x01
FLAG_LONG = 0x02
FLAG_COMMAND = 0x04


def greeting(as_server: bool = False) -> bytes:
    """64-byte ZMTP greeting: signature, version 3.0, NULL mechanism."""
    sig = b"\xff" + b"\x00" * 8 + b"\x7f"
    ver = bytes([3, 0])
    mech = b"NULL" + b"\x00" * 16
    return sig + ver + mech + (b"\x01" if as_server else b"\x00") \
        + b"\x00" * 31


def encode_frame(body: bytes, more: bool = False,
                 command: bool = False) -> bytes:
    flags = (FLAG_MORE if more else 0) | (FLAG_COMMAND if command else 0)
    if len(body) > 255:
        return bytes([flags | FLAG_LONG]) + struct.pack(">Q", len(body)) \
            + body
    return bytes([flags, len(body)]) + body


def encode_command(name: bytes, metadata: List[Tuple[bytes, bytes]] = (),
                   body: bytes = b"") -> bytes:
    """Command frame: name-size + name + (metadata properties | raw body)."""
    payload = bytes([len(name)]) + name
    for k, v in metadata:
        payload += bytes([len(k)]) + k + struct.pack(">I", len(v)) + v
    payload += body
    return encode_frame(payload, command=True)


def parse_metadata(body: bytes) -> dict:
    props = {}
    i = 0
    while i < len(body):
        nlen = body[i]
        name = body[i + 1:i + 1 + nlen]
        i += 1 + nlen
        vlen = struct.unpack(">I", body[i:i + 4])[0]
        props[name.decode("ascii", "replace")] = body[i + 4:i + 4 + vlen]
        i += 4 + vlen
    return props


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed during ZMTP exchange")
        buf += chunk
    return buf


def read_frame(sock: socket.socket) -> Tuple[int, bytes]:
    """Read one frame; returns (flags, body)."""
    flags = _recv_exact(sock, 1)[0]
    if flags & FLAG_LONG:
        size = struct.unpack(">Q", _recv_exact(sock, 8))[0]
    else:
        size = _recv_exact(sock, 1)[0]
    if size > (1 << 30):
        raise ConnectionError(f"oversized ZMTP frame ({size} bytes)")
    return flags, _recv_exact(sock, size)


def handshake(sock: socket.socket, socket_type: bytes,
              expect: Tuple[bytes, ...]) -> dict:
    """Exchange greetings + NULL READY commands. Returns the peer's
    metadata. `socket_type` is ours (b"PUB"/b"SUB"); `expect` lists the
    peer types we accept."""
    sock.sendall(greeting())
    # read the peer greeting tolerantly: libzmq sends the 10-byte
    # signature first, then the rest — but always 64 bytes total
    peer = _recv_exact(sock, 64)
    if peer[0] != 0xFF or peer[9] & 0x01 != 0x01:
        raise ConnectionError("not a ZMTP peer (bad signature)")
    major = peer[10]
    if major < 3:
        raise ConnectionError(f"unsupported ZMTP major version {major}")
    mech = peer[12:32].rstrip(b"\x00")
    if mech != b"NULL":
        raise ConnectionError(f"unsupported ZMTP mechanism {mech!r}")
    sock.sendall(encode_command(b"READY",
                                [(b"Socket-Type", socket_type)]))
    flags, body = read_frame(sock)
    if not flags & FLAG_COMMAND:
        raise ConnectionError("expected READY command frame")
    nlen = body[0]
    name = body[1:1 + nlen]
    if name != b"READY":
        raise ConnectionError(f"expected READY, got {name!r}")
    meta = parse_metadata(body[1 + nlen:])
    ptype = meta.get("Socket-Type", b"")
    if expect and ptype not in expect:
        raise ConnectionError(f"peer socket type {ptype!r} not in {expect}")
    return meta


class PubSocket:
    """ZMTP PUB server: accepts SUB/XSUB peers, tracks per-peer topic
    subscriptions, sends multipart messages to matching peers."""

    def __init__(self, host: str = "0.0.0.0", port: int = 0):
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(16)
        self.port = self._srv.getsockname()[1]
        # sock -> set of topic prefixes
        self._peers: dict = {}
        self._lock = threading.Lock()
        self._stop = False
        self._threads: List[threading.Thread] = []
        t = threading.Thread(target=self._accept_loop, daemon=True,
                             name="zmtp-pub-accept")
        t.start()
        self._threads.append(t)

    def _accept_loop(self):
        self._srv.settimeout(0.5)
        while not self._stop:
            try:
                conn, _ = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            t = threading.Thread(target=self._peer_loop, args=(conn,),
                                 daemon=True, name="zmtp-pub-peer")
            t.start()
            self._threads.append(t)

    def _peer_loop(self, conn: socket.socket):
        try:
            conn.settimeout(5.0)
            handshake(conn, b"PUB", (b"SUB", b"XSUB"))
        except (ConnectionError, OSError, socket.timeout):
            conn.close()
            return
        with self._lock:
            self._peers[conn] = set()
        conn.settimeout(0.5)
        # subscription read loop: ZMTP 3.0 sends \x01<topic>/\x00<topic>
        # message frames; 3.1 peers send SUBSCRIBE/CANCEL commands
        while not self._stop:
            try:
                flags, body = read_frame(conn)
            except socket.timeout:
                continue
            except (ConnectionError, OSError):
                break
            if flags & FLAG_COMMAND:
                nlen = body[0]
                name, rest = body[1:1 + nlen], body[1 + nlen:]
                if name == b"SUBSCRIBE":
                    self._subscribe(conn, rest, True)
                elif name == b"CANCEL":
                    self._subscribe(conn, rest, False)
                continue
            if body[:1] == b"\x01":
                self._subscribe(conn, body[1:], True)
            elif body[:1] == b"\x00":
                self._subscribe(conn, body[1:], False)
        with self._lock:
            self._peers.pop(conn, None)
        conn.close()

    def _subscribe(self, conn, topic: bytes, on: bool):
        with self._lock:
            subs = self._peers.get(conn)
            if subs is None:
                return
            if on:
                subs.add(topic)
            else:
                subs.discard(topic)

    @property
    def num_subscribers(self) -> int:
        with self._lock:
            return len(self._peers)

    @property
    def num_subscriptions(self) -> int:
        """Total topic subscriptions across peers (a peer only receives
        messages after its subscribe frame is processed — ZMQ slow-joiner
        semantics; poll this before publishing in tests)."""
        with self._lock:
            return sum(len(s) for s in self._peers.values())

    def send_multipart(self, parts: List[bytes]) -> int:
        """Send to every peer subscribed to a prefix of parts[0] (the
        topic frame). Returns the number of peers reached."""
        if not parts:
            return 0
        topic = parts[0]
        wire = b"".join(
            encode_frame(p, more=(i < len(parts) - 1))
            for i, p in enumerate(parts))
        sent = 0
        with self._lock:
            dead = []
            for conn, subs in self._peers.items():
                if not any(topic.startswith(t) for t in subs):
                    continue
                try:
                    conn.sendall(wire)
                    sent += 1
                except OSError:
                    dead.append(conn)
            for c in dead:
                self._peers.pop(c, None)
                c.close()
        return sent

    def close(self):
        self._stop = True
        try:
            self._srv.close()
        except OSError:
            pass
        with self._lock:
            for c in self._peers:
                try:
                    c.close()
                except OSError:
                    pass
            self._peers.clear()


class SubSocket:
    """ZMTP SUB client: connects to a PUB server, subscribes to topic
    prefixes, delivers multipart messages to a callback."""

    def __init__(self, host: str, port: int,
                 topics: Tuple[bytes, ...] = (b"",),
                 on_message: Optional[Callable[[List[bytes]], None]] = None):
        self._sock = socket.create_connection((host, port), timeout=5)
        handshake(self._sock, b"SUB", (b"PUB", b"XPUB"))
        for t in topics:
            self._sock.sendall(encode_frame(b"\x01" + t))
        self.on_message = on_message
        self.messages: List[List[bytes]] = []
        self._stop = False
        self._thread = threading.Thread(target=self._recv_loop, daemon=True,
                                        name="zmtp-sub-recv")
        self._thread.start()

    def _recv_loop(self):
        self._sock.settimeout(0.5)
        parts: List[bytes] = []
        while not self._stop:
            try:
                flags, body = read_frame(self._sock)
            except socket.timeout:
                continue
            except (ConnectionError, OSError):
                return
            if flags & FLAG_COMMAND:
                continue  # PING etc. — ignore
            parts.append(body)
            if not flags & FLAG_MORE:
                self.messages.append(parts)
                if self.on_message:
                    self.on_message(parts)
                parts = []

    def close(self):
        self._stop = True
        try:
            self._sock.close()
        except OSError:
            pass
