"""ZMTP 3.0 wire-protocol tests: a raw socket acting as an unmodified
libzmq/pyzmq peer (byte-level greeting, READY handshake, subscription,
frame decoding) must interoperate with the KV-event publisher.

Reference contract: vLLM KV events over ZMQ :5557 consumed by external
EPP builds (pkg/model/interface.go:430-437, consts.go:143-145).
"""
import socket
import struct
import time

import msgpack
import pytest

from kaito_amd.engine import zmtp
from kaito_amd.engine.kv_events import KVEventPublisher, KVEventSubscriber


def _libzmq_greeting() -> bytes:
    # exactly what libzmq 4.x sends for a NULL-mechanism client
    return (b"\xff" + b"\x00" * 8 + b"\x7f" + bytes([3, 0])
            + b"NULL" + b"\x00" * 16 + b"\x00" + b"\x00" * 31)


def _recv_exact(s, n):
    buf = b""
    while len(buf) < n:
        c = s.recv(n - len(buf))
        assert c, "peer closed"
        buf += c
    return buf


def _read_frame(s):
    flags = _recv_exact(s, 1)[0]
    if flags & zmtp.FLAG_LONG:
        size = struct.unpack(">Q", _recv_exact(s, 8))[0]
    else:
        size = _recv_exact(s, 1)[0]
    return flags, _recv_exact(s, size)


def _handshake_as_raw_sub(port, subscribe: bytes = b"",
                          split_greeting: bool = True,
                          use_31_command: bool = False):
    """Perform the client side of the ZMTP handshake with raw bytes, the
    way libzmq does it (signature first, then the rest)."""
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    g = _libzmq_greeting()
    if split_greeting:
        # libzmq sends the 10-byte signature, waits, then sends the rest
        s.sendall(g[:10])
        time.sleep(0.02)
        s.sendall(g[10:])
    else:
        s.sendall(g)
    peer = _recv_exact(s, 64)
    assert peer[0] == 0xFF and peer[9] == 0x7F, "bad server signature"
    assert peer[10] == 3, "server must offer ZMTP 3.x"
    assert peer[12:16] == b"NULL", "server must offer NULL mechanism"
    # send READY(Socket-Type: SUB), read server READY
    s.sendall(zmtp.encode_command(b"READY", [(b"Socket-Type", b"SUB")]))
    flags, body = _read_frame(s)
    assert flags & zmtp.FLAG_COMMAND
    nlen = body[0]
    assert body[1:1 + nlen] == b"READY"
    meta = zmtp.parse_metadata(body[1 + nlen:])
    assert meta["Socket-Type"] == b"PUB"
    if use_31_command:
        s.sendall(zmtp.encode_command(b"SUBSCRIBE", body=subscribe))
    else:
        s.sendall(zmtp.encode_frame(b"\x01" + subscribe))
    return s


def _read_multipart(s):
    parts = []
    while True:
        flags, body = _read_frame(s)
        if flags & zmtp.FLAG_COMMAND:
            continue
        parts.append(body)
        if not flags & zmtp.FLAG_MORE:
            return parts


def _wait_subs(pub, n, timeout=3.0):
    """Wait for n processed SUBSCRIPTIONS (not just connections): ZMQ PUB
    drops messages for peers whose subscribe frame hasn't landed yet."""
    deadline = time.monotonic() + timeout
    while pub._pub.num_subscriptions < n and time.monotonic() < deadline:
        time.sleep(0.02)
    assert pub._pub.num_subscriptions >= n


def test_raw_zmq_peer_receives_events():
    """A raw byte-level ZMTP SUB peer (split greeting, 3.0 subscription)
    receives the multipart [topic, seq, msgpack] the publisher emits."""
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    s = _handshake_as_raw_sub(pub.port)
    _wait_subs(pub, 1)
    pub.block_stored([7, 8])
    parts = _read_multipart(s)
    assert parts[0] == b"kv-events"
    assert struct.unpack(">Q", parts[1])[0] == 0
    evt = msgpack.unpackb(parts[2])
    assert evt["event"] == "BlockStored"
    assert evt["block_hashes"] == [7, 8]
    s.close()
    pub.close()


def test_zmtp31_subscribe_command_accepted():
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    s = _handshake_as_raw_sub(pub.port, use_31_command=True)
    _wait_subs(pub, 1)
    pub.block_removed([3])
    parts = _read_multipart(s)
    assert msgpack.unpackb(parts[2])["event"] == "BlockRemoved"
    s.close()
    pub.close()


def test_topic_prefix_filtering():
    """A peer subscribed to a non-matching topic gets nothing; a
    prefix-matching subscription gets the message (ZMQ PUB semantics)."""
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    s_no = _handshake_as_raw_sub(pub.port, subscribe=b"other")
    s_yes = _handshake_as_raw_sub(pub.port, subscribe=b"kv-")
    _wait_subs(pub, 2)
    assert pub._pub.send_multipart(
        [b"kv-events", b"\x00" * 8, b"x"]) == 1
    parts = _read_multipart(s_yes)
    assert parts[0] == b"kv-events"
    s_no.settimeout(0.3)
    with pytest.raises(socket.timeout):
        s_no.recv(1)
    s_no.close()
    s_yes.close()
    pub.close()


def test_long_frame_encoding():
    """Payloads >255 bytes must use the 8-byte-length long-frame form."""
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    s = _handshake_as_raw_sub(pub.port)
    _wait_subs(pub, 1)
    pub.block_stored(list(range(1000)))
    # topic frame (short), seq frame (short), payload frame (long)
    f1, b1 = _read_frame(s)
    assert not f1 & zmtp.FLAG_LONG and f1 & zmtp.FLAG_MORE
    assert b1 == b"kv-events"
    f2, _ = _read_frame(s)
    assert not f2 & zmtp.FLAG_LONG and f2 & zmtp.FLAG_MORE
    f3, b3 = _read_frame(s)
    assert f3 & zmtp.FLAG_LONG and not f3 & zmtp.FLAG_MORE
    assert len(b3) > 255
    assert msgpack.unpackb(b3)["block_hashes"] == list(range(1000))
    s.close()
    pub.close()


def test_non_zmtp_peer_rejected():
    """Garbage bytes on the socket must not crash the publisher and must
    not register a subscriber."""
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    s = socket.create_connection(("127.0.0.1", pub.port), timeout=2)
    s.sendall(b"GET / HTTP/1.1\r\n\r\n" + b"\x00" * 64)
    time.sleep(0.3)
    assert pub._pub.num_subscribers == 0
    pub.block_stored([1])  # no crash
    s.close()
    pub.close()


def test_own_sub_client_against_own_pub():
    """The in-repo SubSocket (used by the routing side) handshakes with
    the publisher through the same wire protocol."""
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    sub = KVEventSubscriber(port=pub.port)
    _wait_subs(pub, 1)
    pub.block_stored([1])
    pub.all_cleared()
    deadline = time.monotonic() + 3
    while len(sub.events) < 2 and time.monotonic() < deadline:
        time.sleep(0.02)
    assert [e["event"] for e in sub.events] == ["BlockStored",
                                                "AllBlocksCleared"]
    assert sub.seqs == [0, 1]
    sub.close()
    pub.close()
