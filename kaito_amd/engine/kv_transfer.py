"""Prefill/decode disaggregated KV transfer.

The reference wires NIXL (UCX/GPU-direct) between prefill and decode pods
(preset_inferences.go:1082-1105; inference_api.py:506-512). The MI355X
equivalent: same-node transfers ride torch.distributed send/recv over
RCCL (xGMI p2p); cross-node falls back to the TCP path. The side channel
carries (request_id, token_ids, block-layout metadata); payload is the
per-layer KV of the prefilled tokens.

Flow:
  prefill engine: run prefill → extract_kv(seq) → connector.send(...)
  decode engine:  connector.recv() → allocate blocks → inject_kv(...)
                  → sequence continues as decode-only (sched_len = prompt).
"""
from __future__ import annotations

import pickle
import socket
import struct
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch


@dataclass
class KVPayload:
    request_id: str
    token_ids: List[int]
    # per-layer (k, v) tensors [kvh, n_tokens, head_dim]
    layers: List[Tuple[torch.Tensor, torch.Tensor]]
    first_token: Optional[int] = None   # token sampled by the prefill side


def extract_kv(kv_caches, block_table: List[int], n_tokens: int,
               block_size: int) -> List[Tuple[torch.Tensor, torch.Tensor]]:
    """Gather a sequence's KV for transfer: [kvh, n_tokens, d] per layer."""
    nb = (n_tokens + block_size - 1) // block_size
    dev = kv_caches[0][0].device
    blocks = torch.tensor(block_table[:nb], dtype=torch.long, device=dev)
    out = []
    for kc, vc in kv_caches:
        kvh, d = kc.shape[1], kc.shape[3]
        k = kc[blocks].permute(1, 0, 2, 3).reshape(kvh, nb * block_size, d)
        v = vc[blocks].permute(1, 0, 2, 3).reshape(kvh, nb * block_size, d)
        out.append((k[:, :n_tokens].contiguous(),
                    v[:, :n_tokens].contiguous()))
    return out


def inject_kv(kv_caches, block_table: List[int],
              layers: List[Tuple[torch.Tensor, torch.Tensor]],
              block_size: int) -> None:
    """Scatter transferred KV into the receiving pool's blocks."""
    n = layers[0][0].shape[1]
    nb = (n + block_size - 1) // block_size
    dev = kv_caches[0][0].device
    blocks = torch.tensor(block_table[:nb], dtype=torch.long, device=dev)
    pad = nb * block_size
    for (kc, vc), (k, v) in zip(kv_caches, layers):
        kvh, d = kc.shape[1], kc.shape[3]
        kp = torch.zeros(kvh, pad, d, dtype=kc.dtype, device=dev)
        vp = torch.zeros_like(kp)
        kp[:, :n] = k.to(dev)
        vp[:, :n] = v.to(dev)
        kc[blocks] = kp.reshape(kvh, nb, block_size, d).permute(1, 0, 2, 3)
        vc[blocks] = vp.reshape(kvh, nb, block_size, d).permute(1, 0, 2, 3)


# --------------------------------------------------------------- connectors
class P2PGroupConnector:
    """torch.distributed send/recv between prefill and decode ranks
    (RCCL over xGMI when both ranks share the node; gloo in CPU tests)."""

    def __init__(self, peer_rank: int, group=None):
        import torch.distributed as dist
        self.dist = dist
        self.peer = peer_rank
        self.group = group

    def send(self, payload: KVPayload) -> None:
        meta = {
            "request_id": payload.request_id,
            "token_ids": payload.token_ids,
            "first_token": payload.first_token,
            "layers": len(payload.layers),
            "shape": list(payload.layers[0][0].shape),
            "dtype": str(payload.layers[0][0].dtype),
        }
        blob = pickle.dumps(meta)
        hdr = torch.tensor([len(blob)], dtype=torch.long)
        self.dist.send(hdr, self.peer, group=self.group)
        self.dist.send(torch.frombuffer(bytearray(blob), dtype=torch.uint8),
                       self.peer, group=self.group)
        for k, v in payload.layers:
            self.dist.send(k.cpu() if not k.is_cuda else k, self.peer,
                           group=self.group)
            self.dist.send(v.cpu() if not v.is_cuda else v, self.peer,
                           group=self.group)

    def recv(self) -> KVPayload:
        hdr = torch.zeros(1, dtype=torch.long)
        self.dist.recv(hdr, self.peer, group=self.group)
        blob = torch.zeros(int(hdr.item()), dtype=torch.uint8)
        self.dist.recv(blob, self.peer, group=self.group)
        meta = pickle.loads(bytes(blob.numpy().tobytes()))
        shape = meta["shape"]
        dt = getattr(torch, meta["dtype"].split(".")[-1])
        layers = []
        for _ in range(meta["layers"]):
            k = torch.zeros(*shape, dtype=dt)
            v = torch.zeros(*shape, dtype=dt)
            self.dist.recv(k, self.peer, group=self.group)
            self.dist.recv(v, self.peer, group=self.group)
            layers.append((k, v))
        return KVPayload(meta["request_id"], meta["token_ids"], layers,
                         meta["first_token"])


class TCPConnector:
    """Cross-node fallback side channel (the NIXL side-channel analog,
    env KAITO_KV_SIDE_CHANNEL_HOST/PORT in the reference)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 server: bool = False):
        self.server = server
        if server:
            self._srv = socket.socket()
            self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._srv.bind((host, port))
            self._srv.listen(4)
            self.port = self._srv.getsockname()[1]
            self._conn = None
        else:
            self._conn = socket.create_connection((host, port), timeout=30)
            self.port = port

    def _ensure(self):
        if self.server and self._conn is None:
            self._conn, _ = self._srv.accept()
        return self._conn

    def send(self, payload: KVPayload) -> None:
        conn = self._ensure()
        data = pickle.dumps({
            "request_id": payload.request_id,
            "token_ids": payload.token_ids,
            "first_token": payload.first_token,
            "layers": [(k.cpu(), v.cpu()) for k, v in payload.layers],
        })
        conn.sendall(struct.pack(">Q", len(data)) + data)

    def recv(self) -> KVPayload:
        conn = self._ensure()

        def read(n):
            buf = b""
            while len(buf) < n:
                chunk = conn.recv(n - len(buf))
                if not chunk:
                    raise ConnectionError("kv side channel closed")
                buf += chunk
            return buf

        n = struct.unpack(">Q", read(8))[0]
        meta = pickle.loads(read(n))
        return KVPayload(meta["request_id"], meta["token_ids"],
                         meta["layers"], meta["first_token"])
