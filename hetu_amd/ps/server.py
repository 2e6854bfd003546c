"""Parameter-server transport: standalone PS server processes + client.

Reference parity: hetu/v1/ps-lite — dedicated server nodes with network
"vans" (zmq_van.h / ibverbs_van.h), `KVWorker::Push/Pull` routing key
ranges to servers, and server-side sparse optimizers
(PSFhandle_embedding.cc).  MI355X-native shape: the van is a
length-prefixed TCP socket protocol (torch.save framing); keys shard
across servers by `id % num_servers`; each server applies its sparse
optimizer (sgd / adagrad) on push, so workers never hold optimizer state
(the HET/recommendation workload path; dense training uses the collective
engine instead).
"""
from __future__ import annotations

import io
import socket
import socketserver
import struct
import threading
from typing import Dict, List, Optional, Tuple

import torch


def _send_msg(sock: socket.socket, obj) -> None:
    buf = io.BytesIO()
    torch.save(obj, buf)
    data = buf.getvalue()
    sock.sendall(struct.pack("!Q", len(data)) + data)


def _recv_msg(sock: socket.socket):
    hdr = b""
    while len(hdr) < 8:
        c = sock.recv(8 - len(hdr))
        if not c:
            raise ConnectionError("peer closed")
        hdr += c
    (n,) = struct.unpack("!Q", hdr)
    data = b""
    while len(data) < n:
        c = sock.recv(min(1 << 20, n - len(data)))
        if not c:
            raise ConnectionError("peer closed")
        data += c
    return torch.load(io.BytesIO(data), weights_only=False)


class _Table:
    def __init__(self, num: int, dim: int, optimizer: str = "sgd",
                 lr: float = 0.1, eps: float = 1e-10, init_std: float = 0.01,
                 seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.rows = torch.randn(num, dim, generator=g) * init_std
        self.optimizer = optimizer
        self.lr = lr
        self.eps = eps
        if optimizer == "adagrad":
            self.acc = torch.zeros(num, dim)
        self.lock = threading.Lock()

    def pull(self, ids: torch.Tensor) -> torch.Tensor:
        with self.lock:
            return self.rows[ids].clone()

    def push(self, ids: torch.Tensor, grads: torch.Tensor) -> None:
        with self.lock:
            if self.optimizer == "adagrad":
                self.acc.index_add_(0, ids, grads * grads)
                denom = self.acc[ids].sqrt() + self.eps
                self.rows.index_add_(0, ids, -self.lr * grads / denom)
            else:
                self.rows.index_add_(0, ids, -self.lr * grads)


class PSServer:
    """One server shard: serves PULL/PUSH/REGISTER/SAVE over TCP."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.tables: Dict[str, _Table] = {}
        outer = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                while True:
                    try:
                        msg = _recv_msg(self.request)
                    except (ConnectionError, OSError):
                        return
                    op = msg["op"]
                    if op == "register":
                        outer.tables.setdefault(
                            msg["name"],
                            _Table(msg["num"], msg["dim"],
                                   msg.get("optimizer", "sgd"),
                                   msg.get("lr", 0.1),
                                   msg.get("eps", 1e-10),
                                   msg.get("init_std", 0.01),
                                   msg.get("seed", 0)))
                        _send_msg(self.request, {"ok": True})
                    elif op == "pull":
                        rows = outer.tables[msg["name"]].pull(msg["ids"])
                        _send_msg(self.request, {"rows": rows})
                    elif op == "push":
                        outer.tables[msg["name"]].push(msg["ids"],
                                                       msg["grads"])
                        _send_msg(self.request, {"ok": True})
                    elif op == "state":
                        t = outer.tables[msg["name"]]
                        _send_msg(self.request, {"rows": t.rows.clone()})
                    elif op == "stop":
                        _send_msg(self.request, {"ok": True})
                        threading.Thread(target=outer.stop,
                                         daemon=True).start()
                        return

        class Srv(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self._srv = Srv((host, port), Handler)
        self.port = self._srv.server_address[1]
        self._thread = threading.Thread(target=self._srv.serve_forever,
                                        daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._srv.shutdown()
        self._srv.server_close()


class PSClient:
    """KVWorker (reference kv_app.h): pull/push with keys sharded across
    servers by `id % num_servers` (the van routing)."""

    def __init__(self, servers: List[Tuple[str, int]]):
        self.socks: List[socket.socket] = []
        for host, port in servers:
            s = socket.create_connection((host, port), timeout=60)
            self.socks.append(s)
        self.n = len(self.socks)

    def _rpc(self, si: int, msg):
        _send_msg(self.socks[si], msg)
        return _recv_msg(self.socks[si])

    def register(self, name: str, num: int, dim: int, **kw):
        for si in range(self.n):
            # each shard owns ceil(num/n) rows (ids with id % n == si)
            import math
            n_local = int(math.ceil((num - si) / self.n)) if num > si else 0
            self._rpc(si, {"op": "register", "name": name, "num": n_local,
                           "dim": dim, "seed": kw.pop("seed", 0) + si,
                           **kw})

    def pull(self, name: str, ids: torch.Tensor) -> torch.Tensor:
        out: Optional[torch.Tensor] = None
        for si in range(self.n):
            mask = (ids % self.n) == si
            if not bool(mask.any()):
                continue
            local = ids[mask] // self.n
            rows = self._rpc(si, {"op": "pull", "name": name,
                                  "ids": local})["rows"]
            if out is None:
                out = torch.empty(ids.shape[0], rows.shape[1],
                                  dtype=rows.dtype)
            out[mask] = rows
        return out

    def push(self, name: str, ids: torch.Tensor, grads: torch.Tensor):
        for si in range(self.n):
            mask = (ids % self.n) == si
            if not bool(mask.any()):
                continue
            self._rpc(si, {"op": "push", "name": name,
                           "ids": ids[mask] // self.n,
                           "grads": grads[mask]})

    def stop_servers(self):
        for si in range(self.n):
            try:
                self._rpc(si, {"op": "stop"})
            except Exception:  # noqa: BLE001
                pass

    def close(self):
        for s in self.socks:
            try:
                s.close()
            except OSError:
                pass
