"""Dedicated replay-server node — 3-tier mode parity.

The reference ships a standalone PER host (`APE_X/ReplayServer.py` +
learner-side `Replay_Server`): actors push experience to it,
it builds batches and pushes them to a second Redis, the learner pulls
pre-built batches and sends priority updates back, with FLAG_* flow
control. In this framework the 2-tier GPU-resident PER is the production
path (the PER lives in the learner's HBM), but the 3-tier topology is still
useful when actors are remote and the learner host's CPUs are scarce — so
here it is, over one TCP port instead of two Redis servers:

  actors  --K_EXP records-->  ReplayServer (CPU TorchPER)
  learner --K_BATCH_REQ-->    ReplayServer --K_BATCH(cols,idx,w)--> learner
  learner --K_UPDATE(idx,p)-> ReplayServer
  learner --K_LEN_REQ-->      ReplayServer (warmup gating = FLAG_BATCH)

Backpressure parity: the learner pulls batches on demand (no queue to
overflow), replacing the reference's FLAG_ENOUGH/throttle loop; trim
(FLAG_REMOVE / remove_to_fit) is inherent to the ring-evicting store.
"""

from __future__ import annotations

import pickle
import socket
import threading
from typing import Optional

import numpy as np
import torch

from ..actors.transport import RecordCodec
from ..actors.tcp_transport import _recv, _send, K_EXP
from .per import TorchPER

K_BATCH_REQ = 10
K_BATCH = 11
K_UPDATE = 12
K_LEN_REQ = 13
K_LEN = 14


class ReplayServer:
    """Standalone PER host process body."""

    def __init__(self, codec: RecordCodec, capacity: int,
                 host: str = "0.0.0.0", port: int = 6380):
        self.codec = codec
        schema = codec.schema
        self.per = TorchPER(capacity, schema)
        self._lock = threading.Lock()
        self.host, self.port = host, port
        self._srv: Optional[socket.socket] = None
        self._stop = threading.Event()

    def start(self):
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((self.host, self.port))
        self.port = self._srv.getsockname()[1]
        self._srv.listen(256)
        threading.Thread(target=self._accept_loop, daemon=True).start()
        return self

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                conn, _ = self._srv.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            threading.Thread(target=self._serve, args=(conn,), daemon=True).start()

    def _serve(self, conn: socket.socket):
        with conn:
            while not self._stop.is_set():
                msg = _recv(conn)
                if msg is None:
                    return
                kind, payload = msg
                if kind == K_EXP:
                    raw = np.frombuffer(payload, dtype=np.uint8).reshape(
                        -1, self.codec.record_size
                    )
                    rec = raw.view(self.codec.np_dtype).reshape(-1)
                    cols, prio = self.codec.unpack(rec)
                    tcols = {k: torch.from_numpy(v.copy()) for k, v in cols.items()}
                    with self._lock:
                        self.per.push(tcols, torch.from_numpy(prio.copy()))
                elif kind == K_BATCH_REQ:
                    k, beta = pickle.loads(payload)
                    with self._lock:
                        data, idx, w = self.per.sample(k, beta)
                    blob = pickle.dumps(
                        ({n: t.numpy() for n, t in data.items()},
                         idx.numpy(), w.numpy()),
                        protocol=pickle.HIGHEST_PROTOCOL,
                    )
                    _send(conn, K_BATCH, blob)
                elif kind == K_UPDATE:
                    idx_np, prio_np = pickle.loads(payload)
                    with self._lock:
                        self.per.update(torch.from_numpy(idx_np),
                                        torch.from_numpy(prio_np))
                elif kind == K_LEN_REQ:
                    with self._lock:
                        n = len(self.per)
                    _send(conn, K_LEN, pickle.dumps(n))

    def stop(self):
        self._stop.set()
        if self._srv is not None:
            self._srv.close()


class RemoteReplay:
    """Learner-side client — the reference's ``Replay_Server`` role: sample
    pre-built batches from the replay node, send deferred priority updates."""

    def __init__(self, host: str, port: int):
        self.sock = socket.create_connection((host, port), timeout=120)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self.capacity = 1 << 60  # owned by the server
        self.device = torch.device("cpu")

    def sample(self, k: int, beta: float, with_data: bool = True):
        _send(self.sock, K_BATCH_REQ, pickle.dumps((k, beta)))
        kind, payload = _recv(self.sock)
        assert kind == K_BATCH
        cols_np, idx_np, w_np = pickle.loads(payload)
        cols = {n: torch.from_numpy(v) for n, v in cols_np.items()}
        return cols, torch.from_numpy(idx_np), torch.from_numpy(w_np)

    def update(self, idx: torch.Tensor, prios: torch.Tensor):
        _send(self.sock, K_UPDATE,
              pickle.dumps((idx.cpu().numpy(), prios.float().cpu().numpy())))

    def __len__(self) -> int:
        _send(self.sock, K_LEN_REQ, b"")
        kind, payload = _recv(self.sock)
        assert kind == K_LEN
        return pickle.loads(payload)

    @property
    def total_priority(self) -> float:
        return float(len(self))  # informational only on the client side

    def close(self):
        self.sock.close()
