"""TCP transport — multi-host actor fleets (reference parity: the Redis
server reached over the network, README.md:62-77 runs actors on separate
GCP VMs / terminals).

Same endpoint interface as the shared-memory transport (transport.py):
actors push fixed-size records and poll versioned weights; the learner
drains records and publishes weights. Wire protocol: length-prefixed
messages; experience records travel as raw structured-array bytes (no
per-item pickle), control messages as pickled dicts.

    learner:  srv = TcpTransportServer(codec, port=6379); srv.start()
              endpoint = srv.endpoint()          # drain()/publish()/...
    actor:    ep = TcpActorEndpoint("learner-host", 6379, codec)
              player = ApexPlayer(cfg, idx, transport=ep)
"""

from __future__ import annotations

import pickle
import socket
import struct
import threading
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

from .transport import RecordCodec

_HDR = struct.Struct("<BI")  # msg kind, payload length
K_EXP = 1
K_FETCH = 2
K_WEIGHTS = 3
K_REWARD = 4
K_NONE = 5


def _send(sock: socket.socket, kind: int, payload: bytes) -> None:
    sock.sendall(_HDR.pack(kind, len(payload)) + payload)


def _recv_exact(sock: socket.socket, n: int) -> Optional[bytes]:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            return None
        buf.extend(chunk)
    return bytes(buf)


def _recv(sock: socket.socket) -> Optional[Tuple[int, bytes]]:
    hdr = _recv_exact(sock, _HDR.size)
    if hdr is None:
        return None
    kind, length = _HDR.unpack(hdr)
    payload = _recv_exact(sock, length) if length else b""
    if payload is None:
        return None
    return kind, payload


class TcpTransportServer:
    """Learner-side server: accepts actor connections, buffers experience,
    serves weight snapshots."""

    def __init__(self, codec: RecordCodec, host: str = "0.0.0.0",
                 port: int = 6379):
        self.codec = codec
        self.host = host
        self.port = port
        self._lock = threading.Lock()
        self._records: List[np.ndarray] = []
        self._rewards: List[float] = []
        self._weights: Optional[bytes] = None
        self._version = 0
        self._srv: Optional[socket.socket] = None
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()

    def start(self):
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((self.host, self.port))
        self.port = self._srv.getsockname()[1]
        self._srv.listen(1024)
        t = threading.Thread(target=self._accept_loop, daemon=True)
        t.start()
        self._threads.append(t)
        return self

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                conn, _ = self._srv.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            t = threading.Thread(target=self._serve, args=(conn,), daemon=True)
            t.start()
            self._threads.append(t)

    def _serve(self, conn: socket.socket):
        with conn:
            while not self._stop.is_set():
                msg = _recv(conn)
                if msg is None:
                    return
                kind, payload = msg
                if kind == K_EXP:
                    rec = np.frombuffer(payload, dtype=np.uint8).reshape(
                        -1, self.codec.record_size
                    ).copy()
                    with self._lock:
                        self._records.append(rec)
                elif kind == K_FETCH:
                    have = pickle.loads(payload)
                    with self._lock:
                        blob, ver = self._weights, self._version
                    if blob is None or ver == have:
                        _send(conn, K_NONE, b"")
                    else:
                        # u64 version prefix lets the actor skip unchanged
                        # snapshots on later polls (multi-MB state_dicts)
                        _send(conn, K_WEIGHTS, struct.pack("<Q", ver) + blob)
                elif kind == K_REWARD:
                    with self._lock:
                        self._rewards.append(pickle.loads(payload))

    # -- learner endpoint --------------------------------------------------
    def endpoint(self) -> "TcpLearnerEndpoint":
        return TcpLearnerEndpoint(self)

    def stop(self):
        self._stop.set()
        if self._srv is not None:
            self._srv.close()


class TcpLearnerEndpoint:
    def __init__(self, server: TcpTransportServer):
        self.server = server

    def drain(self):
        with self.server._lock:
            chunks, self.server._records = self.server._records, []
        if not chunks:
            return None
        raw = np.concatenate(chunks, axis=0)
        rec = raw.view(self.server.codec.np_dtype).reshape(-1)
        return self.server.codec.unpack(rec)

    def publish(self, obj: Any):
        blob = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
        with self.server._lock:
            self.server._weights = blob
            self.server._version += 1

    def drain_rewards(self) -> List[float]:
        with self.server._lock:
            out, self.server._rewards = self.server._rewards, []
        return out


class TcpActorEndpoint:
    """Actor-side endpoint; interface-compatible with ActorEndpoint."""

    def __init__(self, host: str, port: int, codec: RecordCodec,
                 idx: int = 0):
        self.codec = codec
        self.idx = idx
        self.sock = socket.create_connection((host, port), timeout=60)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._last_version_payload = None
        self._have = -1  # server-side version of the cached snapshot

    def push(self, columns: Dict[str, np.ndarray],
             priorities: Optional[np.ndarray] = None):
        rec = self.codec.pack(columns, priorities)
        _send(self.sock, K_EXP, rec.tobytes())

    def fetch(self):
        _send(self.sock, K_FETCH, pickle.dumps(self._have))
        msg = _recv(self.sock)
        if msg is None:
            return None
        kind, payload = msg
        if kind != K_WEIGHTS:
            # K_NONE: nothing published yet, or our cached version is
            # current — no multi-MB re-download
            return self._last_version_payload
        (self._have,) = struct.unpack_from("<Q", payload)
        self._last_version_payload = pickle.loads(payload[8:])
        return self._last_version_payload

    def push_reward(self, _idx, reward: float, eps: float = 0.0):
        _send(self.sock, K_REWARD, pickle.dumps(float(reward)))

    def close(self):
        self.sock.close()


class SplitActorEndpoint:
    """3-tier mode actor endpoint: experience goes to the replay-server
    node, weights/rewards to the learner's transport server (the reference
    splits these across its two Redis servers the same way)."""

    def __init__(self, exp_endpoint: TcpActorEndpoint,
                 weight_endpoint: TcpActorEndpoint):
        self.exp = exp_endpoint
        self.weights = weight_endpoint
        self.idx = weight_endpoint.idx

    def push(self, columns, priorities=None):
        self.exp.push(columns, priorities)

    def fetch(self):
        return self.weights.fetch()

    def push_reward(self, idx, reward, eps=0.0):
        self.weights.push_reward(idx, reward, eps)

    def close(self):
        self.exp.close()
        self.weights.close()
