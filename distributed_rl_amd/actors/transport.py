"""Actor <-> learner transport: shared-memory rings + versioned weight bus.

MI355X-native replacement for the reference's Redis wire protocol
(SURVEY.md §2.7): the ``experience``/``trajectory`` Redis lists become
per-actor SPSC rings in POSIX shared memory (fixed-size records, no pickle),
and the ``state_dict``/``count`` keys become a seqlock'd weight snapshot
segment. The learner drains every ring into pinned staging tensors and
hipMemcpyAsync's them into the GPU-resident replay on a side stream.

Concurrency model: each ring has exactly one producer (an actor process) and
one consumer (the learner's ingest thread). Counters are 8-byte aligned
little-endian uint64; the producer writes payload THEN head, the consumer
reads head THEN payload (x86-TSO makes this ordering sufficient without
fences). The weight bus is a classic seqlock: odd version = write in
progress; readers retry on version change.
"""

from __future__ import annotations

import json
import os
import pickle
import struct
import time
import uuid
from multiprocessing import shared_memory
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

# ---------------------------------------------------------------------------
# Record codec: columns dict <-> fixed-size byte records via numpy structured
# dtypes (schema mirrors replay.Schema shapes)
# ---------------------------------------------------------------------------

_TORCH2NP = {
    torch.uint8: np.uint8,
    torch.int32: np.int32,
    torch.int64: np.int64,
    torch.float32: np.float32,
    torch.float16: np.float16,
}


class RecordCodec:
    def __init__(self, schema: Dict[str, Tuple[Tuple[int, ...], torch.dtype]],
                 with_priority: bool = True):
        fields = []
        for name, (shape, dtype) in schema.items():
            fields.append((name, _TORCH2NP[dtype], shape if shape else ()))
        if with_priority:
            fields.append(("priority", np.float32, ()))
        self.np_dtype = np.dtype(fields)
        self.schema = schema
        self.with_priority = with_priority

    @property
    def record_size(self) -> int:
        return self.np_dtype.itemsize

    def pack(self, columns: Dict[str, np.ndarray],
             priorities: Optional[np.ndarray] = None) -> np.ndarray:
        n = len(next(iter(columns.values())))
        rec = np.empty(n, dtype=self.np_dtype)
        for name in self.schema:
            rec[name] = columns[name]
        if self.with_priority:
            rec["priority"] = priorities
        return rec

    def unpack(self, rec: np.ndarray) -> Tuple[Dict[str, np.ndarray], Optional[np.ndarray]]:
        cols = {name: np.ascontiguousarray(rec[name]) for name in self.schema}
        prio = np.ascontiguousarray(rec["priority"]) if self.with_priority else None
        return cols, prio


# ---------------------------------------------------------------------------
# SPSC ring over a shared-memory segment
# ---------------------------------------------------------------------------

_HDR = 64  # [head u64][tail u64][drops u64][pad]


class SpscRing:
    def __init__(self, name: str, record_size: int, slots: int,
                 create: bool = False):
        self.record_size = record_size
        self.slots = slots
        size = _HDR + record_size * slots
        self.shm = shared_memory.SharedMemory(name=name, create=create, size=size)
        self.name = name
        self.buf = self.shm.buf
        if create:
            self.buf[:_HDR] = b"\x00" * _HDR
        self._data = np.frombuffer(self.buf, dtype=np.uint8, offset=_HDR,
                                   count=record_size * slots).reshape(slots, record_size)

    # counters -----------------------------------------------------------
    def _get_u64(self, off: int) -> int:
        return struct.unpack_from("<Q", self.buf, off)[0]

    def _set_u64(self, off: int, v: int) -> None:
        struct.pack_into("<Q", self.buf, off, v)

    @property
    def head(self) -> int:
        return self._get_u64(0)

    @property
    def tail(self) -> int:
        return self._get_u64(8)

    @property
    def drops(self) -> int:
        return self._get_u64(16)

    def __len__(self) -> int:
        return self.head - self.tail

    # producer -----------------------------------------------------------
    def push_records(self, rows: np.ndarray) -> int:
        """Write up to len(rows) records; returns the number written (the rest
        are dropped, counted in `drops` — backpressure telemetry)."""
        raw = rows.view(np.uint8).reshape(len(rows), self.record_size)
        head, tail = self.head, self.tail
        free = self.slots - (head - tail)
        n = min(len(rows), free)
        if n > 0:
            start = head % self.slots
            first = min(n, self.slots - start)
            self._data[start : start + first] = raw[:first]
            if n > first:
                self._data[: n - first] = raw[first:n]
        # payload before head (publish)
        self._set_u64(0, head + n)
        if n < len(rows):
            self._set_u64(16, self.drops + (len(rows) - n))
        return n

    # consumer -----------------------------------------------------------
    def peek_records(self, max_n: int = 1 << 30):
        """Zero-copy consume: returns (views, n) where views are 1-2 uint8
        slices of the ring storage covering n records. SPSC contract: the
        consumer owns [tail, tail+n) until it calls advance(n) — copy out of
        the views FIRST, then advance."""
        head, tail = self.head, self.tail
        n = min(head - tail, max_n)
        if n <= 0:
            return None, 0
        start = tail % self.slots
        first = min(n, self.slots - start)
        views = [self._data[start : start + first]]
        if n > first:
            views.append(self._data[: n - first])
        return views, n

    def advance(self, n: int) -> None:
        self._set_u64(8, self.tail + n)

    def pop_records(self, max_n: int = 1 << 30) -> Optional[np.ndarray]:
        head, tail = self.head, self.tail
        n = min(head - tail, max_n)
        if n <= 0:
            return None
        out = np.empty((n, self.record_size), dtype=np.uint8)
        start = tail % self.slots
        first = min(n, self.slots - start)
        out[:first] = self._data[start : start + first]
        if n > first:
            out[first:] = self._data[: n - first]
        self._set_u64(8, tail + n)
        return out

    def close(self, unlink: bool = False):
        self._data = None
        self.buf = None
        self.shm.close()
        if unlink:
            try:
                self.shm.unlink()
            except FileNotFoundError:
                pass


# ---------------------------------------------------------------------------
# Seqlock weight bus
# ---------------------------------------------------------------------------


class WeightBus:
    """Versioned weight snapshot in shared memory.

    Preserves the reference's semantics for the ``state_dict`` /
    ``target_state_dict`` / ``count`` Redis keys (APE_X/Learner.py:152-154,
    207-216; APE_X/Player.py:113-133): the learner publishes
    {count, state_dict, target_state_dict?} and actors poll; the target
    version is derived count//TARGET_FREQUENCY on the actor side.
    """

    _SEQ_OFF = 0
    _LEN_OFF = 8
    _PAYLOAD = 16

    def __init__(self, name: str, capacity: int, create: bool = False):
        self.shm = shared_memory.SharedMemory(
            name=name, create=create, size=self._PAYLOAD + capacity
        )
        self.name = name
        self.capacity = capacity
        if create:
            self.shm.buf[: self._PAYLOAD] = b"\x00" * self._PAYLOAD

    def _get_u64(self, off: int) -> int:
        return struct.unpack_from("<Q", self.shm.buf, off)[0]

    def _set_u64(self, off: int, v: int) -> None:
        struct.pack_into("<Q", self.shm.buf, off, v)

    def publish(self, obj: Any) -> None:
        blob = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
        if len(blob) > self.capacity:
            raise ValueError(
                f"weight payload {len(blob)} > bus capacity {self.capacity}"
            )
        seq = self._get_u64(self._SEQ_OFF)
        self._set_u64(self._SEQ_OFF, seq + 1)  # odd: writing
        self.shm.buf[self._PAYLOAD : self._PAYLOAD + len(blob)] = blob
        self._set_u64(self._LEN_OFF, len(blob))
        self._set_u64(self._SEQ_OFF, seq + 2)  # even: stable

    def fetch(self, retries: int = 64) -> Optional[Any]:
        for _ in range(retries):
            s1 = self._get_u64(self._SEQ_OFF)
            if s1 == 0:
                return None  # nothing published yet
            if s1 & 1:
                time.sleep(0.001)
                continue
            n = self._get_u64(self._LEN_OFF)
            blob = bytes(self.shm.buf[self._PAYLOAD : self._PAYLOAD + n])
            s2 = self._get_u64(self._SEQ_OFF)
            if s1 == s2:
                return pickle.loads(blob)
        return None

    @property
    def version(self) -> int:
        return self._get_u64(self._SEQ_OFF) // 2

    def close(self, unlink: bool = False):
        self.shm.close()
        if unlink:
            try:
                self.shm.unlink()
            except FileNotFoundError:
                pass


# ---------------------------------------------------------------------------
# Session rendezvous (replaces Redis server addresses in cfg)
# ---------------------------------------------------------------------------


class TransportSession:
    """A named transport session = 1 weight bus + N experience rings.

    The learner creates the session (writing a manifest JSON under
    ``transport_dir``); actors attach by session name. Same role split as the
    reference's learner-flushes-Redis startup (APE_X/Learner.py:41-43).
    """

    def __init__(self, transport_dir: str, codec: RecordCodec,
                 num_rings: int, ring_slots: int = 256,
                 weight_capacity: int = 64 << 20,
                 session: Optional[str] = None, create: bool = False):
        self.dir = transport_dir
        self.codec = codec
        os.makedirs(transport_dir, exist_ok=True)
        self.manifest_path = os.path.join(transport_dir, "session.json")
        if create:
            self.session = session or uuid.uuid4().hex[:8]
            manifest = {
                "session": self.session,
                "num_rings": num_rings,
                "ring_slots": ring_slots,
                "record_size": codec.record_size,
                "weight_capacity": weight_capacity,
            }
            with open(self.manifest_path, "w") as f:
                json.dump(manifest, f)
        else:
            with open(self.manifest_path) as f:
                manifest = json.load(f)
            self.session = manifest["session"]
            num_rings = manifest["num_rings"]
            ring_slots = manifest["ring_slots"]
            weight_capacity = manifest["weight_capacity"]
            assert manifest["record_size"] == codec.record_size, (
                "schema mismatch between learner and actor"
            )
        self.num_rings = num_rings
        self.ring_slots = ring_slots
        self.weight_capacity = weight_capacity
        self._create = create
        self.weight_bus = WeightBus(
            f"drl_{self.session}_w", weight_capacity, create=create
        )
        self.rings: Dict[int, SpscRing] = {}
        self.reward_rings: Dict[int, SpscRing] = {}
        self.reward_codec = RecordCodec(
            {"reward": ((), torch.float32), "eps": ((), torch.float32)},
            with_priority=False,
        )
        if create:
            for i in range(num_rings):
                self.rings[i] = self._make_ring(i, create=True)
                self.reward_rings[i] = self._make_reward_ring(i, create=True)

    def _make_ring(self, i: int, create: bool) -> SpscRing:
        return SpscRing(
            f"drl_{self.session}_r{i}", self.codec.record_size, self.ring_slots,
            create=create,
        )

    def _make_reward_ring(self, i: int, create: bool) -> SpscRing:
        return SpscRing(
            f"drl_{self.session}_t{i}", self.reward_codec.record_size, 1024,
            create=create,
        )

    def ring(self, i: int) -> SpscRing:
        if i not in self.rings:
            self.rings[i] = self._make_ring(i, create=False)
        return self.rings[i]

    def reward_ring(self, i: int) -> SpscRing:
        if i not in self.reward_rings:
            self.reward_rings[i] = self._make_reward_ring(i, create=False)
        return self.reward_rings[i]

    def push_reward(self, actor_idx: int, reward: float, eps: float = 0.0):
        rec = self.reward_codec.pack(
            {"reward": np.array([reward], np.float32),
             "eps": np.array([eps], np.float32)}
        )
        self.reward_ring(actor_idx).push_records(rec)

    def drain_rewards_with_eps(self) -> List[tuple]:
        """[(episode_reward, actor_eps)] — eps rides each record so the
        learner can report the near-greedy mean (the reference's Reward
        scalar gates on eps < 0.05, APE_X/Player.py:272-277)."""
        out: List[tuple] = []
        for i in range(self.num_rings):
            r = self.reward_ring(i).pop_records()
            if r is not None:
                rec = r.view(self.reward_codec.np_dtype).reshape(-1)
                out.extend((float(a), float(b))
                           for a, b in zip(rec["reward"], rec["eps"]))
        return out

    def drain_rewards(self) -> List[float]:
        out: List[float] = []
        for i in range(self.num_rings):
            r = self.reward_ring(i).pop_records()
            if r is not None:
                rec = r.view(self.reward_codec.np_dtype).reshape(-1)
                out.extend(float(x) for x in rec["reward"])
        return out

    # -- learner side -----------------------------------------------------
    def drain(self, max_per_ring: int = 1 << 30, ring_ids=None
              ) -> Optional[Tuple[Dict[str, np.ndarray], Optional[np.ndarray]]]:
        chunks = []
        for i in (range(self.num_rings) if ring_ids is None else ring_ids):
            r = self.ring(i).pop_records(max_per_ring)
            if r is not None:
                chunks.append(r)
        if not chunks:
            return None
        raw = np.concatenate(chunks, axis=0)
        rec = raw.view(self.codec.np_dtype).reshape(-1)
        return self.codec.unpack(rec)

    def close(self):
        for r in self.rings.values():
            r.close(unlink=self._create)
        for r in self.reward_rings.values():
            r.close(unlink=self._create)
        self.weight_bus.close(unlink=self._create)
        if self._create:
            try:
                os.remove(self.manifest_path)
            except FileNotFoundError:
                pass


class ActorEndpoint:
    """Actor-side view of a TransportSession: push to MY ring, fetch weights.

    Mirrors the interface of InprocPipe so Players are transport-agnostic."""

    def __init__(self, session: TransportSession, idx: int):
        self.session = session
        self.idx = idx

    def push(self, columns: Dict[str, np.ndarray], priorities: np.ndarray):
        rec = self.session.codec.pack(columns, priorities)
        self.session.ring(self.idx).push_records(rec)

    def fetch(self):
        return self.session.weight_bus.fetch()

    def push_reward(self, _idx, reward: float, eps: float = 0.0):
        self.session.push_reward(self.idx, reward, eps)


class LearnerEndpoint:
    """Learner-side view: drain (a partition of) the rings, publish weights.

    For learner data-parallelism, rank r of world w drains rings
    i % w == r — each GPU replica owns a disjoint actor subset and its own
    HBM replay shard (SURVEY §2.6 build implication c)."""

    def __init__(self, session: TransportSession, rank: int = 0,
                 world_size: int = 1):
        self.session = session
        self.ring_ids = [
            i for i in range(session.num_rings) if i % world_size == rank
        ]

    def drain(self):
        return self.session.drain(ring_ids=self.ring_ids)

    @property
    def record_dtype(self) -> np.dtype:
        return self.session.codec.np_dtype

    def drain_views(self, max_per_ring: int = 1 << 30):
        """Zero-copy drain: yields (views, n, advance) per non-empty ring.
        The caller must copy out of the shm views and THEN call advance(n)
        (the single host copy goes shm -> pinned staging directly; see
        ApexLearner.ingest). max_per_ring bounds a sweep so a large backlog
        (e.g. after the graph-capture pause) cannot turn one drain call
        into a multi-second monolith."""
        for i in self.ring_ids:
            ring = self.session.ring(i)
            views, n = ring.peek_records(max_per_ring)
            if n:
                yield views, n, ring.advance

    def publish(self, obj):
        self.session.weight_bus.publish(obj)

    def drain_rewards(self):
        return self.session.drain_rewards()

    def drain_rewards_with_eps(self):
        return self.session.drain_rewards_with_eps()

    def total_drops(self) -> int:
        """Backpressure telemetry: rows the actors pushed that the rings
        had no space for (SPSC drop counters)."""
        return sum(self.session.ring(i).drops for i in self.ring_ids)

    def total_pushed(self) -> int:
        """Rows the fleet has ever written (ring head counters) —
        distinguishes idle actors from a stalled drain."""
        return sum(self.session.ring(i).head for i in self.ring_ids)


# ---------------------------------------------------------------------------
# In-process pipe (tests / single-process integration)
# ---------------------------------------------------------------------------


class InprocPipe:
    def __init__(self):
        import threading

        self._lock = threading.Lock()
        self._items: List[Tuple[Dict[str, np.ndarray], Optional[np.ndarray]]] = []
        self.weights: Optional[Any] = None
        self.version = 0

    def push(self, columns, priorities=None):
        with self._lock:
            self._items.append((columns, priorities))

    def drain(self):
        with self._lock:
            items, self._items = self._items, []
        if not items:
            return None
        cols = {
            k: np.concatenate([np.asarray(c[0][k]) for c in items])
            for k in items[0][0]
        }
        if items[0][1] is not None:
            prio = np.concatenate([np.asarray(c[1]) for c in items])
        else:
            prio = None
        return cols, prio

    def publish(self, obj):
        self.weights = obj
        self.version += 1

    def fetch(self):
        return self.weights

    def push_reward(self, actor_idx, reward, eps=0.0):
        with self._lock:
            self._rewards = getattr(self, "_rewards", [])
            self._rewards.append(float(reward))

    def drain_rewards(self):
        with self._lock:
            out = getattr(self, "_rewards", [])
            self._rewards = []
        return out
