"""Prioritized experience replay — device-resident, ring-evicting.

Semantic contract (from /root/reference APE_X/ReplayMemory.py
and the inferred ``baseline.PER`` surface):
  * proportional sampling  P(i) = p_i / sum_j p_j  (alpha is applied by the
    producer before push — APE_X/Player.py:135-159),
  * IS weights  w_i = (1/(n * P_i))^beta / max_w, max_w taken buffer-wide
    (the weight of the minimum-priority element; APE_X/ReplayMemory.py:64-67),
  * batched deferred priority updates (update(idx, prios)),
  * capacity eviction (remove_to_fit -> here: ring overwrite, oldest first).

Two implementations share this interface:
  * TorchPER (this file): pure torch ops; runs on CPU (tests, GPU-less actors)
    and is the numerics oracle for the HIP path.
  * HipSumTreePER (gpu_per.py): lock-free sum-tree in HBM3E driven by the
    hand-written gfx950 kernels in ops/hip/ — the production learner path.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

Schema = Dict[str, Tuple[Tuple[int, ...], torch.dtype]]


class ReplayBase:
    """Ring-buffer storage for named columns + priority machinery interface."""

    def __init__(self, capacity: int, schema: Schema, device: str = "cpu"):
        self.capacity = int(capacity)
        self.device = torch.device(device)
        self.schema = schema
        self.data: Dict[str, torch.Tensor] = {}
        for name, (shape, dtype) in schema.items():
            self.data[name] = torch.empty(
                (self.capacity, *shape), dtype=dtype, device=self.device
            )
        self.write_pos = 0
        self.count = 0  # total pushed (monotonic)

    def __len__(self) -> int:
        return min(self.count, self.capacity)

    # -- storage ---------------------------------------------------------
    def _ring_indices(self, n: int) -> torch.Tensor:
        idx = (torch.arange(n, device=self.device) + self.write_pos) % self.capacity
        return idx

    def _write_columns(self, columns: Dict[str, torch.Tensor], idx: torch.Tensor):
        for name, col in columns.items():
            dst = self.data[name]
            dst.index_copy_(0, idx, col.to(self.device, dst.dtype, non_blocking=True))

    def push(self, columns: Dict[str, torch.Tensor], priorities: torch.Tensor):
        n = priorities.shape[0]
        if n == 0:
            return
        if n > self.capacity:
            # oversized batch (e.g. a big ingest drain into a small ring):
            # only the LAST `capacity` rows survive a ring write anyway
            columns = {k: v[n - self.capacity :] for k, v in columns.items()}
            priorities = priorities[n - self.capacity :]
            n = self.capacity
        idx = self._ring_indices(n)
        self._write_columns(columns, idx)
        self._set_priorities(idx, priorities.to(self.device, torch.float32))
        self.write_pos = (self.write_pos + n) % self.capacity
        self.count += n

    def gather(self, idx: torch.Tensor) -> Dict[str, torch.Tensor]:
        if self.device.type == "cuda" and len(self.data) <= 8:
            from ..ops import hip_ext

            ext = hip_ext(required=False)
            if ext is not None and hasattr(ext, "gather_rows"):
                # fused multi-column row gather: ONE launch instead of the
                # per-column index_select chain (8-10 kernels per sample)
                srcs = list(self.data.values())
                dsts = [torch.empty((idx.numel(), *c.shape[1:]),
                                    dtype=c.dtype, device=c.device)
                        for c in srcs]
                ext.gather_rows(idx.to(torch.int64), srcs, dsts)
                return dict(zip(self.data.keys(), dsts))
        return {name: col.index_select(0, idx) for name, col in self.data.items()}

    # -- checkpoint (optional; SURVEY §5.4 "PER state optional") ----------
    def _leaf_priorities(self) -> torch.Tensor:
        """Priorities of the filled region, ring order (subclass hook)."""
        raise NotImplementedError

    def state_dict(self) -> Dict[str, object]:
        """Serializable replay contents (filled region only, CPU tensors).
        Opt-in via the learner's ``checkpoint_replay`` flag — Ape-X's 100k
        uint8 frames are ~5.6 GB on disk, so this is NOT saved by default."""
        n = len(self)
        return {
            "n": n,
            "write_pos": self.write_pos,
            "count": self.count,
            "data": {k: v[:n].to("cpu") for k, v in self.data.items()},
            "priorities": self._leaf_priorities()[:n].to("cpu"),
        }

    def load_state_dict(self, state: Dict[str, object]) -> None:
        n = int(state["n"])
        if n > self.capacity:
            raise ValueError("checkpointed replay larger than capacity")
        for k, col in state["data"].items():
            self.data[k][:n].copy_(col.to(self.device))
        idx = torch.arange(n, device=self.device)
        self._set_priorities(idx, state["priorities"].to(self.device,
                                                         torch.float32))
        self.write_pos = int(state["write_pos"])
        self.count = int(state["count"])

    # -- priority machinery (implemented by subclasses) -------------------
    def _set_priorities(self, idx: torch.Tensor, prios: torch.Tensor):
        raise NotImplementedError

    def update(self, idx: torch.Tensor, prios: torch.Tensor):
        raise NotImplementedError

    def sample(self, k: int, beta: float, with_data: bool = True):
        raise NotImplementedError

    @property
    def total_priority(self) -> float:
        raise NotImplementedError


class TorchPER(ReplayBase):
    """Pure-torch proportional PER (CPU path / oracle)."""

    def __init__(self, capacity: int, schema: Schema, device: str = "cpu",
                 stratified: bool = True, generator: Optional[torch.Generator] = None):
        super().__init__(capacity, schema, device)
        self.priorities = torch.zeros(capacity, dtype=torch.float32, device=self.device)
        self.stratified = stratified
        self.generator = generator

    def _set_priorities(self, idx, prios):
        self.priorities.index_copy_(0, idx, prios)

    def _leaf_priorities(self):
        return self.priorities

    def update(self, idx, prios):
        idx = idx.to(self.device)
        prios = prios.to(self.device, torch.float32)
        self.priorities.index_copy_(0, idx, prios)

    @property
    def total_priority(self) -> float:
        return float(self.priorities.sum())

    def sample(self, k: int, beta: float, with_data: bool = True):
        n = len(self)
        if n == 0:
            raise RuntimeError("sampling from empty replay")
        p = self.priorities
        total = p.sum()
        if self.stratified:
            # stratified inverse-CDF sampling — the same scheme the HIP
            # sum-tree kernel uses, so distributions match exactly in law.
            cdf = torch.cumsum(p, dim=0)
            u = (
                torch.arange(k, device=self.device, dtype=torch.float32)
                + torch.rand(k, device=self.device, generator=self.generator)
            ) / k * total
            idx = torch.searchsorted(cdf, u.contiguous()).clamp(max=n - 1)
        else:
            idx = torch.multinomial(p, k, replacement=True, generator=self.generator)
        probs = p.index_select(0, idx) / total
        # buffer-wide max weight = weight of the min nonzero priority
        p_valid = p[p > 0]
        min_prob = p_valid.min() / total
        max_w = (1.0 / (n * min_prob)) ** beta
        w = (1.0 / (n * probs.clamp_min(1e-12))) ** beta / max_w
        data = self.gather(idx) if with_data else None
        return data, idx, w.to(torch.float32)


def make_apex_schema(frame_shape=(4, 84, 84), state_dtype=torch.uint8) -> Schema:
    """Ape-X n-step transition columns (APE_X/Player.py:252-261 wire tuple)."""
    return {
        "state": (frame_shape, state_dtype),
        "action": ((), torch.int32),
        "reward": ((), torch.float32),
        "next_state": (frame_shape, state_dtype),
        "done": ((), torch.float32),
    }


def make_r2d2_schema(seq_len=80, frame_shape=(4, 84, 84), hidden=512,
                     state_dtype=torch.uint8) -> Schema:
    """R2D2 sequence columns: h0 pair + 80-step (s, a, r) + done
    (R2D2/Player.py:311-319)."""
    return {
        "h0": ((2, hidden), torch.float32),
        "states": ((seq_len, *frame_shape), state_dtype),
        "actions": ((seq_len,), torch.int32),
        "rewards": ((seq_len,), torch.float32),
        "done": ((), torch.float32),
    }


class FifoReplay(ReplayBase):
    """Uniform FIFO replay (IMPALA; IMPALA/ReplayMemory.py:21). Device-resident
    ring, uniform sampling, no priorities."""

    def __init__(self, capacity: int, schema: Schema, device: str = "cpu",
                 generator: Optional[torch.Generator] = None):
        super().__init__(capacity, schema, device)
        self.generator = generator

    def _set_priorities(self, idx, prios):
        pass

    def _leaf_priorities(self):
        return torch.ones(self.capacity, device=self.device)

    def push(self, columns: Dict[str, torch.Tensor], n: Optional[int] = None):
        first = next(iter(columns.values()))
        cnt = first.shape[0] if n is None else n
        super().push(columns, torch.ones(cnt, device=self.device))

    def sample(self, k: int, beta: float = 0.0, with_data: bool = True):
        n = len(self)
        idx = torch.randint(0, n, (k,), device=self.device, generator=self.generator)
        data = self.gather(idx) if with_data else None
        w = torch.ones(k, device=self.device)
        return data, idx, w
