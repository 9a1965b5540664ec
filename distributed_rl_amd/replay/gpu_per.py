"""HBM3E-resident lock-free sum-tree PER — the production learner replay.

Replaces the reference's Redis-list + host-thread replay pipeline
(APE_X/ReplayMemory.py:19-167): experiences live in device memory as typed
columns (288 GB HBM3E per MI355X), priorities live in a binary sum-tree
(float[2P], P = next pow2 of capacity) updated by atomicExch/atomicAdd
kernels so concurrent ingest (side stream) and priority updates (compute
stream) compose without locks. Sampling is a stratified inverse-CDF descent,
one lane per sample. All ops are fixed-shape device kernels -> the whole
train step is hipGraph-capturable.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..ops import hip_ext
from .per import ReplayBase, Schema


def _next_pow2(n: int) -> int:
    p = 1
    while p < n:
        p <<= 1
    return p


class HipSumTreePER(ReplayBase):
    def __init__(self, capacity: int, schema: Schema, device: str = "cuda",
                 seed: int = 0x5EED):
        super().__init__(capacity, schema, device)
        self.ext = hip_ext(required=True)
        self.P = _next_pow2(self.capacity)
        # tree[0] unused, root tree[1], leaves tree[P : P+capacity]
        self.tree = torch.zeros(2 * self.P, dtype=torch.float32, device=self.device)
        self.seed = torch.tensor([seed], dtype=torch.int64, device=self.device)
        self._min_bits = torch.empty(1, dtype=torch.int32, device=self.device)
        self._inf_bits = torch.full(
            (1,), 0x7F800000, dtype=torch.int32, device=self.device
        )

    # -- priority machinery ------------------------------------------------
    def _set_priorities(self, idx: torch.Tensor, prios: torch.Tensor):
        self.ext.sumtree_update(
            self.tree, idx.to(self.device, torch.int64).contiguous(),
            prios.to(self.device, torch.float32).contiguous(), self.P
        )

    def update(self, idx: torch.Tensor, prios: torch.Tensor):
        self._set_priorities(idx, prios)

    def _leaf_priorities(self):
        return self.tree[self.P : self.P + self.capacity]

    @property
    def total_priority(self) -> float:
        return float(self.tree[1])

    def sample(self, k: int, beta: float, with_data: bool = True,
               out: Optional[Tuple[torch.Tensor, torch.Tensor]] = None):
        n = len(self)
        if n == 0:
            raise RuntimeError("sampling from empty replay")
        if out is None:
            idx = torch.empty(k, dtype=torch.int64, device=self.device)
            prob = torch.empty(k, dtype=torch.float32, device=self.device)
            w = torch.empty(k, dtype=torch.float32, device=self.device)
        else:
            idx, prob, w = out
        self.ext.bump_seed(self.seed)
        self.ext.sumtree_sample(self.tree, self.P, n, k, self.seed, idx, prob)
        self._min_bits.copy_(self._inf_bits)
        self.ext.leaf_min_pos(self.tree, self.P, n, self._min_bits)
        self.ext.per_weights(prob, self._min_bits, self.tree, n, beta, w)
        data = self.gather(idx) if with_data else None
        return data, idx, w
