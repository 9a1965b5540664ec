"""bf16 compute / fp32 master mixed precision with flat gradient buffers.

Instead of autocast (which re-casts every weight on every forward — ~29
cast kernels/step measured on the Ape-X model, profiles/), we keep a
persistent bf16 *compute* replica of the fp32 *master* model:

  forward/backward run on the bf16 replica -> grads land in ONE flat bf16
  buffer (param.grad are views) -> [optional RCCL all-reduce on the bf16
  flat buffer — half the xGMI bytes of fp32] -> one fused cast into the
  master's flat fp32 grad buffer -> optimizer.step() on fp32 -> one fused
  cast back into the bf16 replica's flat param buffer.

Every arrow is a single kernel (or one collective), all fixed-shape ->
hipGraph-capturable.
"""

from __future__ import annotations

import copy
from typing import Optional

import torch
import torch.distributed as dist


def _flatten_like(params, dtype, device) -> torch.Tensor:
    total = sum(p.numel() for p in params)
    return torch.zeros(total, dtype=dtype, device=device)


def _is_cl(p: torch.Tensor) -> bool:
    return p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last) \
        and not p.is_contiguous()


def _to_flat(p: torch.Tensor) -> torch.Tensor:
    """Elements of p in its own memory order."""
    if _is_cl(p):
        return p.permute(0, 2, 3, 1).reshape(-1)
    return p.reshape(-1)


def _view_like(flat_slice: torch.Tensor, p: torch.Tensor) -> torch.Tensor:
    """A view of flat_slice with p's logical shape AND memory format."""
    if _is_cl(p):
        N, C, H, W = p.shape
        return flat_slice.view(N, H, W, C).permute(0, 3, 1, 2)
    return flat_slice.view_as(p)


class MixedPrecisionTrainer:
    def __init__(self, master: torch.nn.Module,
                 group: Optional["dist.ProcessGroup"] = None,
                 compute_dtype: torch.dtype = torch.bfloat16):
        self.master = master
        dev = next(master.parameters()).device
        self.compute = copy.deepcopy(master).to(compute_dtype)
        self.compute_dtype = compute_dtype
        self.m_params = [p for p in master.parameters() if p.requires_grad]
        self.c_params = [p for p in self.compute.parameters() if p.requires_grad]
        assert len(self.m_params) == len(self.c_params)

        # flat param buffer for the compute replica (params become views,
        # preserving each param's memory format, e.g. channels_last convs)
        total = sum(p.numel() for p in self.c_params)
        self.flat_cparam = torch.empty(total, dtype=compute_dtype, device=dev)
        off = 0
        with torch.no_grad():
            for p in self.c_params:
                n = p.numel()
                self.flat_cparam[off : off + n].copy_(_to_flat(p))
                p.data = _view_like(self.flat_cparam[off : off + n], p)
                off += n
        # flat master params (views) so the downcast is one kernel
        self.flat_mparam = _flatten_like(self.m_params, torch.float32, dev)
        off = 0
        with torch.no_grad():
            for p in self.m_params:
                n = p.numel()
                self.flat_mparam[off : off + n].copy_(_to_flat(p).float())
                p.data = _view_like(self.flat_mparam[off : off + n], p)
                off += n
        # flat grad buffers; compute grads accumulate in-place
        self.flat_cgrad = _flatten_like(self.c_params, compute_dtype, dev)
        self.flat_mgrad = _flatten_like(self.m_params, torch.float32, dev)
        off = 0
        for p, mp in zip(self.c_params, self.m_params):
            n = p.numel()
            p.grad = _view_like(self.flat_cgrad[off : off + n], p)
            mp.grad = _view_like(self.flat_mgrad[off : off + n], mp)
            off += n
        self.group = group
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1

    # -- per-step plumbing -------------------------------------------------
    def zero_grads(self):
        self.flat_cgrad.zero_()

    def reduce_and_upcast(self):
        """bf16 all-reduce (if distributed) then one cast to fp32 grads."""
        if self.world > 1:
            dist.all_reduce(self.flat_cgrad, group=self.group)
            self.flat_mgrad.copy_(self.flat_cgrad)
            self.flat_mgrad.mul_(1.0 / self.world)
        else:
            self.flat_mgrad.copy_(self.flat_cgrad)

    def sync_compute_params(self):
        """fp32 master -> bf16 compute replica (one cast kernel)."""
        self.flat_cparam.copy_(self.flat_mparam)

    def broadcast_master(self):
        if self.world > 1:
            dist.broadcast(self.flat_mparam, src=0, group=self.group)
            self.sync_compute_params()
