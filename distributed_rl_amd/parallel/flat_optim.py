"""Fused flat-buffer optimizers (K12) for MixedPrecisionTrainer masters.

torch's capturable foreach RMSprop spends ~170 us/step on the 1.7M-param
Ape-X model (six multi_tensor_apply passes, profiles/); these run the whole
update as ONE memory-bound kernel per dtype group over the flat fp32 master
param/grad buffers. Math matches torch.optim (verified in
tests/test_gpu_kernels.py). hipGraph-capturable (Adam's step counter is a
device scalar bumped in-kernel).
"""

from __future__ import annotations

from typing import Any, Dict, List

import torch

from ..ops import hip_ext


class _FlatOptimBase:
    def __init__(self):
        self.param_groups: List[Dict[str, Any]] = [{}]  # capturable-flag shim

    def zero_grad(self, set_to_none: bool = False):
        for g in self.grads:
            g.zero_()


class FlatRMSprop(_FlatOptimBase):
    def __init__(self, params_flat, grads_flat, lr=1e-2, alpha=0.99, eps=1e-8,
                 weight_decay=0.0, momentum=0.0, centered=False):
        super().__init__()
        self.params = list(params_flat)
        self.grads = list(grads_flat)
        self.lr, self.alpha, self.eps = float(lr), float(alpha), float(eps)
        self.wd, self.mu = float(weight_decay), float(momentum)
        self.centered = bool(centered)
        self.sq = [torch.zeros_like(p) for p in self.params]
        self.ga = [torch.zeros_like(p) if centered else torch.empty(0)
                   for p in self.params]
        self.mom = [torch.zeros_like(p) if momentum > 0 else torch.empty(0)
                    for p in self.params]

    def step(self):
        ext = hip_ext()
        for p, g, sq, ga, mom in zip(self.params, self.grads, self.sq,
                                     self.ga, self.mom):
            ext.rmsprop_step(p, g, sq, ga, mom, self.lr, self.alpha, self.eps,
                             self.wd, self.mu, self.centered, self.mu > 0)

    def state_dict(self):
        return {"kind": "flat_rmsprop", "sq": self.sq, "ga": self.ga,
                "mom": self.mom}

    def load_state_dict(self, sd):
        for dst, src in zip(self.sq, sd["sq"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.ga, sd["ga"]):
            if dst.numel():
                dst.copy_(src.to(dst.device))
        for dst, src in zip(self.mom, sd["mom"]):
            if dst.numel():
                dst.copy_(src.to(dst.device))


class FlatAdam(_FlatOptimBase):
    def __init__(self, params_flat, grads_flat, lr=1e-3, betas=(0.9, 0.999),
                 eps=1e-8, weight_decay=0.0):
        super().__init__()
        self.params = list(params_flat)
        self.grads = list(grads_flat)
        self.lr, self.eps = float(lr), float(eps)
        self.b1, self.b2 = float(betas[0]), float(betas[1])
        self.wd = float(weight_decay)
        self.m = [torch.zeros_like(p) for p in self.params]
        self.v = [torch.zeros_like(p) for p in self.params]
        dev = self.params[0].device
        self.t = [torch.zeros(1, device=dev) for _ in self.params]

    def step(self):
        ext = hip_ext()
        for p, g, m, v, t in zip(self.params, self.grads, self.m, self.v,
                                 self.t):
            ext.adam_step(p, g, m, v, t, self.lr, self.b1, self.b2, self.eps,
                          self.wd)

    def state_dict(self):
        return {"kind": "flat_adam", "m": self.m, "v": self.v, "t": self.t}

    def load_state_dict(self, sd):
        for name in ("m", "v", "t"):
            for dst, src in zip(getattr(self, name), sd[name]):
                dst.copy_(src.to(dst.device))


def make_flat_optimizer(optim_info: Dict[str, Any], mp):
    """Build a fused flat optimizer over an mp trainer's master buffers.
    Returns None if the optimizer kind has no fused version."""
    name = str(optim_info.get("name", "")).lower()
    params = [g.flat_mparam for g in mp.groups]
    grads = [g.flat_mgrad for g in mp.groups]
    if name == "rmsprop":
        return FlatRMSprop(
            params, grads, lr=float(optim_info.get("lr", 1e-3)),
            alpha=float(optim_info.get("alpha", 0.99)),
            eps=float(optim_info.get("eps", 1e-8)),
            weight_decay=float(optim_info.get("decay", 0.0)),
            momentum=float(optim_info.get("momentum", 0.0)),
            centered=bool(optim_info.get("centered", False)),
        )
    if name == "adam":
        return FlatAdam(
            params, grads, lr=float(optim_info.get("lr", 1e-3)),
            eps=float(optim_info.get("eps", 1e-8)),
            weight_decay=float(optim_info.get("decay", 0.0)),
        )
    return None
