"""bf16 compute / fp32 master mixed precision with flat gradient buffers.

Instead of autocast (which re-casts every weight on every forward — ~29
cast kernels/step measured on the Ape-X model, profiles/), we keep a
persistent *compute* replica of the fp32 *master* model:

  forward/backward run on the compute replica -> grads land in flat
  per-dtype buffers (param.grad are views) -> [optional RCCL all-reduce on
  the flat buffers — bf16 halves the xGMI bytes] -> one fused cast per
  dtype group into the master's flat fp32 grad buffer -> optimizer.step()
  on fp32 -> one fused cast per group back into the compute replica.

A ``keep_fp32`` predicate pins chosen parameters to fp32 in the compute
replica (e.g. R2D2's LSTM, where the MIOpen fp32 RNN path is 2.4x faster
than bf16 on MI355X) — the master layout is grouped [bf16-group |
fp32-group] so each sync is still one contiguous cast/copy per group.
Every step op is a fixed-shape kernel -> hipGraph-capturable.
"""

from __future__ import annotations

import copy
from typing import Callable, Optional

import torch
import torch.distributed as dist


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


class _Group:
    """One dtype group: flat compute param/grad buffers + the master's
    corresponding flat fp32 region."""

    def __init__(self, dtype, c_params, m_params, device):
        self.dtype = dtype
        self.c_params = c_params
        self.m_params = m_params
        total = sum(p.numel() for p in c_params)
        self.flat_cparam = torch.empty(total, dtype=dtype, device=device)
        self.flat_cgrad = torch.zeros(total, dtype=dtype, device=device)
        self.flat_mparam = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_mgrad = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        with torch.no_grad():
            for cp, mp in zip(c_params, m_params):
                n = cp.numel()
                self.flat_cparam[off : off + n].copy_(_to_flat(cp))
                self.flat_mparam[off : off + n].copy_(_to_flat(mp).float())
                cp.data = _view_like(self.flat_cparam[off : off + n], cp)
                mp.data = _view_like(self.flat_mparam[off : off + n], mp)
                cp.grad = _view_like(self.flat_cgrad[off : off + n], cp)
                mp.grad = _view_like(self.flat_mgrad[off : off + n], mp)
                off += n


class MixedPrecisionTrainer:
    def __init__(self, master: torch.nn.Module,
                 group: Optional["dist.ProcessGroup"] = None,
                 compute_dtype: torch.dtype = torch.bfloat16,
                 keep_fp32: Optional[Callable[[str, torch.Tensor], bool]] = None,
                 param_order: Optional[list] = None):
        self.master = master
        dev = next(master.parameters()).device
        self.compute = copy.deepcopy(master)
        # cast only the non-pinned params of the compute replica
        named_master = [(n, p) for n, p in master.named_parameters()
                        if p.requires_grad]
        if param_order:
            # callers may pin chosen params to the FRONT of the flat buffers
            # (in the given order) so adjacent params can be viewed as one
            # fused tensor (see flat_view)
            rank = {n: i for i, n in enumerate(param_order)}
            named_master.sort(key=lambda np_: rank.get(np_[0], len(rank)))
        self._name_order = [n for n, _ in named_master]
        named_compute = dict(self.compute.named_parameters())
        lo_c, lo_m, hi_c, hi_m = [], [], [], []
        for name, mp_ in named_master:
            cp = named_compute[name]
            if keep_fp32 is not None and keep_fp32(name, mp_):
                hi_c.append(cp)
                hi_m.append(mp_)
            else:
                cp.data = cp.data.to(compute_dtype)
                lo_c.append(cp)
                lo_m.append(mp_)
        self.compute_dtype = compute_dtype
        self.groups = []
        if lo_c:
            self.groups.append(_Group(compute_dtype, lo_c, lo_m, dev))
        if hi_c:
            self.groups.append(_Group(torch.float32, hi_c, hi_m, dev))
        self.pg = group
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1

    def flat_view(self, names: list, of: str = "param") -> torch.Tensor:
        """Contiguous flat view spanning the given (adjacent) params in the
        group that holds them. ``of``: 'param' or 'grad'."""
        named = dict(self.master.named_parameters())
        ident = {id(p): n for n, p in named.items()}
        for g in self.groups:
            offs = {}
            off = 0
            for cp, mp_ in zip(g.c_params, g.m_params):
                n = ident.get(id(mp_))
                if n is not None:
                    offs[n] = (off, mp_.numel())
                off += cp.numel()
            if all(n in offs for n in names):
                start = min(offs[n][0] for n in names)
                end = max(offs[n][0] + offs[n][1] for n in names)
                assert end - start == sum(offs[n][1] for n in names), (
                    f"params {names} are not adjacent in the flat buffer"
                )
                buf = g.flat_cparam if of == "param" else g.flat_cgrad
                return buf[start:end]
        raise KeyError(f"params {names} not found in one group")

    # -- convenience views (single-group fast paths used by callers) -------
    @property
    def flat_cparam(self):
        assert len(self.groups) == 1
        return self.groups[0].flat_cparam

    @property
    def flat_cgrad(self):
        assert len(self.groups) == 1
        return self.groups[0].flat_cgrad

    @property
    def flat_mgrad(self):
        assert len(self.groups) == 1
        return self.groups[0].flat_mgrad

    @property
    def flat_mparam(self):
        assert len(self.groups) == 1
        return self.groups[0].flat_mparam

    def all_mgrads(self):
        return [g.flat_mgrad for g in self.groups]

    # -- per-step plumbing -------------------------------------------------
    def zero_grads(self):
        for g in self.groups:
            g.flat_cgrad.zero_()

    def direct_grads(self, loss, params_override=None):
        """Backward WITHOUT AccumulateGrad: torch.autograd.grad returns each
        parameter's gradient as the tensor its producing op wrote (no
        ``.grad += new`` add kernel per parameter — ~12 x 4.9 us per Ape-X
        step measured), then ONE cat writes the flat compute-grad buffer.
        Equivalent to zero_grads() + loss.backward() for single-visit
        parameters (every model here); downstream (all-reduce / upcast /
        optimizer) is unchanged.

        params_override: the actual autograd leaves when the forward used
        fused views (e.g. Ape-X's w1 spanning the two pinned dueling-stream
        weights) — their flattened concatenation must equal the flat-buffer
        layout (single-group trainers only)."""
        if params_override is not None:
            assert len(self.groups) == 1
            grads = torch.autograd.grad(loss, params_override)
            flats = []
            for p, gr in zip(params_override, grads):
                if gr.dim() == 4 and _is_cl(p) and not _is_cl(gr):
                    gr = gr.contiguous(memory_format=torch.channels_last)
                flats.append(_to_flat(gr))
            torch.cat(flats, out=self.groups[0].flat_cgrad)
            return
        params = [cp for g in self.groups for cp in g.c_params]
        grads = torch.autograd.grad(loss, params)
        i = 0
        for g in self.groups:
            n = len(g.c_params)
            flats = []
            for cp, gr in zip(g.c_params, grads[i : i + n]):
                if gr.dim() == 4 and _is_cl(cp) and not _is_cl(gr):
                    gr = gr.contiguous(memory_format=torch.channels_last)
                flats.append(_to_flat(gr))
            i += n
            torch.cat(flats, out=g.flat_cgrad)

    def allreduce_grads(self):
        """all-reduce the compute-dtype grad flats (eager; NEVER inside a
        hipGraph capture — collectives are not capturable)."""
        if self.world > 1:
            for g in self.groups:
                dist.all_reduce(g.flat_cgrad, group=self.pg)

    def allreduce_grads_async(self):
        """Launch the grad all-reduce without blocking the caller; returns
        the work handles (call ``wait()`` on each before reading grads).
        With the RCCL backend the collective is enqueued on the comm stream
        and overlaps whatever the compute stream runs next (the north-star
        overlap: next replay sample + priority update run during the
        all-reduce — SURVEY.md §2.9 C1)."""
        if self.world > 1:
            return [dist.all_reduce(g.flat_cgrad, group=self.pg, async_op=True)
                    for g in self.groups]
        return []

    def upcast_grads(self):
        """compute-dtype grad flats -> fp32 master grads (+ 1/world)."""
        for g in self.groups:
            g.flat_mgrad.copy_(g.flat_cgrad)
            if self.world > 1:
                g.flat_mgrad.mul_(1.0 / self.world)

    def reduce_and_upcast(self):
        """all-reduce compute grads (if distributed) then cast to fp32."""
        self.allreduce_grads()
        self.upcast_grads()

    def sync_compute_params(self):
        """fp32 master -> compute replica (one cast kernel per group)."""
        for g in self.groups:
            g.flat_cparam.copy_(g.flat_mparam)

    def broadcast_master(self):
        if self.world > 1:
            for g in self.groups:
                dist.broadcast(g.flat_mparam, src=0, group=self.pg)
            self.sync_compute_params()
