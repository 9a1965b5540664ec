"""Op dispatch: HIP extension on device tensors, torch_ref on CPU.

Policy (contract with the round driver): on a GPU box the hand-written HIP
path MUST be the one that runs — if a device tensor reaches an op and the
gfx950 extension is missing, we raise instead of silently falling back to
eager PyTorch. Set DRL_ALLOW_EAGER=1 to override (debug only).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import torch_ref  # noqa: F401

_EXT = None
_EXT_TRIED = False


def hip_ext(required: bool = True):
    """Return the loaded _drl_hip extension module (or None)."""
    global _EXT, _EXT_TRIED
    if _EXT is None and not _EXT_TRIED:
        _EXT_TRIED = True
        from . import build as _build

        _EXT = _build.load_prebuilt()
        if _EXT is None and torch.cuda.is_available():
            # On a GPU box with no prebuilt .so: try building once (hipcc is
            # present in the image), then give up loudly.
            try:
                _EXT = _build.build()
            except Exception as e:  # pragma: no cover
                if required:
                    raise RuntimeError(
                        "gfx950 HIP extension missing and build failed: %s" % e
                    )
    if _EXT is None and required and not _allow_eager():
        raise RuntimeError(
            "device tensor reached a fused op but the _drl_hip extension is "
            "not built; run `python -m distributed_rl_amd.ops.build` "
            "(or set DRL_ALLOW_EAGER=1 to debug with eager torch)"
        )
    return _EXT


def _allow_eager() -> bool:
    return os.environ.get("DRL_ALLOW_EAGER", "0") == "1"


def _use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    ext = hip_ext(required=not _allow_eager())
    return ext is not None


# ---------------------------------------------------------------------------
# K1 — dequant
# ---------------------------------------------------------------------------


def dequant_frames(x_u8: torch.Tensor, dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    if _use_hip(x_u8):
        out = torch.empty(x_u8.shape, dtype=dtype, device=x_u8.device)
        hip_ext().dequant(x_u8.contiguous(), out)
        return out
    return torch_ref.dequant_frames(x_u8, dtype)


def dequant_frames_nhwc(x_u8: torch.Tensor) -> torch.Tensor:
    """(N,4,H,W) uint8 -> channels_last bf16, fused /255 (K1 NHWC variant)."""
    if _use_hip(x_u8):
        out = torch.empty(
            x_u8.shape, dtype=torch.bfloat16, device=x_u8.device,
            memory_format=torch.channels_last,
        )
        hip_ext().dequant_nhwc(x_u8.contiguous(), out)
        return out
    return torch_ref.dequant_frames(x_u8, torch.float32).to(
        memory_format=torch.channels_last
    )


# ---------------------------------------------------------------------------
# K2 — fused MFMA conv (NHWC bf16, fused dequant/bias/ReLU) with aten backward
# ---------------------------------------------------------------------------


def conv_supported(H, W, C, KH, KW, S, COUT, u8: bool) -> bool:
    ext = hip_ext(required=False)
    if ext is None:
        return False
    return bool(ext.conv_fwd_supported(H, W, C, KH, KW, S, COUT, u8))


class _FusedConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, chw_out=False):
        ext = hip_ext()
        N, C, H, W = x.shape
        COUT, _, KH, KW = weight.shape
        P = (H - KH) // stride + 1
        Q = (W - KW) // stride + 1
        b = bias if bias is not None else torch.empty(0, device=x.device)
        if chw_out:
            # flatten fused into the epilogue: 2-D (N, COUT*P*Q) out in
            # logical-NCHW element order (= torch.flatten of the 4-D out)
            out = torch.empty(N, COUT * P * Q, dtype=torch.bfloat16,
                              device=x.device)
            ext.conv_fwd_chw(x, weight, b, out, stride)
        else:
            out = torch.empty(
                N, COUT, P, Q, dtype=torch.bfloat16, device=x.device,
                memory_format=torch.channels_last,
            )
            ext.conv_fwd(x, weight, b, out, stride)
        ctx.save_for_backward(x, weight, out)
        ctx.stride = stride
        ctx.has_bias = bias is not None
        ctx.chw = chw_out
        ctx.pq = (P, Q)
        return out

    @staticmethod
    def backward(ctx, gout):
        x, weight, out = ctx.saved_tensors
        stride = ctx.stride
        COUT = weight.shape[0]
        if ctx.chw:
            # one kernel: relu-mask + CHW->NHWC transpose
            P, Q = ctx.pq
            N = gout.shape[0]
            masked = torch.empty(N, COUT, P, Q, dtype=torch.bfloat16,
                                 device=gout.device,
                                 memory_format=torch.channels_last)
            hip_ext().relu_mask_bwd_chw(gout.contiguous(), out, masked,
                                        COUT, P * Q)
            gout = masked
        else:
            if not gout.is_contiguous(memory_format=torch.channels_last):
                gout = gout.contiguous(memory_format=torch.channels_last)
            # fused ReLU mask (one kernel instead of compare+mul)
            masked = torch.empty_like(gout)
            hip_ext().relu_mask_bwd(gout, out, masked)
            gout = masked
        need_x = ctx.needs_input_grad[0]
        need_w = ctx.needs_input_grad[1]
        need_b = ctx.has_bias and ctx.needs_input_grad[2]
        COUT, C, KH, KW = weight.shape
        ext = hip_ext()
        use_own_wrw = need_w and bool(
            ext.conv_wrw_supported(x.shape[2], x.shape[3], C, KH, KW, stride,
                                   COUT, x.dtype == torch.uint8)
        )
        gw = gb = None
        if use_own_wrw:
            # hand-written MFMA weight-grad (conv_mfma.hip): fp32 workspace
            # accumulated by atomics, cast to the bf16 channels_last grad.
            # single zero-fill for both workspaces (one FillFunctor launch
            # instead of two — these run every step inside the graph)
            K = KH * KW * C
            flat_ws = torch.zeros(COUT * (K + 1), dtype=torch.float32,
                                  device=x.device)
            ws = flat_ws[: COUT * K].view(COUT, K)
            gb_ws = (flat_ws[COUT * K :] if need_b
                     else torch.empty(0, device=x.device))
            ext.conv_wrw(x, gout, ws, gb_ws, stride)
            gw = ws.view(COUT, KH, KW, C).permute(0, 3, 1, 2).to(torch.bfloat16)
            if need_b:
                gb = gb_ws.to(torch.bfloat16)
        gi = None
        # Hand-written dgrad exists and is oracle-tested, but MEASURED
        # SLOWER than MIOpen's igemm at these geometries (tools/
        # gpu_dgrad_bench.py: 133 vs 28 us at 20x20 s2, 42 vs 33 at 9x9
        # s1) — dispatch keeps the faster library kernel; DRL_OWN_DGRAD=1
        # forces ours (profiles/r02 notes).
        use_own_dgrad = (
            os.environ.get("DRL_OWN_DGRAD", "0") == "1"
            and need_x and x.dtype == torch.bfloat16
            and hasattr(ext, "conv_dgrad_supported")
            and bool(ext.conv_dgrad_supported(x.shape[2], x.shape[3], C, KH,
                                              KW, stride, COUT))
        )
        if use_own_dgrad:
            # hand-written MFMA dgrad (conv_mfma.hip): masked-tap gather
            # against a per-step transposed weight copy — no MIOpen on the
            # hot path
            gi = torch.empty_like(x)
            w_t = torch.empty(weight.numel(), dtype=torch.bfloat16,
                              device=x.device)
            ext.conv_dgrad(gout, weight, w_t, gi, stride)
        if (need_x and not use_own_dgrad) or (need_w and not use_own_wrw):
            if x.dtype == torch.uint8:
                # NHWC u8 -> bf16 for the aten path (own wrw reads u8 directly)
                xf = dequant_frames(x.permute(0, 2, 3, 1), torch.bfloat16)
                xf = xf.permute(0, 3, 1, 2)
            else:
                xf = x
            gi2, gw2, gb2 = torch.ops.aten.convolution_backward(
                gout, xf, weight,
                [COUT] if ctx.has_bias else None,
                [stride, stride], [0, 0], [1, 1], False, [0, 0], 1,
                [need_x and not use_own_dgrad, need_w and not use_own_wrw,
                 need_b and not use_own_wrw],
            )
            if gi is None:
                gi = gi2
            if gw is None:
                gw, gb = gw2, gb2
        return (gi if need_x else None), gw, gb, None, None


def fused_conv_relu(x, weight, bias, stride: int, chw_out: bool = False):
    """One fused conv+bias+ReLU layer (u8 or bf16 channels_last input).
    chw_out=True additionally fuses the trailing flatten into the epilogue
    (returns 2-D (N, COUT*P*Q) in logical-NCHW order)."""
    return _FusedConvFn.apply(x, weight, bias, stride, chw_out)


def conv_chw_supported(H, W, C, KH, KW, S, COUT) -> bool:
    ext = hip_ext(required=False)
    return ext is not None and hasattr(ext, "conv_fwd_chw_supported") and \
        bool(ext.conv_fwd_chw_supported(H, W, C, KH, KW, S, COUT))


def linear_relu_supported(K, N) -> bool:
    ext = hip_ext(required=False)
    return ext is not None and hasattr(ext, "linear_relu_supported") and \
        bool(ext.linear_relu_supported(K, N))


class _LinearReluFn(torch.autograd.Function):
    """Own MFMA Linear+bias+ReLU (the MLP trunk GEMM; hipBLASLt ran it at
    ~17.6 us + a separate ReLU). Backward: fused relu-mask, then plain
    bf16 GEMMs for dX/dW and a reduce for db."""

    @staticmethod
    def forward(ctx, x, w, b):
        ext = hip_ext()
        out = torch.empty(x.shape[0], w.shape[0], dtype=torch.bfloat16,
                          device=x.device)
        ext.linear_relu(x.contiguous(), w.contiguous(),
                        b if b is not None else torch.empty(0, device=x.device),
                        out)
        ctx.save_for_backward(x, w, out)
        ctx.has_bias = b is not None
        return out

    @staticmethod
    def backward(ctx, g):
        x, w, out = ctx.saved_tensors
        masked = torch.empty_like(g)
        hip_ext().relu_mask_bwd(g.contiguous(), out, masked)
        dx = masked.mm(w) if ctx.needs_input_grad[0] else None
        dw = masked.t().mm(x) if ctx.needs_input_grad[1] else None
        db = (masked.float().sum(0).to(torch.bfloat16)
              if ctx.has_bias and ctx.needs_input_grad[2] else None)
        return dx, dw, db


def fused_linear_relu(x, w, b):
    return _LinearReluFn.apply(x, w, b)


def seq_transpose(src: torch.Tensor) -> torch.Tensor:
    """(B, T, ...) -> (T, B, ...) whole-row block copy (the torch
    permute+contiguous of R2D2's 72 MB u8 frame block ran at ~1.5 TB/s)."""
    if _use_hip(src) and (src[0, 0].numel() * src.element_size()) % 16 == 0:
        dst = torch.empty((src.shape[1], src.shape[0], *src.shape[2:]),
                          dtype=src.dtype, device=src.device)
        hip_ext().seq_transpose_rows(src.contiguous(), dst)
        return dst
    return src.transpose(0, 1).contiguous()


# ---------------------------------------------------------------------------
# K3 — fused dueling-head epilogue (A + V) - mean(A)
# ---------------------------------------------------------------------------


class _DuelingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, adv, val):
        ext = hip_ext()
        B, A = adv.shape
        out = torch.empty(B, A, dtype=torch.float32, device=adv.device)
        ext.dueling_fwd(adv.float().contiguous(),
                        val.float().reshape(B).contiguous(), out)
        ctx.A = A
        return out

    @staticmethod
    def backward(ctx, g):
        ext = hip_ext()
        B, A = g.shape
        g = g.contiguous()
        gadv = torch.empty_like(g)
        gval = torch.empty(B, dtype=torch.float32, device=g.device)
        ext.dueling_bwd(g, gadv, gval)
        return gadv, gval.unsqueeze(1)


def dueling_head(adv: torch.Tensor, val: torch.Tensor) -> torch.Tensor:
    """out = (adv + val) - mean(adv) (reference dueling graph nodes)."""
    if _use_hip(adv):
        return _DuelingFn.apply(adv, val)
    return (adv + val) - adv.mean(dim=-1, keepdim=True)


# ---------------------------------------------------------------------------
# K4 — fused n-step double-DQN loss as an autograd Function
# ---------------------------------------------------------------------------


class _DQNLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q_s, q_sp_on, q_sp_tg, actions, rewards, dones, weights,
                gamma_n, alpha):
        ext = hip_ext()
        B, A = q_s.shape
        dev = q_s.device
        acc2 = torch.zeros(2, dtype=torch.float32, device=dev)  # one fill
        loss, qmean = acc2[0], acc2[1]
        prio = torch.empty(B, dtype=torch.float32, device=dev)
        coef = torch.empty(B, dtype=torch.float32, device=dev)
        ext.dqn_loss_fwd(
            q_s.contiguous(), q_sp_on.contiguous(), q_sp_tg.contiguous(),
            actions.contiguous(), rewards.contiguous(), dones.contiguous(),
            weights.contiguous(), float(gamma_n), float(alpha),
            acc2[0:1], prio, coef, acc2[1:2],
        )
        ctx.save_for_backward(coef, actions)
        ctx.shape = (B, A)
        ctx.in_dtype = q_s.dtype
        ctx.mark_non_differentiable(prio, qmean)
        return loss, prio, qmean

    @staticmethod
    def backward(ctx, gout, _gprio, _gqm):
        coef, actions = ctx.saved_tensors
        B, A = ctx.shape
        grad_q = torch.empty(B, A, dtype=ctx.in_dtype, device=coef.device)
        hip_ext().dqn_loss_bwd(coef, actions, gout.reshape(1).contiguous(), grad_q)
        return grad_q, None, None, None, None, None, None, None, None


def nstep_dqn_loss(q_s, q_sp_on, q_sp_tg, actions, rewards, dones, weights,
                   gamma: float, n_step: int, alpha: float,
                   with_value_stat: bool = False):
    """Returns (loss, new_priorities[, mean-of-row-max value stat]).
    On GPU q tensors may be bf16 or fp32 (the kernel reads both)."""
    if _use_hip(q_s):
        loss, prio, qmean = _DQNLossFn.apply(
            q_s, q_sp_on, q_sp_tg, actions.long(),
            rewards.float(), dones.float(), weights.float(),
            gamma ** n_step, alpha,
        )
        return (loss, prio, qmean) if with_value_stat else (loss, prio)
    loss, prio = torch_ref.nstep_dqn_loss(
        q_s.float(), q_sp_on.float(), q_sp_tg.float(), actions, rewards.float(),
        dones.float(), weights.float(), gamma, n_step, alpha
    )
    if with_value_stat:
        return loss, prio, q_s.detach().float().max(1).values.mean()
    return loss, prio


# ---------------------------------------------------------------------------
# K3+K4 fused — whole-head dueling n-step double-DQN loss (round 2):
# the dueling epilogue of all three forwards lives inside the loss kernel
# and the backward writes (g_adv, g_val) in closed form — no dueling_fwd,
# no casts, no dueling_bwd, no dqn_loss_bwd launches.
# ---------------------------------------------------------------------------


class _DuelingDQNLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, adv_s, val_s, adv_on, val_on, adv_tg, val_tg, actions,
                rewards, dones, weights, gamma_n, alpha):
        ext = hip_ext()
        B, A = adv_s.shape
        dev = adv_s.device
        acc2 = torch.zeros(2, dtype=torch.float32, device=dev)
        prio = torch.empty(B, dtype=torch.float32, device=dev)
        coef = torch.empty(B, dtype=torch.float32, device=dev)
        ext.dueling_dqn_loss_fwd(
            adv_s.contiguous(), val_s.reshape(B).contiguous(),
            adv_on.contiguous(), val_on.reshape(B).contiguous(),
            adv_tg.contiguous(), val_tg.reshape(B).contiguous(),
            actions.contiguous(), rewards.contiguous(), dones.contiguous(),
            weights.contiguous(), float(gamma_n), float(alpha),
            acc2[0:1], prio, coef, acc2[1:2],
        )
        ctx.save_for_backward(coef, actions)
        ctx.shape = (B, A)
        ctx.in_dtype = adv_s.dtype
        # every non-loss output must be non-differentiable: a grad_fn-
        # carrying stat held in a graphed stepper's static_out keeps the
        # whole autograd graph alive across steps, and the stale
        # AccumulateGrad stream then breaks hipGraph capture (observed as
        # a core dump at R2D2 capture time)
        loss_out, qmean = acc2[0], acc2[1]
        ctx.mark_non_differentiable(prio, qmean)
        return loss_out, prio, qmean

    @staticmethod
    def backward(ctx, gout, _gprio, _gqm):
        coef, actions = ctx.saved_tensors
        B, A = ctx.shape
        g_adv = torch.empty(B, A, dtype=ctx.in_dtype, device=coef.device)
        g_val = torch.empty(B, 1, dtype=ctx.in_dtype, device=coef.device)
        hip_ext().dueling_dqn_loss_bwd(coef, actions,
                                       gout.reshape(1).contiguous(), g_adv,
                                       g_val)
        return (g_adv, g_val) + (None,) * 10


def dueling_nstep_dqn_loss(adv_s, val_s, adv_on, val_on, adv_tg, val_tg,
                           actions, rewards, dones, weights, gamma: float,
                           n_step: int, alpha: float):
    """Fused dueling + n-step double-DQN loss; returns (loss, prio, qmean).
    adv_*: (B, A); val_*: (B, 1). Only the (adv_s, val_s) pair is
    differentiable. On CPU composes dueling_head + nstep_dqn_loss (oracle)."""
    if _use_hip(adv_s):
        return _DuelingDQNLossFn.apply(
            adv_s, val_s, adv_on, val_on, adv_tg, val_tg, actions.long(),
            rewards.float(), dones.float(), weights.float(),
            gamma ** n_step, alpha,
        )
    q_s = dueling_head(adv_s.float(), val_s.float())
    q_on = dueling_head(adv_on.float(), val_on.float())
    q_tg = dueling_head(adv_tg.float(), val_tg.float())
    return nstep_dqn_loss(q_s, q_on, q_tg, actions, rewards, dones, weights,
                          gamma, n_step, alpha, with_value_stat=True)


# ---------------------------------------------------------------------------
# K3+K4+heads fused (round 2b): the dueling head projections join the loss
# kernel; backward is closed-form dh + one reduction kernel for the head
# weight/bias grads. Geometry-locked to hidden=512/A=6 (the shipped cfg).
# ---------------------------------------------------------------------------


def has_dueling_q_loss(hidden: int, actions: int) -> bool:
    ext = hip_ext(required=False)
    return (ext is not None and hasattr(ext, "dueling_q_loss_fwd")
            and hidden == 512 and actions == 6)


class _DuelingQLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h_s, wa, ba, wv, bv, h_on, h_tg, wa_t, ba_t, wv_t, bv_t,
                actions, rewards, dones, weights, gamma_n, alpha):
        ext = hip_ext()
        B = h_s.shape[0]
        dev = h_s.device
        acc2 = torch.zeros(2, dtype=torch.float32, device=dev)
        prio = torch.empty(B, dtype=torch.float32, device=dev)
        coef = torch.empty(B, dtype=torch.float32, device=dev)
        ext.dueling_q_loss_fwd(
            h_s.contiguous(), h_on.contiguous(), h_tg.contiguous(),
            wa.contiguous(), ba.contiguous(), wv.reshape(-1).contiguous(),
            bv.contiguous(), wa_t.contiguous(), ba_t.contiguous(),
            wv_t.reshape(-1).contiguous(), bv_t.contiguous(),
            actions.contiguous(), rewards.contiguous(), dones.contiguous(),
            weights.contiguous(), float(gamma_n), float(alpha),
            acc2[0:1], prio, coef, acc2[1:2],
        )
        ctx.save_for_backward(coef, actions, wa, wv, h_s)
        loss_out, qmean = acc2[0], acc2[1]
        ctx.mark_non_differentiable(prio, qmean)
        return loss_out, prio, qmean

    @staticmethod
    def backward(ctx, gout, _gprio, _gqm):
        coef, actions, wa, wv, h_s = ctx.saved_tensors
        ext = hip_ext()
        B = h_s.shape[0]
        dev = h_s.device
        dh = torch.empty(B, 1024, dtype=torch.bfloat16, device=dev)
        dwa = torch.empty(6, 512, dtype=torch.bfloat16, device=dev)
        dba = torch.empty(6, dtype=torch.bfloat16, device=dev)
        dwv = torch.empty(1, 512, dtype=torch.bfloat16, device=dev)
        dbv = torch.empty(1, dtype=torch.bfloat16, device=dev)
        ext.dueling_q_loss_bwd(coef, actions, gout.reshape(1).contiguous(),
                               wa.contiguous(), wv.reshape(-1).contiguous(),
                               h_s.contiguous(), dh, dwa, dba, dwv, dbv)
        return (dh, dwa, dba, dwv, dbv) + (None,) * 12


def has_r2d2_seq_loss() -> bool:
    ext = hip_ext(required=False)
    return ext is not None and hasattr(ext, "r2d2_loss_fwd")


class _R2D2SeqLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q_train, q_tgt, actions, rewards, done, weights,
                burn_in, n_step, gamma, alpha, eta, use_rescaling):
        ext = hip_ext()
        T, B, A = q_tgt.shape
        W = T - 1 - burn_in
        dev = q_tgt.device
        td = torch.empty(W, B, dtype=torch.float32, device=dev)
        stats = torch.zeros(3, dtype=torch.float32, device=dev)
        ext.r2d2_loss_fwd(q_train, q_tgt, actions, rewards, done, weights,
                          burn_in, n_step, float(gamma), bool(use_rescaling),
                          td, stats)
        prio = torch.empty(B, dtype=torch.float32, device=dev)
        ext.r2d2_prio(td, float(alpha), float(eta), prio)
        ctx.save_for_backward(td, actions, weights)
        ctx.meta = (T, burn_in, tuple(q_train.shape))
        # stats views must be non-differentiable: a grad_fn-carrying stat
        # in a graphed stepper's static_out keeps the autograd graph alive
        # across steps and the stale AccumulateGrad stream then breaks
        # hipGraph capture (core dump at R2D2 capture, r2d2_live.log)
        loss_out, value, td_abs = stats[0], stats[1], stats[2]
        ctx.mark_non_differentiable(prio, value, td_abs)
        return loss_out, prio, value, td_abs

    @staticmethod
    def backward(ctx, gout, _gp, _gv, _gt):
        td, actions, weights = ctx.saved_tensors
        T, m, qshape = ctx.meta
        dq = torch.empty(qshape, dtype=torch.float32, device=td.device)
        hip_ext().r2d2_loss_bwd(td, actions, weights,
                                gout.reshape(1).contiguous(), T, m, dq)
        return (dq,) + (None,) * 11


def r2d2_sequence_loss(q_train, q_tgt, actions, rewards, done, weights,
                       burn_in: int, n_step: int, gamma: float, alpha: float,
                       eta: float, use_rescaling: bool):
    """Fused R2D2 target construction + IS loss + eta-mix priority.
    q_train: (T-burn_in, B, A) fp32 differentiable; q_tgt: (T, B, A) fp32;
    actions: (T, B) int32; rewards (T, B); done/weights (B).
    Returns (loss, prio, value_stat, td_abs_stat)."""
    return _R2D2SeqLossFn.apply(
        q_train.contiguous(), q_tgt.contiguous(),
        actions.to(torch.int32).contiguous(), rewards.float().contiguous(),
        done.float().contiguous(), weights.float().contiguous(),
        burn_in, n_step, gamma, alpha, eta, use_rescaling,
    )


def dueling_q_head_loss(h_s, wa, ba, wv, bv, h_on, h_tg, wa_t, ba_t, wv_t,
                        bv_t, actions, rewards, dones, weights, gamma: float,
                        n_step: int, alpha: float):
    """Heads + dueling + n-step double-DQN loss fully fused.
    h_*: (B, 1024) = [adv-stream | val-stream] post-ReLU hidden.
    Only (h_s, wa, ba, wv, bv) are differentiable."""
    if _use_hip(h_s):
        return _DuelingQLossFn.apply(
            h_s, wa, ba, wv, bv, h_on, h_tg, wa_t, ba_t, wv_t, bv_t,
            actions.long(), rewards.float(), dones.float(), weights.float(),
            gamma ** n_step, alpha,
        )
    import torch.nn.functional as F

    def heads(h, wa_, ba_, wv_, bv_):
        return (F.linear(h[:, :512].float(), wa_.float(), ba_.float()),
                F.linear(h[:, 512:].float(), wv_.float(), bv_.float()))

    adv_s, val_s = heads(h_s, wa, ba, wv, bv)
    adv_on, val_on = heads(h_on, wa, ba, wv, bv)
    adv_tg, val_tg = heads(h_tg, wa_t, ba_t, wv_t, bv_t)
    return dueling_nstep_dqn_loss(adv_s, val_s, adv_on, val_on, adv_tg,
                                  val_tg, actions, rewards, dones, weights,
                                  gamma, n_step, alpha)


# ---------------------------------------------------------------------------
# K6 / K7
# ---------------------------------------------------------------------------


def value_rescale(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    if _use_hip(x) and not x.requires_grad:
        y = torch.empty_like(x, dtype=torch.float32)
        hip_ext().value_rescale(x.float().contiguous(), y, eps)
        return y
    return torch_ref.value_rescale(x, eps)


def inv_value_rescale(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    if _use_hip(x) and not x.requires_grad:
        y = torch.empty_like(x, dtype=torch.float32)
        hip_ext().inv_value_rescale(x.float().contiguous(), y, eps)
        return y
    return torch_ref.inv_value_rescale(x, eps)


def sequence_priority(td_abs: torch.Tensor, alpha: float, eta: float = 0.9) -> torch.Tensor:
    if _use_hip(td_abs):
        T, B = td_abs.shape
        out = torch.empty(B, dtype=torch.float32, device=td_abs.device)
        hip_ext().seq_priority(td_abs.float().contiguous(), eta, alpha, out)
        return out
    return torch_ref.sequence_priority(td_abs, alpha, eta)


# ---------------------------------------------------------------------------
# K8 — V-trace
# ---------------------------------------------------------------------------


def vtrace(behavior_log_prob, target_log_prob, rewards, values, bootstrap_value,
           not_done, gamma, rho_bar=1.0, c_bar=1.0, lam=1.0):
    if _use_hip(rewards):
        T, B = rewards.shape
        dev = rewards.device
        vs = torch.empty(T, B, dtype=torch.float32, device=dev)
        pg_adv = torch.empty_like(vs)
        rho_c = torch.empty_like(vs)
        hip_ext().vtrace(
            behavior_log_prob.float().contiguous(),
            target_log_prob.float().contiguous(), rewards.float().contiguous(),
            values.float().contiguous(), bootstrap_value.float().contiguous(),
            not_done.float().contiguous(), gamma, rho_bar, c_bar, lam,
            vs, pg_adv, rho_c,
        )
        return vs, pg_adv, rho_c
    return torch_ref.vtrace(
        behavior_log_prob, target_log_prob, rewards, values, bootstrap_value,
        not_done, gamma, rho_bar, c_bar, lam
    )


def vtrace_bt(mu_probs, target_log_prob, rewards, values, bootstrap_value,
              not_done, gamma, rho_bar=1.0, c_bar=1.0, lam=1.0):
    """V-trace on (B, T) row-major tensors with raw behavior probabilities.

    Same math as :func:`vtrace` but the learner's natural layout goes straight
    in and comes straight out — on GPU this removes the 4 input transposes, the
    mu.log() pass and the 2 output transposes per IMPALA step. Returns
    (vs, pg_adv), both (B, T)."""
    if _use_hip(rewards):
        B, T = rewards.shape
        dev = rewards.device
        vs = torch.empty(B, T, dtype=torch.float32, device=dev)
        pg_adv = torch.empty_like(vs)
        hip_ext().vtrace_bt(
            mu_probs.float().contiguous(), target_log_prob.float().contiguous(),
            rewards.float().contiguous(), values.float().contiguous(),
            bootstrap_value.float().contiguous(), not_done.float().contiguous(),
            gamma, rho_bar, c_bar, lam, vs, pg_adv,
        )
        return vs, pg_adv
    vs_T, pg_T, _ = torch_ref.vtrace(
        mu_probs.log().t().contiguous(), target_log_prob.t().contiguous(),
        rewards.t().contiguous(), values.t().contiguous(), bootstrap_value,
        not_done, gamma, rho_bar, c_bar, lam
    )
    return vs_T.t(), pg_T.t()


# ---------------------------------------------------------------------------
# K9 — fused IMPALA policy objective
# ---------------------------------------------------------------------------


def policy_softmax_stats(logits, actions):
    """One fused pass over (N, A) logits -> (log_pi_a, pi, H, mean_entropy),
    all detached (K9 part 1; feeds V-trace before the objective exists)."""
    ext = hip_ext()
    N, A = logits.shape
    dev = logits.device
    zbuf = torch.zeros(N + 2, device=dev)  # one fill for all three
    obj, ent, dummy = zbuf[0:1], zbuf[1:2], zbuf[2:]
    logpa = torch.empty(N, device=dev)
    pi = torch.empty(N, A, device=dev)
    H = torch.empty(N, device=dev)
    ext.policy_loss_fwd(logits.detach().contiguous(), actions.contiguous(),
                        dummy, 0.0, obj, ent, logpa, pi, H)
    return logpa, pi, H, ent.squeeze(0)


class _PolicyObjFn(torch.autograd.Function):
    """obj = mean(log_pi_a * adv) + er * mean(H), differentiable in logits
    via the closed-form softmax gradient (K9 part 2)."""

    @staticmethod
    def forward(ctx, logits, logpa, pi, H, actions, adv, er):
        ctx.save_for_backward(pi, H, actions, adv)
        ctx.er = er
        return (logpa * adv).mean() + er * H.mean()

    @staticmethod
    def backward(ctx, gobj):
        ext = hip_ext()
        pi, H, actions, adv = ctx.saved_tensors
        dlogits = torch.empty_like(pi)
        ext.policy_loss_bwd(pi, H, actions, adv,
                            gobj.reshape(1).contiguous(), ctx.er, 1.0, dlogits)
        return dlogits, None, None, None, None, None, None


class _ImpalaLossFn(torch.autograd.Function):
    """Whole IMPALA loss on the raw head output out = (B*(T+1), A+1):
    loss = -(mean(logpa*adv) + er*mean_H) + 0.5*mean((v_t - vs)^2),
    one forward kernel and ONE backward kernel that writes the full d out
    (policy grad, critic grad and the zero bootstrap/T rows together) —
    no SliceBackward zeros+copy+accumulate chains
    (IMPALA/Learner.py:95-119 math)."""

    @staticmethod
    def forward(ctx, out, v_t, logpa, pi, H, actions, adv, vs, mean_H, er, T):
        ext = hip_ext()
        dev = out.device
        v_c = v_t.contiguous().view(-1)
        vs_c = vs.contiguous().view(-1)
        loss = torch.empty(1, device=dev)
        obj = torch.empty(1, device=dev)
        critic = torch.empty(1, device=dev)
        ext.impala_loss_fwd(logpa, adv, mean_H.reshape(1).contiguous(),
                            v_c, vs_c, er, loss, obj, critic)
        ctx.save_for_backward(pi, H, actions, adv, v_c, vs_c)
        ctx.er = er
        ctx.T = T
        ctx.out_shape = out.shape
        ctx.set_materialize_grads(False)
        ctx.mark_non_differentiable(obj, critic)
        return loss.squeeze(0), obj.squeeze(0), critic.squeeze(0)

    @staticmethod
    def backward(ctx, gloss, _gobj, _gcritic):
        ext = hip_ext()
        pi, H, actions, adv, v_c, vs_c = ctx.saved_tensors
        if gloss is None:
            return (None,) * 11
        T = ctx.T
        A = ctx.out_shape[1] - 1
        B = ctx.out_shape[0] // (T + 1)
        g = gloss.reshape(1).contiguous()
        dout = torch.empty(ctx.out_shape, dtype=torch.float32,
                           device=v_c.device)
        ext.impala_out_bwd(pi, H, actions, adv, v_c, vs_c, g, B, T, A,
                           ctx.er, dout)
        return (dout, None, None, None, None, None, None, None, None, None,
                None)


def impala_fused_loss(out, v_t, stats, actions, adv, vs, entropy_coef, T):
    """GPU-only fused total loss on the raw (B*(T+1), A+1) head output;
    returns (loss, obj, critic) scalars with loss differentiable in ``out``.
    ``v_t`` is the detached (B, T) value slice (also fed to V-trace);
    ``stats`` is the policy_softmax_stats tuple computed for V-trace."""
    logpa, pi, H, ent = stats
    return _ImpalaLossFn.apply(out, v_t, logpa, pi, H, actions.long(),
                               adv.float().contiguous(), vs, ent,
                               entropy_coef, T)


def impala_policy_objective(logits, actions, adv, entropy_coef,
                            stats=None):
    """obj_actor = mean(log pi(a) * adv) + er * mean(H); returns
    (obj, entropy). ``stats`` is the policy_softmax_stats tuple when it was
    already computed for V-trace."""
    if _use_hip(logits):
        if stats is None:
            stats = policy_softmax_stats(logits, actions.long())
        logpa, pi, H, ent = stats
        obj = _PolicyObjFn.apply(logits, logpa, pi, H, actions.long(),
                                 adv.float(), entropy_coef)
        return obj, ent
    log_pi = torch.log_softmax(logits.float(), dim=-1)
    pi = log_pi.exp()
    entropy = -(pi * log_pi).sum(-1).mean()
    log_pi_a = log_pi.gather(1, actions.long().unsqueeze(1)).squeeze(1)
    obj = (log_pi_a * adv).mean() + entropy_coef * entropy
    return obj, entropy.detach()


# ---------------------------------------------------------------------------
# K11 — fused grad clip over a flat buffer
# ---------------------------------------------------------------------------


def clip_flat_grad_(flat: torch.Tensor, max_norm: float,
                    sqsum_buf: Optional[torch.Tensor] = None) -> None:
    if _use_hip(flat):
        if sqsum_buf is None:
            sqsum_buf = torch.zeros(1, dtype=torch.float32, device=flat.device)
        else:
            sqsum_buf.zero_()
        hip_ext().grad_clip(flat, max_norm, sqsum_buf)
        return
    norm = flat.norm(2)
    scale = max_norm / (norm + 1e-6)
    if norm > max_norm:
        flat.mul_(scale)
