"""Manual LSTM sequence (K5-lite): host-driven recurrence over fused HIP
cell kernels.

Why not MIOpen's RNN (what nn.LSTM dispatches to on ROCm): (a) it rejects
hipGraph capture (hipErrorStreamCaptureUnsupported), so the R2D2 step can't
be graphed around it; (b) its per-step kernels leave the input projection
inside the loop. Here the input projection x @ W_ih^T + b is hoisted into
ONE GEMM over all T timesteps, the loop body is one hh-addmm + one fused
cell kernel (ops/hip/drl_kernels.hip lstm_cell_*), and backward is one
fused cell-bwd kernel + one GEMM per step with the weight grads batched
into two big GEMMs at the end. Everything is fixed-shape -> capturable.

Uses nn.LSTM's own parameters (weight_ih_l0 / weight_hh_l0 / bias_*_l0) so
state_dict parity with the eager/CPU path is exact.
"""

from __future__ import annotations

from typing import Tuple

import torch


class _ManualLSTMSeq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, h0, c0, w_ih, w_hh, b_ih, b_hh):
        # x: (T, B, IN); h0/c0: (B, H)
        from ..ops import hip_ext

        ext = hip_ext()
        T, B, IN = x.shape
        H = w_hh.shape[1]
        dev = x.device
        bias = b_ih + b_hh
        if x.dtype == torch.bfloat16:
            # bf16 input projection (DRL_LSTM_BF16_IN): the (T*B, IN) GEMM
            # runs on bf16 MFMA (fp32 accumulate inside hipBLASLt) instead
            # of fp32 — ~8x the rate at these shapes — and the fp32 upcast
            # happens on the 4H-wide gates instead of the IN-wide input.
            # Recurrence, cell state and cell kernels stay fp32.
            w_ih_c = w_ih.to(torch.bfloat16)
            xp = torch.mm(x.reshape(T * B, IN), w_ih_c.t()).float()
            xp.add_(bias)
            xp = xp.view(T, B, 4 * H)
        else:
            xp = torch.addmm(bias, x.reshape(T * B, IN),
                             w_ih.t()).view(T, B, 4 * H)
        hs = torch.empty(T + 1, B, H, device=dev)
        cs = torch.empty(T + 1, B, H, device=dev)
        acts = torch.empty(T, B, 4 * H, device=dev)
        tanhc = torch.empty(T, B, H, device=dev)
        hs[0] = h0
        cs[0] = c0
        # Default: in-place addmm_ + fused cell kernel. Two alternatives
        # were built and MEASURED SLOWER at H=512/B=32 (profiles/): the
        # one-kernel fused step (DRL_LSTM_FUSED) and the whole-sequence
        # persistent kernel with agent-scope grid barriers
        # (DRL_LSTM_PERSISTENT) — both are issue-bound on a 32-block grid
        # (128 waves on 1024 SIMDs); the hipBLASLt addmm keeps more of the
        # chip busy. Revisit with a wider decomposition (split-K + 2-phase
        # barrier) next round.
        import os as _os

        # K5 v2 (round 2, default): ONE kernel per timestep — bf16-MFMA hh
        # GEMM fused with the cell. The recurrence state (c), gates and
        # saves stay fp32; h additionally keeps a bf16 shadow that feeds the
        # next step's MFMA. Halves the launch count of the hot loop.
        use_v2 = (
            x.is_cuda and H == 512 and B <= 32
            and _os.environ.get("DRL_LSTM_BF16_HH", "1") == "1"
            and hasattr(ext, "lstm_step_fwd_bf16")
        )
        h_bfs = None
        used = False
        ctx.persist = False
        if use_v2:
            w_bf = w_hh.detach().to(torch.bfloat16).contiguous()
            h_bfs = torch.empty(T + 1, B, H, dtype=torch.bfloat16, device=dev)
            h_bfs[0].copy_(h0)
            # K5 v3 (persistent whole-sequence, grid barriers): one launch
            # for the entire recurrence instead of T
            if _os.environ.get("DRL_LSTM_PERSISTENT_BF16", "1") == "1" and \
                    hasattr(ext, "lstm_seq_fwd_bf16"):
                ctr = torch.zeros(1, dtype=torch.int32, device=dev)
                used = bool(ext.lstm_seq_fwd_bf16(
                    xp, cs[0], w_bf, hs, cs, h_bfs, acts, tanhc, ctr))
                ctx.persist = used
            if not used:
                used = True
                for t in range(T):
                    if not ext.lstm_step_fwd_bf16(
                            xp[t], h_bfs[t], cs[t], w_bf, hs[t + 1], cs[t + 1],
                            h_bfs[t + 1], acts[t], tanhc[t]):
                        used = False
                        break
        if not used and _os.environ.get("DRL_LSTM_PERSISTENT", "0") == "1":
            ctr = torch.zeros(1, dtype=torch.int32, device=dev)
            used = bool(ext.lstm_seq_persistent(
                xp, hs, cs, w_hh.contiguous(), acts, tanhc, ctr))
        if not used:
            h_bfs = None
            w_hh_t = w_hh.t()
            for t in range(T):
                gates = xp[t].reshape(B, 4 * H)
                gates.addmm_(hs[t], w_hh_t)
                ext.lstm_cell_fwd(gates, cs[t], hs[t + 1], cs[t + 1], acts[t],
                                  tanhc[t])
        ctx.v2 = h_bfs is not None
        if ctx.v2:
            ctx.save_for_backward(x, hs, cs, acts, tanhc, w_ih, w_hh, h_bfs)
        else:
            ctx.save_for_backward(x, hs, cs, acts, tanhc, w_ih, w_hh)
        ctx.dims = (T, B, IN, H)
        return hs[1:].clone(), hs[T].clone(), cs[T].clone()

    @staticmethod
    def backward(ctx, gout, gh_T, gc_T):
        from ..ops import hip_ext

        ext = hip_ext()
        if ctx.v2:
            x, hs, cs, acts, tanhc, w_ih, w_hh, h_bfs = ctx.saved_tensors
        else:
            x, hs, cs, acts, tanhc, w_ih, w_hh = ctx.saved_tensors
            h_bfs = None
        T, B, IN, H = ctx.dims
        dev = x.device
        dgates_all = torch.empty(T, B, 4 * H, device=dev)
        gout = gout.contiguous()
        if ctx.v2:
            # one fused kernel per step: dh GEMM (bf16 MFMA) + cell-bwd —
            # or, when the forward used the persistent kernel, ONE launch
            # for the whole reversed scan
            dg_bf_all = torch.empty(T, B, 4 * H, dtype=torch.bfloat16,
                                    device=dev)
            w_t_bf = w_hh.t().contiguous().to(torch.bfloat16)
            dh_init = gh_T.contiguous()
            done = False
            if ctx.persist and hasattr(ext, "lstm_seq_bwd_bf16"):
                ctr = torch.zeros(1, dtype=torch.int32, device=dev)
                dc = torch.empty(B, H, device=dev)
                done = bool(ext.lstm_seq_bwd_bf16(
                    dh_init, gout, gc_T.contiguous(), w_t_bf, acts, tanhc,
                    cs, dgates_all, dg_bf_all, dc, ctr))
            if not done:
                dc = gc_T.contiguous().clone()
                dc_next = torch.empty(B, H, device=dev)
                empty_bf = torch.empty(0, dtype=torch.bfloat16, device=dev)
                for t in range(T - 1, -1, -1):
                    dg_prev = dg_bf_all[t + 1] if t < T - 1 else empty_bf
                    ext.lstm_step_bwd_bf16(dg_prev, dh_init, gout[t], dc,
                                           w_t_bf, acts[t], tanhc[t], cs[t],
                                           dgates_all[t], dg_bf_all[t],
                                           dc_next)
                    dc, dc_next = dc_next, dc
            dh = dgates_all[0].mm(w_hh)  # grad wrt h0 (fp32, once per seq)
            dg_flat = dgates_all.reshape(T * B, 4 * H)
            dg_bf = dg_bf_all.reshape(T * B, 4 * H)
            dw_hh = dg_bf.t().mm(h_bfs[:-1].reshape(T * B, H)).float()
        else:
            dh = gh_T.contiguous().clone()
            dc = gc_T.contiguous().clone()
            dc_next = torch.empty(B, H, device=dev)
            for t in range(T - 1, -1, -1):
                # the per-step output grad is folded into the cell-bwd kernel
                ext.lstm_cell_bwd(dh.contiguous(), gout[t], dc, acts[t],
                                  tanhc[t], cs[t], dgates_all[t], dc_next)
                dc, dc_next = dc_next, dc
                dh = dgates_all[t].mm(w_hh)
            dg_flat = dgates_all.reshape(T * B, 4 * H)
            dg_bf = None
            dw_hh = dg_flat.t().mm(hs[:-1].reshape(T * B, H))
        db = dg_flat.sum(0)
        if x.dtype == torch.bfloat16:
            # bf16 weight/input grads (fp32-accumulated inside the GEMM,
            # rounded to bf16 on output — same precision class as the conv
            # trunk's bf16 grads that feed the same fp32 master upcast)
            if dg_bf is None:
                dg_bf = dg_flat.to(torch.bfloat16)
            dw_ih = dg_bf.t().mm(x.reshape(T * B, IN)).float()
            dx = dg_bf.mm(w_ih.to(torch.bfloat16)).view(T, B, IN)
        else:
            dw_ih = dg_flat.t().mm(x.reshape(T * B, IN))
            dx = dg_flat.mm(w_ih).view(T, B, IN)
        return dx, dh, dc, dw_ih, dw_hh, db, db


def manual_lstm_seq(x: torch.Tensor, state: Tuple[torch.Tensor, torch.Tensor],
                    lstm: torch.nn.LSTM):
    """Run a single-layer LSTM over (T, B, IN) with fused cell kernels.

    Returns (out (T,B,H), (h_T (1,B,H), c_T (1,B,H))) like nn.LSTM."""
    assert lstm.num_layers == 1 and not lstm.bidirectional
    h0, c0 = state
    x_in = x if x.dtype == torch.bfloat16 else x.float()
    out, hT, cT = _ManualLSTMSeq.apply(
        x_in, h0[0].float(), c0[0].float(), lstm.weight_ih_l0,
        lstm.weight_hh_l0, lstm.bias_ih_l0, lstm.bias_hh_l0,
    )
    return out, (hT.unsqueeze(0), cT.unsqueeze(0))
