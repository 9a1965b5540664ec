"""GPU end-to-end steps for IMPALA and R2D2 learners."""

import copy

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_impala_gpu_step():
    from distributed_rl_amd.algos.impala import ImpalaLearner
    from distributed_rl_amd.config import Config, load_config

    raw = copy.deepcopy(load_config("impala").raw)
    raw["BATCHSIZE"] = 16
    cfg = Config(raw=raw)
    learner = ImpalaLearner(cfg, device=DEV, enable_tb=False,
                            replay_capacity=256)
    B, T = 32, cfg.unroll_step
    cols = {
        "states": torch.randint(0, 255, (B, T + 1, 4, 84, 84),
                                dtype=torch.uint8, device=DEV),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32, device=DEV),
        "mu": torch.full((B, T), 1 / 6, device=DEV),
        "rewards": torch.randn(B, T, device=DEV),
        "not_done": torch.ones(B, device=DEV),
    }
    learner.push_trajectories(cols)
    for _ in range(3):
        stats = learner.step()
    torch.cuda.synchronize()
    assert np.isfinite(float(stats["loss"]))
    assert 0 < float(stats["entropy"]) <= np.log(6) + 1e-3


def test_impala_gpu_graphed_step():
    from distributed_rl_amd.algos.impala import ImpalaLearner
    from distributed_rl_amd.config import Config, load_config

    raw = copy.deepcopy(load_config("impala").raw)
    raw["BATCHSIZE"] = 16
    cfg = Config(raw=raw)
    learner = ImpalaLearner(cfg, device=DEV, enable_tb=False,
                            replay_capacity=128)
    B, T = 64, cfg.unroll_step
    cols = {
        "states": torch.randint(0, 255, (B, T + 1, 4, 84, 84),
                                dtype=torch.uint8, device=DEV),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32, device=DEV),
        "mu": torch.full((B, T), 1 / 6, device=DEV),
        "rewards": torch.randn(B, T, device=DEV),
        "not_done": torch.ones(B, device=DEV),
    }
    learner.push_trajectories(cols)
    stepper = learner.make_graphed_step()
    for _ in range(3):
        out = stepper()
    torch.cuda.synchronize()
    assert np.isfinite(float(out["loss"]))
    assert learner.step_count == 3


def test_r2d2_gpu_step():
    from distributed_rl_amd.algos.r2d2 import R2D2Learner
    from distributed_rl_amd.config import Config, load_config

    raw = copy.deepcopy(load_config("r2d2").raw)
    raw["BATCHSIZE"] = 8
    raw["FIXED_TRAJECTORY"] = 32
    raw["MEM"] = 8
    cfg = Config(raw=raw)
    learner = R2D2Learner(cfg, device=DEV, enable_tb=False,
                          replay_capacity=64)
    B, T, H = 16, 32, 512
    cols = {
        "h0": torch.randn(B, 2, H, device=DEV) * 0.01,
        "states": torch.randint(0, 255, (B, T, 4, 84, 84), dtype=torch.uint8,
                                device=DEV),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32, device=DEV),
        "rewards": torch.randn(B, T, device=DEV),
        "done": torch.zeros(B, device=DEV),
    }
    learner.push_sequences(cols, torch.rand(B, device=DEV) + 0.1)
    for _ in range(2):
        stats = learner.step()
    torch.cuda.synchronize()
    assert np.isfinite(float(stats["loss"]))
    assert learner.replay.total_priority > 0


def test_manual_lstm_matches_nn_lstm():
    from distributed_rl_amd.models.manual_lstm import manual_lstm_seq

    torch.manual_seed(11)
    T, B, IN, H = 12, 5, 64, 32
    lstm = torch.nn.LSTM(IN, H).to(DEV)
    x = torch.randn(T, B, IN, device=DEV, requires_grad=True)
    h0 = torch.randn(1, B, H, device=DEV)
    c0 = torch.randn(1, B, H, device=DEV)
    out, (hT, cT) = manual_lstm_seq(x, (h0, c0), lstm)
    ref_out, (rhT, rcT) = lstm(x, (h0, c0))
    assert torch.allclose(out, ref_out, atol=1e-4), (out - ref_out).abs().max()
    assert torch.allclose(hT, rhT, atol=1e-4)
    assert torch.allclose(cT, rcT, atol=1e-4)
    # backward vs autograd through nn.LSTM
    g = torch.randn_like(out)
    out.backward(g)
    gx = x.grad.clone()
    gw = lstm.weight_hh_l0.grad.clone()
    x.grad = None
    for p in lstm.parameters():
        p.grad = None
    ref_out2, _ = lstm(x, (h0, c0))
    ref_out2.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-3), (gx - x.grad).abs().max()
    assert torch.allclose(gw, lstm.weight_hh_l0.grad, atol=1e-3)


def test_manual_lstm_bf16_input_close_to_fp32():
    """Round-2 prototype (DRL_LSTM_BF16_IN): bf16 input-projection GEMMs,
    fp32 recurrence. Values/grads must track the fp32 path within bf16
    rounding (outputs are tanh-bounded, so absolute tolerance is safe)."""
    from distributed_rl_amd.models.manual_lstm import manual_lstm_seq

    torch.manual_seed(13)
    T, B, IN, H = 8, 4, 64, 32
    lstm = torch.nn.LSTM(IN, H).to(DEV)
    h0 = torch.randn(1, B, H, device=DEV)
    c0 = torch.randn(1, B, H, device=DEV)
    x32 = torch.randn(T, B, IN, device=DEV)
    xbf = x32.to(torch.bfloat16).requires_grad_(True)
    x = x32.clone().requires_grad_(True)

    out, (hT, cT) = manual_lstm_seq(x, (h0, c0), lstm)
    g = torch.randn_like(out)
    out.backward(g)
    gw_ih = lstm.weight_ih_l0.grad.clone()
    gx = x.grad.clone()
    for p in lstm.parameters():
        p.grad = None

    out_bf, (hT_bf, cT_bf) = manual_lstm_seq(xbf, (h0, c0), lstm)
    assert out_bf.dtype == torch.float32  # recurrence stays fp32
    out_bf.backward(g)
    assert xbf.grad.dtype == torch.bfloat16

    assert torch.allclose(out_bf, out, atol=0.05), (out_bf - out).abs().max()
    assert torch.allclose(hT_bf, hT, atol=0.05)
    assert torch.allclose(cT_bf, cT, atol=0.08)
    assert torch.allclose(xbf.grad.float(), gx, atol=0.08,
                          rtol=0.05), (xbf.grad.float() - gx).abs().max()
    rel = (lstm.weight_ih_l0.grad - gw_ih).abs().max() / gw_ih.abs().max()
    assert rel < 0.05, rel


def test_r2d2_gpu_graphed_step():
    from distributed_rl_amd.algos.r2d2 import R2D2Learner
    from distributed_rl_amd.config import Config, load_config

    raw = copy.deepcopy(load_config("r2d2").raw)
    raw["BATCHSIZE"] = 8
    raw["FIXED_TRAJECTORY"] = 32
    raw["MEM"] = 8
    cfg = Config(raw=raw)
    learner = R2D2Learner(cfg, device=DEV, enable_tb=False, replay_capacity=64)
    B, T, H = 32, 32, 512
    cols = {
        "h0": torch.randn(B, 2, H, device=DEV) * 0.01,
        "states": torch.randint(0, 255, (B, T, 4, 84, 84), dtype=torch.uint8,
                                device=DEV),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32, device=DEV),
        "rewards": torch.randn(B, T, device=DEV),
        "done": torch.zeros(B, device=DEV),
    }
    learner.push_sequences(cols, torch.rand(B, device=DEV) + 0.1)
    stepper = learner.make_graphed_step()
    for _ in range(3):
        out = stepper()
    torch.cuda.synchronize()
    assert np.isfinite(float(out["loss"]))
    assert learner.step_count == 3


def test_manual_lstm_bf16_hh_close_to_fp32():
    """K5 v2 (one kernel per step, bf16-MFMA hh GEMM): only the GEMM
    operands (h, dgates, W) are bf16-rounded; gates/cell/saves stay fp32.
    Requires H=512 (the R2D2 production width). Values and grads must track
    the fp32 2-kernel path within bf16 rounding."""
    import os

    from distributed_rl_amd.models.manual_lstm import manual_lstm_seq

    torch.manual_seed(17)
    T, B, IN, H = 16, 8, 96, 512
    lstm = torch.nn.LSTM(IN, H).to(DEV)
    h0 = torch.randn(1, B, H, device=DEV) * 0.5
    c0 = torch.randn(1, B, H, device=DEV) * 0.5
    x_a = torch.randn(T, B, IN, device=DEV, requires_grad=True)
    x_b = x_a.detach().clone().requires_grad_(True)
    g = torch.randn(T, B, H, device=DEV)

    os.environ["DRL_LSTM_BF16_HH"] = "0"
    try:
        out_ref, (hT_ref, cT_ref) = manual_lstm_seq(x_a, (h0, c0), lstm)
        out_ref.backward(g)
        gw_hh = lstm.weight_hh_l0.grad.clone()
        gb = lstm.bias_ih_l0.grad.clone()
        gx = x_a.grad.clone()
        for p in lstm.parameters():
            p.grad = None
    finally:
        os.environ["DRL_LSTM_BF16_HH"] = "1"

    out, (hT, cT) = manual_lstm_seq(x_b, (h0, c0), lstm)
    assert out.dtype == torch.float32
    out.backward(g)

    assert torch.allclose(out, out_ref, atol=0.06), (out - out_ref).abs().max()
    assert torch.allclose(hT, hT_ref, atol=0.06)
    assert torch.allclose(cT, cT_ref, atol=0.10)
    assert torch.allclose(x_b.grad, gx, atol=0.10, rtol=0.05), \
        (x_b.grad - gx).abs().max()
    rel_w = (lstm.weight_hh_l0.grad - gw_hh).abs().max() / gw_hh.abs().max()
    assert rel_w < 0.06, rel_w
    rel_b = (lstm.bias_ih_l0.grad - gb).abs().max() / gb.abs().max()
    assert rel_b < 0.06, rel_b


def test_lstm_step_bf16_kernels_match_composition():
    """Exact-ish oracle for the fused kernels: one step's fwd/bwd vs the
    same math composed from torch ops with identically bf16-rounded
    operands (differences only from MFMA vs torch accumulation order)."""
    from distributed_rl_amd.ops import hip_ext

    ext = hip_ext(required=True)
    torch.manual_seed(23)
    B, H = 32, 512
    xp = torch.randn(B, 4 * H, device=DEV)
    h = torch.randn(B, H, device=DEV)
    c = torch.randn(B, H, device=DEV)
    w_hh = torch.randn(4 * H, H, device=DEV) * 0.05
    h_bf = h.to(torch.bfloat16)
    w_bf = w_hh.to(torch.bfloat16)

    h_out = torch.empty(B, H, device=DEV)
    c_out = torch.empty(B, H, device=DEV)
    h_bf_out = torch.empty(B, H, dtype=torch.bfloat16, device=DEV)
    acts = torch.empty(B, 4 * H, device=DEV)
    tanhc = torch.empty(B, H, device=DEV)
    assert ext.lstm_step_fwd_bf16(xp, h_bf, c, w_bf, h_out, c_out, h_bf_out,
                                  acts, tanhc)
    # composition oracle with the same bf16-rounded operands
    gates = xp + h_bf.float().mm(w_bf.float().t())
    i = torch.sigmoid(gates[:, :H])
    f = torch.sigmoid(gates[:, H:2 * H])
    gg = torch.tanh(gates[:, 2 * H:3 * H])
    o = torch.sigmoid(gates[:, 3 * H:])
    c_ref = f * c + i * gg
    h_ref = o * torch.tanh(c_ref)
    assert torch.allclose(c_out, c_ref, atol=2e-3), (c_out - c_ref).abs().max()
    assert torch.allclose(h_out, h_ref, atol=2e-3)
    assert torch.allclose(h_bf_out.float(), h_ref, atol=8e-3)

    # backward kernel vs composition
    dg_prev = torch.randn(B, 4 * H, device=DEV).to(torch.bfloat16)
    gout = torch.randn(B, H, device=DEV)
    dc_in = torch.randn(B, H, device=DEV)
    w_t_bf = w_hh.t().contiguous().to(torch.bfloat16)
    dgates = torch.empty(B, 4 * H, device=DEV)
    dg_bf = torch.empty(B, 4 * H, dtype=torch.bfloat16, device=DEV)
    dc_out = torch.empty(B, H, device=DEV)
    assert ext.lstm_step_bwd_bf16(dg_prev, gout, gout, dc_in, w_t_bf, acts,
                                  tanhc, c, dgates, dg_bf, dc_out)
    dh_ref = dg_prev.float().mm(w_bf.float()) + gout
    do_ = dh_ref * tanhc
    dct = dc_in + dh_ref * o * (1 - tanhc * tanhc)
    d_ref = torch.cat([
        dct * gg * i * (1 - i),
        dct * c * f * (1 - f),
        dct * i * (1 - gg * gg),
        do_ * o * (1 - o),
    ], dim=1)
    assert torch.allclose(dgates, d_ref, atol=2e-3), (dgates - d_ref).abs().max()
    assert torch.allclose(dc_out, dct * f, atol=2e-3)


def test_r2d2_fused_seq_loss_matches_torch():
    """Fused R2D2 target/loss/priority kernels vs the torch composition
    (nstep_recurrent_targets + IS loss + sequence_priority) on identical
    inputs — values, priorities and the q_train gradient."""
    from distributed_rl_amd import ops
    from distributed_rl_amd.algos.r2d2 import ETA, nstep_recurrent_targets

    torch.manual_seed(3)
    T, B, A, m, n_step, gamma = 16, 8, 6, 4, 5, 0.997
    q_train = (torch.randn(T - m, B, A, device=DEV) * 2).requires_grad_(True)
    q_train2 = q_train.detach().clone().requires_grad_(True)
    q_tgt = torch.randn(T, B, A, device=DEV) * 2
    actions = torch.randint(0, A, (T, B), device=DEV, dtype=torch.int32)
    rewards = torch.randn(T, B, device=DEV)
    done = (torch.rand(B, device=DEV) < 0.3).float()
    weights = torch.rand(B, device=DEV) + 0.5

    loss, prio, value, td_abs = ops.r2d2_sequence_loss(
        q_train, q_tgt, actions, rewards, done, weights, m, n_step, gamma,
        0.9, ETA, True)
    loss.backward()

    q_full = torch.cat([q_tgt[:m], q_train2], dim=0)
    td, q_taken, _ = nstep_recurrent_targets(
        q_full, q_tgt, actions.long(), rewards, done, m, n_step, gamma, True)
    loss2 = 0.5 * (weights * td.pow(2).mean(dim=0)).mean()
    prio2 = ops.sequence_priority(td.detach().abs(), 0.9, ETA)
    loss2.backward()
    torch.cuda.synchronize()

    assert torch.allclose(loss, loss2, atol=1e-4, rtol=1e-4), \
        (float(loss), float(loss2))
    assert torch.allclose(prio, prio2, atol=1e-4, rtol=1e-3)
    assert torch.allclose(value, q_taken.detach().mean(), atol=1e-4)
    assert torch.allclose(td_abs, td.detach().abs().mean(), atol=1e-4)
    assert torch.allclose(q_train.grad, q_train2.grad, atol=1e-5), \
        (q_train.grad - q_train2.grad).abs().max()
