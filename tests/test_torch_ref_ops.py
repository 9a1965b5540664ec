"""Numerics of the pure-torch reference ops (the oracles for the HIP kernels)."""

import math

import torch

from distributed_rl_amd.ops import torch_ref as R


def test_nstep_dqn_loss_manual():
    torch.manual_seed(0)
    B, A, n = 4, 6, 3
    gamma = 0.99
    q_s = torch.randn(B, A, requires_grad=True)
    q_sp_on = torch.randn(B, A)
    q_sp_tg = torch.randn(B, A)
    act = torch.randint(0, A, (B,))
    rew = torch.randn(B)
    done = torch.tensor([0.0, 1.0, 0.0, 0.0])
    w = torch.rand(B)
    loss, prio = R.nstep_dqn_loss(q_s, q_sp_on, q_sp_tg, act, rew, done, w,
                                  gamma, n, alpha=0.6)
    # manual row 1 (terminal): target = rew only
    td1 = (rew[1] - q_s[1, act[1]]).clamp(-1, 1)
    assert torch.allclose(prio[1], (td1.abs() + 1e-7) ** 0.6)
    # manual row 0
    a_star = q_sp_on[0].argmax()
    tgt0 = rew[0] + gamma ** 3 * q_sp_tg[0, a_star]
    td0 = (tgt0 - q_s[0, act[0]]).clamp(-1, 1)
    assert torch.allclose(prio[0], (td0.abs() + 1e-7) ** 0.6)
    loss.backward()
    assert q_s.grad is not None
    # gradient only at chosen actions
    mask = torch.zeros(B, A)
    mask[torch.arange(B), act] = 1.0
    assert torch.all((q_s.grad != 0) <= (mask > 0))


def test_value_rescale_roundtrip():
    x = torch.linspace(-50, 50, 101)
    y = R.inv_value_rescale(R.value_rescale(x))
    assert torch.allclose(y, x, atol=1e-3, rtol=1e-4)


def test_vtrace_on_policy_equals_mc_return():
    """With pi == mu, rho = c = 1 and vs_t telescopes to the full
    discounted return + bootstrap."""
    torch.manual_seed(1)
    T, B = 20, 5
    gamma = 0.99
    logp = torch.log_softmax(torch.randn(T, B, 3), -1)[..., 0]
    rew = torch.randn(T, B)
    values = torch.randn(T, B)
    boot = torch.randn(B)
    not_done = torch.ones(B)
    vs, pg_adv, rho = R.vtrace(logp, logp, rew, values, boot, not_done, gamma)
    assert torch.allclose(rho, torch.ones_like(rho))
    ret = boot.clone()
    expect = torch.empty(T, B)
    for t in reversed(range(T)):
        ret = rew[t] + gamma * ret
        expect[t] = ret
    assert torch.allclose(vs, expect, atol=1e-4, rtol=1e-4)


def test_vtrace_terminal_masks_bootstrap():
    T, B = 4, 2
    logp = torch.zeros(T, B)
    rew = torch.zeros(T, B)
    values = torch.zeros(T, B)
    boot = torch.full((B,), 10.0)
    not_done = torch.tensor([1.0, 0.0])
    vs, _, _ = R.vtrace(logp, logp, rew, values, boot, not_done, 0.9)
    assert vs[0, 0] > 5.0  # bootstrap flows back
    assert abs(vs[0, 1]) < 1e-6  # terminal: no bootstrap


def test_vtrace_clipping_caps_rho():
    T, B = 6, 3
    b_logp = torch.full((T, B), -3.0)
    t_logp = torch.zeros(T, B)  # ratio e^3 >> 1
    rew = torch.rand(T, B)
    values = torch.rand(T, B)
    vs, pg, rho = R.vtrace(b_logp, t_logp, rew, values, torch.rand(B),
                           torch.ones(B), 0.99, rho_bar=1.0, c_bar=1.0)
    assert torch.allclose(rho, torch.ones_like(rho))


def test_sequence_priority():
    td = torch.tensor([[1.0, 0.0], [3.0, 2.0], [2.0, 0.0]])  # (T=3, B=2)
    p = R.sequence_priority(td, alpha=1.0, eta=0.9)
    assert torch.allclose(p[0], torch.tensor(0.9 * 3.0 + 0.1 * 2.0))
    assert torch.allclose(p[1], torch.tensor(0.9 * 2.0 + 0.1 * (2.0 / 3.0)))


def test_fold_nstep_reward():
    r = torch.tensor([[1.0, 1.0, 1.0]])
    out = R.fold_nstep_reward(r, 0.5)
    assert torch.allclose(out, torch.tensor([1.75]))


def test_impala_loss_shapes_and_entropy():
    torch.manual_seed(2)
    N, A = 64, 6
    logits = torch.randn(N, A, requires_grad=True)
    values = torch.randn(N, requires_grad=True)
    actions = torch.randint(0, A, (N,))
    pg_adv = torch.randn(N)
    vs = torch.randn(N)
    total, obj, critic, ent = R.impala_loss(logits, values, actions, pg_adv, vs, 0.01)
    assert 0 < ent.item() <= math.log(A) + 1e-5
    total.backward()
    assert logits.grad is not None and values.grad is not None


def test_vtrace_bt_cpu_matches_tmajor_composition():
    """The (B,T)-layout wrapper (CPU fallback path) equals the T-major
    oracle with explicit transposes and mu.log()."""
    from distributed_rl_amd import ops

    torch.manual_seed(4)
    B, T = 6, 12
    mu = torch.rand(B, T).clamp_min(1e-3)
    tlogp = -torch.rand(B, T)
    rew = torch.randn(B, T)
    val = torch.randn(B, T)
    boot = torch.randn(B)
    nd = (torch.rand(B) < 0.7).float()
    vs, pg = ops.vtrace_bt(mu, tlogp, rew, val, boot, nd, 0.97,
                           rho_bar=1.0, c_bar=1.0, lam=0.9)
    vs_r, pg_r, _ = R.vtrace(mu.log().t().contiguous(), tlogp.t().contiguous(),
                             rew.t().contiguous(), val.t().contiguous(),
                             boot, nd, 0.97, rho_bar=1.0, c_bar=1.0, lam=0.9)
    assert torch.allclose(vs, vs_r.t(), atol=1e-5)
    assert torch.allclose(pg, pg_r.t(), atol=1e-5)
