"""HIP kernel numerics vs the plain-PyTorch fp32 references (torch_ref)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(scope="module")
def ext():
    from distributed_rl_amd.ops import hip_ext

    return hip_ext(required=True)


def test_dequant_bf16(ext):
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    x = torch.randint(0, 256, (33, 4, 84, 84), dtype=torch.uint8, device=DEV)
    y = ops.dequant_frames(x, torch.bfloat16)
    ref = torch_ref.dequant_frames(x.float().cpu()).to(torch.bfloat16)
    assert y.dtype == torch.bfloat16
    assert torch.allclose(y.float().cpu(), ref.float(), atol=1 / 255)
    y32 = ops.dequant_frames(x, torch.float32)
    assert torch.allclose(y32.cpu(), torch_ref.dequant_frames(x.cpu().float()) / 1.0,
                          atol=1e-7)


def test_sumtree_total_and_update(ext):
    from distributed_rl_amd.replay.gpu_per import HipSumTreePER

    schema = {"x": ((), torch.float32)}
    per = HipSumTreePER(1000, schema, DEV)
    prios = torch.rand(1000, device=DEV) + 0.01
    per.push({"x": torch.arange(1000.0, device=DEV)}, prios)
    assert abs(per.total_priority - float(prios.sum())) < 1e-2
    # update a subset (with duplicate indices — atomicExch path)
    idx = torch.tensor([3, 3, 500, 999], device=DEV)
    newp = torch.tensor([5.0, 7.0, 1.0, 2.0], device=DEV)
    per.update(idx, newp)
    torch.cuda.synchronize()
    expect = prios.clone()
    expect[3] = 7.0  # last write wins
    expect[500] = 1.0
    expect[999] = 2.0
    assert abs(per.total_priority - float(expect.sum())) < 1e-2
    leaf = per.tree[per.P + 3].item()
    assert leaf == 7.0


def test_sumtree_sampling_distribution(ext):
    from distributed_rl_amd.replay.gpu_per import HipSumTreePER

    schema = {"x": ((), torch.float32)}
    per = HipSumTreePER(4, schema, DEV)
    prios = torch.tensor([1.0, 2.0, 3.0, 4.0], device=DEV)
    per.push({"x": torch.arange(4.0, device=DEV)}, prios)
    counts = torch.zeros(4, device=DEV)
    for _ in range(40):
        _, idx, _ = per.sample(1000, beta=0.4)
        counts += torch.bincount(idx, minlength=4).float()
    freq = (counts / counts.sum()).cpu()
    expect = (prios / prios.sum()).cpu()
    assert torch.allclose(freq, expect, atol=0.02), (freq, expect)


def test_sumtree_is_weights_vs_oracle(ext):
    from distributed_rl_amd.replay.gpu_per import HipSumTreePER
    from distributed_rl_amd.ops import torch_ref

    schema = {"x": ((), torch.float32)}
    per = HipSumTreePER(64, schema, DEV)
    prios = torch.rand(64, device=DEV) + 0.1
    per.push({"x": torch.zeros(64, device=DEV)}, prios)
    beta = 0.4
    _, idx, w = per.sample(256, beta=beta)
    torch.cuda.synchronize()
    probs = prios[idx] / prios.sum()
    expect_max_w = (1.0 / (64 * (prios.min() / prios.sum()))) ** beta
    expect = (1.0 / (64 * probs)) ** beta / expect_max_w
    assert torch.allclose(w, expect, rtol=1e-3), (w[:5], expect[:5])


def test_dqn_loss_matches_ref(ext):
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    torch.manual_seed(0)
    B, A, n = 64, 6, 3
    gamma, alpha = 0.99, 0.6
    q_s = torch.randn(B, A, device=DEV, requires_grad=True)
    q_sp_on = torch.randn(B, A, device=DEV)
    q_sp_tg = torch.randn(B, A, device=DEV)
    act = torch.randint(0, A, (B,), device=DEV)
    rew = torch.randn(B, device=DEV)
    done = (torch.rand(B, device=DEV) < 0.2).float()
    w = torch.rand(B, device=DEV)

    loss, prio = ops.nstep_dqn_loss(q_s, q_sp_on, q_sp_tg, act, rew, done, w,
                                    gamma, n, alpha)
    loss.backward()
    g_hip = q_s.grad.clone()

    q_s2 = q_s.detach().cpu().requires_grad_(True)
    loss_ref, prio_ref = torch_ref.nstep_dqn_loss(
        q_s2, q_sp_on.cpu(), q_sp_tg.cpu(), act.cpu(), rew.cpu(), done.cpu(),
        w.cpu(), gamma, n, alpha)
    loss_ref.backward()
    assert abs(loss.item() - loss_ref.item()) < 1e-5
    assert torch.allclose(prio.cpu(), prio_ref, atol=1e-5)
    assert torch.allclose(g_hip.cpu(), q_s2.grad, atol=1e-6)


def test_vtrace_matches_ref(ext):
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    torch.manual_seed(1)
    T, B = 20, 32
    blogp = -torch.rand(T, B, device=DEV)
    tlogp = -torch.rand(T, B, device=DEV)
    rew = torch.randn(T, B, device=DEV)
    val = torch.randn(T, B, device=DEV)
    boot = torch.randn(B, device=DEV)
    nd = (torch.rand(B, device=DEV) < 0.8).float()
    vs, pg, rho = ops.vtrace(blogp, tlogp, rew, val, boot, nd, 0.99)
    vs_r, pg_r, rho_r = torch_ref.vtrace(
        blogp.cpu(), tlogp.cpu(), rew.cpu(), val.cpu(), boot.cpu(), nd.cpu(), 0.99)
    assert torch.allclose(vs.cpu(), vs_r, atol=1e-4)
    assert torch.allclose(pg.cpu(), pg_r, atol=1e-4)
    assert torch.allclose(rho.cpu(), rho_r, atol=1e-5)


def test_vtrace_bt_matches_ref(ext):
    """(B,T)-layout V-trace (raw mu probs, in-kernel log) vs the T-major
    oracle composition."""
    from distributed_rl_amd import ops

    torch.manual_seed(2)
    B, T = 48, 20
    mu = torch.rand(B, T, device=DEV).clamp_min(1e-3)
    tlogp = -torch.rand(B, T, device=DEV)
    rew = torch.randn(B, T, device=DEV)
    val = torch.randn(B, T, device=DEV)
    boot = torch.randn(B, device=DEV)
    nd = (torch.rand(B, device=DEV) < 0.8).float()
    vs, pg = ops.vtrace_bt(mu, tlogp, rew, val, boot, nd, 0.99,
                           rho_bar=1.0, c_bar=1.0, lam=0.95)
    vs_c, pg_c = ops.vtrace_bt(mu.cpu(), tlogp.cpu(), rew.cpu(), val.cpu(),
                               boot.cpu(), nd.cpu(), 0.99,
                               rho_bar=1.0, c_bar=1.0, lam=0.95)
    assert vs.shape == (B, T) and pg.shape == (B, T)
    assert torch.allclose(vs.cpu(), vs_c, atol=1e-4)
    assert torch.allclose(pg.cpu(), pg_c, atol=1e-4)


def test_impala_fused_loss_matches_torch(ext):
    """One-kernel total loss (pg obj + entropy + critic MSE) + one-kernel
    whole-head backward vs the plain torch composition on the raw
    (B*(T+1), A+1) head output."""
    import torch.nn.functional as F
    from distributed_rl_amd import ops

    torch.manual_seed(3)
    B, T, A = 16, 16, 6
    N = B * T
    out = torch.randn(B * (T + 1), A + 1, device=DEV, requires_grad=True)
    actions = torch.randint(0, A, (N,), device=DEV)
    adv = torch.randn(N, device=DEV)
    vs = torch.randn(B, T, device=DEV)
    er = 0.01
    logits3 = out.view(B, T + 1, A + 1)[:, :T, :A]
    logits_flat = logits3.detach().reshape(N, A).contiguous()
    v_t = out.view(B, T + 1, A + 1)[:, :T, A].detach().contiguous()
    stats = ops.policy_softmax_stats(logits_flat, actions)
    loss, obj, critic = ops.impala_fused_loss(out, v_t, stats, actions, adv,
                                              vs, er, T)
    loss.backward()

    out2 = out.detach().clone().requires_grad_(True)
    logits2 = out2.view(B, T + 1, A + 1)[:, :T, :A].reshape(N, A)
    v2 = out2.view(B, T + 1, A + 1)[:, :T, A]
    log_pi = torch.log_softmax(logits2, -1)
    pi = log_pi.exp()
    entropy = -(pi * log_pi).sum(-1).mean()
    logpa = log_pi.gather(1, actions.long().unsqueeze(1)).squeeze(1)
    obj_r = (logpa * adv).mean() + er * entropy
    critic_r = 0.5 * F.mse_loss(v2, vs)
    loss_r = -obj_r + critic_r
    loss_r.backward()

    assert abs(loss.item() - loss_r.item()) < 1e-4
    assert abs(obj.item() - obj_r.item()) < 1e-4
    assert abs(critic.item() - critic_r.item()) < 1e-4
    assert torch.allclose(out.grad, out2.grad, atol=1e-5)


def test_value_rescale_matches_ref(ext):
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    x = torch.linspace(-40, 40, 4096, device=DEV)
    y = ops.value_rescale(x)
    assert torch.allclose(y.cpu(), torch_ref.value_rescale(x.cpu()), atol=1e-5)
    z = ops.inv_value_rescale(y)
    assert torch.allclose(z.cpu(), x.cpu(), atol=1e-2)


def test_seq_priority_matches_ref(ext):
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    td = torch.rand(80, 32, device=DEV)
    p = ops.sequence_priority(td, alpha=0.9)
    p_ref = torch_ref.sequence_priority(td.cpu(), alpha=0.9)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-5)


def test_grad_clip_matches_torch(ext):
    from distributed_rl_amd import ops

    x = torch.randn(100_000, device=DEV) * 3
    ref = x.clone()
    norm = ref.norm(2)
    max_norm = 40.0
    if norm > max_norm:
        ref.mul_(max_norm / (norm + 1e-6))
    ops.clip_flat_grad_(x, max_norm)
    assert torch.allclose(x, ref, rtol=1e-5)
    # below the clip threshold: untouched
    y = torch.randn(1000, device=DEV) * 0.001
    y0 = y.clone()
    ops.clip_flat_grad_(y, max_norm)
    assert torch.equal(y, y0)


def test_dueling_fused_vs_eager(ext):
    from distributed_rl_amd import ops

    torch.manual_seed(9)
    B, A = 128, 6
    adv = torch.randn(B, A, device=DEV, requires_grad=True)
    val = torch.randn(B, 1, device=DEV, requires_grad=True)
    out = ops.dueling_head(adv, val)
    ref = (adv + val) - adv.mean(dim=-1, keepdim=True)
    assert torch.allclose(out, ref, atol=1e-6)
    g = torch.randn_like(out)
    out.backward(g, retain_graph=False)
    ga, gv = adv.grad.clone(), val.grad.clone()
    adv.grad = None
    val.grad = None
    ref2 = (adv + val) - adv.mean(dim=-1, keepdim=True)
    ref2.backward(g)
    assert torch.allclose(ga, adv.grad, atol=1e-5)
    assert torch.allclose(gv, val.grad, atol=1e-5)


def test_model_dueling_fused_matches(ext):
    from distributed_rl_amd.config import load_config
    from distributed_rl_amd.models import BaseAgent

    net = BaseAgent(load_config("ape_x").model_info).to(DEV)
    assert net._dueling  # pattern detected
    x = torch.rand(4, 4, 84, 84, device=DEV)
    y_gpu = net.forward([x])[0]
    y_cpu = net.cpu().forward([x.cpu()])[0]
    assert torch.allclose(y_gpu.cpu(), y_cpu, atol=1e-4)


def test_policy_objective_fused_vs_torch(ext):
    from distributed_rl_amd import ops

    torch.manual_seed(13)
    N, A = 256, 6
    er = 0.01
    logits = torch.randn(N, A, device=DEV, requires_grad=True)
    actions = torch.randint(0, A, (N,), device=DEV)
    adv = torch.randn(N, device=DEV)

    stats = ops.policy_softmax_stats(logits, actions)
    logpa, pi, H, ent = stats
    ref_logpi = torch.log_softmax(logits.detach(), -1)
    assert torch.allclose(logpa, ref_logpi.gather(1, actions.unsqueeze(1)
                                                  ).squeeze(1), atol=1e-5)
    assert torch.allclose(pi, ref_logpi.exp(), atol=1e-5)
    ref_H = -(ref_logpi.exp() * ref_logpi).sum(-1)
    assert torch.allclose(H, ref_H, atol=1e-5)
    assert abs(ent.item() - ref_H.mean().item()) < 1e-5

    obj, ent2 = ops.impala_policy_objective(logits, actions, adv, er,
                                            stats=stats)
    (-obj).backward()
    g_fused = logits.grad.clone()

    logits2 = logits.detach().clone().requires_grad_(True)
    lp = torch.log_softmax(logits2, -1)
    p2 = lp.exp()
    entropy = -(p2 * lp).sum(-1).mean()
    lpa = lp.gather(1, actions.unsqueeze(1)).squeeze(1)
    obj_ref = (lpa * adv).mean() + er * entropy
    assert abs(obj.item() - obj_ref.item()) < 1e-5
    (-obj_ref).backward()
    assert torch.allclose(g_fused, logits2.grad, atol=1e-5), (
        (g_fused - logits2.grad).abs().max())


def test_fused_rmsprop_matches_torch(ext):
    torch.manual_seed(21)
    n = 100_000
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)
    from distributed_rl_amd.parallel.flat_optim import FlatRMSprop

    opt = FlatRMSprop([p], [g], lr=6.25e-5, alpha=0.95, eps=1.5e-7,
                      weight_decay=0.0, momentum=0.0, centered=True)
    p_ref = torch.nn.Parameter(p.clone())
    ref = torch.optim.RMSprop([p_ref], lr=6.25e-5, alpha=0.95, eps=1.5e-7,
                              centered=True, foreach=True)
    for it in range(5):
        gcur = torch.randn(n, device=DEV, generator=torch.Generator(DEV
                           ).manual_seed(it))
        g.copy_(gcur)
        p_ref.grad = gcur.clone()
        opt.step()
        ref.step()
    torch.cuda.synchronize()
    assert torch.allclose(p, p_ref.detach(), atol=1e-6), (
        (p - p_ref.detach()).abs().max())


def test_fused_adam_matches_torch(ext):
    torch.manual_seed(22)
    n = 50_000
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)
    from distributed_rl_amd.parallel.flat_optim import FlatAdam

    opt = FlatAdam([p], [g], lr=1e-4, eps=1e-3)
    p_ref = torch.nn.Parameter(p.clone())
    ref = torch.optim.Adam([p_ref], lr=1e-4, eps=1e-3, foreach=True)
    for it in range(5):
        gcur = torch.randn(n, device=DEV, generator=torch.Generator(DEV
                           ).manual_seed(100 + it))
        g.copy_(gcur)
        p_ref.grad = gcur.clone()
        opt.step()
        ref.step()
    torch.cuda.synchronize()
    assert torch.allclose(p, p_ref.detach(), atol=1e-6), (
        (p - p_ref.detach()).abs().max())


def test_sumtree_deep_tree_large_capacity(ext):
    """Tree depth / large-capacity path (BASELINE config 4 sizes the replay
    for 288 GB HBM): 4M-slot tree, distribution still proportional."""
    from distributed_rl_amd.replay.gpu_per import HipSumTreePER

    cap = 4_000_000
    per = HipSumTreePER(cap, {"x": ((), torch.float32)}, DEV)
    n = 300_000
    prios = torch.rand(n, device=DEV) + 0.01
    per.push({"x": torch.zeros(n, device=DEV)}, prios)
    torch.cuda.synchronize()
    assert abs(per.total_priority - float(prios.sum())) / float(prios.sum()) < 1e-3
    data, idx, w = per.sample(4096, beta=0.4)
    torch.cuda.synchronize()
    assert idx.max() < n and idx.min() >= 0
    # high-priority items sampled more often than low: split-half check
    hi = prios[idx] > prios.median()
    assert hi.float().mean() > 0.55


def test_dqn_loss_bf16_inputs(ext):
    """The fused loss reads bf16 Q tensors directly (no cast kernels)."""
    from distributed_rl_amd import ops
    from distributed_rl_amd.ops import torch_ref

    torch.manual_seed(31)
    B, A = 128, 6
    q_s = torch.randn(B, A, device=DEV).to(torch.bfloat16).requires_grad_(True)
    q_on = torch.randn(B, A, device=DEV).to(torch.bfloat16)
    q_tg = torch.randn(B, A, device=DEV).to(torch.bfloat16)
    act = torch.randint(0, A, (B,), device=DEV)
    rew = torch.randn(B, device=DEV)
    done = torch.zeros(B, device=DEV)
    w = torch.rand(B, device=DEV)
    loss, prio, qmean = ops.nstep_dqn_loss(q_s, q_on, q_tg, act, rew, done, w,
                                           0.99, 3, 0.6, with_value_stat=True)
    loss.backward()
    assert q_s.grad.dtype == torch.bfloat16
    loss_ref, prio_ref = torch_ref.nstep_dqn_loss(
        q_s.detach().float().cpu(), q_on.float().cpu(), q_tg.float().cpu(),
        act.cpu(), rew.cpu(), done.cpu(), w.cpu(), 0.99, 3, 0.6)
    assert abs(loss.item() - loss_ref.item()) < 1e-3
    assert torch.allclose(prio.cpu(), prio_ref, atol=1e-3)
    qm_ref = q_s.detach().float().max(1).values.mean()
    assert abs(qmean.item() - qm_ref.item()) < 1e-3


def test_linear_relu_matches_torch():
    """Own MFMA Linear+bias+ReLU vs F.relu(F.linear) on identical bf16
    operands, forward + backward."""
    import torch.nn.functional as F

    from distributed_rl_amd import ops

    if not ops.linear_relu_supported(3136, 1024):
        import pytest

        pytest.skip("no instantiation")
    torch.manual_seed(12)
    dev = "cuda:0"
    M, K, N = 512, 3136, 1024
    x = (torch.randn(M, K, device=dev) * 0.5).to(torch.bfloat16
                                                 ).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    w = (torch.randn(N, K, device=dev) * 0.02).to(torch.bfloat16
                                                  ).requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b = torch.randn(N, device=dev).to(torch.bfloat16).requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)

    out = ops.fused_linear_relu(x, w, b)
    ref = F.relu(F.linear(x2, w2, b2))
    torch.cuda.synchronize()
    assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2), \
        (out.float() - ref.float()).abs().max()

    g = torch.randn_like(ref)
    out.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=2e-2,
                          rtol=2e-2)
    rel = (w.grad.float() - w2.grad.float()).abs().max() / \
        w2.grad.float().abs().max()
    assert rel < 0.05, rel
    assert torch.allclose(b.grad.float(), b2.grad.float(), atol=2e-2,
                          rtol=2e-2)
