"""Phase-by-phase kernel exerciser to localize GPU faults.

Run with AMD_SERIALIZE_KERNEL=3 HIP_LAUNCH_BLOCKING=1 so each launch is
synchronous and the first faulting kernel is identified by the last-printed
phase marker.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

def main():
    dev = "cuda:0"
    from distributed_rl_amd.ops import hip_ext

    ext = hip_ext(required=True)
    sync = torch.cuda.synchronize

    print("phase: dequant", flush=True)
    x = torch.randint(0, 256, (8, 4, 84, 84), dtype=torch.uint8, device=dev)
    out = torch.empty(x.shape, dtype=torch.bfloat16, device=dev)
    ext.dequant(x, out)
    sync()
    print("  ok mean=", out.float().mean().item(), flush=True)

    print("phase: sumtree_update", flush=True)
    P = 1024
    tree = torch.zeros(2 * P, device=dev)
    idx = torch.arange(512, dtype=torch.int64, device=dev)
    prio = torch.rand(512, device=dev) + 0.01
    ext.sumtree_update(tree, idx, prio, P)
    sync()
    print("  ok total=", tree[1].item(), "expect", prio.sum().item(), flush=True)

    print("phase: sumtree_sample", flush=True)
    seed = torch.tensor([123], dtype=torch.int64, device=dev)
    oi = torch.empty(256, dtype=torch.int64, device=dev)
    op = torch.empty(256, device=dev)
    ext.bump_seed(seed)
    ext.sumtree_sample(tree, P, 512, 256, seed, oi, op)
    sync()
    print("  ok idx range", oi.min().item(), oi.max().item(), flush=True)

    print("phase: leaf_min", flush=True)
    mb = torch.full((1,), 0x7F800000, dtype=torch.int32, device=dev)
    ext.leaf_min_pos(tree, P, 512, mb)
    sync()
    print("  ok", flush=True)

    print("phase: per_weights", flush=True)
    w = torch.empty(256, device=dev)
    ext.per_weights(op, mb, tree, 512, 0.4, w)
    sync()
    print("  ok wmax=", w.max().item(), flush=True)

    print("phase: dqn_loss", flush=True)
    B, A = 64, 6
    q_s = torch.randn(B, A, device=dev)
    act = torch.randint(0, A, (B,), dtype=torch.int64, device=dev)
    loss = torch.zeros(1, device=dev)
    pr = torch.empty(B, device=dev)
    coef = torch.empty(B, device=dev)
    qm = torch.zeros(1, device=dev)
    ext.dqn_loss_fwd(q_s, torch.randn(B, A, device=dev),
                     torch.randn(B, A, device=dev), act,
                     torch.randn(B, device=dev), torch.zeros(B, device=dev),
                     torch.rand(B, device=dev), 0.97, 0.6, loss, pr, coef, qm)
    sync()
    print("  fwd ok loss=", loss.item(), flush=True)
    gq = torch.empty(B, A, device=dev)
    ext.dqn_loss_bwd(coef, act, torch.ones(1, device=dev), gq)
    sync()
    print("  bwd ok", flush=True)

    print("phase: vtrace", flush=True)
    T, Bv = 20, 32
    z = torch.zeros(T, Bv, device=dev)
    vs = torch.empty(T, Bv, device=dev)
    pg = torch.empty_like(vs)
    rho = torch.empty_like(vs)
    ext.vtrace(z, z, torch.rand(T, Bv, device=dev), torch.rand(T, Bv, device=dev),
               torch.rand(Bv, device=dev), torch.ones(Bv, device=dev), 0.99, 1.0,
               1.0, 1.0, vs, pg, rho)
    sync()
    print("  ok", flush=True)

    print("phase: vtrace_bt", flush=True)
    Bv2, Tv2 = 32, 20
    mu = torch.rand(Bv2, Tv2, device=dev).clamp_min(1e-3)
    tlp = -torch.rand(Bv2, Tv2, device=dev)
    vs_bt = torch.empty(Bv2, Tv2, device=dev)
    pg_bt = torch.empty_like(vs_bt)
    ext.vtrace_bt(mu, tlp, torch.randn(Bv2, Tv2, device=dev),
                  torch.randn(Bv2, Tv2, device=dev),
                  torch.randn(Bv2, device=dev), torch.ones(Bv2, device=dev),
                  0.99, 1.0, 1.0, 1.0, vs_bt, pg_bt)
    sync()
    print("  ok", flush=True)

    print("phase: impala_loss_fwd + out_bwd", flush=True)
    Ni, Ai = Bv2 * Tv2, 6
    logpa = -torch.rand(Ni, device=dev)
    adv = torch.randn(Ni, device=dev)
    mh = torch.rand(1, device=dev)
    l3 = torch.empty(1, device=dev)
    o3 = torch.empty(1, device=dev)
    c3 = torch.empty(1, device=dev)
    ext.impala_loss_fwd(logpa, adv, mh, vs_bt.view(-1), pg_bt.view(-1),
                        0.01, l3, o3, c3)
    sync()
    pi_s = torch.softmax(torch.randn(Ni, Ai, device=dev), -1).contiguous()
    H_s = torch.rand(Ni, device=dev)
    acts = torch.randint(0, Ai, (Ni,), dtype=torch.int64, device=dev)
    dout = torch.empty(Bv2 * (Tv2 + 1), Ai + 1, device=dev)
    ext.impala_out_bwd(pi_s, H_s, acts, adv, vs_bt.view(-1), pg_bt.view(-1),
                       torch.ones(1, device=dev), Bv2, Tv2, Ai, 0.01, dout)
    sync()
    print("  ok loss=", l3.item(), flush=True)

    print("phase: rescale", flush=True)
    xv = torch.linspace(-5, 5, 1024, device=dev)
    yv = torch.empty_like(xv)
    ext.value_rescale(xv, yv, 1e-3)
    ext.inv_value_rescale(yv, xv, 1e-3)
    sync()
    print("  ok", flush=True)

    print("phase: seq_priority", flush=True)
    td = torch.rand(80, 32, device=dev)
    sp = torch.empty(32, device=dev)
    ext.seq_priority(td, 0.9, 0.9, sp)
    sync()
    print("  ok", flush=True)

    print("phase: grad_clip", flush=True)
    g = torch.randn(100_000, device=dev) * 10
    sq = torch.zeros(1, device=dev)
    ext.grad_clip(g, 40.0, sq)
    sync()
    print("  ok norm->", g.norm().item(), flush=True)

    print("phase: conv_autocast", flush=True)
    m = torch.nn.Conv2d(4, 32, 8, 4).to(dev)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(torch.rand(8, 4, 84, 84, device=dev))
    y.float().sum().backward()
    sync()
    print("  ok", flush=True)

    print("phase: full model fwd/bwd", flush=True)
    from distributed_rl_amd.config import load_config
    from distributed_rl_amd.models import BaseAgent

    net = BaseAgent(load_config("ape_x").model_info).to(dev)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        q = net.forward([torch.rand(32, 4, 84, 84, device=dev)])[0]
    q.float().sum().backward()
    sync()
    print("  ok", flush=True)

    print("phase: learner step", flush=True)
    from distributed_rl_amd.algos.ape_x import ApexLearner
    from distributed_rl_amd.config import load_config as lc

    learner = ApexLearner(lc("ape_x"), device=dev, enable_tb=False,
                          batch_size=32, replay_capacity=2048)
    Bf = 512
    cols = {
        "state": torch.randint(0, 256, (Bf, 4, 84, 84), dtype=torch.uint8, device=dev),
        "action": torch.randint(0, 6, (Bf,), dtype=torch.int32, device=dev),
        "reward": torch.rand(Bf, device=dev),
        "next_state": torch.randint(0, 256, (Bf, 4, 84, 84), dtype=torch.uint8,
                                    device=dev),
        "done": torch.zeros(Bf, device=dev),
    }
    learner.push_experience(cols, torch.rand(Bf, device=dev) + 0.1)
    sync()
    print("  pushed", flush=True)
    st = learner.step()
    sync()
    print("  ok loss=", float(st["loss"]), flush=True)
    print("ALL PHASES OK", flush=True)


if __name__ == "__main__":
    main()
