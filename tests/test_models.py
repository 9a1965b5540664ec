import torch

from distributed_rl_amd.config import load_config
from distributed_rl_amd.models import BaseAgent, get_optim


def test_ape_x_model_shapes():
    cfg = load_config("ape_x")
    net = BaseAgent(cfg.model_info)
    x = torch.rand(5, 4, 84, 84)
    out = net.forward([x])
    assert len(out) == 1
    assert out[0].shape == (5, 6)


def test_dueling_identity():
    """Output must equal (A + V) - mean(A) of the two heads."""
    cfg = load_config("ape_x")
    net = BaseAgent(cfg.model_info)
    x = torch.rand(3, 4, 84, 84)
    feat = net.nodes["module00"](x)
    adv = net.nodes["module02"](feat)
    val = net.nodes["module02_1"](feat)
    expect = (adv + val) - adv.mean(dim=-1, keepdim=True)
    got = net.forward([x])[0]
    assert torch.allclose(got, expect, atol=1e-6)


def test_impala_model_shapes():
    cfg = load_config("impala")
    net = BaseAgent(cfg.model_info)
    out = net.forward([torch.rand(7, 4, 84, 84)])[0]
    assert out.shape == (7, 7)  # 6 logits + 1 value


def test_r2d2_model_shapes_and_cell_state():
    cfg = load_config("r2d2")
    net = BaseAgent(cfg.model_info)
    assert net.has_lstm
    seq, batch = 8, 3
    x = torch.rand(seq * batch, 4, 84, 84)
    hint = torch.tensor([seq, batch, -1])
    net.zeroCellState(batch)
    out = net.forward([x, hint])[0]
    assert out.shape == (seq * batch, 6)
    h, c = net.getCellState()
    assert h.shape == (1, batch, 512)
    # state persists and is replaceable
    net.setCellState((h * 0, c * 0))
    net.detachCellState()
    out2 = net.forward([x, hint])[0]
    assert out2.shape == (seq * batch, 6)


def test_update_parameter_hard_and_polyak():
    cfg = load_config("ape_x")
    a = BaseAgent(cfg.model_info)
    b = BaseAgent(cfg.model_info)
    b.updateParameter(a, 1.0)
    for p, q in zip(a.parameters(), b.parameters()):
        assert torch.equal(p, q)
    with torch.no_grad():
        for p in a.parameters():
            p.add_(1.0)
    b_before = [p.clone() for p in b.parameters()]
    b.updateParameter(a, 0.5)
    for p, q, q0 in zip(a.parameters(), b.parameters(), b_before):
        assert torch.allclose(q, 0.5 * p + 0.5 * q0, atol=1e-6)


def test_norm_utilities():
    cfg = load_config("impala")
    net = BaseAgent(cfg.model_info)
    out = net.forward([torch.rand(2, 4, 84, 84)])[0]
    out.sum().backward()
    n = net.calculateNorm()
    assert n.item() > 0
    net.clippingNorm(1e-6)
    assert net.calculateNorm().item() <= 1.1e-6


def test_optimizer_factory():
    cfg = load_config("ape_x")
    net = BaseAgent(cfg.model_info)
    opt = get_optim(cfg.optim_info, net)
    assert isinstance(opt, torch.optim.RMSprop)
    assert abs(opt.defaults["lr"] - 6.25e-5) < 1e-12
    assert opt.defaults["centered"] is True
    cfg2 = load_config("r2d2")
    opt2 = get_optim(cfg2.optim_info, net)
    assert isinstance(opt2, torch.optim.Adam)
    assert abs(opt2.defaults["eps"] - 1e-3) < 1e-12


def test_state_dict_roundtrip():
    cfg = load_config("r2d2")
    a = BaseAgent(cfg.model_info)
    b = BaseAgent(cfg.model_info)
    b.load_state_dict(a.state_dict())
    for p, q in zip(a.parameters(), b.parameters()):
        assert torch.equal(p, q)


def test_impala_resnet_model():
    cfg = load_config("impala_resnet")
    net = BaseAgent(cfg.model_info)
    out = net.forward([torch.rand(3, 4, 84, 84)])[0]
    assert out.shape == (3, 7)
    # residual path actually contributes
    from distributed_rl_amd.models.base_agent import ResidualBlock

    blocks = [m for m in net.modules() if isinstance(m, ResidualBlock)]
    assert len(blocks) == 6  # 3 sections x 2 blocks


def test_reference_name_aliases():
    """The reference's camelCase surface (baseline.baseAgent / getOptim)
    resolves to the same objects."""
    from distributed_rl_amd.models import (
        BaseAgent, baseAgent, get_optim, getOptim,
    )

    assert baseAgent is BaseAgent
    assert getOptim is get_optim
