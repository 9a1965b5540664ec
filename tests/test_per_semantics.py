"""PER semantic contract tests against the torch oracle."""

import torch

from distributed_rl_amd.replay import TorchPER, FifoReplay, make_apex_schema


def small_schema():
    return {"x": ((), torch.float32)}


def test_push_and_len_ring_eviction():
    per = TorchPER(8, small_schema())
    per.push({"x": torch.arange(6.0)}, torch.ones(6))
    assert len(per) == 6
    per.push({"x": torch.arange(6.0, 12.0)}, torch.ones(6))
    assert len(per) == 8  # capacity
    # oldest entries overwritten: slots 0..3 hold items 8..11
    assert per.data["x"][0].item() == 8.0
    assert per.data["x"][3].item() == 11.0
    assert per.data["x"][4].item() == 4.0


def test_proportional_sampling_distribution():
    g = torch.Generator().manual_seed(0)
    per = TorchPER(4, small_schema(), generator=g)
    prios = torch.tensor([1.0, 2.0, 3.0, 4.0])
    per.push({"x": torch.arange(4.0)}, prios)
    counts = torch.zeros(4)
    n_draws = 40000
    _, idx, _ = per.sample(n_draws, beta=0.4)
    counts += torch.bincount(idx, minlength=4).float()
    freq = counts / counts.sum()
    expect = prios / prios.sum()
    assert torch.allclose(freq, expect, atol=0.02), (freq, expect)


def test_is_weights_formula():
    per = TorchPER(4, small_schema())
    prios = torch.tensor([1.0, 2.0, 3.0, 4.0])
    per.push({"x": torch.arange(4.0)}, prios)
    beta = 0.4
    _, idx, w = per.sample(1000, beta=beta)
    n = 4
    probs = prios[idx] / prios.sum()
    max_w = (1.0 / (n * (prios.min() / prios.sum()))) ** beta
    expect = (1.0 / (n * probs)) ** beta / max_w
    assert torch.allclose(w, expect.float(), rtol=1e-5)
    assert w.max() <= 1.0 + 1e-5  # min-priority element carries weight 1


def test_priority_update_shifts_distribution():
    g = torch.Generator().manual_seed(1)
    per = TorchPER(4, small_schema(), generator=g)
    per.push({"x": torch.arange(4.0)}, torch.ones(4))
    per.update(torch.tensor([2]), torch.tensor([100.0]))
    _, idx, _ = per.sample(2000, beta=0.4)
    freq2 = (idx == 2).float().mean().item()
    assert freq2 > 0.9


def test_apex_schema_columns():
    per = TorchPER(16, make_apex_schema())
    B = 4
    cols = {
        "state": torch.randint(0, 255, (B, 4, 84, 84), dtype=torch.uint8),
        "action": torch.randint(0, 6, (B,), dtype=torch.int32),
        "reward": torch.rand(B),
        "next_state": torch.randint(0, 255, (B, 4, 84, 84), dtype=torch.uint8),
        "done": torch.zeros(B),
    }
    per.push(cols, torch.rand(B) + 0.1)
    data, idx, w = per.sample(8, beta=0.4)
    assert data["state"].shape == (8, 4, 84, 84)
    assert data["state"].dtype == torch.uint8
    assert data["action"].shape == (8,)
    assert w.shape == (8,)


def test_fifo_uniform():
    g = torch.Generator().manual_seed(2)
    rep = FifoReplay(8, small_schema(), generator=g)
    rep.push({"x": torch.arange(8.0)})
    data, idx, w = rep.sample(4000)
    freq = torch.bincount(idx, minlength=8).float() / 4000
    assert (freq > 0.08).all() and (freq < 0.17).all()
    assert (w == 1).all()
