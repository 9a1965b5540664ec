"""R2D2: n-step recurrent target math vs brute force, and e2e CPU slice."""

import copy

import numpy as np
import pytest
import torch

from distributed_rl_amd.actors.transport import InprocPipe
from distributed_rl_amd.algos.r2d2 import (
    R2D2Learner, R2D2Player, nstep_recurrent_targets,
)
from distributed_rl_amd.config import Config, load_config
from distributed_rl_amd.ops import torch_ref


def small_cfg():
    raw = copy.deepcopy(load_config("r2d2").raw)
    raw["BATCHSIZE"] = 4
    raw["REPLAY_MEMORY_LEN"] = 128
    raw["BUFFER_SIZE"] = 4
    raw["N"] = 2
    raw["FIXED_TRAJECTORY"] = 16
    raw["MEM"] = 4
    return Config(raw=raw)


def brute_force_targets(q_on, q_tg, actions, rewards, done, burn, n, gamma,
                        rescale):
    """Literal per-t implementation of the documented target."""
    T, B, A = q_on.shape
    h = torch_ref.value_rescale if rescale else (lambda x: x)
    hinv = torch_ref.inv_value_rescale if rescale else (lambda x: x)
    a_star = q_on.argmax(2)
    boot = hinv(q_tg.gather(2, a_star.unsqueeze(2)).squeeze(2))
    boot = torch.cat([boot[:-1], (boot[-1] * (1 - done)).unsqueeze(0)])
    out = torch.zeros(T - 1 - burn, B)
    for j, t in enumerate(range(burn, T - 1)):
        nt = min(n, T - 1 - t)
        acc = torch.zeros(B)
        for i in range(nt - 1, -1, -1):
            acc = rewards[t + i] + gamma * acc
        out[j] = h(acc + gamma ** nt * boot[t + nt])
    return out


@pytest.mark.parametrize("rescale", [True, False])
def test_nstep_recurrent_targets_vs_brute_force(rescale):
    torch.manual_seed(0)
    T, B, A = 16, 3, 6
    q_on = torch.randn(T, B, A)
    q_tg = torch.randn(T, B, A)
    actions = torch.randint(0, A, (T, B))
    rewards = torch.randn(T, B)
    done = torch.tensor([0.0, 1.0, 0.0])
    td, q_taken, targets = nstep_recurrent_targets(
        q_on, q_tg, actions, rewards, done, burn_in=4, n_step=5, gamma=0.997,
        use_rescaling=rescale)
    expect = brute_force_targets(q_on, q_tg, actions, rewards, done, 4, 5,
                                 0.997, rescale)
    assert torch.allclose(targets, expect, atol=1e-4), (targets - expect).abs().max()
    manual_q = q_on[4 : T - 1].gather(2, actions[4 : T - 1].unsqueeze(2)).squeeze(2)
    assert torch.allclose(td, expect - manual_q, atol=1e-4)


def test_r2d2_end_to_end_cpu():
    cfg = small_cfg()
    pipe = InprocPipe()
    learner = R2D2Learner(cfg, device="cpu", transport=pipe, enable_tb=False)
    learner.publish_weights(include_target=True)
    player = R2D2Player(cfg, idx=0, transport=pipe, env_kind="synthetic")
    player.run(max_env_steps=200)
    assert player.weight_version == 0
    n = learner.ingest()
    assert n >= 5  # (200-16)/8 overlapping windows
    losses = []
    for _ in range(3):
        stats = learner.step()
        losses.append(float(stats["loss"]))
    assert all(np.isfinite(l) for l in losses)
    assert learner.replay.total_priority > 0


def test_r2d2_sequence_overlap_emission():
    cfg = small_cfg()
    pipe = InprocPipe()
    player = R2D2Player(cfg, idx=0, transport=pipe, env_kind="synthetic",
                        seed=7)
    player.run(max_env_steps=3 * cfg.fixed_trajectory)
    cols, prio = pipe.drain()
    T = cfg.fixed_trajectory
    assert cols["states"].shape[1:] == (T, 4, 84, 84)
    assert cols["h0"].shape[1:] == (2, 512)
    assert (prio > 0).all()
    # overlapping windows: consecutive sequences share T/2 frames
    n_seq = cols["states"].shape[0]
    assert n_seq >= 4
    a = cols["states"][0][T // 2]
    b = cols["states"][1][0]
    assert np.array_equal(a, b)


def test_r2d2_checkpoint_roundtrip(tmp_path):
    cfg = small_cfg()
    learner = R2D2Learner(cfg, device="cpu", enable_tb=False,
                          run_root=str(tmp_path))
    B, T, H = 6, cfg.fixed_trajectory, 512
    cols = {
        "h0": torch.zeros(B, 2, H),
        "states": torch.randint(0, 255, (B, T, 4, 84, 84), dtype=torch.uint8),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32),
        "rewards": torch.randn(B, T),
        "done": torch.zeros(B),
    }
    learner.push_sequences(cols, torch.rand(B) + 0.1)
    learner.step()
    p = learner.save_checkpoint()
    l2 = R2D2Learner(cfg, device="cpu", enable_tb=False, run_root=str(tmp_path))
    l2.resume(p.replace("weight.pth", "resume.pt"))
    assert l2.step_count == 1
    for a, b in zip(l2.model.parameters(), learner.model.parameters()):
        assert torch.equal(a, b)


def test_nstep_recurrent_targets_random_geometries():
    """The vectorized target math must match the brute force for every
    (T, burn_in, n_step) combination, including truncated-tail windows."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=40, deadline=None)
    @given(T=st.integers(6, 24), burn=st.integers(0, 8), n=st.integers(1, 8),
           rescale=st.booleans(), seed=st.integers(0, 10_000))
    def check(T, burn, n, rescale, seed):
        if burn >= T - 1:
            return
        torch.manual_seed(seed)
        B, A = 3, 4
        q_on = torch.randn(T, B, A)
        q_tg = torch.randn(T, B, A)
        actions = torch.randint(0, A, (T, B))
        rewards = torch.randn(T, B)
        done = (torch.rand(B) < 0.5).float()
        td, q_taken, targets = nstep_recurrent_targets(
            q_on, q_tg, actions, rewards, done, burn_in=burn, n_step=n,
            gamma=0.97, use_rescaling=rescale)
        expect = brute_force_targets(q_on, q_tg, actions, rewards, done,
                                     burn, n, 0.97, rescale)
        assert targets.shape == expect.shape
        assert torch.allclose(targets, expect, atol=1e-5), (
            T, burn, n, rescale, (targets - expect).abs().max())

    check()


def test_r2d2_terminal_sequence_emitted_off_boundary():
    """An episode whose length is NOT a window boundary still produces a
    done=1 sequence covering the last T steps (R2D2/Player.py:37-47
    terminal emission)."""
    from distributed_rl_amd.actors.env import SyntheticEnv

    cfg = small_cfg()
    T = cfg.fixed_trajectory  # 16; overlap 8 -> boundaries at 16, 24, 32...
    pipe = InprocPipe()
    env = SyntheticEnv(seed=3, episode_len=T + 5)  # 21: off-boundary
    player = R2D2Player(cfg, idx=0, transport=pipe, env=env)
    player.run(max_env_steps=T + 5)  # exactly one episode
    cols, prio = pipe.drain()
    done = cols["done"].ravel()
    assert done[-1] == 1.0, done  # terminal window emitted
    # it covers the LAST T steps: its first frame equals the boundary
    # window's frame at offset 5 (windows: [0,16) done=0, [5,21) done=1)
    assert np.array_equal(cols["states"][-1][0], cols["states"][0][5])
