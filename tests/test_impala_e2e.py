"""IMPALA single-process vertical slice on CPU (BASELINE.json config 1:
'IMPALA 1 CPU learner + 2 CPU actors on synthetic 84x84 frames')."""

import copy

import numpy as np
import torch

from distributed_rl_amd.actors.transport import InprocPipe
from distributed_rl_amd.algos.impala import ImpalaLearner, ImpalaPlayer
from distributed_rl_amd.config import Config, load_config


def small_cfg():
    raw = copy.deepcopy(load_config("impala").raw)
    raw["BATCHSIZE"] = 4
    raw["REPLAY_MEMORY_LEN"] = 256
    return Config(raw=raw)


def test_impala_two_actors_end_to_end():
    cfg = small_cfg()
    pipe = InprocPipe()
    learner = ImpalaLearner(cfg, device="cpu", transport=pipe, enable_tb=False,
                            publish_every=5)
    learner.publish_weights()
    players = [ImpalaPlayer(cfg, idx=i, transport=pipe, env_kind="synthetic")
               for i in range(2)]
    for p in players:
        p.run(max_env_steps=200)
        assert p.weight_version == 0
    n = learner.ingest()
    assert n >= 2 * (200 // cfg.unroll_step) - 4
    losses = []
    for _ in range(4):
        stats = learner.step()
        losses.append(float(stats["loss"]))
    assert all(np.isfinite(l) for l in losses)
    # entropy of a 6-action policy
    assert 0 < float(stats["entropy"]) <= np.log(6) + 1e-4
    # publish cadence hit at step 5
    learner.step()
    assert pipe.fetch()["count"] == 5


def test_impala_trajectory_layout():
    cfg = small_cfg()
    pipe = InprocPipe()
    p = ImpalaPlayer(cfg, idx=0, transport=pipe, env_kind="synthetic")
    p.run(max_env_steps=100)
    cols, _ = pipe.drain()
    T = cfg.unroll_step
    assert cols["states"].shape[1:] == (T + 1, 4, 84, 84)
    assert cols["actions"].shape[1] == T
    assert cols["mu"].shape[1] == T
    assert (cols["mu"] > 0).all() and (cols["mu"] <= 1).all()
    assert set(np.unique(cols["not_done"])) <= {0.0, 1.0}


def test_impala_learning_signal():
    """Gradient flows and losses stay finite over repeated steps on a
    fixed buffer."""
    torch.manual_seed(0)
    cfg = small_cfg()
    learner = ImpalaLearner(cfg, device="cpu", enable_tb=False)
    B, T = 8, cfg.unroll_step
    cols = {
        "states": torch.randint(0, 255, (B, T + 1, 4, 84, 84), dtype=torch.uint8),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32),
        "mu": torch.full((B, T), 1 / 6, dtype=torch.float32),
        "rewards": torch.randn(B, T),
        "not_done": torch.ones(B),
    }
    learner.push_trajectories(cols)
    before = [p.clone() for p in learner.model.parameters()]
    for _ in range(3):
        stats = learner.step()
    changed = any(
        not torch.equal(p, q) for p, q in zip(before, learner.model.parameters())
    )
    assert changed
    assert np.isfinite(float(stats["loss"]))


def test_impala_resnet_learner_cpu():
    raw = copy.deepcopy(load_config("impala_resnet").raw)
    raw["BATCHSIZE"] = 2
    cfg = Config(raw=raw)
    learner = ImpalaLearner(cfg, device="cpu", enable_tb=False,
                            replay_capacity=16)
    B, T = 4, cfg.unroll_step
    cols = {
        "states": torch.randint(0, 255, (B, T + 1, 4, 84, 84), dtype=torch.uint8),
        "actions": torch.randint(0, 6, (B, T), dtype=torch.int32),
        "mu": torch.full((B, T), 1 / 6, dtype=torch.float32),
        "rewards": torch.randn(B, T),
        "not_done": torch.ones(B),
    }
    learner.push_trajectories(cols)
    stats = learner.step()
    assert np.isfinite(float(stats["loss"]))
