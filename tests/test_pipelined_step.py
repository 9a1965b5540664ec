"""The overlapped (pipelined) stepper must be a pure reordering: the RCCL/
gloo all-reduce runs concurrently with the priority update of batch t and
the sample of batch t+1, but the op order seen by the replay
(update-then-sample) and the optimizer math are identical to the sequential
stepper — so whole trajectories must match bit-for-bit.

This is the CPU rehearsal of the north-star C1 overlap;
the GPU variant (4-graph capture + comm-stream collective) runs the same
ordering and is covered by tests/test_gpu_algos.py.
"""

import copy
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as tmp

from distributed_rl_amd.config import Config, load_config


def _small_cfg(alg):
    raw = copy.deepcopy(load_config(alg).raw)
    if alg == "ape_x":
        raw["BATCHSIZE"] = 8
        raw["REPLAY_MEMORY_LEN"] = 128
    elif alg == "impala":
        raw["BATCHSIZE"] = 4
        raw["REPLAY_MEMORY_LEN"] = 64
    else:  # r2d2
        raw["BATCHSIZE"] = 2
        raw["REPLAY_MEMORY_LEN"] = 16
        raw["BUFFER_SIZE"] = 2
        raw["FIXED_TRAJECTORY"] = 16
        raw["MEM"] = 4
    return Config(raw=raw)


def _build(alg, cfg, world=1, rank=0, data_seed=42):
    g = torch.Generator().manual_seed(data_seed)
    if alg == "ape_x":
        from distributed_rl_amd.algos.ape_x import ApexLearner

        learner = ApexLearner(cfg, device="cpu", enable_tb=False,
                              world_size=world, rank=rank)
        B = 64
        cols = {
            "state": torch.randint(0, 255, (B, 4, 84, 84), dtype=torch.uint8,
                                   generator=g),
            "action": torch.randint(0, 6, (B,), dtype=torch.int32,
                                    generator=g),
            "reward": torch.rand(B, generator=g),
            "next_state": torch.randint(0, 255, (B, 4, 84, 84),
                                        dtype=torch.uint8, generator=g),
            "done": torch.zeros(B),
        }
        learner.push_experience(cols, torch.rand(B, generator=g) + 0.1)
    elif alg == "impala":
        from distributed_rl_amd.algos.impala import ImpalaLearner

        learner = ImpalaLearner(cfg, device="cpu", enable_tb=False,
                                world_size=world, rank=rank)
        B, T = 8, cfg.unroll_step
        learner.push_trajectories({
            "states": torch.randint(0, 255, (B, T + 1, 4, 84, 84),
                                    dtype=torch.uint8, generator=g),
            "actions": torch.randint(0, 6, (B, T), dtype=torch.int32,
                                     generator=g),
            "mu": torch.full((B, T), 1 / 6),
            "rewards": torch.randn(B, T, generator=g),
            "not_done": torch.ones(B),
        })
    else:
        from distributed_rl_amd.algos.r2d2 import R2D2Learner

        learner = R2D2Learner(cfg, device="cpu", enable_tb=False,
                              world_size=world, rank=rank)
        B, T, H = 4, cfg.fixed_trajectory, 512
        learner.push_sequences({
            "h0": torch.zeros(B, 2, H),
            "states": torch.randint(0, 255, (B, T, 4, 84, 84),
                                    dtype=torch.uint8, generator=g),
            "actions": torch.randint(0, 6, (B, T), dtype=torch.int32,
                                     generator=g),
            "rewards": torch.randn(B, T, generator=g),
            "done": torch.zeros(B),
        }, torch.rand(B, generator=g) + 0.1)
    return learner


def _param_vec(learner):
    return torch.cat([p.detach().double().reshape(-1)
                      for p in learner.model.parameters()])


@pytest.mark.parametrize("alg", ["ape_x", "impala", "r2d2"])
def test_pipelined_equals_sequential(alg):
    cfg = _small_cfg(alg)
    steps = 3

    torch.manual_seed(7)
    seq = _build(alg, cfg)
    for _ in range(steps):
        seq.step()

    torch.manual_seed(7)
    pip = _build(alg, cfg)
    stepper = pip.make_pipelined_step()
    for _ in range(steps):
        stepper()

    a, b = _param_vec(seq), _param_vec(pip)
    assert torch.equal(a, b), (a - b).abs().max()


def _worker(rank, world, port, alg, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from distributed_rl_amd.parallel import attach_reducer

        cfg = _small_cfg(alg)
        steps = 2
        # sequential learner first (matching collective order across ranks);
        # identical init on both ranks, DIFFERENT replay data per rank
        torch.manual_seed(7)
        seq = _build(alg, cfg, world=world, rank=rank, data_seed=1000 + rank)
        attach_reducer(seq)
        h0 = _param_vec(seq).clone()
        torch.manual_seed(100 + rank)  # per-rank sampling streams
        for _ in range(steps):
            seq.step()

        torch.manual_seed(7)
        pip = _build(alg, cfg, world=world, rank=rank, data_seed=1000 + rank)
        attach_reducer(pip)
        torch.manual_seed(100 + rank)
        stepper = pip.make_pipelined_step()
        for _ in range(steps):
            stepper()

        same = torch.equal(_param_vec(seq), _param_vec(pip))
        moved = not torch.equal(_param_vec(pip), h0)
        # ranks bit-identical after the async all-reduce?
        v = _param_vec(pip)
        gathered = [torch.empty_like(v) for _ in range(world)]
        dist.all_gather(gathered, v)
        in_sync = torch.equal(gathered[0], gathered[1])
        if rank == 0:
            result_q.put((same, in_sync, moved))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("alg,port", [("ape_x", 29641), ("impala", 29642),
                                      ("r2d2", 29643)])
def test_pipelined_world2_gloo(alg, port):
    """world-2 gloo: the async-overlap stepper (a) reproduces the sequential
    stepper's trajectory exactly, (b) keeps replicas bit-identical, and
    (c) actually trains."""
    ctx = tmp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, alg, q))
             for r in range(2)]
    for p in procs:
        p.start()
    same, in_sync, moved = q.get(timeout=240)
    for p in procs:
        p.join(30)
    assert same, "pipelined trajectory diverged from sequential"
    assert in_sync, "replicas diverged under the pipelined stepper"
    assert moved, "no learning happened"
