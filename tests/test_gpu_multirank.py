"""Multi-rank execution on real hardware. With only one GPU on the box,
two ranks share cuda:0: RCCL refuses same-device ranks, so the collective
backend falls back to gloo (CUDA tensors, host-staged) — the point is to
exercise the 4-graph overlapped stepper + process-group interplay
(capture around an eager collective, comm-stream ordering, replica sync)
on hardware before the driver's 8-GPU scale run.
"""

import copy
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as tmp

from distributed_rl_amd.config import Config, load_config

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    # both ranks share cuda:0 (1-GPU box)
    backend = "gloo"
    try:
        dist.init_process_group(backend, rank=rank, world_size=world)
        from distributed_rl_amd.algos.ape_x import ApexLearner
        from distributed_rl_amd.parallel import attach_reducer

        torch.manual_seed(1234 + rank)
        raw = copy.deepcopy(load_config("ape_x").raw)
        raw["BATCHSIZE"] = 32
        raw["REPLAY_MEMORY_LEN"] = 512
        cfg = Config(raw=raw)
        learner = ApexLearner(cfg, device="cuda:0", enable_tb=False,
                              world_size=world, rank=rank)
        attach_reducer(learner)
        g = torch.Generator(device="cuda:0").manual_seed(rank)
        B = 512
        cols = {
            "state": torch.randint(0, 255, (B, 4, 84, 84), dtype=torch.uint8,
                                   device="cuda:0", generator=g),
            "action": torch.randint(0, 6, (B,), dtype=torch.int32,
                                    device="cuda:0", generator=g),
            "reward": torch.rand(B, device="cuda:0", generator=g),
            "next_state": torch.randint(0, 255, (B, 4, 84, 84),
                                        dtype=torch.uint8, device="cuda:0",
                                        generator=g),
            "done": torch.zeros(B, device="cuda:0"),
        }
        learner.push_experience(cols, torch.ones(B, device="cuda:0"))
        # ring full -> 4-graph overlapped capture path
        stepper = learner.make_graphed_step()
        for _ in range(5):
            stepper()
        torch.cuda.synchronize()
        h = float(sum(p.double().sum() for p in learner.model.parameters()))
        gathered = [None] * world
        dist.all_gather_object(gathered, h)
        graphs = len(learner._graph) if isinstance(learner._graph, tuple) else 1
        if rank == 0:
            result_q.put(("ok", gathered, graphs))
    except Exception as e:  # pragma: no cover
        if rank == 0:
            result_q.put(("err", repr(e), None))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_overlapped_stepper_world2_one_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    ctx = tmp.get_context("spawn")
    q = ctx.Queue()
    port = 29651
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    status, payload, graphs = q.get(timeout=540)
    for p in procs:
        p.join(60)
    assert status == "ok", payload
    assert graphs == 4, f"expected the 4-graph overlapped capture, got {graphs}"
    # gloo all-reduce has host staging; replicas must still agree closely
    assert abs(payload[0] - payload[1]) < 1e-4, payload


@pytest.mark.timeout(600)
def test_snapshot_state_dict_matches_slow_path():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from distributed_rl_amd.algos.ape_x import ApexLearner

    raw = copy.deepcopy(load_config("ape_x").raw)
    raw["REPLAY_MEMORY_LEN"] = 128
    learner = ApexLearner(Config(raw=raw), device="cuda:0", enable_tb=False)
    fast = learner.snapshot_state_dict()
    slow = {k: v.detach().to("cpu", torch.float32)
            for k, v in learner.model.state_dict().items()}
    assert set(fast) == set(slow)
    for k in slow:
        assert fast[k].dtype == torch.float32
        assert fast[k].shape == slow[k].shape
        assert torch.equal(fast[k], slow[k]), k
        assert fast[k].is_contiguous(), k
