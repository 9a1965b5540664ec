#!/usr/bin/env python3
"""Flagship benchmark: Ape-X DQN learner frames/sec on MI355X (BASELINE.json).

Runs the complete learner hot loop — GPU sum-tree PER sample -> uint8
dequant -> 3 network passes (online s, online s', target s') -> fused TD
loss -> backward -> (RCCL all-reduce at N>1) -> optimizer step -> PER
priority update — on synthetic 84x84x4 uint8 frames and random-init weights
(no network access for datasets), model exactly cfg/ape_x.json (dueling
Atari CNN), compute dtype bf16.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N ... bench.py --gpus N ...

Metric: whole-job learner frames/sec = steps * per_gpu_batch * N / elapsed,
elapsed = MAX over ranks of barrier+synchronized timed region.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_rl_amd.algos.ape_x import ApexLearner
from distributed_rl_amd.config import load_config
from distributed_rl_amd.parallel import init_distributed, attach_reducer


def prefill(learner: ApexLearner, n_items: int, chunk: int = 8192, seed: int = 0):
    dev = learner.replay.device
    g = torch.Generator(device=dev)
    g.manual_seed(seed)
    remaining = n_items
    while remaining > 0:
        b = min(chunk, remaining)
        cols = {
            "state": torch.randint(0, 256, (b, 4, 84, 84), dtype=torch.uint8,
                                   device=dev, generator=g),
            "action": torch.randint(0, 6, (b,), dtype=torch.int32, device=dev,
                                    generator=g),
            "reward": torch.rand(b, device=dev, generator=g) * 2 - 1,
            "next_state": torch.randint(0, 256, (b, 4, 84, 84), dtype=torch.uint8,
                                        device=dev, generator=g),
            "done": (torch.rand(b, device=dev, generator=g) < 0.02).float(),
        }
        prio = torch.rand(b, device=dev, generator=g).clamp_min(1e-3)
        learner.push_experience(cols, prio)
        remaining -= b


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--batch", type=int, default=512,
                    help="per-GPU learner batch (Ape-X paper uses 512)")
    ap.add_argument("--replay", type=int, default=0,
                    help="replay capacity override (default: cfg REPLAY_MEMORY_LEN)")
    ap.add_argument("--cfg", type=str, default="ape_x")
    ap.add_argument("--graph", type=str, default="auto",
                    choices=["auto", "on", "off"],
                    help="hipGraph-capture the train step")
    args = ap.parse_args()

    rank, local_rank, world = init_distributed()
    has_cuda = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if has_cuda else "cpu"
    if world > 1:
        assert world == args.gpus or args.gpus == 1, (
            f"launched with WORLD_SIZE={world} but --gpus={args.gpus}"
        )
    n_gpus = world if world > 1 else (args.gpus if has_cuda else 0)

    cfg = load_config(args.cfg)
    learner = ApexLearner(
        cfg, device=device, rank=rank, world_size=world, enable_tb=False,
        batch_size=args.batch,
        replay_capacity=args.replay or cfg.replay_memory_len,
    )
    attach_reducer(learner)
    prefill(learner, len_target(learner), seed=1234 + rank)

    use_graph = args.graph == "on" or (args.graph == "auto" and has_cuda)
    stepper = learner.step
    if use_graph and has_cuda:
        try:
            stepper = make_graphed_stepper(learner)
        except Exception as e:
            print(f"# graph capture failed ({e}); falling back to eager",
                  file=sys.stderr)
            stepper = learner.step
            use_graph = False

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if has_cuda:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        stepper()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        stepper()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device if has_cuda else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t)

    frames = args.steps * args.batch * max(n_gpus, 1)
    fps = frames / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    if rank == 0:
        out = {
            "metric": "learner frames/sec (whole node), Ape-X DQN Atari CNN",
            "value": round(fps, 1),
            "unit": "frames/s",
            "n_gpus": max(n_gpus, 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "ape_x dueling Atari CNN (cfg/ape_x.json)",
                "global_batch": args.batch * max(n_gpus, 1),
                "seq_len": 4,
                "parallelism": f"dp{max(n_gpus, 1)}",
                "replay": "gpu sum-tree PER" if has_cuda else "cpu PER",
                "graph": bool(use_graph),
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()


def len_target(learner) -> int:
    # fill the whole ring so sampling covers capacity (bounded for CPU runs)
    cap = learner.replay.capacity
    return min(cap, 100_000 if learner.device.type == "cuda" else 2_048)


def make_graphed_stepper(learner: ApexLearner):
    return learner.make_graphed_step()


if __name__ == "__main__":
    main()
