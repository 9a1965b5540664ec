#!/usr/bin/env python3
"""Benchmark harness. Default (the driver contract / BASELINE.json headline):
Ape-X DQN learner frames/sec on MI355X, batch 512/GPU, bf16, synthetic
84x84x4 frames, GPU-resident sum-tree PER, hipGraph-captured step.

  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N ... bench.py --gpus N ...

Timed region per step = the COMPLETE learner iteration: PER sample ->
(fused-dequant) forwards -> fused TD loss -> backward -> [RCCL all-reduce at
N>1] -> optimizer step -> PER priority update. value = steps * batch * N /
elapsed, elapsed = MAX over ranks between barrier+synchronize fences.

--alg impala / r2d2 run the same protocol on the other two learners
(secondary benches; numbers recorded in profiles/).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_rl_amd.config import load_config
from distributed_rl_amd.parallel import init_distributed, attach_reducer


def prefill_apex(learner, n_items: int, chunk: int = 8192, seed: int = 0):
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
            "next_state": torch.randint(0, 256, (b, 4, 84, 84),
                                        dtype=torch.uint8, device=dev,
                                        generator=g),
            "done": (torch.rand(b, device=dev, generator=g) < 0.02).float(),
        }
        prio = torch.rand(b, device=dev, generator=g).clamp_min(1e-3)
        learner.push_experience(cols, prio)
        remaining -= b


def build_apex(cfg, device, rank, world, args):
    from distributed_rl_amd.algos.ape_x import ApexLearner

    learner = ApexLearner(
        cfg, device=device, rank=rank, world_size=world, enable_tb=False,
        batch_size=args.batch,
        replay_capacity=args.replay or cfg.replay_memory_len,
        replay_state_dtype=(torch.float16 if args.replay_dtype == "fp16"
                            else None),
    )
    attach_reducer(learner)
    cap = learner.replay.capacity
    n = min(cap, 100_000 if learner.device.type == "cuda" else 2_048)
    prefill_apex(learner, n, seed=1234 + rank)
    frames_per_step = args.batch
    return learner, frames_per_step


def build_impala(cfg, device, rank, world, args):
    from distributed_rl_amd.algos.impala import ImpalaLearner

    batch = args.batch if args.batch != 512 or cfg.alg != "IMPALA" else 64
    learner = ImpalaLearner(
        cfg, device=device, rank=rank, world_size=world, enable_tb=False,
        batch_size=batch, replay_capacity=args.replay or 2048,
    )
    attach_reducer(learner)
    dev = learner.replay.device
    g = torch.Generator(device=dev)
    g.manual_seed(99 + rank)
    T = cfg.unroll_step
    cap = learner.replay.capacity
    filled = 0
    while filled < min(cap, 2048):
        b = min(256, cap - filled)
        cols = {
            "states": torch.randint(0, 256, (b, T + 1, 4, 84, 84),
                                    dtype=torch.uint8, device=dev, generator=g),
            "actions": torch.randint(0, 6, (b, T), dtype=torch.int32,
                                     device=dev, generator=g),
            "mu": torch.rand(b, T, device=dev, generator=g) * 0.9 + 0.05,
            "rewards": torch.randn(b, T, device=dev, generator=g),
            "not_done": torch.ones(b, device=dev),
        }
        learner.push_trajectories(cols)
        filled += b
    return learner, batch * T


def build_r2d2(cfg, device, rank, world, args):
    from distributed_rl_amd.algos.r2d2 import R2D2Learner

    batch = args.batch if args.batch != 512 or cfg.alg != "R2D2" else 32
    learner = R2D2Learner(
        cfg, device=device, rank=rank, world_size=world, enable_tb=False,
        batch_size=batch, replay_capacity=args.replay or 1024,
    )
    attach_reducer(learner)
    dev = learner.replay.device
    g = torch.Generator(device=dev)
    g.manual_seed(7 + rank)
    T = cfg.fixed_trajectory
    cap = learner.replay.capacity
    filled = 0
    while filled < min(cap, 512):
        b = min(64, cap - filled)
        cols = {
            "h0": torch.randn(b, 2, 512, device=dev, generator=g) * 0.01,
            "states": torch.randint(0, 256, (b, T, 4, 84, 84),
                                    dtype=torch.uint8, device=dev, generator=g),
            "actions": torch.randint(0, 6, (b, T), dtype=torch.int32,
                                     device=dev, generator=g),
            "rewards": torch.randn(b, T, device=dev, generator=g),
            "done": (torch.rand(b, device=dev, generator=g) < 0.1).float(),
        }
        learner.push_sequences(cols, torch.rand(b, device=dev, generator=g) + 0.1)
        filled += b
    return learner, batch * T


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--batch", type=int, default=512,
                    help="per-GPU learner batch (Ape-X paper uses 512)")
    ap.add_argument("--replay", type=int, default=0,
                    help="replay capacity override")
    ap.add_argument("--alg", "--cfg", dest="cfg", type=str, default="ape_x")
    ap.add_argument("--graph", type=str, default="auto",
                    choices=["auto", "on", "off"])
    ap.add_argument("--replay-dtype", type=str, default="u8",
                    choices=["u8", "fp16"],
                    help="Ape-X replay frame storage (fp16 = BASELINE "
                         "config-5 compression option)")
    args = ap.parse_args()

    rank, local_rank, world = init_distributed()
    has_cuda = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if has_cuda else "cpu"
    if world == 1 and args.gpus > 1:
        # N>1 must come through torchrun (one rank per GPU); a single process
        # only ever drives one GPU — never multiply frames by an unused N.
        print(f"# --gpus {args.gpus} without torchrun: clamping to 1",
              file=sys.stderr)
        args.gpus = 1
    n_gpus = world if world > 1 else (args.gpus if has_cuda else 0)

    cfg = load_config(args.cfg)
    builders = {"APE_X": build_apex, "IMPALA": build_impala, "R2D2": build_r2d2}
    learner, frames_per_step = builders[cfg.alg](cfg, device, rank, world, args)

    use_graph = args.graph == "on" or (
        args.graph == "auto" and has_cuda and hasattr(learner, "make_graphed_step")
    )
    stepper = learner.step
    if use_graph and has_cuda:
        try:
            stepper = learner.make_graphed_step()
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
    if world > 1:
        t = torch.tensor([elapsed], device=device if has_cuda else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t)

    frames = args.steps * frames_per_step * max(n_gpus, 1)
    fps = frames / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    metric_name = {
        "APE_X": "learner frames/sec (whole node), Ape-X DQN Atari CNN",
        "IMPALA": "learner frames/sec (whole node), IMPALA V-trace Atari CNN",
        "R2D2": "learner frames/sec (whole node), R2D2 LSTM seq80 burn-in20",
    }[cfg.alg]
    if rank == 0:
        out = {
            "metric": metric_name,
            "value": round(fps, 1),
            "unit": "frames/s",
            "n_gpus": max(n_gpus, 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("bf16" if has_cuda else "fp32")
            if cfg.alg != "R2D2" else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"{cfg.alg.lower()} net ({os.path.basename(cfg.path)})",
                "global_batch": learner.batch_size * max(n_gpus, 1),
                "seq_len": {"APE_X": 4, "IMPALA": cfg.unroll_step,
                            "R2D2": cfg.fixed_trajectory}[cfg.alg],
                "parallelism": f"dp{max(n_gpus, 1)}",
                "replay": (("gpu sum-tree PER" if has_cuda else "cpu PER")
                           + ("/fp16" if args.replay_dtype == "fp16" else ""))
                if cfg.alg != "IMPALA" else "uniform fifo",
                "graph": bool(use_graph),
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
