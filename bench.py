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
        transport=getattr(args, "_transport", None),
    )
    attach_reducer(learner)
    if (learner.device.type == "cuda"
            and getattr(learner, "_fast_fwd", None) is None
            and os.environ.get("DRL_ALLOW_SLOW_DUELING") != "1"):
        # the fused dueling forward is worth ~15%; a silent fallback would
        # quietly degrade the headline number
        raise RuntimeError(
            "dueling fast-forward inactive for this cfg — set "
            "DRL_ALLOW_SLOW_DUELING=1 to bench the slow path"
        )
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
        transport=getattr(args, "_transport", None),
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
        transport=getattr(args, "_transport", None),
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
    ap.add_argument("--with-actors", type=int, default=0,
                    help="spawn K synthetic-env CPU actor processes feeding "
                         "the replay DURING the timed region (BASELINE "
                         "config-2 whole-node mode); 0 = learner-only")
    ap.add_argument("--envs-per-proc", type=int, default=1,
                    help="virtual actors per fleet process (batched shared-"
                         "model inference; Ape-X only)")
    ap.add_argument("--burn-in", type=int, default=None,
                    help="R2D2 burn-in override (BASELINE config 4 = 40)")
    ap.add_argument("--transport-dir", default=None)
    args = ap.parse_args()

    # cap intra-op threads: the default (ncores) oversubscribes the
    # box against the actor fleet and the ingest pack
    torch.set_num_threads(min(16, os.cpu_count() or 16))
    rank, local_rank, world = init_distributed()
    has_cuda = torch.cuda.is_available()
    dev_idx = local_rank % torch.cuda.device_count() if has_cuda else 0
    device = f"cuda:{dev_idx}" if has_cuda else "cpu"
    if world == 1 and args.gpus > 1:
        # N>1 must come through torchrun (one rank per GPU); a single process
        # only ever drives one GPU — never multiply frames by an unused N.
        print(f"# --gpus {args.gpus} without torchrun: clamping to 1",
              file=sys.stderr)
        args.gpus = 1
    n_gpus = world if world > 1 else (args.gpus if has_cuda else 0)

    cfg = load_config(args.cfg)
    if args.burn_in is not None:
        cfg.raw["MEM"] = int(args.burn_in)
    if args.with_actors > 0 and args.replay == 0:
        # size the ring to the prefill so it is exactly full at capture time:
        # mid-bench pushes then wrap (overwrite-oldest), keeping the baked
        # n_valid of the captured sample kernel exact
        args.replay = 100_000

    # --- optional whole-node mode: shm transport + synthetic actor fleet ---
    session = fleet = None
    if args.with_actors > 0:
        import tempfile

        from distributed_rl_amd.actors.fleet import ActorFleet
        from distributed_rl_amd.actors.transport import (
            LearnerEndpoint, RecordCodec, TransportSession,
        )
        from distributed_rl_amd.algos import get_wire_schema

        schema, with_prio = get_wire_schema(cfg)
        codec = RecordCodec(schema, with_priority=with_prio)
        tdir = args.transport_dir or tempfile.mkdtemp(prefix="drl_bench_")
        if rank == 0:
            # rings sized to absorb the graph-capture pause (ingest thread
            # stops for a few seconds; 256 slots measured 33k drops)
            session = TransportSession(tdir, codec, num_rings=args.with_actors,
                                       ring_slots=1024, create=True)
        if world > 1:
            torch.distributed.barrier()
        if rank != 0:
            session = TransportSession(tdir, codec, num_rings=args.with_actors,
                                       create=False)
        args._transport = LearnerEndpoint(session, rank=rank, world_size=world)
        if rank == 0:
            fleet = ActorFleet(args.cfg, args.with_actors, tdir,
                               env_kind="synthetic", respawn_on_exit=False,
                               envs_per_proc=args.envs_per_proc)

    builders = {"APE_X": build_apex, "IMPALA": build_impala, "R2D2": build_r2d2}
    learner, frames_per_step = builders[cfg.alg](cfg, device, rank, world, args)

    if fleet is not None:
        learner.publish_weights(include_target=True)
        fleet.start()
        # ingest runs on its own daemon thread (the production run-loop
        # shape); wait for the fleet to warm up (actor processes import
        # torch before their first env step) so the timed region measures
        # steady-state ingest, not an idle transport
        learner.start_ingest_thread()
        t_warm = time.perf_counter()
        warm_target = max(8192, 32 * args.with_actors)
        # big fleets take minutes to finish 200+ concurrent torch imports —
        # wait longer so the timed region sees the steady state
        warm_deadline = 180 if args.with_actors <= 128 else 420
        while learner.ingested_total < warm_target \
                and time.perf_counter() - t_warm < warm_deadline:
            time.sleep(0.05)
        dt_warm = time.perf_counter() - t_warm
        print(f"# fleet warm: {learner.ingested_total} rows in {dt_warm:.1f}s "
              f"({learner.ingested_total / max(dt_warm, 1e-9):.0f} rows/s), "
              f"{fleet.alive_count()}/{fleet.num_procs} actor procs alive",
              file=sys.stderr)
        learner.stop_ingest_thread()  # capture below needs quiet streams

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
            stepper = (learner.make_pipelined_step() if world > 1
                       and hasattr(learner, "make_pipelined_step")
                       else learner.step)
            use_graph = False
    elif world > 1 and hasattr(learner, "make_pipelined_step"):
        # eager pipelined step: same overlap ordering without graphs
        stepper = learner.make_pipelined_step()

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if has_cuda:
            torch.cuda.synchronize()

    if args.with_actors > 0:
        learner.start_ingest_thread()  # concurrent with the timed region
        # drain the backlog that built up during the graph-capture pause so
        # the timed region measures steady-state ingest
        t_bk = time.perf_counter()
        while time.perf_counter() - t_bk < 90:
            gap = (learner.transport.total_pushed()
                   - learner.transport.total_drops()
                   - learner.ingested_total)
            if gap < 4096:
                break
            time.sleep(0.2)
    for i in range(args.warmup):
        stepper()
    barrier_sync()
    ingested0 = learner.ingested_total if args.with_actors else 0
    t0 = time.perf_counter()
    for i in range(args.steps):
        stepper()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    ingested = (learner.ingested_total - ingested0) if args.with_actors else 0
    if args.with_actors > 0:
        drops = (learner.transport.total_drops()
                 if hasattr(learner.transport, "total_drops") else -1)
        pushed = (learner.transport.total_pushed()
                  if hasattr(learner.transport, "total_pushed") else -1)
        alive = (learner._ingest_thread is not None
                 and learner._ingest_thread.is_alive())
        print(f"# transport: {ingested} rows ingested in timed region, "
              f"{pushed} pushed / {drops} dropped total, "
              f"ingest_thread_alive={alive}", file=sys.stderr)
        learner.stop_ingest_thread()
    if fleet is not None:
        fleet.stop()
    if session is not None:
        session.close()
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
        "R2D2": f"learner frames/sec (whole node), R2D2 LSTM "
                f"seq{cfg.fixed_trajectory} burn-in{cfg.burn_in}",
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
                "actors": args.with_actors,
                "ingested_rows": ingested if args.with_actors else None,
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
