#!/usr/bin/env python3
"""Learner entry point — API parity with the reference's ``python
run_learner.py`` (/root/reference/run_learner.py:15-18), with the algorithm
selected by --alg / DRL_CFG instead of a source edit
(reference configuration.py:11-13).

Single GPU:      python run_learner.py --alg ape_x
Learner DP:      torchrun --standalone --local-addr 127.0.0.1 \
                   --nproc-per-node 8 run_learner.py --alg ape_x
(actors attach via run_actor.py using the same --transport-dir)
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from distributed_rl_amd.actors.transport import (
    LearnerEndpoint, RecordCodec, TransportSession,
)
from distributed_rl_amd.algos import get_learner_cls, get_wire_schema
from distributed_rl_amd.config import load_config
from distributed_rl_amd.parallel import attach_reducer, init_distributed


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--alg", "--cfg", dest="cfg", default=None,
                    help="ape_x | r2d2 | impala | path to cfg json")
    ap.add_argument("--max-steps", type=int, default=1_000_000)
    ap.add_argument("--device", default=None)
    ap.add_argument("--num-actors", type=int, default=None,
                    help="number of actor rings to create (default cfg N)")
    ap.add_argument("--transport-dir", default=None)
    ap.add_argument("--ring-slots", type=int, default=256)
    ap.add_argument("--resume", default=None, help="checkpoint to resume from")
    ap.add_argument("--no-transport", action="store_true",
                    help="run without an actor session (debug)")
    ap.add_argument("--tcp-port", type=int, default=None,
                    help="serve actors over TCP on this port (multi-host "
                         "mode) instead of shared-memory rings")
    ap.add_argument("--replay-server", default=None,
                    help="host:port of a dedicated replay-server node "
                         "(3-tier mode, run_replay_server.py)")
    args = ap.parse_args()

    rank, local_rank, world = init_distributed()
    cfg = load_config(args.cfg)
    device = args.device
    if device is None:
        device = (
            f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
        )

    transport = None
    session = None
    tcp_srv = None
    if not args.no_transport:
        schema, with_prio = get_wire_schema(cfg)
        codec = RecordCodec(schema, with_priority=with_prio)
        if args.tcp_port is not None:
            from distributed_rl_amd.actors.tcp_transport import TcpTransportServer

            tcp_srv = TcpTransportServer(codec, port=args.tcp_port + rank).start()
            transport = tcp_srv.endpoint()
        else:
            tdir = args.transport_dir or cfg.transport_dir
            n_act = args.num_actors or cfg.num_actors
            if rank == 0:
                session = TransportSession(tdir, codec, num_rings=n_act,
                                           ring_slots=args.ring_slots,
                                           create=True)
            if world > 1:
                torch.distributed.barrier()
            if rank != 0:
                session = TransportSession(tdir, codec, num_rings=n_act,
                                           create=False)
            transport = LearnerEndpoint(session, rank=rank, world_size=world)

    learner_kw = {}
    if args.replay_server:
        from distributed_rl_amd.replay.server import RemoteReplay

        host, port = args.replay_server.rsplit(":", 1)
        learner_kw["replay"] = RemoteReplay(host, int(port))
    learner = get_learner_cls(cfg.alg)(
        cfg, device=device, rank=rank, world_size=world, transport=transport,
        **learner_kw,
    )
    attach_reducer(learner)
    if args.resume:
        learner.resume(args.resume)
    try:
        learner.run(max_steps=args.max_steps)
    finally:
        if session is not None:
            session.close()
        if tcp_srv is not None:
            tcp_srv.stop()
        if world > 1:
            torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
