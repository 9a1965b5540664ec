#!/usr/bin/env python3
"""Dedicated replay-server node (3-tier mode) — the reference's
``APE_X/ReplayServer.py`` role. Actors point their experience
stream here; the learner samples batches from here with
``run_learner.py --replay-server host:port``.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_rl_amd.actors.transport import RecordCodec
from distributed_rl_amd.algos import get_wire_schema
from distributed_rl_amd.config import load_config
from distributed_rl_amd.replay.server import ReplayServer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--alg", "--cfg", dest="cfg", default=None)
    ap.add_argument("--port", type=int, default=6380)
    ap.add_argument("--capacity", type=int, default=None)
    args = ap.parse_args()
    cfg = load_config(args.cfg)
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    srv = ReplayServer(codec, args.capacity or cfg.replay_memory_len,
                       port=args.port).start()
    print(f"[replay-server] {cfg.alg} PER capacity "
          f"{args.capacity or cfg.replay_memory_len} on :{srv.port}", flush=True)
    try:
        while True:
            time.sleep(5)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
