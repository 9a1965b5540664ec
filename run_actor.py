#!/usr/bin/env python3
"""Actor fleet entry point — API parity with the reference's
``python run_actor.py --num-worker N --start-idx K``
(/root/reference/run_actor.py:21-33), with multiprocessing + shared-memory
rings instead of Ray + Redis.

The learner (run_learner.py) must be started first: it creates the
transport session the actors attach to.
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_rl_amd.actors.fleet import ActorFleet
from distributed_rl_amd.config import load_config


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--num-worker", "--num-workers", dest="num_worker",
                    type=int, default=None, help="actor count (default cfg N)")
    ap.add_argument("--start-idx", type=int, default=0)
    ap.add_argument("--alg", "--cfg", dest="cfg", default=None)
    ap.add_argument("--transport-dir", default=None)
    ap.add_argument("--env", default="auto",
                    help="auto | atari | synthetic")
    ap.add_argument("--max-env-steps", type=int, default=1 << 60)
    ap.add_argument("--no-respawn", action="store_true")
    ap.add_argument("--envs-per-proc", type=int, default=1,
                    help="virtual actors per OS process (batched shared-"
                         "model inference; Ape-X only)")
    ap.add_argument("--tcp", default="",
                    help="learner host:port for TCP mode (multi-host)")
    args = ap.parse_args()

    cfg = load_config(args.cfg)
    tdir = args.transport_dir or cfg.transport_dir
    n = args.num_worker or cfg.num_actors
    if not args.tcp:
        # wait for the learner's session manifest
        manifest = os.path.join(tdir, "session.json")
        t0 = time.time()
        while not os.path.exists(manifest):
            if time.time() - t0 > 300:
                raise TimeoutError(f"no learner session at {manifest}")
            time.sleep(0.5)

    fleet = ActorFleet(args.cfg or cfg.alg.lower(), n, tdir,
                       start_idx=args.start_idx, env_kind=args.env,
                       max_env_steps=args.max_env_steps,
                       respawn_on_exit=not args.no_respawn, tcp=args.tcp,
                       envs_per_proc=args.envs_per_proc)
    fleet.start()
    print(f"[run_actor] {n} actors running (start_idx={args.start_idx})",
          flush=True)
    try:
        fleet.supervise()
    except KeyboardInterrupt:
        fleet.stop()


if __name__ == "__main__":
    main()
