"""Full-system live run: spawned actor fleet + shm transport + learner
run() loop (ingest thread, graph capture, publish cadence) — the
production topology in one command, used for stability/learning evidence:

  python tools/live_run.py --alg impala --actors 16 --max-steps 12000 \
      --set MAX_REPLAY_REUSE=6 --set REPLAY_MEMORY_LEN=512

Console telemetry (each learner's 500-step block) is the record; grep it
into profiles/.
"""

import argparse
import copy
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--alg", default="impala")
    ap.add_argument("--actors", type=int, default=8)
    ap.add_argument("--max-steps", type=int, default=12_000)
    ap.add_argument("--env", default="synthetic")
    ap.add_argument("--device", default=None)
    ap.add_argument("--batch", type=int, default=None)
    ap.add_argument("--set", action="append", default=[],
                    help="cfg override KEY=VALUE (VALUE json-parsed)")
    args = ap.parse_args()

    from distributed_rl_amd.actors.fleet import ActorFleet
    from distributed_rl_amd.actors.transport import (
        LearnerEndpoint, RecordCodec, TransportSession,
    )
    from distributed_rl_amd.algos import get_learner_cls, get_wire_schema
    from distributed_rl_amd.config import Config, load_config

    # cap intra-op threads: the default (ncores) oversubscribes the
    # box against the actor fleet and the ingest pack
    torch.set_num_threads(min(16, os.cpu_count() or 16))
    torch.manual_seed(0)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    raw = copy.deepcopy(load_config(args.alg).raw)
    raw["N"] = args.actors
    for kv in args.set:
        k, v = kv.split("=", 1)
        try:
            raw[k] = json.loads(v)
        except json.JSONDecodeError:
            raw[k] = v
    cfg = Config(raw=raw)

    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    tdir = tempfile.mkdtemp(prefix=f"drl_live_{cfg.alg.lower()}_")
    session = TransportSession(tdir, codec, num_rings=args.actors,
                               ring_slots=256, create=True)
    transport = LearnerEndpoint(session)
    kw = {"batch_size": args.batch} if args.batch else {}
    learner = get_learner_cls(cfg.alg)(cfg, device=device, transport=transport,
                                       enable_tb=False, run_root=tdir, **kw)
    learner.publish_weights(include_target=hasattr(learner, "target"))

    cfg_path = os.path.join(tdir, "live_cfg.json")
    with open(cfg_path, "w") as f:
        json.dump(raw, f)
    fleet = ActorFleet(cfg_path, args.actors, tdir, env_kind=args.env,
                       respawn_on_exit=True)
    fleet.start()
    try:
        learner.run(max_steps=args.max_steps)
    finally:
        learner.stop_ingest_thread()
        fleet.stop()
        session.close()


if __name__ == "__main__":
    main()
