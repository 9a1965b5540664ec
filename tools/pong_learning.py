"""Pong learning run — the reference's integration test, re-created.

The reference's only success criterion is ALE Pong reward climbing from
-21 (/root/reference/APE_X/Player.py:272-277, Learner.py:226-231). ALE
cannot be installed in this image, so this drives Ape-X on the from-scratch
pixel Pong (actors/pong.py) through the REAL production stack: spawned CPU
actor processes -> shm SPSC rings -> learner ingest -> PER -> train loop ->
seqlock weight bus back to the actors.

  python tools/pong_learning.py --actors 6 --max-steps 40000

Prints the learner's 500-step console blocks (reward = mean episode return
drained from the fleet); the curve is the deliverable (profiles/).
"""

import argparse
import copy
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--actors", type=int, default=6)
    ap.add_argument("--max-steps", type=int, default=40_000)
    ap.add_argument("--device", default=None)
    ap.add_argument("--batch", type=int, default=None)
    ap.add_argument("--train-per-ingest", type=float, default=0.0,
                    help="cap learner steps per ingested transition "
                         "(0 = unpaced, the reference's behavior); e.g. 8 "
                         "keeps replay reuse bounded when the learner is "
                         "much faster than the fleet (GPU)")
    args = ap.parse_args()

    from distributed_rl_amd.actors.fleet import ActorFleet
    from distributed_rl_amd.actors.transport import (
        LearnerEndpoint, RecordCodec, TransportSession,
    )
    from distributed_rl_amd.algos import get_wire_schema
    from distributed_rl_amd.algos.ape_x import ApexLearner
    from distributed_rl_amd.config import Config, load_config

    # cap intra-op threads: the default (ncores) oversubscribes the
    # box against the actor fleet and the ingest pack
    torch.set_num_threads(min(16, os.cpu_count() or 16))
    torch.manual_seed(0)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    raw = copy.deepcopy(load_config("ape_x").raw)
    raw.update({
        "REPLAY_MEMORY_LEN": 100_000,
        "BUFFER_SIZE": 2_000,
        "BATCHSIZE": args.batch or (64 if device == "cpu" else 512),
        "N": args.actors,
    })
    cfg = Config(raw=raw)

    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    tdir = tempfile.mkdtemp(prefix="drl_pong_")
    session = TransportSession(tdir, codec, num_rings=args.actors,
                               ring_slots=512, create=True)
    transport = LearnerEndpoint(session)
    learner = ApexLearner(cfg, device=device, transport=transport,
                          enable_tb=False, run_root=tdir)
    learner.publish_weights(include_target=True)

    # write cfg override so fleet actors build the same Config
    cfg_path = os.path.join(tdir, "pong_cfg.json")
    import json

    with open(cfg_path, "w") as f:
        json.dump(raw, f)
    fleet = ActorFleet(cfg_path, args.actors, tdir, env_kind="pong",
                       respawn_on_exit=True)
    fleet.start()
    try:
        if args.train_per_ingest > 0:
            _paced_run(learner, args.max_steps, args.train_per_ingest)
        else:
            learner.run(max_steps=args.max_steps)
    finally:
        fleet.stop()
        session.close()


def _paced_run(learner, max_steps, train_per_ingest):
    """learner.run with a replay-reuse cap: at most train_per_ingest
    learner-step-batches per ingested transition (used on GPU, where the
    learner outruns a small CPU fleet by orders of magnitude)."""
    import time

    learner.wait_memory()
    learner.publish_weights(include_target=True)
    ingested = len(learner.replay)
    budget = 0.0
    while learner.step_count < max_steps:
        with learner._ingest_lock:
            got = learner.ingest()
        ingested += got
        budget += got * train_per_ingest / learner.batch_size
        if budget < 1.0:
            time.sleep(0.002)
            continue
        budget -= 1.0
        stats = learner.step()
        if learner.step_count % learner.LOG_EVERY == 0:
            learner._log_block(stats)


if __name__ == "__main__":
    main()
