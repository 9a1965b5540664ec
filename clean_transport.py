#!/usr/bin/env python3
"""Flush stale transport state — the reference's ``delete_redis.py``
equivalent (/root/reference/delete_redis.py:5-18 flushes both Redis
servers; here we unlink the session's shared-memory segments and manifest).
"""

import argparse
import json
import os
import sys
from multiprocessing import shared_memory

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_rl_amd.config import load_config


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--transport-dir", default=None)
    args = ap.parse_args()
    tdir = args.transport_dir or load_config().transport_dir
    manifest = os.path.join(tdir, "session.json")
    if not os.path.exists(manifest):
        print(f"no session at {manifest}")
        return
    with open(manifest) as f:
        m = json.load(f)
    session = m["session"]
    removed = 0
    names = [f"drl_{session}_w"]
    for i in range(m["num_rings"]):
        names += [f"drl_{session}_r{i}", f"drl_{session}_t{i}"]
    for name in names:
        try:
            shm = shared_memory.SharedMemory(name=name)
            shm.close()
            shm.unlink()
            removed += 1
        except FileNotFoundError:
            pass
    os.remove(manifest)
    print(f"removed session {session}: {removed} shm segments")


if __name__ == "__main__":
    main()
