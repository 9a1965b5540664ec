"""3-tier replay-server mode: actors -> TCP PER node -> learner batches."""

import copy

import numpy as np
import pytest
import torch

from distributed_rl_amd.actors.tcp_transport import TcpActorEndpoint
from distributed_rl_amd.actors.transport import RecordCodec
from distributed_rl_amd.algos.ape_x import ApexLearner
from distributed_rl_amd.config import Config, load_config
from distributed_rl_amd.replay import make_apex_schema
from distributed_rl_amd.replay.server import RemoteReplay, ReplayServer


@pytest.mark.timeout(180)
def test_replay_server_roundtrip_and_learner_step():
    codec = RecordCodec(make_apex_schema())
    srv = ReplayServer(codec, capacity=4096, host="127.0.0.1", port=0).start()
    try:
        actor = TcpActorEndpoint("127.0.0.1", srv.port, codec)
        n = 128
        cols = {
            "state": np.random.randint(0, 255, (n, 4, 84, 84), dtype=np.uint8),
            "action": np.random.randint(0, 6, n).astype(np.int32),
            "reward": np.random.rand(n).astype(np.float32),
            "next_state": np.random.randint(0, 255, (n, 4, 84, 84),
                                            dtype=np.uint8),
            "done": np.zeros(n, np.float32),
        }
        actor.push(cols, np.random.rand(n).astype(np.float32) + 0.1)
        remote = RemoteReplay("127.0.0.1", srv.port)
        import time

        t0 = time.time()
        while len(remote) < n and time.time() - t0 < 30:
            time.sleep(0.05)
        assert len(remote) == n
        data, idx, w = remote.sample(16, beta=0.4)
        assert data["state"].shape == (16, 4, 84, 84)
        assert w.shape == (16,)
        remote.update(idx, torch.full((16,), 5.0))
        # learner drives the remote replay end to end
        raw = copy.deepcopy(load_config("ape_x").raw)
        raw["BATCHSIZE"] = 8
        learner = ApexLearner(Config(raw=raw), device="cpu", enable_tb=False,
                              replay=remote)
        stats = learner.step()
        assert np.isfinite(float(stats["loss"]))
        actor.close()
        remote.close()
    finally:
        srv.stop()


@pytest.mark.timeout(180)
def test_replay_server_r2d2_sequences():
    """The PER node is schema-generic: R2D2 sequence records (h0 riding the
    record) round-trip and drive a learner step (R2D2/ReplayServer.py parity
    — without its double-push defect)."""
    from distributed_rl_amd.algos.r2d2 import R2D2Learner
    from distributed_rl_amd.replay import make_r2d2_schema

    raw = copy.deepcopy(load_config("r2d2").raw)
    raw.update({"BATCHSIZE": 4, "BUFFER_SIZE": 4, "N": 2,
                "FIXED_TRAJECTORY": 16, "MEM": 4, "UNROLL_STEP": 3})
    cfg = Config(raw=raw)
    T, H = cfg.fixed_trajectory, 512
    codec = RecordCodec(make_r2d2_schema(seq_len=T, hidden=H))
    srv = ReplayServer(codec, capacity=256, host="127.0.0.1", port=0).start()
    try:
        actor = TcpActorEndpoint("127.0.0.1", srv.port, codec)
        n = 16
        cols = {
            "h0": np.random.randn(n, 2, H).astype(np.float32) * 0.01,
            "states": np.random.randint(0, 255, (n, T, 4, 84, 84),
                                        dtype=np.uint8),
            "actions": np.random.randint(0, 6, (n, T)).astype(np.int32),
            "rewards": np.random.randn(n, T).astype(np.float32),
            "done": np.zeros(n, np.float32),
        }
        actor.push(cols, np.random.rand(n).astype(np.float32) + 0.1)
        remote = RemoteReplay("127.0.0.1", srv.port)
        import time

        t0 = time.time()
        while len(remote) < n and time.time() - t0 < 30:
            time.sleep(0.05)
        assert len(remote) == n
        data, idx, w = remote.sample(4, beta=0.4)
        assert data["h0"].shape == (4, 2, H)
        assert data["states"].shape == (4, T, 4, 84, 84)
        learner = R2D2Learner(cfg, device="cpu", enable_tb=False,
                              replay=remote)
        stats = learner.step()
        assert np.isfinite(float(stats["loss"]))
        actor.close()
        remote.close()
    finally:
        srv.stop()
