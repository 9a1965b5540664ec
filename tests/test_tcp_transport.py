"""TCP transport: record + weight + reward round trip, and an Ape-X player
driving it end-to-end (multi-host parity path)."""

import copy

import numpy as np
import pytest
import torch

from distributed_rl_amd.actors.tcp_transport import (
    TcpActorEndpoint, TcpTransportServer,
)
from distributed_rl_amd.actors.transport import RecordCodec
from distributed_rl_amd.algos.ape_x import ApexLearner, ApexPlayer
from distributed_rl_amd.config import Config, load_config
from distributed_rl_amd.replay import make_apex_schema


@pytest.mark.timeout(120)
def test_tcp_roundtrip():
    codec = RecordCodec(make_apex_schema())
    srv = TcpTransportServer(codec, host="127.0.0.1", port=0).start()
    try:
        ep = TcpActorEndpoint("127.0.0.1", srv.port, codec)
        le = srv.endpoint()
        assert ep.fetch() is None
        le.publish({"count": 3, "state_dict": {"w": torch.ones(2)}})
        got = ep.fetch()
        assert got["count"] == 3
        # unchanged version: second poll is served from the client cache
        # (server answers K_NONE, no blob re-transfer)
        have = ep._have
        assert have >= 1
        got2 = ep.fetch()
        assert got2 is got and ep._have == have
        le.publish({"count": 4, "state_dict": {"w": torch.ones(2)}})
        got3 = ep.fetch()
        assert got3["count"] == 4 and ep._have == have + 1
        n = 4
        cols = {
            "state": np.zeros((n, 4, 84, 84), np.uint8),
            "action": np.arange(n, dtype=np.int32),
            "reward": np.ones(n, np.float32),
            "next_state": np.zeros((n, 4, 84, 84), np.uint8),
            "done": np.zeros(n, np.float32),
        }
        ep.push(cols, np.full(n, 0.5, np.float32))
        ep.push_reward(0, 7.25)
        import time

        deadline = time.time() + 10
        out = None
        while out is None and time.time() < deadline:
            out = le.drain()
        assert out is not None
        got_cols, got_prio = out
        assert got_cols["action"].tolist() == [0, 1, 2, 3]
        assert np.allclose(got_prio, 0.5)
        deadline = time.time() + 10
        rewards = []
        while not rewards and time.time() < deadline:
            rewards = le.drain_rewards()
        assert rewards == [7.25]
        ep.close()
    finally:
        srv.stop()


@pytest.mark.timeout(180)
def test_tcp_apex_player_end_to_end():
    raw = copy.deepcopy(load_config("ape_x").raw)
    raw.update({"REPLAY_MEMORY_LEN": 1024, "BUFFER_SIZE": 16, "BATCHSIZE": 8,
                "N": 2})
    cfg = Config(raw=raw)
    codec = RecordCodec(make_apex_schema())
    srv = TcpTransportServer(codec, host="127.0.0.1", port=0).start()
    try:
        learner = ApexLearner(cfg, device="cpu", transport=srv.endpoint(),
                              enable_tb=False)
        learner.publish_weights(include_target=True)
        ep = TcpActorEndpoint("127.0.0.1", srv.port, codec)
        player = ApexPlayer(cfg, idx=0, transport=ep, env_kind="synthetic")
        player.run(max_env_steps=150)
        assert player.weight_version == 0
        import time

        got = 0
        deadline = time.time() + 30
        while got < 32 and time.time() < deadline:
            got += learner.ingest()
            time.sleep(0.02)
        assert got >= 32
        stats = learner.step()
        assert np.isfinite(float(stats["loss"]))
        ep.close()
    finally:
        srv.stop()
