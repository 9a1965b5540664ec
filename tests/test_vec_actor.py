"""Vectorized multi-env actor (envs_per_proc > 1): M virtual actors per
OS process share one model and take their eps-greedy argmax from a single
batched forward (algos.ape_x.run_apex_vec, actors.fleet._vec_actor_main).
The reference pins one Ray process per env (run_actor.py:46-55); this
mode divides the process count while preserving per-virtual-actor
semantics (own eps slot, rng stream, n-step buffer, ring, telemetry)."""

import copy
import json
import time

import numpy as np
import pytest
import torch

from distributed_rl_amd.actors.env import SyntheticEnv
from distributed_rl_amd.actors.fleet import ActorFleet
from distributed_rl_amd.actors.transport import (
    InprocPipe, LearnerEndpoint, RecordCodec, TransportSession,
)
from distributed_rl_amd.algos import (
    get_learner_cls, get_vec_runner, get_wire_schema,
)
from distributed_rl_amd.algos.ape_x import ApexPlayer, run_apex_vec
from distributed_rl_amd.config import Config, load_config


def _small_cfg(**over):
    raw = copy.deepcopy(load_config("ape_x").raw)
    raw.update({"REPLAY_MEMORY_LEN": 4096, "BUFFER_SIZE": 32, "BATCHSIZE": 8,
                "N": 4})
    raw.update(over)
    return Config(raw=raw)


def test_vec_runner_registry():
    assert get_vec_runner("APE_X") is run_apex_vec
    assert get_vec_runner("ape_x") is run_apex_vec
    assert get_vec_runner("R2D2") is None
    assert get_vec_runner("IMPALA") is None


def test_fleet_refuses_vec_without_runner(tmp_path):
    raw = copy.deepcopy(load_config("r2d2").raw)
    p = tmp_path / "r2d2.json"
    p.write_text(json.dumps(raw))
    with pytest.raises(ValueError, match="vectorized"):
        ActorFleet(str(p), 4, str(tmp_path), envs_per_proc=2)


def test_vec_matches_greedy_per_player_actions():
    """With eps=0 the batched argmax must equal each player's own act()."""
    cfg = _small_cfg()
    torch.manual_seed(0)
    players = [
        ApexPlayer(cfg, idx=i, transport=InprocPipe(),
                   env=SyntheticEnv(seed=i, episode_len=50))
        for i in range(3)
    ]
    lead = players[0]
    for p in players[1:]:
        p.model.load_state_dict(lead.model.state_dict())
        p.eps = 0.0
    lead.eps = 0.0
    states = [p.env.reset() for p in players]
    solo = [p.act(s) for p, s in zip(players, states)]
    x = torch.from_numpy(np.stack(states)).float().div_(255.0)
    with torch.no_grad():
        batched = lead.model.forward([x])[0].argmax(1).tolist()
    assert solo == batched


def test_vec_loop_pushes_all_rings():
    """run_apex_vec drives M players end-to-end: every virtual actor's
    transport receives transitions and reward telemetry flows."""
    cfg = _small_cfg()
    torch.manual_seed(0)
    pipe = InprocPipe()
    players = [
        ApexPlayer(cfg, idx=i, transport=pipe,
                   env=SyntheticEnv(seed=i, episode_len=40))
        for i in range(3)
    ]
    # distinct per-player push counting: thin transport wrapper
    counts = [0, 0, 0]

    class _Counting:
        def __init__(self, j):
            self.j = j

        def push(self, cols, prio=None):
            counts[self.j] += len(cols["action"])
            return pipe.push(cols, prio)

        def push_reward(self, idx, reward, eps=0.0):
            return pipe.push_reward(idx, reward, eps)

        def fetch(self):
            return pipe.fetch()

    for j, p in enumerate(players):
        p.transport = _Counting(j)
    run_apex_vec(players, max_env_steps=120)
    assert all(c > 0 for c in counts), counts
    # shared model object (not copies)
    assert players[1].model is players[0].model
    assert players[2].target is players[0].target


@pytest.mark.timeout(180)
def test_vec_fleet_process_integration(tmp_path):
    """2 processes x 2 virtual actors = 4 rings feeding a CPU learner."""
    cfg = _small_cfg()
    cfg_path = tmp_path / "apex_small.json"
    cfg_path.write_text(json.dumps(cfg.raw))
    tdir = str(tmp_path / "transport")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = TransportSession(tdir, codec, num_rings=4, ring_slots=512,
                               create=True)
    fleet = ActorFleet(str(cfg_path), 4, tdir, env_kind="synthetic",
                       max_env_steps=400, respawn_on_exit=False,
                       envs_per_proc=2)
    assert fleet.num_procs == 2
    try:
        endpoint = LearnerEndpoint(session)
        learner = get_learner_cls("APE_X")(
            cfg, device="cpu", transport=endpoint, enable_tb=False,
        )
        learner.publish_weights(include_target=True)
        fleet.start()
        got = 0
        t0 = time.time()
        # wait for BOTH processes: every ring must have pushed (the second
        # proc can deliver 64 rows before the first finishes importing
        # torch, so gating on `got` alone races process startup)
        while time.time() - t0 < 150:
            got += learner.ingest()
            pushed = [session.ring(i).head for i in range(4)]
            if got >= 64 and all(n > 0 for n in pushed):
                break
            time.sleep(0.05)
        assert got >= 64, f"only {got} transitions arrived"
        assert all(n > 0 for n in pushed), pushed
        for _ in range(3):
            stats = learner.step()
        assert float(stats["loss"]) == float(stats["loss"])  # finite
        fleet.join(60)
    finally:
        fleet.stop()
        session.close()
