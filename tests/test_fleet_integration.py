"""Multiprocess integration: real actor processes over shm rings feeding a
CPU learner (run_learner/run_actor path without the CLIs)."""

import copy
import json
import os
import time

import pytest
import torch

from distributed_rl_amd.actors.fleet import ActorFleet
from distributed_rl_amd.actors.transport import (
    LearnerEndpoint, RecordCodec, TransportSession,
)
from distributed_rl_amd.algos import get_learner_cls, get_wire_schema
from distributed_rl_amd.config import Config, load_config


def _small_cfg_file(tmp_path, alg="ape_x", **over):
    raw = copy.deepcopy(load_config(alg).raw)
    raw.update({"REPLAY_MEMORY_LEN": 4096, "BUFFER_SIZE": 32, "BATCHSIZE": 8,
                "N": 2})
    raw.update(over)
    p = tmp_path / f"{alg}_small.json"
    p.write_text(json.dumps(raw))
    return str(p), Config(raw=raw)


@pytest.mark.timeout(180)
def test_apex_fleet_two_actors(tmp_path):
    cfg_path, cfg = _small_cfg_file(tmp_path)
    tdir = str(tmp_path / "transport")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = TransportSession(tdir, codec, num_rings=2, ring_slots=512,
                               create=True)
    fleet = ActorFleet(cfg_path, 2, tdir, env_kind="synthetic",
                       max_env_steps=400, respawn_on_exit=False)
    try:
        endpoint = LearnerEndpoint(session)
        learner = get_learner_cls("APE_X")(
            cfg, device="cpu", transport=endpoint, enable_tb=False,
        )
        learner.publish_weights(include_target=True)
        fleet.start()
        got = 0
        t0 = time.time()
        while got < 64 and time.time() - t0 < 120:
            got += learner.ingest()
            time.sleep(0.05)
        assert got >= 64, f"only {got} transitions arrived"
        for _ in range(3):
            stats = learner.step()
        assert float(stats["loss"]) == float(stats["loss"])  # finite
        fleet.join(60)
    finally:
        fleet.stop()
        session.close()


@pytest.mark.timeout(180)
def test_impala_fleet_roundtrip(tmp_path):
    cfg_path, cfg = _small_cfg_file(tmp_path, alg="impala", BATCHSIZE=4)
    tdir = str(tmp_path / "transport")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = TransportSession(tdir, codec, num_rings=2, ring_slots=64,
                               create=True)
    fleet = ActorFleet(cfg_path, 2, tdir, env_kind="synthetic",
                       max_env_steps=200, respawn_on_exit=False)
    try:
        endpoint = LearnerEndpoint(session)
        learner = get_learner_cls("IMPALA")(
            cfg, device="cpu", transport=endpoint, enable_tb=False,
            publish_every=2,
        )
        learner.publish_weights()
        fleet.start()
        got = 0
        t0 = time.time()
        while got < 8 and time.time() - t0 < 120:
            got += learner.ingest()
            time.sleep(0.05)
        assert got >= 8
        stats = learner.step()
        assert float(stats["loss"]) == float(stats["loss"])
        fleet.join(60)
    finally:
        fleet.stop()
        session.close()


@pytest.mark.timeout(240)
def test_r2d2_fleet_roundtrip(tmp_path):
    cfg_path, cfg = _small_cfg_file(
        tmp_path, alg="r2d2", BATCHSIZE=2, FIXED_TRAJECTORY=16, MEM=4,
        REPLAY_MEMORY_LEN=128, BUFFER_SIZE=4,
    )
    tdir = str(tmp_path / "transport")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = TransportSession(tdir, codec, num_rings=2, ring_slots=64,
                               create=True)
    fleet = ActorFleet(cfg_path, 2, tdir, env_kind="synthetic",
                       max_env_steps=120, respawn_on_exit=False)
    try:
        endpoint = LearnerEndpoint(session)
        learner = get_learner_cls("R2D2")(
            cfg, device="cpu", transport=endpoint, enable_tb=False,
        )
        learner.publish_weights(include_target=True)
        fleet.start()
        got = 0
        t0 = time.time()
        while got < 4 and time.time() - t0 < 180:
            got += learner.ingest()
            time.sleep(0.1)
        assert got >= 4, f"only {got} sequences arrived"
        stats = learner.step()
        assert float(stats["loss"]) == float(stats["loss"])
        fleet.join(60)
    finally:
        fleet.stop()
        session.close()


@pytest.mark.timeout(240)
def test_fleet_respawns_killed_actor(tmp_path):
    """Failure handling: a killed actor process is respawned
    by the supervisor and the fleet returns to full strength."""
    cfg_path, cfg = _small_cfg_file(tmp_path)
    tdir = str(tmp_path / "transport")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = TransportSession(tdir, codec, num_rings=2, ring_slots=64,
                               create=True)
    fleet = ActorFleet(cfg_path, 2, tdir, env_kind="synthetic",
                       max_env_steps=1 << 30, respawn_on_exit=True)
    try:
        fleet.start()
        t0 = time.time()
        while fleet.alive_count() < 2 and time.time() - t0 < 60:
            time.sleep(0.2)
        assert fleet.alive_count() == 2
        victim = fleet.procs[0]
        victim.kill()
        victim.join(20)
        # one supervision pass replaces it
        import threading

        th = threading.Thread(target=fleet.supervise, kwargs={"poll_s": 0.5},
                              daemon=True)
        th.start()
        t0 = time.time()
        while time.time() - t0 < 60:
            p0 = fleet.procs[0]
            if p0 is not None and p0.is_alive() and p0.pid != victim.pid:
                break
            time.sleep(0.2)
        p0 = fleet.procs[0]
        assert p0 is not None and p0.is_alive() and p0.pid != victim.pid
    finally:
        fleet.respawn = False
        fleet.stop()
        session.close()
