import numpy as np
import torch

from distributed_rl_amd.actors.transport import (
    RecordCodec, SpscRing, WeightBus, TransportSession, ActorEndpoint,
    LearnerEndpoint, InprocPipe,
)
from distributed_rl_amd.replay import make_apex_schema


def test_codec_roundtrip():
    codec = RecordCodec(make_apex_schema())
    n = 5
    cols = {
        "state": np.random.randint(0, 255, (n, 4, 84, 84), dtype=np.uint8),
        "action": np.random.randint(0, 6, n).astype(np.int32),
        "reward": np.random.rand(n).astype(np.float32),
        "next_state": np.random.randint(0, 255, (n, 4, 84, 84), dtype=np.uint8),
        "done": np.zeros(n, np.float32),
    }
    prio = np.random.rand(n).astype(np.float32)
    rec = codec.pack(cols, prio)
    cols2, prio2 = codec.unpack(rec)
    assert np.array_equal(cols2["state"], cols["state"])
    assert np.array_equal(cols2["action"], cols["action"])
    assert np.allclose(prio2, prio)


def test_spsc_ring_wrap_and_drop():
    rs = 16
    ring = SpscRing("drl_test_ring_a", rs, 4, create=True)
    try:
        rows = np.arange(3 * rs, dtype=np.uint8).reshape(3, rs)
        assert ring.push_records(rows) == 3
        out = ring.pop_records()
        assert np.array_equal(out, rows)
        # wrap across the boundary
        rows2 = np.arange(4 * rs, dtype=np.uint8).reshape(4, rs) + 7
        assert ring.push_records(rows2) == 4
        # full now: next push drops
        assert ring.push_records(rows2[:2]) == 0
        assert ring.drops == 2
        out2 = ring.pop_records(2)
        assert np.array_equal(out2, rows2[:2])
        out3 = ring.pop_records()
        assert np.array_equal(out3, rows2[2:])
        assert ring.pop_records() is None
    finally:
        ring.close(unlink=True)


def test_weight_bus_seqlock():
    bus = WeightBus("drl_test_bus_a", 1 << 20, create=True)
    try:
        assert bus.fetch() is None
        payload = {"count": 7, "state_dict": {"w": torch.ones(3)}}
        bus.publish(payload)
        got = bus.fetch()
        assert got["count"] == 7
        assert torch.equal(got["state_dict"]["w"], torch.ones(3))
        bus.publish({"count": 8})
        assert bus.fetch()["count"] == 8
        assert bus.version == 2
    finally:
        bus.close(unlink=True)


def test_transport_session_end_to_end(tmp_path):
    codec = RecordCodec(make_apex_schema())
    learner_side = TransportSession(str(tmp_path), codec, num_rings=2,
                                    ring_slots=8, weight_capacity=1 << 20,
                                    create=True)
    try:
        actor_side = TransportSession(str(tmp_path), codec, num_rings=2,
                                      create=False)
        ep0 = ActorEndpoint(actor_side, 0)
        ep1 = ActorEndpoint(actor_side, 1)
        le = LearnerEndpoint(learner_side)
        n = 3
        cols = {
            "state": np.zeros((n, 4, 84, 84), np.uint8),
            "action": np.arange(n, dtype=np.int32),
            "reward": np.ones(n, np.float32),
            "next_state": np.zeros((n, 4, 84, 84), np.uint8),
            "done": np.zeros(n, np.float32),
        }
        ep0.push(cols, np.full(n, 0.5, np.float32))
        ep1.push(cols, np.full(n, 0.25, np.float32))
        got_cols, got_prio = le.drain()
        assert got_cols["action"].shape == (2 * n,)
        assert set(np.unique(got_prio)) == {0.25, 0.5}
        assert le.drain() is None
        le.publish({"count": 1, "state_dict": {}})
        assert ep0.fetch()["count"] == 1
        ep0.push_reward(0, 3.5)
        assert le.drain_rewards() == [3.5]
        actor_side.close()
    finally:
        learner_side.close()


def test_inproc_pipe():
    pipe = InprocPipe()
    pipe.push({"x": np.ones(2)}, np.ones(2))
    pipe.push({"x": np.zeros(3)}, np.zeros(3))
    cols, prio = pipe.drain()
    assert cols["x"].shape == (5,)
    assert pipe.drain() is None
    pipe.publish({"count": 3})
    assert pipe.fetch()["count"] == 3


def test_learner_endpoint_ring_partitioning(tmp_path):
    """Learner-DP partitioning: rank r of world w drains rings i % w == r."""
    codec = RecordCodec(make_apex_schema())
    sess = TransportSession(str(tmp_path), codec, num_rings=4, ring_slots=8,
                            weight_capacity=1 << 16, create=True)
    try:
        n = 1
        def push(i, action):
            cols = {
                "state": np.zeros((n, 4, 84, 84), np.uint8),
                "action": np.full(n, action, np.int32),
                "reward": np.zeros(n, np.float32),
                "next_state": np.zeros((n, 4, 84, 84), np.uint8),
                "done": np.zeros(n, np.float32),
            }
            ActorEndpoint(sess, i).push(cols, np.ones(n, np.float32))
        for i in range(4):
            push(i, action=i)
        le0 = LearnerEndpoint(sess, rank=0, world_size=2)
        le1 = LearnerEndpoint(sess, rank=1, world_size=2)
        c0, _ = le0.drain()
        c1, _ = le1.drain()
        assert sorted(c0["action"].tolist()) == [0, 2]
        assert sorted(c1["action"].tolist()) == [1, 3]
    finally:
        sess.close()


def test_pack_rows_matches_numpy_unpack():
    """C++ ingest pack (AoS shm records -> SoA staging) vs the numpy
    structured-dtype reference — host code, runs on CPU."""
    import numpy as np
    import torch

    from distributed_rl_amd.ops import hip_ext
    from distributed_rl_amd.replay import make_apex_schema

    ext = hip_ext(required=False)
    if ext is None or not hasattr(ext, "pack_rows"):
        import pytest

        pytest.skip("extension not built")
    from distributed_rl_amd.actors.transport import RecordCodec

    codec = RecordCodec(make_apex_schema(), with_priority=True)
    rng = np.random.default_rng(3)
    n = 37
    cols = {
        "state": rng.integers(0, 255, (n, 4, 84, 84), dtype=np.uint8),
        "action": rng.integers(0, 6, (n,), dtype=np.int32),
        "reward": rng.random(n, dtype=np.float32),
        "next_state": rng.integers(0, 255, (n, 4, 84, 84), dtype=np.uint8),
        "done": rng.random(n, dtype=np.float32),
    }
    prio = rng.random(n, dtype=np.float32)
    rec = codec.pack(cols, prio)
    raw = torch.from_numpy(np.ascontiguousarray(rec).view(np.uint8).reshape(-1))

    names = list(codec.schema) + ["priority"]
    f = codec.np_dtype.fields
    offs = [int(f[k][1]) for k in names]
    sizes = [int(f[k][0].itemsize) for k in names]
    dsts = [torch.zeros((64, *codec.schema[k][0]),
                        dtype=dict(codec.schema)[k][1])
            for k in codec.schema] + [torch.zeros(64)]
    ext.pack_rows(raw, codec.record_size, offs, sizes, dsts, 5)

    ref_cols, ref_prio = codec.unpack(rec)
    for i, k in enumerate(codec.schema):
        got = dsts[i][5 : 5 + n].numpy()
        assert np.array_equal(got.reshape(ref_cols[k].shape), ref_cols[k]), k
        assert not dsts[i][:5].any() and not dsts[i][5 + n :].any()
    assert np.array_equal(dsts[-1][5 : 5 + n].numpy(), ref_prio)
