"""Property-based tests (hypothesis) for the wire codec, SPSC ring and PER
invariants — the pieces where silent corruption would be hardest to spot."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from distributed_rl_amd.actors.transport import RecordCodec, SpscRing
from distributed_rl_amd.replay import TorchPER

_DTYPES = [torch.uint8, torch.int32, torch.float32, torch.float16]


@st.composite
def schemas(draw):
    n_cols = draw(st.integers(1, 4))
    schema = {}
    for i in range(n_cols):
        nd = draw(st.integers(0, 2))
        shape = tuple(draw(st.integers(1, 5)) for _ in range(nd))
        schema[f"c{i}"] = (shape, draw(st.sampled_from(_DTYPES)))
    return schema


@settings(max_examples=30, deadline=None)
@given(schema=schemas(), n=st.integers(1, 7), with_prio=st.booleans())
def test_codec_roundtrip_random_schema(schema, n, with_prio):
    codec = RecordCodec(schema, with_priority=with_prio)
    rng = np.random.default_rng(0)
    cols = {}
    for name, (shape, dtype) in schema.items():
        if dtype == torch.uint8:
            cols[name] = rng.integers(0, 255, (n, *shape)).astype(np.uint8)
        elif dtype == torch.int32:
            cols[name] = rng.integers(-5, 5, (n, *shape)).astype(np.int32)
        elif dtype == torch.float16:
            cols[name] = rng.random((n, *shape)).astype(np.float16)
        else:
            cols[name] = rng.random((n, *shape)).astype(np.float32)
    prio = rng.random(n).astype(np.float32) if with_prio else None
    rec = codec.pack(cols, prio)
    # survive a raw-bytes round trip (the wire representation)
    raw = rec.tobytes()
    rec2 = np.frombuffer(raw, dtype=codec.np_dtype)
    cols2, prio2 = codec.unpack(rec2)
    for name in schema:
        assert np.array_equal(cols2[name], cols[name]), name
    if with_prio:
        assert np.allclose(prio2, prio)


@settings(max_examples=20, deadline=None)
@given(ops=st.lists(st.integers(1, 9), min_size=1, max_size=40),
       slots=st.integers(2, 16))
def test_spsc_ring_fifo_order_random_ops(ops, slots):
    """Alternating pushes/pops of random sizes never lose or reorder
    records (drops allowed only when full, and counted)."""
    rs = 8
    ring = SpscRing(f"drl_prop_{abs(hash(tuple(ops))) % 99999}_{slots}", rs,
                    slots, create=True)
    try:
        next_val = 0
        expect = []
        got = []
        for i, k in enumerate(ops):
            if i % 2 == 0:
                rows = np.arange(next_val, next_val + k,
                                 dtype=np.uint64).view(np.uint8).reshape(k, rs)
                wrote = ring.push_records(rows)
                expect.extend(range(next_val, next_val + wrote))
                next_val += k  # dropped tail values are never retried here
            else:
                out = ring.pop_records(k)
                if out is not None:
                    got.extend(out.view(np.uint64).ravel().tolist())
        out = ring.pop_records()
        if out is not None:
            got.extend(out.view(np.uint64).ravel().tolist())
        assert got == expect[: len(got)]
        assert len(got) == len(expect)
    finally:
        ring.close(unlink=True)


@settings(max_examples=15, deadline=None)
@given(prios=st.lists(st.floats(0.01, 100.0), min_size=2, max_size=32),
       beta=st.floats(0.1, 1.0))
def test_per_weight_invariants(prios, beta):
    """IS weights are in (0, 1] and the min-priority element has weight 1."""
    per = TorchPER(len(prios), {"x": ((), torch.float32)})
    t = torch.tensor(prios, dtype=torch.float32)
    per.push({"x": torch.zeros(len(prios))}, t)
    _, idx, w = per.sample(64, beta=beta)
    assert (w > 0).all() and (w <= 1.0 + 1e-4).all()
    # sampling the argmin priority directly yields weight ~1
    amin = int(t.argmin())
    probs = t / t.sum()
    max_w = (1.0 / (len(prios) * probs[amin])) ** beta
    w_amin = (1.0 / (len(prios) * probs[amin])) ** beta / max_w
    assert abs(w_amin - 1.0) < 1e-5
