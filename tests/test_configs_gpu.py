"""BASELINE config 4 at size on the GPU (reduce_core + consolidate path):
1M-row stream, ~1M distinct keys, SUM(f64)+COUNT — bit-exact vs oracle,
plus distinct (zero-agg reduce) parity."""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def assert_same(a, b, label):
    for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                      err_msg=f"{label}: {what}")


def test_config4_reduce_1m_rows():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(71)
    aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                          is_float=0, nullable=0),
            abi.Aggregate(func=abi.MZ_AGG_SUM_F64, off=0, width=8,
                          is_float=1, nullable=0)]
    spec = abi.reduce_spec(aggs, abi.schema(1, 8))
    gop, oop = g.reduce_create(spec), o.reduce_create(spec)
    for step in range(3):
        n = 1_000_000
        keys = rng.integers(0, 1_000_000, n).astype(np.int64)
        xs = rng.uniform(0, 1000, n)
        diffs = (np.ones(n, np.int64) if step == 0
                 else rng.choice([-1, 1], n).astype(np.int64))
        u = abi.make_updates(keys, xs.view(np.uint8),
                             np.full(n, step, np.uint64), diffs, step,
                             step + 1)
        assert_same(g.reduce_push(gop, u), o.reduce_push(oop, u),
                    f"config4 step {step}")


def test_distinct_parity():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(73)
    spec = abi.reduce_spec([], abi.schema(1, 0))
    gop, oop = g.reduce_create(spec), o.reduce_create(spec)
    for step in range(4):
        n = 20_000
        keys = rng.integers(0, 2_000, n).astype(np.int64)
        diffs = rng.choice([-1, 1], n).astype(np.int64)
        u = abi.make_updates(keys, None, np.full(n, step, np.uint64),
                             diffs, step, step + 1)
        assert_same(g.reduce_push(gop, u), o.reduce_push(oop, u),
                    f"distinct step {step}")
