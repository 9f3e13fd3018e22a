"""FlatMap / key-prep closure over a stream (no lookup): oracle semantics
+ GPU parity."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def scenario(ctx):
    rng = np.random.default_rng(55)
    n = 3000
    keys = rng.integers(-100, 100, n).astype(np.int64)
    vals = rng.integers(0, 50, (n, 2)).astype(np.int64)
    diffs = rng.choice([-2, -1, 1, 2], n).astype(np.int64)
    u = abi.make_updates(keys, vals.view(np.uint8),
                         np.arange(n, dtype=np.uint64) % 3, diffs, 0, 3)
    # filter val[0] > 10; re-key to val[1]; out val = (old key, val[0])
    cl = abi.closure(
        [abi.filt(abi.MZ_SRC_VAL_STREAM, 0, 8, abi.MZ_CMP_GT, 10)],
        [abi.field(abi.MZ_SRC_VAL_STREAM, 8, 8)],
        [abi.field(abi.MZ_SRC_KEY, 0, 8),
         abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8)],
        abi.schema(1, 16))
    return ctx.map(abi.schema(1, 16), u, cl), (keys, vals, diffs)


def test_oracle_map():
    res, (keys, vals, diffs) = scenario(OracleCtx())
    okeys, ovals, otimes, odiffs = res
    m = vals[:, 0] > 10
    assert len(otimes) == m.sum()
    ov = ovals.view(np.int64).reshape(-1, 2)
    np.testing.assert_array_equal(okeys, vals[m, 1])
    np.testing.assert_array_equal(ov[:, 0], keys[m])
    np.testing.assert_array_equal(ov[:, 1], vals[m, 0])
    np.testing.assert_array_equal(odiffs, diffs[m])


@pytest.mark.gpu
def test_gpu_map_parity():
    from materialize_amd._ffi import GpuCtx
    a, _ = scenario(GpuCtx())
    b, _ = scenario(OracleCtx())
    for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                      err_msg=what)
