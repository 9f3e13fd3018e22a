"""Peek/readback surface (SURVEY §8f.2 — handle_peek/process_peeks
analog): per-key (val, summed diff) reads as of a time, on the oracle
(CPU) and GPU-vs-oracle parity (gpu mark)."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def build(ctx):
    sch = abi.schema(1, 8)
    arr = ctx.arr_create(sch)
    keys = np.array([1, 1, 2, 3, 3, 3], np.int64)
    vals = np.array([10, 11, 20, 30, 30, 31], np.int64)
    times = np.array([0, 1, 0, 0, 2, 1], np.uint64)
    diffs = np.array([1, 1, 1, 1, -1, 2], np.int64)
    ctx.arr_insert(arr, abi.make_updates(keys, vals.view(np.uint8), times,
                                         diffs, 0, 3))
    return arr


def rows(res):
    keys, vals, times, diffs = res
    n = len(times)
    v = vals.view(np.int64)
    return sorted((int(keys[i]), int(v[i]), int(diffs[i]))
                  for i in range(n))


def test_peek_oracle():
    ctx = OracleCtx()
    arr = build(ctx)
    # at t=0: (1,10,1), (2,20,1), (3,30,1)
    assert rows(ctx.peek(arr, [1, 2, 3, 9], 0)) == \
        [(1, 10, 1), (2, 20, 1), (3, 30, 1)]
    # at t=2: 3->30 cancelled (diff +1 at 0, -1 at 2); 3->31 has diff 2
    assert rows(ctx.peek(arr, [1, 3], 2)) == \
        [(1, 10, 1), (1, 11, 1), (3, 31, 2)]


@pytest.mark.gpu
def test_peek_gpu_parity():
    from materialize_amd._ffi import GpuCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(91)
    sch = abi.schema(1, 8)
    ga, oa = g.arr_create(sch), o.arr_create(sch)
    for t in range(4):
        n = 5000
        keys = rng.integers(0, 500, n).astype(np.int64)
        vals = rng.integers(0, 20, (n, 1)).astype(np.int64)
        diffs = rng.choice([-1, 1], n).astype(np.int64)
        u = abi.make_updates(keys, vals.view(np.uint8),
                             np.full(n, t, np.uint64), diffs, t, t + 1)
        g.arr_insert(ga, u)
        o.arr_insert(oa, u)
    probe = rng.integers(0, 600, 200).astype(np.int64)
    for t in (0, 2, 3):
        rg = g.peek(ga, probe, t)
        ro = o.peek(oa, probe, t)
        for a, b, what in zip(rg, ro, ("keys", "vals", "times", "diffs")):
            np.testing.assert_array_equal(a.view(np.uint8), b.view(np.uint8),
                                          err_msg=f"peek t={t}: {what}")
