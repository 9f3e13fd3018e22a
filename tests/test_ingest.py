"""Pinned-buffer ingest feeder: the host->HBM hand-off
produces exactly the batch that direct staging does."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from materialize_amd.ingest import PinnedFeeder


def batch(rng, n, t, vb=8):
    keys = rng.integers(0, 50, (n, 1)).astype(np.int64)
    vals = rng.integers(0, 5, (n, vb)).astype(np.uint8)
    times = np.full(n, t, np.uint64)
    diffs = rng.integers(-2, 3, n).astype(np.int64)
    return keys, vals, times, diffs


def test_cpu_fallback_feeds_oracle():
    from pyoracle import OracleCtx
    rng = np.random.default_rng(3)
    sch = abi.schema(1, 8)
    fd = PinnedFeeder(sch, capacity_rows=4096, device=None)
    ctx = OracleCtx()
    a_fed, a_ref = ctx.arr_create(sch), ctx.arr_create(sch)
    for t in range(4):
        keys, vals, times, diffs = batch(rng, 1000, t)
        u, ready = fd.stage(keys, vals, times, diffs, t, t + 1)
        ready()
        ctx.arr_insert(a_fed, u)
        ctx.arr_insert(a_ref, abi.make_updates(keys, vals, times, diffs,
                                               t, t + 1))
    sf, sr = ctx.arr_stats(a_fed), ctx.arr_stats(a_ref)
    assert sf == sr
    ctx.close()


@pytest.mark.gpu
def test_pinned_path_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    rng = np.random.default_rng(5)
    sch = abi.schema(1, 8)
    g, o = GpuCtx(), OracleCtx()
    fd = PinnedFeeder(sch, capacity_rows=4096, device="cuda:0", depth=2)
    assert fd.gpu, "pinned path must be active on a GPU box"
    ga, oa = g.arr_create(sch), o.arr_create(sch)
    cl_kf = [abi.field(abi.MZ_SRC_KEY, 0, 8)]
    cl_vf = [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)]
    cl = abi.closure([], cl_kf, cl_vf, abi.schema(1, 8))
    for t in range(5):
        keys, vals, times, diffs = batch(rng, 2000, t)
        u, ready = fd.stage(keys, vals, times, diffs, t, t + 1)
        ready()
        g.arr_insert(ga, u)
        fd.mark_consumed()  # synchronous push: consumed on return
        o.arr_insert(oa, abi.make_updates(keys, vals, times, diffs,
                                          t, t + 1))
        pk = rng.integers(0, 50, (300, 1)).astype(np.int64)
        pu = abi.make_updates(pk, None, np.full(300, t, np.uint64),
                              np.ones(300, np.int64), t, t + 1)
        rg, ro = g.halfjoin(ga, pu, 0, True, cl), \
            o.halfjoin(oa, pu, 0, True, cl)
        for x, y in zip(rg, ro):
            np.testing.assert_array_equal(x, y)
    g.close()
    o.close()

@pytest.mark.gpu
def test_slot_reuse_without_consume_raises():
    """Lifetime contract: staging past the ring depth without
    mark_consumed() must raise, not overwrite in-flight buffers."""
    from materialize_amd._ffi import GpuCtx
    rng = np.random.default_rng(7)
    sch = abi.schema(1, 8)
    g = GpuCtx()
    fd = PinnedFeeder(sch, capacity_rows=1024, device="cuda:0", depth=2)
    assert fd.gpu
    for t in range(2):
        keys, vals, times, diffs = batch(rng, 100, t)
        fd.stage(keys, vals, times, diffs, t, t + 1)
    keys, vals, times, diffs = batch(rng, 100, 2)
    with pytest.raises(RuntimeError):
        fd.stage(keys, vals, times, diffs, 2, 3)
    fd.mark_consumed()
    u, ready = fd.stage(keys, vals, times, diffs, 2, 3)  # now fits
    ready()
    g.close()
