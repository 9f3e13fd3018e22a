"""Operator lifecycle: resident-table compaction/growth under churn
(zero-count rows reclaimed, capacity doubles when live + incoming
exceeds it — render/threshold.rs erases zeroed entries; reduce trace
compaction drops empty accums) and full teardown of drop entry points
(no leaked arrangements / state tables; dropped ops vanish from the
ctx registries so mz_gpu_sync never touches them again)."""
import os

import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def _mku(keys, vals, t, diffs):
    n = len(keys)
    return abi.make_updates(
        np.asarray(keys, np.int64), vals,
        np.full(n, t, np.uint64), np.asarray(diffs, np.int64), t, t + 1)


def test_threshold_grows_past_initial_capacity():
    """Churn far beyond a tiny initial capacity: zeroed rows are
    reclaimed and the table grows; results stay bit-exact vs oracle."""
    os.environ["MZ_GPU_THR_CAP"] = "64"
    try:
        from materialize_amd._ffi import GpuCtx
        from pyoracle import OracleCtx
        g, o = GpuCtx(), OracleCtx()
        sch = abi.schema(1, 8)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        rng = np.random.default_rng(11)
        live = {}
        for t in range(12):
            # insert 40 fresh records, retract ~40 older ones to zero
            ins = np.arange(t * 40, t * 40 + 40, dtype=np.int64)
            vals = np.asarray(ins % 7, np.int64).view(np.uint8)
            keys, vv, dd = list(ins), [vals.reshape(40, 8)], [np.ones(40)]
            for k in list(live)[:40]:
                keys.append(k)
                vv.append(live.pop(k).reshape(1, 8))
                dd.append([-1.0])
            for k in ins:
                live[int(k)] = np.asarray([k % 7], np.int64).view(np.uint8)
            keys = np.asarray(keys, np.int64)
            vals = np.concatenate([np.atleast_2d(v) for v in vv])
            diffs = np.concatenate([np.atleast_1d(d) for d in dd]) \
                .astype(np.int64)
            a = g.threshold_push(gop, _mku(keys, vals, t, diffs))
            b = o.threshold_push(oop, _mku(keys, vals, t, diffs))
            for x, y in zip(a, b):
                np.testing.assert_array_equal(x.view(np.uint8),
                                              y.view(np.uint8),
                                              err_msg=f"t={t}")
        g.close()
        o.close()
    finally:
        del os.environ["MZ_GPU_THR_CAP"]


def test_reduce_grows_past_initial_capacity():
    os.environ["MZ_GPU_RED_CAP"] = "32"
    try:
        from materialize_amd._ffi import GpuCtx
        from pyoracle import OracleCtx
        g, o = GpuCtx(), OracleCtx()
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                              is_float=0, nullable=0)]
        spec = abi.reduce_spec(aggs, abi.schema(1, 8))
        gop, oop = g.reduce_create(spec), o.reduce_create(spec)
        rng = np.random.default_rng(13)
        for t in range(6):
            n = 100
            keys = (rng.integers(0, 60, n) + 60 * t).astype(np.int64)
            vals = rng.integers(0, 50, n).astype(np.int64) \
                .reshape(-1, 1).view(np.uint8).reshape(n, 8)
            diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
            a = g.reduce_push(gop, _mku(keys, vals, t, diffs))
            b = o.reduce_push(oop, _mku(keys, vals, t, diffs))
            for x, y in zip(a, b):
                np.testing.assert_array_equal(x.view(np.uint8),
                                              y.view(np.uint8),
                                              err_msg=f"t={t}")
        g.close()
        o.close()
    finally:
        del os.environ["MZ_GPU_RED_CAP"]


def test_drop_entry_points_tear_down():
    """Create, use, drop each operator kind; the ctx must stay fully
    usable (sync iterates live registries only)."""
    import ctypes as C

    from materialize_amd._ffi import GpuCtx
    g = GpuCtx()
    sch = abi.schema(1, 8)
    rng = np.random.default_rng(3)
    n = 500
    keys = rng.integers(0, 50, n).astype(np.int64)
    vals = rng.integers(0, 9, n).astype(np.int64) \
        .reshape(-1, 1).view(np.uint8).reshape(n, 8)
    u = _mku(keys, vals, 0, np.ones(n, np.int64))

    # arrangement: insert then drop
    a1 = g.arr_create(sch)
    g.arr_insert(a1, u)
    g.lib.mz_gpu_arr_drop(g.ctx, a1)

    # join over two fresh arrangements, then drop all three
    a2, a3 = g.arr_create(sch), g.arr_create(sch)
    g.arr_insert(a2, u)
    g.arr_insert(a3, u)
    cl = abi.closure([], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
                     [abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8)],
                     abi.schema(1, 8))
    jop = g.join_create(a2, a3, cl)
    g.join_push(jop, 1, u)
    g.lib.mz_gpu_join_drop.argtypes = [C.c_void_p, C.c_void_p]
    g.lib.mz_gpu_join_drop(g.ctx, jop)
    g.lib.mz_gpu_arr_drop(g.ctx, a2)
    g.lib.mz_gpu_arr_drop(g.ctx, a3)

    # threshold + topk + reduce + minmax: push then drop
    th = g.threshold_create(sch)
    g.threshold_push(th, u)
    g.lib.mz_gpu_threshold_drop(g.ctx, th)

    tspec = abi.topk_spec(abi.schema(1, 8), [(0, 8, 0)],
                          offset=0, limit=3)
    tk = g.topk_create(tspec)
    g.topk_push(tk, u)
    g.lib.mz_gpu_topk_drop(g.ctx, tk)

    aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                          is_float=0, nullable=0)]
    rop = g.reduce_create(abi.reduce_spec(aggs, abi.schema(1, 8)))
    g.reduce_push(rop, u)
    g.lib.mz_gpu_reduce_drop.argtypes = [C.c_void_p, C.c_void_p]
    g.lib.mz_gpu_reduce_drop(g.ctx, rop)

    mm = g.minmax_create(abi.schema(1, 8), False, [16])
    g.minmax_push(mm, u)
    g.lib.mz_gpu_minmax_drop(g.ctx, mm)

    # ctx still alive and consistent after all drops
    g.lib.mz_gpu_sync(g.ctx)
    a4 = g.arr_create(sch)
    g.arr_insert(a4, u)
    nb, nu, by = g.arr_stats(a4)
    assert nu > 0
    g.close()
