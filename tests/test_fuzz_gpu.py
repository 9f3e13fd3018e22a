"""Randomized mixed-operation fuzz: a seeded schedule of arrangement
inserts (sync and async), probes, reduces, thresholds and topk pushes is
driven through the HIP engine and the oracle in lockstep; every output
must stay bit-exact. Exercises the per-arrangement lane machinery
(async insert + implicit flush ordering) under irregular interleavings
that the structured dataflow tests never produce."""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def rand_batch(rng, t, nkeys, vb, n):
    keys = rng.integers(-4, nkeys, (n, 1)).astype(np.int64)
    vals = rng.integers(0, 4, (n, vb)).astype(np.uint8) if vb else None
    diffs = rng.integers(-2, 3, n).astype(np.int64)
    return abi.make_updates(keys, vals, np.full(n, t, np.uint64), diffs,
                            t, t + 1)


def assert_same(rg, ro, label):
    for i, (x, y) in enumerate(zip(rg, ro)):
        np.testing.assert_array_equal(x, y, err_msg=f"{label} col {i}")


@pytest.mark.parametrize("seed", [101, 103, 107])
def test_mixed_op_schedule(seed):
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(seed)
    sched_rng = np.random.default_rng(seed + 1)
    VB = 8
    sch = abi.schema(1, VB)
    n_arrs = 3
    garrs = [g.arr_create(sch) for _ in range(n_arrs)]
    oarrs = [o.arr_create(sch) for _ in range(n_arrs)]
    cl = abi.closure([], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
                     [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, VB)],
                     abi.schema(1, VB))
    red = abi.reduce_spec(
        [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                       is_float=0, nullable=0)], abi.schema(1, VB))
    gred, ored = g.reduce_create(red), o.reduce_create(red)
    gthr, othr = g.threshold_create(sch), o.threshold_create(sch)
    tkspec = abi.topk_spec(sch, [(0, 8, 1)], offset=0, limit=2)
    gtk, otk = g.topk_create(tkspec), o.topk_create(tkspec)

    for t in range(24):
        op = sched_rng.integers(0, 6)
        a = int(sched_rng.integers(0, n_arrs))
        n = int(sched_rng.integers(1, 800))
        u = rand_batch(rng, t, 60, VB, n)
        if op == 0:  # async insert (flushed by the next probe or sync)
            g.arr_insert_async(garrs[a], u)
            o.arr_insert_async(oarrs[a], u)
        elif op == 1:  # sync insert
            g.arr_insert(garrs[a], u)
            o.arr_insert(oarrs[a], u)
        elif op == 2:  # probe (halfjoin le)
            assert_same(g.halfjoin(garrs[a], u, VB, True, cl),
                        o.halfjoin(oarrs[a], u, VB, True, cl),
                        f"t{t} probe arr{a}")
        elif op == 3:
            assert_same(g.reduce_push(gred, u), o.reduce_push(ored, u),
                        f"t{t} reduce")
        elif op == 4:
            assert_same(g.threshold_push(gthr, u),
                        o.threshold_push(othr, u), f"t{t} threshold")
        else:
            # topk requires non-negative accumulated multiplicities:
            # feed insert-only batches
            ku = abi.make_updates(
                rng.integers(0, 30, (n, 1)).astype(np.int64),
                rng.integers(0, 3, (n, VB)).astype(np.uint8),
                np.full(n, t, np.uint64), np.ones(n, np.int64), t, t + 1)
            assert_same(g.topk_push(gtk, ku), o.topk_push(otk, ku),
                        f"t{t} topk")
    # end-state check: every arrangement drained through a final probe
    for a in range(n_arrs):
        pu = rand_batch(rng, 24, 60, VB, 500)
        assert_same(g.halfjoin(garrs[a], pu, VB, True, cl),
                    o.halfjoin(oarrs[a], pu, VB, True, cl),
                    f"final probe arr{a}")
    g.close()
    o.close()
