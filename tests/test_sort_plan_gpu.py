"""Cached-sort-plan redo path: a device-staged insert whose column
ranges escape the cached bounds is flagged by the device coverage check
and synchronously rebuilt at the flush (before anything installs) — the
arrangement's contents stay bit-exact to the oracle's."""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def _dev_updates(keys, vals, times, diffs, lower, upper):
    import torch
    from materialize_amd._abi import make_updates_from_torch
    kt = torch.from_numpy(np.ascontiguousarray(keys, np.int64)).cuda()
    vt = torch.from_numpy(
        np.ascontiguousarray(vals, np.uint8).reshape(-1)).cuda()
    tt = torch.from_numpy(
        np.ascontiguousarray(times, np.int64)).cuda()
    dt = torch.from_numpy(np.ascontiguousarray(diffs, np.int64)).cuda()
    u = make_updates_from_torch(kt, vt, tt, dt, lower, upper)
    return u, (kt, vt, tt, dt)


def test_plan_mismatch_redo_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    sch = abi.schema(1, 8)
    ga, oa = g.arr_create(sch), o.arr_create(sch)
    rng = np.random.default_rng(59)
    keep = []
    # batch 0 primes the cache on a NARROW range; batch 1 explodes the
    # key and val ranges (guaranteed coverage-check mismatch -> redo);
    # batch 2 reprimes and stays cached
    ranges = [(0, 1000, 0, 50), (-2**40, 2**40, -2**30, 2**30),
              (-2**40, 2**40, -2**30, 2**30)]
    for t, (klo, khi, vlo, vhi) in enumerate(ranges):
        n = 4000
        keys = rng.integers(klo, khi, n).astype(np.int64)
        vals = rng.integers(vlo, vhi, n).astype(np.int64) \
            .reshape(-1, 1).view(np.uint8).reshape(n, 8)
        diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
        times = np.full(n, t, np.int64)
        du, refs = _dev_updates(keys, vals, times, diffs, t, t + 1)
        keep.append(refs)  # device inputs must outlive the flush
        g.arr_insert_async(ga, du)
        g.arr_flush(ga)
        o.arr_insert(oa, abi.make_updates(keys, vals,
                                          times.view(np.uint64), diffs,
                                          t, t + 1))
    # identity probe of every key present: arrangement contents compare
    cl = abi.closure([], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
                     [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
                     abi.schema(1, 8))
    pk = np.unique(np.concatenate(
        [rng.integers(r[0], r[1], 2000) for r in ranges]).astype(np.int64))
    pu = abi.make_updates(pk, None, np.full(len(pk), 3, np.uint64),
                          np.ones(len(pk), np.int64), 3, 4)
    rg = g.halfjoin(ga, pu, 0, True, cl)
    ro = o.halfjoin(oa, pu, 0, True, cl)
    for x, y, what in zip(rg, ro, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(np.asarray(x).view(np.uint8),
                                      np.asarray(y).view(np.uint8),
                                      err_msg=what)
    g.close()
    o.close()
