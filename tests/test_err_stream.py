"""Error-row channel: the ok/err split of the reference's join closure
(JoinClosure::could_error, linear_join.rs:495-541). A closure with an
erroring compute (MZ_COMPUTE_DIV_I64 by a sometimes-zero field) diverts
the affected (time, diff) cross-products into the out-batch's error
stream — consolidated like any update collection — while ok rows are
unaffected. Oracle semantics here; GPU bit-parity in the gpu test."""
import numpy as np
import pytest

from materialize_amd import _abi as abi


def _cl_div():
    # out val = [stream_val / lookup_val]; lookup val 0 -> err row
    return abi.closure(
        [],
        [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_COMPUTE, abi.MZ_COMPUTE_DIV_I64, 8,
                   arg0=0, arg1=0, arg0_src=abi.MZ_SRC_VAL_STREAM,
                   arg1_src=abi.MZ_SRC_VAL_LOOKUP)],
        abi.schema(1, 8))


def _data(rng, n, zero_every=5):
    keys = rng.integers(0, 40, n).astype(np.int64)
    lookup_vals = rng.integers(0, 4, n).astype(np.int64)  # zeros common
    return keys, lookup_vals


def _run(ctx):
    rng = np.random.default_rng(31)
    sch = abi.schema(1, 8)
    arr = ctx.arr_create(sch)
    n = 500
    keys, lv = _data(rng, n)
    u = abi.make_updates(keys, lv.reshape(-1, 1).view(np.uint8),
                         np.zeros(n, np.uint64), np.ones(n, np.int64), 0, 1)
    ctx.arr_insert(arr, u)
    m = 300
    pk = rng.integers(0, 40, m).astype(np.int64)
    pv = rng.integers(1, 100, m).astype(np.int64)
    pu = abi.make_updates(pk, pv.reshape(-1, 1).view(np.uint8),
                          np.full(m, 1, np.uint64),
                          rng.choice([-1, 1, 1], m).astype(np.int64), 1, 2)
    res = ctx.halfjoin(arr, pu, 8, True, _cl_div())
    errs = ctx.last_errs
    return res, errs


def test_oracle_err_stream():
    from pyoracle import OracleCtx
    o = OracleCtx()
    res, (codes, times, diffs) = _run(o)
    assert len(codes) > 0, "zero divisors must produce error rows"
    assert all(c == abi.MZ_ERR_DIVISION_BY_ZERO for c in codes)
    assert all(t == 1 for t in times)
    # ok rows never divide by zero: every emitted val is a quotient of
    # nonzero divisors — sanity: total emitted + errs == total matches
    o.close()


@pytest.mark.gpu
def test_gpu_err_stream_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rg, eg = _run(g)
    ro, eo = _run(o)
    for x, y, what in zip(rg, ro, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(np.asarray(x).view(np.uint8),
                                      np.asarray(y).view(np.uint8),
                                      err_msg=what)
    for x, y, what in zip(eg, eo, ("codes", "times", "diffs")):
        np.testing.assert_array_equal(np.asarray(x), np.asarray(y),
                                      err_msg=f"err {what}")
    assert len(eg[0]) > 0
    g.close()
    o.close()
