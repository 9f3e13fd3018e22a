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


def _consolidate_host(keys, vals, times, diffs, kw=1, vb=8):
    """Canonical (key, val, time) consolidation of a raw update stream."""
    n = len(times)
    if n == 0:
        return [], [], [], []
    k = np.asarray(keys).reshape(n, kw)
    v = np.asarray(vals).reshape(n, vb) if vb else np.zeros((n, 0),
                                                            np.uint8)
    t = np.asarray(times)
    d = np.asarray(diffs)
    rows = {}
    for i in range(n):
        key = (tuple(int(x) for x in k[i]), bytes(v[i].tobytes()),
               int(t[i]))
        rows[key] = rows.get(key, 0) + int(d[i])
    out = sorted((key, s) for key, s in rows.items() if s != 0)
    return out


@pytest.mark.gpu
def test_gpu_halfjoin2_fused_matches_two_stage_with_errs():
    """The fused two-stage path probe (k_probe_path2) must equal the
    two-call halfjoin sequence — ok rows AND stage-2 error rows — on a
    path whose stage-2 closure divides by a sometimes-zero lookup val."""
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(77)
    # stage 1: lookup val = (k2, carry) ; stage 2: carry / divisor(k2)
    cl1 = abi.closure(
        [],
        [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_LOOKUP, 8, 8)],
        abi.schema(1, 8))
    cl2 = abi.closure(
        [],
        [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_COMPUTE, abi.MZ_COMPUTE_DIV_I64, 8,
                   arg0=0, arg1=0, arg0_src=abi.MZ_SRC_VAL_STREAM,
                   arg1_src=abi.MZ_SRC_VAL_LOOKUP)],
        abi.schema(1, 8))
    n1, n2, m = 400, 60, 300
    k1 = rng.integers(0, 50, n1).astype(np.int64)
    v1 = np.zeros((n1, 16), np.uint8)
    v1[:, 0:8] = rng.integers(0, 60, n1).astype(np.int64) \
        .reshape(-1, 1).view(np.uint8).reshape(n1, 8)
    v1[:, 8:16] = rng.integers(1, 1000, n1).astype(np.int64) \
        .reshape(-1, 1).view(np.uint8).reshape(n1, 8)
    k2 = np.arange(60).astype(np.int64)
    v2 = rng.integers(0, 4, n2).astype(np.int64)  # zeros common
    pk = rng.integers(0, 50, m).astype(np.int64)
    pd = rng.choice([-1, 1, 1], m).astype(np.int64)

    def build(ctx):
        a1, a2 = (ctx.arr_create(abi.schema(1, 16)),
                  ctx.arr_create(abi.schema(1, 8)))
        ctx.arr_insert(a1, abi.make_updates(
            k1, v1, np.zeros(n1, np.uint64), np.ones(n1, np.int64), 0, 1))
        ctx.arr_insert(a2, abi.make_updates(
            k2, v2.reshape(-1, 1).view(np.uint8), np.zeros(n2, np.uint64),
            np.ones(n2, np.int64), 0, 1))
        return a1, a2

    ga1, ga2 = build(g)
    pu = abi.make_updates(pk, None, np.full(m, 1, np.uint64), pd, 1, 2)
    fused = g.halfjoin2_dev(ga1, True, cl1, ga2, True, cl2, pu, 0)
    fk, fv, ft, fd = fused.to_host()
    g_errs = fused.errs_to_host()
    fused.release()
    # reference: two-call sequence on the oracle
    oa1, oa2 = build(o)
    s1 = o.halfjoin(oa1, abi.make_updates(
        pk, None, np.full(m, 1, np.uint64), pd, 1, 2), 0, True, cl1)
    s1k, s1v, s1t, s1d = s1
    u2 = abi.make_updates(np.asarray(s1k, np.int64).reshape(-1),
                          np.asarray(s1v, np.uint8),
                          np.asarray(s1t, np.uint64),
                          np.asarray(s1d, np.int64), 1, 2)
    s2 = o.halfjoin(oa2, u2, 8, True, cl2)
    o_errs = o.last_errs
    assert _consolidate_host(fk, fv, ft, fd) == \
        _consolidate_host(s2[0], s2[1], s2[2], s2[3])
    # err streams are consolidated on both sides already
    for x, y, what in zip(g_errs, o_errs, ("codes", "times", "diffs")):
        np.testing.assert_array_equal(np.asarray(x), np.asarray(y),
                                      err_msg=f"err {what}")
    assert len(g_errs[0]) > 0, "zero divisors must produce error rows"
    g.close()
    o.close()


def test_cmp_fields_signed_widths():
    """MZ_COMPUTE_CMP_FIELDS reads SIGNED ints at the filter's width —
    negative i32 dates must order correctly (oracle semantics)."""
    from pyoracle import OracleCtx
    o = OracleCtx()
    sch_in = abi.schema(1, 16)
    # val = (a i32 @0, b i32 @4, pad)
    cl = abi.closure(
        [abi.filt(abi.MZ_SRC_COMPUTE, abi.MZ_COMPUTE_CMP_FIELDS, 4,
                  abi.MZ_CMP_LT, 0, arg0=0, arg1=4,
                  arg0_src=abi.MZ_SRC_VAL_STREAM,
                  arg1_src=abi.MZ_SRC_VAL_STREAM)],
        [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_STREAM, 0, 4)],
        abi.schema(1, 4))
    pairs = [(-5, -3), (-3, -5), (-1, 2), (2, -1), (7, 7), (0, 1)]
    n = len(pairs)
    v = np.zeros((n, 16), np.uint8)
    v[:, 0:4] = np.array([p[0] for p in pairs], np.int32) \
        .reshape(-1, 1).view(np.uint8).reshape(n, 4)
    v[:, 4:8] = np.array([p[1] for p in pairs], np.int32) \
        .reshape(-1, 1).view(np.uint8).reshape(n, 4)
    u = abi.make_updates(np.arange(n, dtype=np.int64), v,
                         np.zeros(n, np.uint64), np.ones(n, np.int64),
                         0, 1)
    k, _, _, _ = o.map(sch_in, u, cl)
    kept = sorted(int(x) for x in np.asarray(k).reshape(-1))
    want = sorted(i for i, (a, b) in enumerate(pairs) if a < b)
    assert kept == want
    o.close()


def test_render_dec_reduced():
    from materialize_amd.tpch_exact import render_dec_reduced
    assert render_dec_reduced(123400, 4) == "12.34"
    assert render_dec_reduced(123000, 4) == "12.3"
    assert render_dec_reduced(120000, 4) == "12"
    assert render_dec_reduced(5, 4) == "0.0005"
    assert render_dec_reduced(0, 2) == "0"
    assert render_dec_reduced(100, 2) == "1"
    assert render_dec_reduced(101, 2) == "1.01"
