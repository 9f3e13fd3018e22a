"""Merge-probe parity: a SORTED delta stream probing a large arrangement
takes the streaming merge-scan path (k_probe_merge — the sort-merge
restatement of half_join2's cursor seek, delta_join.rs:500) and must
produce results bit-identical to the hash-walk path and to the oracle.
MZ_PROBE_MERGE_MIN_MB=0 forces merge for any batch with >=4096 keys."""
import os

import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def _setup(ctx, rng, n_keys=20000, upds_per_key=2):
    sch = abi.schema(1, 8)
    arr = ctx.arr_create(sch)
    keys = np.repeat(np.arange(n_keys, dtype=np.int64), upds_per_key)
    n = len(keys)
    vals = rng.integers(0, 1 << 30, n).astype(np.int64) \
        .reshape(-1, 1).view(np.uint8).reshape(n, 8)
    diffs = np.ones(n, np.int64)
    u = abi.make_updates(keys, vals, np.zeros(n, np.uint64), diffs, 0, 1)
    ctx.arr_insert(arr, u)
    return arr


def _delta(rng, n_keys, m, t):
    # sorted, consolidated delta: unique ascending keys
    keys = np.sort(rng.choice(n_keys * 2, m, replace=False)).astype(np.int64)
    vals = (keys * 7 % 97).reshape(-1, 1).view(np.uint8).reshape(m, 8)
    diffs = rng.choice([-1, 1], m).astype(np.int64)
    return keys, vals, diffs


def _cl():
    return abi.closure(
        [], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8),
         abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
        abi.schema(1, 16))


def test_merge_vs_hash_vs_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    rng = np.random.default_rng(21)
    n_keys = 20000
    ga, oa = _setup(g, np.random.default_rng(3), n_keys), \
        _setup(o, np.random.default_rng(3), n_keys)
    cl = _cl()
    for t in range(1, 4):
        keys, vals, diffs = _delta(rng, n_keys, 6000, t)
        u = abi.make_updates(keys, vals, np.full(len(keys), t, np.uint64),
                             diffs, t, t + 1, sorted=1)
        os.environ["MZ_PROBE_MERGE_MIN_MB"] = "0"
        try:
            rm = g.halfjoin(ga, u, 8, True, cl)
        finally:
            del os.environ["MZ_PROBE_MERGE_MIN_MB"]
        os.environ["MZ_PROBE_MERGE"] = "0"
        try:
            rh = g.halfjoin(ga, u, 8, True, cl)
        finally:
            del os.environ["MZ_PROBE_MERGE"]
        ro = o.halfjoin(oa, u, 8, True, cl)
        for x, y, what in zip(rm, rh, ("keys", "vals", "times", "diffs")):
            np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                          err_msg=f"merge-vs-hash {what}")
        for x, y, what in zip(rm, ro, ("keys", "vals", "times", "diffs")):
            np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                          err_msg=f"merge-vs-oracle {what}")
    g.close()
    o.close()


def test_merge_probe_multibatch_spine():
    """Sorted delta against a multi-batch spine: big batches merge-probe,
    small ones hash-walk — the union must match the oracle."""
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    sch = abi.schema(1, 8)
    ga, oa = g.arr_create(sch), o.arr_create(sch)
    rng = np.random.default_rng(31)
    # several inserts at different times -> spine with several batches
    for t in range(4):
        n = 15000 if t == 0 else 700
        keys = rng.integers(0, 30000, n).astype(np.int64)
        vals = rng.integers(0, 1000, n).astype(np.int64) \
            .reshape(-1, 1).view(np.uint8).reshape(n, 8)
        diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
        u = abi.make_updates(keys, vals, np.full(n, t, np.uint64), diffs,
                             t, t + 1)
        g.arr_insert(ga, u)
        o.arr_insert(oa, u)
    cl = _cl()
    keys, vals, diffs = _delta(rng, 15000, 5000, 4)
    u = abi.make_updates(keys, vals, np.full(len(keys), 4, np.uint64),
                         diffs, 4, 5, sorted=1)
    os.environ["MZ_PROBE_MERGE_MIN_MB"] = "0"
    try:
        rm = g.halfjoin(ga, u, 8, False, cl)
    finally:
        del os.environ["MZ_PROBE_MERGE_MIN_MB"]
    ro = o.halfjoin(oa, u, 8, False, cl)
    for x, y, what in zip(rm, ro, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                      err_msg=what)
    g.close()
    o.close()


def test_q3_step_dev_parity():
    """The bench path (step_dev: consolidate-once, sorted inserts + merge
    probes) must equal the structured step() path on the oracle."""
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import Q3Dataflow
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    df_g, df_o = Q3Dataflow(g), Q3Dataflow(o)
    gen_g, gen_o = TpchGen(sf=0.01, seed=7), TpchGen(sf=0.01, seed=7)
    df_g.load(gen_g)
    df_o.load(gen_o)
    os.environ["MZ_PROBE_MERGE_MIN_MB"] = "0"
    try:
        for t in range(1, 5):
            churn_g = gen_g.churn(1500)
            churn_o = gen_o.churn(1500)
            upd = {
                "lineitem": churn_g["lineitem"],
                "orders_by_orderkey": churn_g["orders"],
                "orders_by_custkey": churn_g["orders_by_cust"],
            }
            upd_desc = {}
            for name, (kk, vv, dd) in upd.items():
                upd_desc[name] = abi.make_updates(
                    np.ascontiguousarray(kk, np.int64), vv,
                    np.full(len(kk), t, np.uint64),
                    np.ascontiguousarray(dd, np.int64), t, t + 1)
            cg = df_g.step_dev(upd_desc, t)
            _, co = df_o.step(churn_o, t)
            kg, vg, tg, dg = (cg.to_host() if cg is not None
                              else (np.empty(0, np.int64),
                                    np.empty(0, np.uint8),
                                    np.empty(0, np.uint64),
                                    np.empty(0, np.int64)))
            if cg is not None:
                cg.release()
            ko, vo, to, do_ = (co.to_host() if co is not None
                               else (np.empty(0, np.int64),
                                     np.empty(0, np.uint8),
                                     np.empty(0, np.uint64),
                                     np.empty(0, np.int64)))
            if co is not None:
                co.release()
            np.testing.assert_array_equal(np.asarray(kg).view(np.uint8),
                                          np.asarray(ko).view(np.uint8),
                                          err_msg=f"t={t} keys")
            np.testing.assert_array_equal(np.asarray(vg).view(np.uint8),
                                          np.asarray(vo).view(np.uint8),
                                          err_msg=f"t={t} vals")
            np.testing.assert_array_equal(np.asarray(dg),
                                          np.asarray(do_),
                                          err_msg=f"t={t} diffs")
    finally:
        del os.environ["MZ_PROBE_MERGE_MIN_MB"]
    g.close()
    o.close()
