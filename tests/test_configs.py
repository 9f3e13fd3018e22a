"""BASELINE.json configs 1 and 4 as correctness gates.

Config 1: two-table equi-join COUNT(*) over a static CSV, CPU oracle,
one worker (plumbing gate — scaled to 100k rows so the CPU suite stays
fast; the join algorithm is size-independent).
Config 4: GROUP BY SUM over a keyed stream (reduce_core + consolidate
path) — CPU correctness here; GPU parity at size in test_configs_gpu.
"""
import csv
import io

import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def test_config1_csv_equijoin_count():
    rng = np.random.default_rng(17)
    n = 100_000
    a_keys = rng.integers(0, n, n).astype(np.int64)
    b_keys = rng.integers(0, n, n).astype(np.int64)
    # write + read back as CSV (the config's ingest shape)
    buf = io.StringIO()
    w = csv.writer(buf)
    for k in a_keys[:1000]:
        w.writerow([int(k)])
    buf.seek(0)
    back = np.array([int(r[0]) for r in csv.reader(buf)], np.int64)
    assert np.array_equal(back, a_keys[:1000])

    ctx = OracleCtx()
    sch = abi.schema(1, 0)
    a1, a2 = ctx.arr_create(sch), ctx.arr_create(sch)
    # join closure: key -> const 0 (global count), no val
    cl = abi.closure(
        [], [abi.field(abi.MZ_SRC_COMPUTE, abi.MZ_COMPUTE_CONST0, 8)],
        [], abi.schema(1, 0))
    op = ctx.join_create(a1, a2, cl)

    def seal(keys):
        u = abi.make_updates(keys, None, np.zeros(len(keys), np.uint64),
                             np.ones(len(keys), np.int64), 0, 1)
        k, v, t, d = ctx.consolidate(sch, u)
        return abi.make_updates(k, v, t, d, 0, 1)

    ua, ub = seal(a_keys), seal(b_keys)
    ctx.arr_push(a1, ua)
    o1 = ctx.join_push(op, 1, ua)
    ctx.arr_push(a2, ub)
    o2 = ctx.join_push(op, 2, ub)
    # COUNT(*) = sum of diffs of the joined collection
    total = int(o1[3].sum() + o2[3].sum())
    # numpy reference
    ca = np.bincount(a_keys, minlength=n)
    cb = np.bincount(b_keys, minlength=n)
    want = int((ca.astype(np.int64) * cb.astype(np.int64)).sum())
    assert total == want


def test_config4_group_by_sum_oracle():
    """GROUP BY SUM(f64) + SUM(i64) over a keyed stream vs numpy."""
    rng = np.random.default_rng(23)
    n, nkeys = 200_000, 20_000
    keys = rng.integers(0, nkeys, n).astype(np.int64)
    f = rng.uniform(0, 1000, n)
    ctx = OracleCtx()
    aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_F64, off=0, width=8,
                          is_float=1, nullable=0)]
    spec = abi.reduce_spec(aggs, abi.schema(1, 8))
    op = ctx.reduce_create(spec)
    res = ctx.reduce_push(op, abi.make_updates(
        keys, f.view(np.uint8), np.zeros(n, np.uint64),
        np.ones(n, np.int64), 0, 1))
    okeys, ovals, otimes, odiffs = res
    m = len(otimes)
    assert (odiffs == 1).all()
    ovals = ovals.reshape(m, 24)
    got = {int(okeys[i]): np.frombuffer(ovals[i][8:16].tobytes(),
                                        np.float64)[0]
           for i in range(m)}
    # fixed-point truncation error <= count * 2^-24 per group
    counts = np.bincount(keys, minlength=nkeys)
    sums = np.bincount(keys, weights=f, minlength=nkeys)
    for k in range(nkeys):
        if counts[k]:
            assert abs(got[int(k)] - sums[k]) <= counts[k] * 2**-24 + 1e-9
    assert len(got) == int((counts > 0).sum())


def test_distinct_as_zero_agg_reduce():
    """build_distinct (reduce.rs:286-360) == accumulable reduce with zero
    aggregates: emits one unit row per present key, retracts on absence."""
    ctx = OracleCtx()
    spec = abi.reduce_spec([], abi.schema(1, 0))
    op = ctx.reduce_create(spec)
    u = abi.make_updates(np.array([1, 1, 2], np.int64), None,
                         np.zeros(3, np.uint64),
                         np.array([1, 1, 1], np.int64), 0, 1)
    keys, vals, times, diffs = ctx.reduce_push(op, u)
    assert keys.tolist() == [1, 2] and diffs.tolist() == [1, 1]
    # retract one of key 1's two rows: still present, no correction
    u = abi.make_updates(np.array([1], np.int64), None,
                         np.ones(1, np.uint64), np.array([-1], np.int64),
                         1, 2)
    keys, vals, times, diffs = ctx.reduce_push(op, u)
    assert len(times) == 0
    # retract the last row: key disappears
    u = abi.make_updates(np.array([1], np.int64), None,
                         np.full(1, 2, np.uint64), np.array([-1], np.int64),
                         2, 3)
    keys, vals, times, diffs = ctx.reduce_push(op, u)
    assert keys.tolist() == [1] and diffs.tolist() == [-1]
