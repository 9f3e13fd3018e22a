"""Oracle sanity + property tests (CPU, no GPU).

Checks the C++ oracle (oracle/oracle.cpp — restatement of
mz_join_core/reduce/half_join) against naive Python references on small
randomized inputs, plus incremental-equals-recomputed properties.
"""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx

import naive


def upd(keys, vals_i64, times, diffs, lower=0, upper=1, vb=8):
    """Updates with 1-word keys and a single i64 val field (vb=8), or vb=0."""
    keys = np.asarray(keys, dtype=np.int64)
    if vb:
        vals = np.asarray(vals_i64, dtype=np.int64).view(np.uint8)
    else:
        vals = None
    return abi.make_updates(keys, vals, np.asarray(times, np.uint64),
                            np.asarray(diffs, np.int64), lower, upper)


def concat_closure(out_vb):
    """key -> key; out val = val1 ‖ val2 (8 bytes each)."""
    kf = [abi.field(abi.MZ_SRC_KEY, 0, 8)]
    vf = []
    if out_vb >= 8:
        vf.append(abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8))
    if out_vb >= 16:
        vf.append(abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8))
    return abi.closure([], kf, vf, abi.schema(1, out_vb))


def to_tuples(res, kw=1, vb=16):
    keys, vals, times, diffs = res
    n = len(times)
    out = []
    vals = vals.reshape(n, vb) if vb else None
    for i in range(n):
        k = tuple(keys[i * kw:(i + 1) * kw]) if kw > 1 else int(keys[i])
        v = tuple(vals[i].view(np.int64)) if vb else ()
        out.append((k, v, int(times[i]), int(diffs[i])))
    return out


def seal(ctx, sch, u):
    """Consolidate raw updates into sealed (sorted) form."""
    keys, vals, times, diffs = ctx.consolidate(sch, u)
    return abi.make_updates(keys, vals, times, diffs, u.lower, u.upper)


class TestConsolidate:
    def test_basic(self):
        ctx = OracleCtx()
        sch = abi.schema(1, 8)
        u = upd([3, 1, 3, 2, 3], [30, 10, 30, 20, 31], [5, 1, 5, 2, 5],
                [1, 1, -1, 2, 4])
        keys, vals, times, diffs = ctx.consolidate(sch, u)
        got = to_tuples((keys, vals, times, diffs), vb=8)
        assert got == [(1, (10,), 1, 1), (2, (20,), 2, 2), (3, (31,), 5, 4)]

    def test_random_vs_naive(self):
        rng = np.random.default_rng(7)
        ctx = OracleCtx()
        sch = abi.schema(1, 8)
        for trial in range(20):
            n = int(rng.integers(1, 200))
            keys = rng.integers(-5, 5, n)
            vals = rng.integers(0, 3, n)
            times = rng.integers(0, 4, n).astype(np.uint64)
            diffs = rng.integers(-2, 3, n)
            got = to_tuples(ctx.consolidate(sch, upd(keys, vals, times, diffs)),
                            vb=8)
            want = [(k, (v,), t, d) for (k, v, t, d) in naive.consolidate(
                list(zip(keys.tolist(), vals.tolist(), times.tolist(),
                         diffs.tolist())))]
            assert got == want, f"trial {trial}"


class TestLinearJoin:
    def _run_oracle_join(self, ctx, in1, in2, steps):
        """Feed per-step batches to arrangements + join op, side 1 first
        (the reference drain order), collect consolidated output."""
        sch = abi.schema(1, 8)
        a1 = ctx.arr_create(sch)
        a2 = ctx.arr_create(sch)
        cl = concat_closure(16)
        op = ctx.join_create(a1, a2, cl)
        out_all = []
        for t in range(steps):
            b1 = [(k, v, tt, d) for (k, v, tt, d) in in1 if tt == t]
            b2 = [(k, v, tt, d) for (k, v, tt, d) in in2 if tt == t]
            for side, b, arr in ((1, b1, a1), (2, b2, a2)):
                if not b:
                    continue
                ks, vs, ts, ds = zip(*b)
                u = seal(ctx, sch, upd(ks, vs, ts, ds, lower=t, upper=t + 1))
                ctx.arr_push(arr, u)
                out_all.extend(to_tuples(ctx.join_push(op, side, u)))
        return naive.consolidate([(k, v, t, d) for (k, v, t, d) in out_all])

    def test_static_equijoin(self):
        ctx = OracleCtx()
        in1 = [(1, 10, 0, 1), (1, 11, 0, 1), (2, 20, 0, 1), (3, 30, 0, 1)]
        in2 = [(1, 100, 0, 1), (2, 200, 0, 2), (4, 400, 0, 1)]
        got = self._run_oracle_join(ctx, in1, in2, 1)
        want = naive.join_full(in1, in2)
        assert got == want
        # cross product within key 1: 2 results; key 2 diff 2
        assert (1, (10, 100), 0, 1) in got
        assert (2, (20, 200), 0, 2) in got

    def test_incremental_matches_full(self):
        """Incremental (per-timestamp batches, both sides changing) equals
        the naive full join of the complete histories."""
        rng = np.random.default_rng(42)
        ctx = OracleCtx()
        for trial in range(10):
            steps = 4
            n1, n2 = int(rng.integers(1, 60)), int(rng.integers(1, 60))
            mk = lambda n: [(int(rng.integers(0, 8)), int(rng.integers(0, 4)),
                             int(rng.integers(0, steps)),
                             int(rng.integers(-2, 3)) or 1) for _ in range(n)]
            in1, in2 = mk(n1), mk(n2)
            got = self._run_oracle_join(ctx, in1, in2, steps)
            want = naive.join_full(in1, in2)
            assert got == want, f"trial {trial}"

    def test_linear_scan_strategy(self):
        """>=10 edits on both sides at many distinct times forces the
        linear-time-scan strategy (mz_join_core.rs:743); result must equal
        the naive join."""
        ctx = OracleCtx()
        # one key, 12 vals each side, distinct times, single batch at the end
        in1 = [(5, v, t, 1) for t, v in enumerate(range(12))]
        in2 = [(5, 100 + v, t, 1) for t, v in enumerate(range(12))]
        sch = abi.schema(1, 8)
        a1 = ctx.arr_create(sch)
        a2 = ctx.arr_create(sch)
        op = ctx.join_create(a1, a2, concat_closure(16))
        ks, vs, ts, ds = zip(*in1)
        u1 = seal(ctx, sch, upd(ks, vs, ts, ds, lower=0, upper=12))
        ks, vs, ts, ds = zip(*in2)
        u2 = seal(ctx, sch, upd(ks, vs, ts, ds, lower=0, upper=12))
        ctx.arr_push(a1, u1)
        out1 = to_tuples(ctx.join_push(op, 1, u1))
        ctx.arr_push(a2, u2)
        out2 = to_tuples(ctx.join_push(op, 2, u2))
        got = naive.consolidate(out1 + out2)
        want = naive.join_full(in1, in2)
        assert got == want


class TestHalfJoin:
    def test_le_lt(self):
        ctx = OracleCtx()
        sch = abi.schema(1, 8)
        arr = ctx.arr_create(sch)
        u = seal(ctx, sch, upd([1, 1, 1], [100, 101, 102], [0, 1, 2],
                               [1, 1, 1], 0, 3))
        ctx.arr_push(arr, u)
        delta = seal(ctx, sch, upd([1], [7], [1], [1], 1, 2))
        cl = concat_closure(16)
        # le: trace times <= 1 -> vals 100,101 match
        got = to_tuples(ctx.halfjoin(arr, delta, 8, True, cl))
        assert got == [(1, (7, 100), 1, 1), (1, (7, 101), 1, 1)]
        # lt: trace times < 1 -> only val 100
        got = to_tuples(ctx.halfjoin(arr, delta, 8, False, cl))
        assert got == [(1, (7, 100), 1, 1)]


class TestReduce:
    def _spec(self):
        aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                              is_float=0, nullable=0),
                abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                              is_float=0, nullable=0)]
        return abi.reduce_spec(aggs, abi.schema(1, 8))

    def _finalized(self, res, n_aggs=2):
        """Parse reduce corrections into tuples of (key, (aggs...), t, d)."""
        keys, vals, times, diffs = res
        n = len(times)
        out = []
        vals = vals.reshape(n, abi.AGG_SLOT_BYTES * n_aggs)
        for i in range(n):
            row = []
            for a in range(n_aggs):
                slot = vals[i, a * 24:(a + 1) * 24]
                if slot[0]:
                    row.append(None)
                else:
                    lo = int(slot[8:16].view(np.uint64)[0])
                    hi = int(slot[16:24].view(np.int64)[0])
                    row.append(hi * 2**64 + lo)
            out.append((int(keys[i]), tuple(row), int(times[i]),
                        int(diffs[i])))
        return out

    def test_count_sum_and_retraction(self):
        ctx = OracleCtx()
        op = ctx.reduce_create(self._spec())
        # t=0: key 1 -> vals 10, 20; key 2 -> val 5
        out = self._finalized(ctx.reduce_push(
            op, upd([1, 1, 2], [10, 20, 5], [0, 0, 0], [1, 1, 1], 0, 1)))
        assert out == [(1, (2, 30), 0, 1), (2, (1, 5), 0, 1)]
        # t=1: retract (1,10) -> correction: -old +new
        out = self._finalized(ctx.reduce_push(
            op, upd([1], [10], [1], [-1], 1, 2)))
        # consolidated output order = (key, val-bytes, time)
        assert out == [(1, (1, 20), 1, 1), (1, (2, 30), 1, -1)]
        # t=2: remove everything for key 2 -> retraction only
        out = self._finalized(ctx.reduce_push(
            op, upd([2], [5], [2], [-1], 2, 3)))
        assert out == [(2, (1, 5), 2, -1)]

    def test_wrapping_i128_sum(self):
        """aggregates.slt:188-198 golden: sum(bigint) past i64::MAX must be
        the exact wide value (2*i64::MAX = 18446744073709551614)."""
        ctx = OracleCtx()
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                              is_float=0, nullable=0)]
        op = ctx.reduce_create(abi.reduce_spec(aggs, abi.schema(1, 8)))
        m = 2**63 - 1
        out = self._finalized(ctx.reduce_push(
            op, upd([2, 2, 3, 3], [m, m, -m - 1, -m - 1], [0] * 4, [1] * 4,
                    0, 1)), n_aggs=1)
        assert out == [(2, (2 * m,), 0, 1), (3, (-2 * (m + 1),), 0, 1)]

    def test_multi_timestamp_batch(self):
        ctx = OracleCtx()
        op = ctx.reduce_create(self._spec())
        # one batch containing t=0 and t=1 for the same key: corrections
        # must be emitted per timestamp in order.
        out = self._finalized(ctx.reduce_push(
            op, upd([7, 7], [1, 2], [0, 1], [1, 1], 0, 2)))
        assert out == [(7, (1, 1), 0, 1), (7, (1, 1), 1, -1),
                       (7, (2, 3), 1, 1)]


class TestFloatSum:
    def test_fixed_point_exact(self):
        """SUM(float) via the 24-frac-bit fixed-point i128 restatement
        (reduce.rs:1641-1697): deterministic and order-independent."""
        ctx = OracleCtx()
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_F64, off=0, width=8,
                              is_float=1, nullable=0)]
        spec = abi.reduce_spec(aggs, abi.schema(1, 8))
        rng = np.random.default_rng(3)
        xs = rng.uniform(0, 1000, 100)

        def run(order):
            op = ctx.reduce_create(spec)
            vals = xs[order].view(np.int64)
            res = ctx.reduce_push(op, abi.make_updates(
                np.zeros(100, np.int64), vals.view(np.uint8),
                np.zeros(100, np.uint64), np.ones(100, np.int64), 0, 1))
            keys, v, t, d = res
            return v[8:16].tobytes()

        b1 = run(np.arange(100))
        b2 = run(np.arange(100)[::-1])
        assert b1 == b2  # bit-identical regardless of order
        got = np.frombuffer(b1, np.float64)[0]
        # fixed point truncates each addend at 2^-24
        assert abs(got - xs.sum()) < 100 * 2**-24 + 1e-9

    def test_float_specials(self):
        ctx = OracleCtx()
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_F64, off=0, width=8,
                              is_float=1, nullable=0)]
        spec = abi.reduce_spec(aggs, abi.schema(1, 8))
        op = ctx.reduce_create(spec)
        xs = np.array([np.inf, 1.5], np.float64)
        res = ctx.reduce_push(op, abi.make_updates(
            np.zeros(2, np.int64), xs.view(np.uint8),
            np.zeros(2, np.uint64), np.ones(2, np.int64), 0, 1))
        v = np.frombuffer(res[1][8:16].tobytes(), np.float64)[0]
        assert v == np.inf
        # retract the inf -> back to 1.5
        res = ctx.reduce_push(op, abi.make_updates(
            np.zeros(1, np.int64), xs[:1].view(np.uint8),
            np.zeros(1, np.uint64), -np.ones(1, np.int64), 1, 2))
        # find the insertion (+1) correction row
        keys, vals, times, diffs = res
        vals = vals.reshape(len(times), 24)
        ins = [i for i in range(len(times)) if diffs[i] == 1]
        assert len(ins) == 1
        v = np.frombuffer(vals[ins[0]][8:16].tobytes(), np.float64)[0]
        assert v == 1.5


class TestRouteHash:
    def test_shard_determinism(self):
        ctx = OracleCtx()
        h1 = ctx.route_hash([42])
        h2 = ctx.route_hash([42])
        assert h1 == h2
        assert ctx.route_hash([42]) != ctx.route_hash([43])
