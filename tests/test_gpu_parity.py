"""GPU vs oracle bit-exact parity (SURVEY §8c gates).

Every scenario drives the HIP engine (materialize_amd._ffi.GpuCtx) and the
CPU oracle (oracle/pyoracle.OracleCtx) through identical descriptors and
compares raw output columns bit-for-bit. Integer keys/counts must be
bit-exact; SUM(float) is bit-exact too via the fixed-point restatement.
"""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctxs():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g = GpuCtx()
    o = OracleCtx()
    yield g, o
    g.close()
    o.close()


def assert_same(res_g, res_o, label=""):
    kg, vg, tg, dg = res_g
    ko, vo, to, do = res_o
    assert len(tg) == len(to), f"{label}: row count {len(tg)} vs {len(to)}"
    np.testing.assert_array_equal(kg.view(np.int64), ko.view(np.int64),
                                  err_msg=f"{label}: keys")
    np.testing.assert_array_equal(vg, vo, err_msg=f"{label}: vals")
    np.testing.assert_array_equal(tg, to, err_msg=f"{label}: times")
    np.testing.assert_array_equal(dg, do, err_msg=f"{label}: diffs")


def rand_updates(rng, n, kw=1, vb=8, nkeys=50, ntimes=3, lower=0, upper=None):
    keys = rng.integers(-nkeys // 2, nkeys, (n, kw)).astype(np.int64)
    vals = rng.integers(0, 5, (n, vb)).astype(np.uint8)
    times = rng.integers(lower, lower + ntimes, n).astype(np.uint64)
    diffs = rng.integers(-2, 3, n).astype(np.int64)
    return abi.make_updates(keys, vals, times, diffs, lower,
                            upper or (lower + ntimes))


def seal(ctx, sch, u):
    k, v, t, d = ctx.consolidate(sch, u)
    return abi.make_updates(k, v, t, d, u.lower, u.upper)


def concat_cl(vb1, vb2, okw=1):
    kf = [abi.field(abi.MZ_SRC_KEY, 0, 8)]
    vf = []
    if vb1:
        vf.append(abi.field(abi.MZ_SRC_VAL_STREAM, 0, vb1))
    if vb2:
        vf.append(abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, vb2))
    return abi.closure([], kf, vf, abi.schema(okw, vb1 + vb2))


class TestConsolidateParity:
    @pytest.mark.parametrize("kw,vb", [(1, 8), (2, 8), (1, 0), (2, 20),
                                       (1, 12)])
    def test_random(self, ctxs, kw, vb):
        g, o = ctxs
        rng = np.random.default_rng(11)
        sch = abi.schema(kw, vb)
        for n in (1, 7, 100, 5000):
            u = rand_updates(rng, n, kw, vb)
            assert_same(g.consolidate(sch, u), o.consolidate(sch, u),
                        f"consolidate kw={kw} vb={vb} n={n}")

    def test_empty(self, ctxs):
        g, o = ctxs
        sch = abi.schema(1, 8)
        u = abi.make_updates(np.empty(0, np.int64), np.empty(0, np.uint8),
                             np.empty(0, np.uint64), np.empty(0, np.int64),
                             0, 1)
        assert_same(g.consolidate(sch, u), o.consolidate(sch, u), "empty")

    def test_all_cancel(self, ctxs):
        g, o = ctxs
        sch = abi.schema(1, 8)
        keys = np.array([5, 5, 5, 5], np.int64)
        vals = np.tile(np.arange(8, dtype=np.uint8), (4, 1))
        u = abi.make_updates(keys, vals, np.zeros(4, np.uint64),
                             np.array([1, -1, 2, -2], np.int64), 0, 1)
        rg, ro = g.consolidate(sch, u), o.consolidate(sch, u)
        assert len(rg[2]) == 0
        assert_same(rg, ro, "cancel")


class TestSpineMergeParity:
    def test_pool_and_pair_merges(self, ctxs):
        """Many small arr_inserts force the lazy pool's k-way tournament
        merges and geometric pair merges; a halfjoin probe after every
        insert checks the merged spine (hash tables, offsets,
        consolidation) bit-exactly against the oracle."""
        g, o = ctxs
        rng = np.random.default_rng(83)
        sch = abi.schema(1, 8)
        ga, oa = g.arr_create(sch), o.arr_create(sch)
        cl = concat_cl(0, 8)
        for t in range(16):
            n = 1500
            keys = rng.integers(0, 400, (n, 1)).astype(np.int64)
            vals = rng.integers(0, 4, n).astype(np.int64)
            diffs = rng.integers(-2, 3, n).astype(np.int64)
            u = abi.make_updates(keys, vals.view(np.uint8),
                                 np.full(n, t, np.uint64), diffs, t, t + 1)
            g.arr_insert(ga, u)
            o.arr_insert(oa, u)
            pk = rng.integers(0, 400, (200, 1)).astype(np.int64)
            pu = abi.make_updates(pk, None, np.full(200, t, np.uint64),
                                  np.ones(200, np.int64), t, t + 1)
            assert_same(g.halfjoin(ga, pu, 0, True, cl),
                        o.halfjoin(oa, pu, 0, True, cl),
                        f"spine probe t={t}")


class TestJoinParity:
    def _stream(self, ctxs, steps, kw, vb1, vb2, seed, nkeys=30):
        """Run the same per-step batches through GPU and oracle join ops;
        compare every step's output."""
        g, o = ctxs
        rng = np.random.default_rng(seed)
        sch1, sch2 = abi.schema(kw, vb1), abi.schema(kw, vb2)
        cl = concat_cl(vb1, vb2, okw=kw)
        # NOTE: out key = 8B word; for kw=2 keep first word only? use kw out
        cl = abi.closure(
            [], [abi.field(abi.MZ_SRC_KEY, 0, 8 * kw)],
            ([abi.field(abi.MZ_SRC_VAL_STREAM, 0, vb1)] if vb1 else []) +
            ([abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, vb2)] if vb2 else []),
            abi.schema(kw, vb1 + vb2))
        ga1, ga2 = g.arr_create(sch1), g.arr_create(sch2)
        oa1, oa2 = o.arr_create(sch1), o.arr_create(sch2)
        gop = g.join_create(ga1, ga2, cl)
        oop = o.join_create(oa1, oa2, cl)
        for t in range(steps):
            for side, sch, (garr, oarr), vb in ((1, sch1, (ga1, oa1), vb1),
                                                (2, sch2, (ga2, oa2), vb2)):
                n = int(rng.integers(0, 200))
                if n == 0:
                    continue
                u = rand_updates(rng, n, kw, vb, nkeys=nkeys, ntimes=1,
                                 lower=t)
                su_g = seal(g, sch, u)
                su_o = seal(o, sch, u)
                # sanity: sealed forms agree
                g.arr_push(garr, su_g)
                o.arr_push(oarr, su_o)
                rg = g.join_push(gop, side, su_g)
                ro = o.join_push(oop, side, su_o)
                assert_same(rg, ro, f"join step {t} side {side}")
        # maintenance must not change results: merge GPU spine, then join a
        # probe batch on both and compare
        g.arr_maintain(ga2)
        o.arr_maintain(oa2)
        u = rand_updates(rng, 50, kw, vb1, nkeys=nkeys, ntimes=1,
                         lower=steps)
        su_g, su_o = seal(g, sch1, u), seal(o, sch1, u)
        g.arr_push(ga1, su_g)
        o.arr_push(oa1, su_o)
        assert_same(g.join_push(gop, 1, su_g), o.join_push(oop, 1, su_o),
                    "post-maintain join")

    def test_basic(self, ctxs):
        self._stream(ctxs, steps=4, kw=1, vb1=8, vb2=8, seed=1)

    def test_two_word_keys(self, ctxs):
        self._stream(ctxs, steps=3, kw=2, vb1=8, vb2=16, seed=2)

    def test_empty_vals(self, ctxs):
        self._stream(ctxs, steps=3, kw=1, vb1=12, vb2=0, seed=3)

    def test_hot_key(self, ctxs):
        # heavy skew: few keys -> long value chains, cross products
        self._stream(ctxs, steps=3, kw=1, vb1=8, vb2=8, seed=4, nkeys=3)

    def test_filter_closure(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(9)
        sch = abi.schema(1, 8)
        cl = abi.closure(
            [abi.filt(abi.MZ_SRC_VAL_LOOKUP, 0, 8, abi.MZ_CMP_GT, 2)],
            [abi.field(abi.MZ_SRC_KEY, 0, 8)],
            [abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8),
             abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
            abi.schema(1, 16))
        ga1, ga2 = g.arr_create(sch), g.arr_create(sch)
        oa1, oa2 = o.arr_create(sch), o.arr_create(sch)
        gop, oop = g.join_create(ga1, ga2, cl), o.join_create(oa1, oa2, cl)
        vals = rng.integers(0, 6, (100, 1)).astype(np.int64)
        u = abi.make_updates(rng.integers(0, 10, 100).astype(np.int64),
                             vals.view(np.uint8),
                             np.zeros(100, np.uint64),
                             np.ones(100, np.int64), 0, 1)
        su_g, su_o = seal(g, sch, u), seal(o, sch, u)
        g.arr_push(ga2, su_g)
        o.arr_push(oa2, su_o)
        u2 = rand_updates(rng, 40, 1, 8, nkeys=10, ntimes=1)
        s2g, s2o = seal(g, sch, u2), seal(o, sch, u2)
        g.arr_push(ga1, s2g)
        o.arr_push(oa1, s2o)
        assert_same(g.join_push(gop, 1, s2g), o.join_push(oop, 1, s2o),
                    "filtered join")


class TestHalfJoinParity:
    @pytest.mark.parametrize("le", [True, False])
    def test_random(self, ctxs, le):
        g, o = ctxs
        rng = np.random.default_rng(21)
        sch = abi.schema(1, 8)
        garr, oarr = g.arr_create(sch), o.arr_create(sch)
        u = rand_updates(rng, 300, 1, 8, nkeys=40, ntimes=5)
        sg, so = seal(g, sch, u), seal(o, sch, u)
        g.arr_push(garr, sg)
        o.arr_push(oarr, so)
        cl = concat_cl(8, 8)
        delta = rand_updates(rng, 100, 1, 8, nkeys=40, ntimes=5)
        dg, do = seal(g, sch, delta), seal(o, sch, delta)
        assert_same(g.halfjoin(garr, dg, 8, le, cl),
                    o.halfjoin(oarr, do, 8, le, cl), f"halfjoin le={le}")


class TestReduceParity:
    def _spec(self, aggs, vb=16):
        return abi.reduce_spec(aggs, abi.schema(1, vb))

    def test_count_sum(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(31)
        aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                              is_float=0, nullable=1),
                abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                              is_float=0, nullable=1)]
        spec = self._spec(aggs)
        gop, oop = g.reduce_create(spec), o.reduce_create(spec)
        for step in range(4):
            n = 500
            keys = rng.integers(0, 100, n).astype(np.int64)
            vals = np.zeros((n, 16), np.uint8)
            v = rng.integers(-50, 50, n).astype(np.int64)
            vals[:, :8] = v.reshape(-1, 1).view(np.uint8).reshape(n, 8)
            vals[:, 8] = rng.integers(0, 2, n)  # null flags
            diffs = rng.choice([-1, 1, 2], n).astype(np.int64)
            times = np.full(n, step, np.uint64)
            u = abi.make_updates(keys, vals, times, diffs, step, step + 1)
            assert_same(g.reduce_push(gop, u), o.reduce_push(oop, u),
                        f"reduce step {step}")

    def test_float_sum_bitexact(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(37)
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_F64, off=0, width=8,
                              is_float=1, nullable=0)]
        spec = self._spec(aggs, vb=8)
        gop, oop = g.reduce_create(spec), o.reduce_create(spec)
        for step in range(3):
            n = 400
            keys = rng.integers(0, 40, n).astype(np.int64)
            xs = rng.uniform(-1e6, 1e6, n)
            xs[rng.random(n) < 0.02] = np.inf
            xs[rng.random(n) < 0.02] = -np.inf
            diffs = rng.choice([-1, 1], n).astype(np.int64)
            u = abi.make_updates(keys, xs.view(np.uint8),
                                 np.full(n, step, np.uint64), diffs, step,
                                 step + 1)
            assert_same(g.reduce_push(gop, u), o.reduce_push(oop, u),
                        f"float reduce step {step}")

    def test_multi_timestamp(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(41)
        aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                              is_float=0, nullable=0)]
        spec = self._spec(aggs, vb=8)
        gop, oop = g.reduce_create(spec), o.reduce_create(spec)
        n = 300
        keys = rng.integers(0, 30, n).astype(np.int64)
        vals = rng.integers(-9, 9, (n, 1)).astype(np.int64)
        times = rng.integers(0, 4, n).astype(np.uint64)
        diffs = rng.choice([-1, 1], n).astype(np.int64)
        u = abi.make_updates(keys, vals.view(np.uint8), times, diffs, 0, 4)
        assert_same(g.reduce_push(gop, u), o.reduce_push(oop, u),
                    "multi-timestamp reduce")


class TestThresholdParity:
    """build_threshold_basic semantics (render/threshold.rs:34-51):
    corrections with diff = pos(new count) - pos(old count) per record."""

    def test_random_churn(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(53)
        sch = abi.schema(1, 8)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        for step in range(5):
            n = 600
            keys = rng.integers(-5, 40, n).astype(np.int64)
            vals = rng.integers(0, 3, n).astype(np.int64)
            diffs = rng.integers(-2, 3, n).astype(np.int64)
            u = abi.make_updates(keys, vals.view(np.uint8),
                                 np.full(n, step, np.uint64), diffs, step,
                                 step + 1)
            assert_same(g.threshold_push(gop, u), o.threshold_push(oop, u),
                        f"threshold step {step}")

    def test_two_word_keys_wide_vals(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(59)
        kw, vb = 2, 12
        sch = abi.schema(kw, vb)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        for step in range(3):
            n = 400
            keys = rng.integers(-4, 10, (n, kw)).astype(np.int64)
            vals = rng.integers(0, 4, (n, vb)).astype(np.uint8)
            diffs = rng.integers(-3, 4, n).astype(np.int64)
            u = abi.make_updates(keys, vals, np.full(n, step, np.uint64),
                                 diffs, step, step + 1)
            assert_same(g.threshold_push(gop, u), o.threshold_push(oop, u),
                        f"threshold wide step {step}")

    def test_multi_timestamp(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(61)
        sch = abi.schema(1, 8)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        n = 500
        keys = rng.integers(0, 25, n).astype(np.int64)
        vals = rng.integers(0, 2, n).astype(np.int64)
        times = rng.integers(0, 4, n).astype(np.uint64)
        diffs = rng.integers(-2, 3, n).astype(np.int64)
        u = abi.make_updates(keys, vals.view(np.uint8), times, diffs, 0, 4)
        assert_same(g.threshold_push(gop, u), o.threshold_push(oop, u),
                    "threshold multi-timestamp")

    def test_empty(self, ctxs):
        g, o = ctxs
        sch = abi.schema(1, 8)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        u = abi.make_updates(np.zeros((0, 1), np.int64),
                             np.zeros((0, 8), np.uint8),
                             np.zeros(0, np.uint64), np.zeros(0, np.int64),
                             0, 1)
        assert_same(g.threshold_push(gop, u), o.threshold_push(oop, u),
                    "threshold empty")

    def test_no_vals(self, ctxs):
        """vb=0 — the EXCEPT ALL lowering's whole-row-as-key shape."""
        g, o = ctxs
        rng = np.random.default_rng(97)
        sch = abi.schema(1, 0)
        gop, oop = g.threshold_create(sch), o.threshold_create(sch)
        for step in range(3):
            n = 400
            keys = rng.integers(-8, 20, n).astype(np.int64)
            diffs = rng.integers(-2, 3, n).astype(np.int64)
            u = abi.make_updates(keys, None,
                                 np.full(n, step, np.uint64), diffs, step,
                                 step + 1)
            assert_same(g.threshold_push(gop, u), o.threshold_push(oop, u),
                        f"threshold vb0 step {step}")


class TestTopKParity:
    """render_topk Basic plan (top_k.rs:322-418): corrections parity under
    churn across offset/limit/desc shapes."""

    def _churn(self, rng, net, n, nkeys=10, vb=8):
        ks, vs, ds = [], [], []
        for _ in range(n):
            k = int(rng.integers(0, nkeys))
            v = int(rng.integers(-15, 15))
            vbytes = v.to_bytes(8, "little", signed=True)[:vb]
            cur = net.get((k, vbytes), 0)
            if cur > 0 and rng.random() < 0.45:
                d = -int(rng.integers(1, cur + 1))
            else:
                d = int(rng.integers(1, 3))
            net[(k, vbytes)] = cur + d
            ks.append(k)
            vs.append(list(vbytes))
            ds.append(d)
        return (np.array(ks, np.int64).reshape(-1, 1),
                np.array(vs, np.uint8), np.array(ds, np.int64))

    @pytest.mark.parametrize("offset,limit,desc", [
        (0, 3, 0), (0, 1, 1), (2, 4, 0), (0, -1, 0), (3, 2, 1),
    ])
    def test_churn(self, ctxs, offset, limit, desc):
        g, o = ctxs
        rng = np.random.default_rng(67 + offset + limit + desc)
        spec = abi.topk_spec(abi.schema(1, 8), [(0, 8, desc)],
                             offset=offset, limit=limit)
        gop, oop = g.topk_create(spec), o.topk_create(spec)
        net = {}
        for step in range(4):
            ks, vs, ds = self._churn(rng, net, 300)
            u = abi.make_updates(ks, vs, np.full(300, step, np.uint64), ds,
                                 step, step + 1)
            assert_same(g.topk_push(gop, u), o.topk_push(oop, u),
                        f"topk o{offset} l{limit} d{desc} step {step}")

    def test_two_order_cols(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(71)
        spec = abi.topk_spec(abi.schema(1, 8),
                             [(0, 4, 1), (4, 4, 0)], offset=0, limit=3)
        gop, oop = g.topk_create(spec), o.topk_create(spec)
        for step in range(3):
            n = 250
            ks = rng.integers(0, 8, n).astype(np.int64).reshape(-1, 1)
            a = rng.integers(0, 3, n).astype(np.int32)
            b = rng.integers(-6, 6, n).astype(np.int32)
            vals = np.zeros((n, 8), np.uint8)
            vals[:, :4] = a.view(np.uint8).reshape(-1, 4)
            vals[:, 4:] = b.view(np.uint8).reshape(-1, 4)
            u = abi.make_updates(ks, vals, np.full(n, step, np.uint64),
                                 np.ones(n, np.int64), step, step + 1)
            assert_same(g.topk_push(gop, u), o.topk_push(oop, u),
                        f"topk 2-col step {step}")

    def test_multi_timestamp_push(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(73)
        spec = abi.topk_spec(abi.schema(1, 8), [(0, 8, 0)], offset=1,
                             limit=2)
        gop, oop = g.topk_create(spec), o.topk_create(spec)
        net = {}
        cols = [self._churn(rng, net, 200) for _ in range(3)]
        ks = np.concatenate([c[0] for c in cols])
        vs = np.concatenate([c[1] for c in cols])
        ds = np.concatenate([c[2] for c in cols])
        times = np.repeat(np.arange(3, dtype=np.uint64), 200)
        u = abi.make_updates(ks, vs, times, ds, 0, 3)
        assert_same(g.topk_push(gop, u), o.topk_push(oop, u),
                    "topk multi-timestamp")

    def test_two_word_group_keys(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(79)
        spec = abi.topk_spec(abi.schema(2, 8), [(0, 8, 1)], offset=0,
                             limit=2)
        gop, oop = g.topk_create(spec), o.topk_create(spec)
        for step in range(3):
            n = 300
            ks = rng.integers(-3, 5, (n, 2)).astype(np.int64)
            vs = rng.integers(-10, 10, n).astype(np.int64)
            u = abi.make_updates(ks, vs.view(np.uint8),
                                 np.full(n, step, np.uint64),
                                 np.ones(n, np.int64), step, step + 1)
            assert_same(g.topk_push(gop, u), o.topk_push(oop, u),
                        f"topk kw2 step {step}")

    def test_negative_multiplicity_errors(self, ctxs):
        g, o = ctxs
        spec = abi.topk_spec(abi.schema(1, 8), [(0, 8, 0)], offset=0,
                             limit=2)
        gop = g.topk_create(spec)
        u = abi.make_updates(np.array([[1]], np.int64),
                             np.zeros((1, 8), np.uint8),
                             np.zeros(1, np.uint64),
                             np.array([-1], np.int64), 0, 1)
        with pytest.raises(Exception, match="[Nn]egative multiplicities"):
            g.topk_push(gop, u)


class TestRouteHashParity:
    def test_hash_agrees(self, ctxs):
        g, o = ctxs
        rng = np.random.default_rng(51)
        for _ in range(100):
            w = [int(x) for x in rng.integers(0, 2**63, 2)]
            assert g.route_hash(w[:1]) == o.route_hash(w[:1])
            assert g.route_hash(w) == o.route_hash(w)
