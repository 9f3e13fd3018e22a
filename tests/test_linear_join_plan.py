"""LinearJoinPlan rendering surface (compute-types/src/plan/join/
linear_join.rs:27-76; executor = render.LinearJoinOp): a 2-stage chain
join A ⋈ B ⋈ C driven by one plan, with interior JoinStage
arrangements, checked against a naive recompute on the oracle and
bit-exactly GPU-vs-oracle."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from materialize_amd.render import (LinearJoinPlan, LinearStagePlan,
                                    render_join)

F = abi.field


def _plan():
    # A arranged by k2, val=[a i64] (the source_key arrangement matches
    # stage 1's stream key, as the reference requires); B(k2) val=[k3];
    # C(k3) val=[c]
    # stage1: probe B on k2; out key := B.val (k3), val := [a]
    cl1 = abi.closure(
        [], [F(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
        [F(abi.MZ_SRC_VAL_STREAM, 0, 8)],
        abi.schema(1, 8))
    # stage2: probe C on k3; out key := k3, val := [a, c]
    cl2 = abi.closure(
        [], [F(abi.MZ_SRC_KEY, 0, 8)],
        [F(abi.MZ_SRC_VAL_STREAM, 0, 8), F(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
        abi.schema(1, 16))
    return LinearJoinPlan(
        source_relation="A",
        stage_plans=[
            LinearStagePlan("B", cl1, stream_key_words=1,
                            stream_val_bytes=8),
            LinearStagePlan("C", cl2, stream_key_words=1,
                            stream_val_bytes=8),
        ])


def _mk(rng, n, lo, hi, extra):
    keys = rng.integers(lo, hi, n).astype(np.int64)
    v0 = rng.integers(0, extra, n).astype(np.int64)
    return keys, v0


def _naive(A, B, C):
    """dict-based recompute of the chain join result multiset."""
    from collections import Counter
    bmap = {}
    for k2, k3 in zip(*B):
        bmap.setdefault(int(k2), []).append(int(k3))
    cmap = {}
    for k3, c in zip(*C):
        cmap.setdefault(int(k3), []).append(int(c))
    out = Counter()
    ak2, aa = A
    for k2, a in zip(ak2, aa):
        for k3 in bmap.get(int(k2), []):
            for c in cmap.get(int(k3), []):
                out[(int(k3), int(a), int(c))] += 1
    return out


def _run(ctx, seed=5):
    rng = np.random.default_rng(seed)
    arrs = {"A": ctx.arr_create(abi.schema(1, 8)),
            "B": ctx.arr_create(abi.schema(1, 8)),
            "C": ctx.arr_create(abi.schema(1, 8))}
    op = render_join(ctx, arrs, _plan())
    bk, bv = _mk(rng, 300, 0, 50, 60)
    ck, cv = _mk(rng, 200, 0, 60, 100)
    for name, (k, v) in (("B", (bk, bv)), ("C", (ck, cv))):
        u = abi.make_updates(k, v.reshape(-1, 1).view(np.uint8),
                             np.zeros(len(k), np.uint64),
                             np.ones(len(k), np.int64), 0, 1)
        ctx.arr_insert(arrs[name], u)
    # source delta at t=1 (A keyed by k2)
    n = 400
    ak2 = rng.integers(0, 50, n).astype(np.int64)
    aa = rng.integers(0, 1000, n).astype(np.int64)
    av = aa.reshape(-1, 1).view(np.uint8).reshape(n, 8)
    times = np.full(n, 1, np.uint64)
    diffs = np.ones(n, np.int64)
    au = abi.make_updates(ak2, av, times, diffs, 1, 2)
    ctx.arr_insert(arrs["A"], au)
    cols = op.step(1, (ak2, av.reshape(-1), times, diffs))
    return cols, (ak2, bk, bv, ck, cv, aa)


def test_two_stage_plan_oracle_vs_naive():
    from collections import Counter

    from pyoracle import OracleCtx
    ctx = OracleCtx()
    cols, (ak2, bk, bv, ck, cv, aa) = _run(ctx)
    k, v, t, d = cols
    v = np.asarray(v).reshape(-1, 16)
    got = Counter()
    for i in range(len(t)):
        a = int(v[i][:8].view(np.int64)[0])
        c = int(v[i][8:].view(np.int64)[0])
        got[(int(k[i]), a, c)] += int(d[i])
    want = _naive((ak2, aa), (bk, bv), (ck, cv))
    ctx.close()
    assert +got == +want


@pytest.mark.gpu
def test_two_stage_plan_gpu_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    cg, _ = _run(g)
    co, _ = _run(o)
    for x, y, what in zip(cg, co, ("keys", "vals", "times", "diffs")):
        np.testing.assert_array_equal(np.asarray(x).view(np.uint8),
                                      np.asarray(y).view(np.uint8),
                                      err_msg=what)
    g.close()
    o.close()
