import os, sys
REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))
import numpy as np
from materialize_amd import _abi as abi
from materialize_amd._ffi import GpuCtx
from pyoracle import OracleCtx

g, o = GpuCtx(), OracleCtx()
rng = np.random.default_rng(123)
sch = abi.schema(1, 8)
cl = abi.closure([], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
                 [abi.field(abi.MZ_SRC_VAL_STREAM, 0, 8),
                  abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 8)],
                 abi.schema(1, 16))

def updates(n, t):
    keys = rng.integers(0, 50, n).astype(np.int64)
    vals = rng.integers(0, 100, (n, 1)).astype(np.int64)
    diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
    return abi.make_updates(keys, vals.view(np.uint8),
                            np.full(n, t, np.uint64), diffs, t, t + 1)

def seal(ctx, u):
    k, v, t, d = ctx.consolidate(sch, u)
    return abi.make_updates(k, v, t, d, u.lower, u.upper)

ga1, ga2 = g.arr_create(sch), g.arr_create(sch)
oa1, oa2 = o.arr_create(sch), o.arr_create(sch)
gop, oop = g.join_create(ga1, ga2, cl), o.join_create(oa1, oa2, cl)
aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8, is_float=0,
                      nullable=0),
        abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                      is_float=0, nullable=0)]
spec = abi.reduce_spec(aggs, abi.schema(1, 16))
grd, ord_ = g.reduce_create(spec), o.reduce_create(spec)

for t in range(3):
    for side, (garr, oarr) in ((1, (ga1, oa1)), (2, (ga2, oa2))):
        u = updates(200, t)
        sg, so = seal(g, u), seal(o, u)
        g.arr_push(garr, sg)
        o.arr_push(oarr, so)
        rg = g.join_push(gop, side, sg)
        ro = o.join_push(oop, side, so)
        for a, b, what in zip(rg, ro, ("keys", "vals", "times", "diffs")):
            assert np.array_equal(a.view(np.uint8), b.view(np.uint8)), \
                f"join {what} t={t} side={side}"
        if len(rg[2]):
            ru = abi.make_updates(rg[0], rg[1], rg[2], rg[3], t, t + 1)
            og = g.reduce_push(grd, ru)
            ru2 = abi.make_updates(ro[0], ro[1], ro[2], ro[3], t, t + 1)
            oo = o.reduce_push(ord_, ru2)
            for ci, (a, b) in enumerate(zip(og, oo)):
                if not np.array_equal(a.view(np.uint8), b.view(np.uint8)):
                    print(f"MISMATCH t={t} side={side} col={ci}")
                    print("g rows:", len(og[2]), "o rows:", len(oo[2]))
                    ng, no_ = len(og[2]), len(oo[2])
                    gk, gv = og[0], og[1].reshape(ng, -1)
                    ok_, ov = oo[0], oo[1].reshape(no_, -1)
                    for i in range(max(ng, no_)):
                        grow = (int(gk[i]), gv[i].tobytes().hex(), int(og[3][i])) if i < ng else None
                        orow = (int(ok_[i]), ov[i].tobytes().hex(), int(oo[3][i])) if i < no_ else None
                        if grow != orow:
                            print(i, "G", grow)
                            print(i, "O", orow)
                    sys.exit(1)
print("all match")
