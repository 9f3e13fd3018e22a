import os, sys
REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO); sys.path.insert(0, os.path.join(REPO, "oracle"))
import numpy as np
from materialize_amd import _abi as abi
from materialize_amd._ffi import GpuCtx
from pyoracle import OracleCtx

def batch(rng, t, n=200, nullable=False):
    vb = 16
    keys = rng.integers(0, 40, n).astype(np.int64)
    vals = np.zeros((n, vb), np.uint8)
    vals[:, :8] = rng.integers(0, 100, n).astype(np.int64).reshape(-1, 1).view(np.uint8).reshape(n, 8)
    diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
    return abi.make_updates(keys, vals, np.full(n, t, np.uint64), diffs, t, t + 1)

def spec(nullable):
    aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8, is_float=0, nullable=nullable),
            abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8, is_float=0, nullable=nullable)]
    return abi.reduce_spec(aggs, abi.schema(1, 16))

def check(name, g, o, gop, oop, rng):
    u = batch(rng, 0)
    a, b = g.reduce_push(gop, u), o.reduce_push(oop, u)
    same = all(np.array_equal(x.view(np.uint8), y.view(np.uint8)) for x, y in zip(a, b))
    print(name, "MATCH" if same else "MISMATCH")

# (a) fresh ctx, reduce first op, nullable=0
g, o = GpuCtx(), OracleCtx()
check("fresh-reduce-first-nn0", g, o, g.reduce_create(spec(0)), o.reduce_create(spec(0)), np.random.default_rng(5))
g.close(); o.close()
# (b) fresh ctx, reduce first, nullable=1
g, o = GpuCtx(), OracleCtx()
check("fresh-reduce-first-nn1", g, o, g.reduce_create(spec(1)), o.reduce_create(spec(1)), np.random.default_rng(5))
g.close(); o.close()
# (c) fresh ctx, one consolidate first, then reduce
g, o = GpuCtx(), OracleCtx()
rng = np.random.default_rng(5)
u0 = batch(rng, 0, 50)
g.consolidate(abi.schema(1, 16), u0); o.consolidate(abi.schema(1, 16), u0)
check("consolidate-then-reduce", g, o, g.reduce_create(spec(0)), o.reduce_create(spec(0)), np.random.default_rng(5))
g.close(); o.close()
# (d) fresh ctx, reduce first, SECOND push compare too
g, o = GpuCtx(), OracleCtx()
gop, oop = g.reduce_create(spec(0)), o.reduce_create(spec(0))
rng = np.random.default_rng(5)
check("fresh-push1", g, o, gop, oop, rng)
check("fresh-push2-same-t", g, o, gop, oop, rng)
