import os, sys
REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO); sys.path.insert(0, os.path.join(REPO, "oracle"))
import numpy as np
from materialize_amd import _abi as abi
from materialize_amd._ffi import GpuCtx
from pyoracle import OracleCtx

g, o = GpuCtx(), OracleCtx()
aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8, is_float=0, nullable=0),
        abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8, is_float=0, nullable=0)]
spec = abi.reduce_spec(aggs, abi.schema(1, 16))

def batch(rng, t, n=200):
    keys = rng.integers(0, 40, n).astype(np.int64)
    vals = np.zeros((n, 16), np.uint8)
    vals[:, :8] = rng.integers(0, 100, n).astype(np.int64).reshape(-1, 1).view(np.uint8).reshape(n, 8)
    diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
    return abi.make_updates(keys, vals, np.full(n, t, np.uint64), diffs, t, t + 1)

def run(case, times):
    rng = np.random.default_rng(5)
    gop, oop = g.reduce_create(spec), o.reduce_create(spec)
    for i, t in enumerate(times):
        u = batch(rng, t)
        a, b = g.reduce_push(gop, u), o.reduce_push(oop, u)
        same = all(np.array_equal(x.view(np.uint8), y.view(np.uint8)) for x, y in zip(a, b))
        print(case, "push", i, "t=", t, "rows g/o:", len(a[2]), len(b[2]), "MATCH" if same else "MISMATCH")

run("same-time", [0, 0, 0])
run("distinct-times", [0, 1, 2])
