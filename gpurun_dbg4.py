import os, sys
REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO); sys.path.insert(0, os.path.join(REPO, "oracle"))
import numpy as np
from materialize_amd import _abi as abi
from materialize_amd._ffi import GpuCtx
from pyoracle import OracleCtx

g, o = GpuCtx(), OracleCtx()
print("ctx ok", flush=True)
aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8, is_float=0, nullable=0),
        abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8, is_float=0, nullable=0)]
spec = abi.reduce_spec(aggs, abi.schema(1, 16))

rng = np.random.default_rng(5)
n = 200
keys = rng.integers(0, 40, n).astype(np.int64)
v = rng.integers(0, 100, n).astype(np.int64)
vals = np.zeros((n, 16), np.uint8)
vals[:, :8] = v.reshape(-1, 1).view(np.uint8).reshape(n, 8)
diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
def mku():
    return abi.make_updates(keys, vals, np.zeros(n, np.uint64), diffs, 0, 1)

import collections
cnt = collections.Counter(); sm = collections.Counter()
for i in range(n):
    cnt[int(keys[i])] += int(diffs[i]); sm[int(keys[i])] += int(diffs[i]) * int(v[i])

def show(tag, res):
    k, vv, t, d = res
    m = len(t)
    vv = vv.reshape(m, 48)
    rows = {}
    for i in range(m):
        c = int(vv[i][8:16].view(np.int64)[0])
        s = int(vv[i][32:40].view(np.uint64)[0])
        rows[int(k[i])] = (c, s, int(d[i]))
    bad = 0
    for kk in sorted(rows):
        exp = (cnt[kk], sm[kk])
        got = rows[kk][:2]
        if got != exp and bad < 4:
            print(f"{tag} key={kk} got count={got[0]} sum={got[1]} want {exp}", flush=True)
            bad += 1
    if bad == 0:
        print(tag, "all rows correct", flush=True)

for trial in range(3):
    gop = g.reduce_create(spec)
    res = g.reduce_push(gop, mku())
    show(f"gpu-op{trial}", res)
oop = o.reduce_create(spec)
show("oracle", o.reduce_push(oop, mku()))
