"""Hierarchical MIN/MAX reduce (SURVEY §8f.1 — build_bucketed +
ReductionMonoid restatement): oracle vs naive dict model under churn,
including retraction of the current extremum (the case plain accumulation
cannot handle)."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def run_scenario(ctx_factory, is_max, buckets, seed=5, steps=6, n=800,
                 nkeys=40):
    ctx = ctx_factory()
    op = ctx.minmax_create(abi.schema(1, 8), is_max, buckets)
    rng = np.random.default_rng(seed)
    model = {}  # key -> {val: count}
    maintained = {}  # key -> extremum, from corrections
    outs = []
    present = []  # list of (key, val) currently insertable for retraction
    for t in range(steps):
        ks, vs, ds = [], [], []
        for _ in range(n):
            if present and rng.random() < 0.4:
                i = int(rng.integers(0, len(present)))
                k, v = present.pop(i)
                d = -1
            else:
                k = int(rng.integers(0, nkeys))
                v = int(rng.integers(-50, 50))
                d = 1
                present.append((k, v))
            ks.append(k)
            vs.append(v)
            ds.append(d)
            cnts = model.setdefault(k, {})
            cnts[v] = cnts.get(v, 0) + d
            if cnts[v] == 0:
                del cnts[v]
            if not cnts:
                del model[k]
        u = abi.make_updates(np.array(ks, np.int64),
                             np.array(vs, np.int64).view(np.uint8),
                             np.full(n, t, np.uint64),
                             np.array(ds, np.int64), t, t + 1)
        keys, vals, times, diffs = ctx.minmax_push(op, u)
        outs.append((keys.copy(), vals.copy(), times.copy(), diffs.copy()))
        m = len(times)
        vals = vals.view(np.int64)
        order = sorted(range(m), key=lambda i: (int(diffs[i])))
        for i in order:
            k, v, d = int(keys[i]), int(vals[i]), int(diffs[i])
            if d == 1:
                maintained[k] = v
            else:
                assert maintained.pop(k) == v, f"retract mismatch key {k}"
        want = {k: (max(c) if is_max else min(c)) for k, c in model.items()}
        assert maintained == want, f"step {t} ({'max' if is_max else 'min'})"
    return outs


@pytest.mark.parametrize("is_max", [False, True])
@pytest.mark.parametrize("buckets", [[1], [16, 1], [256, 16, 1]])
def test_oracle_minmax(is_max, buckets):
    run_scenario(OracleCtx, is_max, buckets)


@pytest.mark.gpu
@pytest.mark.parametrize("is_max", [False, True])
def test_gpu_minmax_parity(is_max):
    from materialize_amd._ffi import GpuCtx
    a = run_scenario(GpuCtx, is_max, [256, 16, 1], n=2000, nkeys=100)
    b = run_scenario(OracleCtx, is_max, [256, 16, 1], n=2000, nkeys=100)
    for t, (ra, rb) in enumerate(zip(a, b)):
        for x, y, what in zip(ra, rb, ("keys", "vals", "times", "diffs")):
            np.testing.assert_array_equal(x.view(np.uint8), y.view(np.uint8),
                                          err_msg=f"step {t}: {what}")
