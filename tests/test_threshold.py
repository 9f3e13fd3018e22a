"""Threshold operator (SURVEY §8f — build_threshold_basic /
threshold_local, src/compute/src/render/threshold.rs:34-51,75-97):
oracle vs a naive dict model under churn, including records whose net
count crosses zero in both directions and multiplicities > 1 (the
EXCEPT ALL shape, plan/threshold.rs usage)."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def wrap_i64(x):
    return ((x + 2**63) % 2**64) - 2**63


def run_oracle(batches, kw=1, vb=8):
    """Push each batch; return accumulated corrections as a dict
    (key, valbytes) -> net diff, plus the raw per-push outputs."""
    ctx = OracleCtx()
    op = ctx.threshold_create(abi.schema(kw, vb))
    acc = {}
    outs = []
    for (keys, vals, times, diffs, lower, upper) in batches:
        u = abi.make_updates(keys, vals, times, diffs, lower, upper)
        k, v, t, d = ctx.threshold_push(op, u)
        outs.append((k, v, t, d))
        for i in range(len(t)):
            kk = tuple(int(x) for x in k[i * kw:(i + 1) * kw])
            vv = bytes(v[i * vb:(i + 1) * vb])
            acc[(kk, vv)] = acc.get((kk, vv), 0) + int(d[i])
    ctx.close()
    return {r: c for r, c in acc.items() if c != 0}, outs


def naive_threshold(batches, kw=1, vb=8):
    """Net count per (key, val) over ALL updates; keep positive counts."""
    cnt = {}
    for (keys, vals, times, diffs, lower, upper) in batches:
        n = len(diffs)
        for i in range(n):
            kk = tuple(int(x) for x in keys[i])
            vv = bytes(vals[i].tobytes())
            cnt[kk, vv] = wrap_i64(cnt.get((kk, vv), 0) + int(diffs[i]))
    return {r: c for r, c in cnt.items() if c > 0}


def make_batches(seed, steps, n, nkeys=30, nvals=4, vb=8):
    rng = np.random.default_rng(seed)
    batches = []
    for t in range(steps):
        keys = rng.integers(-3, nkeys, (n, 1)).astype(np.int64)
        raw = rng.integers(0, nvals, n).astype(np.int64)
        vals = raw.view(np.uint8).reshape(n, 8)[:, :vb].copy()
        times = np.full(n, t, np.uint64)
        diffs = rng.integers(-2, 3, n).astype(np.int64)
        batches.append((keys, vals, times, diffs, t, t + 1))
    return batches


def test_threshold_kat():
    """Hand-computed known-answer: EXCEPT ALL-style counts.

    Row a: +3 then -1  -> kept with count 2 (delta stream +3, -1)
    Row b: +1 then -2  -> count -1, dropped (delta +1 then -1)
    Row c: -2 then +1  -> count -1, never emitted
    """
    kw, vb = 1, 8
    val = np.zeros((3, vb), np.uint8)
    b1 = (np.array([[1], [2], [3]], np.int64), val,
          np.zeros(3, np.uint64), np.array([3, 1, -2], np.int64), 0, 1)
    b2 = (np.array([[1], [2], [3]], np.int64), val,
          np.ones(3, np.uint64), np.array([-1, -2, 1], np.int64), 1, 2)
    acc, outs = run_oracle([b1, b2], kw, vb)
    assert acc == {((1,), bytes(8)): 2}
    # first push emitted a:+3, b:+1; second a:-1, b:-1, nothing for c
    k1, v1, t1, d1 = outs[0]
    assert sorted(zip(k1.tolist(), d1.tolist())) == [(1, 3), (2, 1)]
    k2, v2, t2, d2 = outs[1]
    assert sorted(zip(k2.tolist(), d2.tolist())) == [(1, -1), (2, -1)]


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_threshold_matches_naive(seed):
    batches = make_batches(seed, steps=5, n=400)
    acc, _ = run_oracle(batches)
    assert acc == naive_threshold(batches)


def test_threshold_multitime_batch():
    """One push carrying several timestamps is processed in time order."""
    b = make_batches(7, steps=3, n=300)
    keys = np.concatenate([x[0] for x in b])
    vals = np.concatenate([x[1] for x in b])
    times = np.concatenate([x[2] for x in b])
    diffs = np.concatenate([x[3] for x in b])
    one = [(keys, vals, times, diffs, 0, 3)]
    acc_one, _ = run_oracle(one)
    acc_many, _ = run_oracle(b)
    assert acc_one == acc_many == naive_threshold(b)
