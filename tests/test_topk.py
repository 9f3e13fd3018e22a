"""TopK operator (SURVEY §8f — render_topk Basic plan:
build_topk / build_topk_negated_stage, src/compute/src/render/top_k.rs:
322-418,614-770): oracle vs a naive model under churn, covering offset,
limit, descending order, multiplicities spanning the limit boundary, and
retraction of rows inside the kept window."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle import OracleCtx


def canon_val_key(vb_bytes):
    """The engine's canonical val order: zero-padded LE u64 words,
    unsigned ascending (oracle.cpp cmp_val)."""
    b = bytes(vb_bytes)
    words = []
    for off in range(0, len(b), 8):
        chunk = b[off:off + 8] + bytes(8 - min(8, len(b) - off))
        words.append(int.from_bytes(chunk, "little"))
    return tuple(words)


def naive_topk_output(counts, order, offset, limit):
    """counts: {(key, valbytes): net}. Returns {(key, valbytes): kept}."""
    groups = {}
    for (k, v), c in counts.items():
        if c == 0:
            continue
        assert c > 0, "naive model expects validated input"
        groups.setdefault(k, []).append((v, c))
    out = {}
    for k, items in groups.items():
        def sort_key(item):
            v, _ = item
            cols = []
            for (off, width, desc) in order:
                x = int.from_bytes(v[off:off + width], "little", signed=True)
                cols.append(-x if desc else x)
            return (tuple(cols), canon_val_key(v))
        items.sort(key=sort_key)
        running = 0
        for v, c in items:
            lo, hi = running, running + c
            running = hi
            wlo = max(lo, offset)
            whi = hi if limit < 0 else min(hi, offset + limit)
            if whi > wlo:
                out[(k, v)] = whi - wlo
    return out


def run_oracle(spec, batches, kw=1, vb=8):
    ctx = OracleCtx()
    op = ctx.topk_create(spec)
    acc = {}
    outs = []
    for (keys, vals, times, diffs, lower, upper) in batches:
        u = abi.make_updates(keys, vals, times, diffs, lower, upper)
        k, v, t, d = ctx.topk_push(op, u)
        outs.append((k, v, t, d))
        for i in range(len(t)):
            kk = tuple(int(x) for x in k[i * kw:(i + 1) * kw])
            vv = bytes(v[i * vb:(i + 1) * vb])
            acc[(kk, vv)] = acc.get((kk, vv), 0) + int(d[i])
    ctx.close()
    return {r: c for r, c in acc.items() if c != 0}, outs


def churn_batches(seed, steps, n, nkeys=12, vb=8, maxdiff=2):
    """Insert/retract churn that never drives a net count negative."""
    rng = np.random.default_rng(seed)
    net = {}
    batches = []
    for t in range(steps):
        ks, vs, ds = [], [], []
        for _ in range(n):
            k = int(rng.integers(0, nkeys))
            v = int(rng.integers(-20, 20))
            vbytes = v.to_bytes(8, "little", signed=True)[:vb]
            cur = net.get(((k,), vbytes), 0)
            if cur > 0 and rng.random() < 0.45:
                d = -int(rng.integers(1, cur + 1))
            else:
                d = int(rng.integers(1, maxdiff + 1))
            net[((k,), vbytes)] = cur + d
            ks.append(k)
            vs.append(list(vbytes))
            ds.append(d)
        batches.append((np.array(ks, np.int64).reshape(-1, 1),
                        np.array(vs, np.uint8),
                        np.full(n, t, np.uint64),
                        np.array(ds, np.int64), t, t + 1))
    counts = {r: c for r, c in net.items() if c != 0}
    return batches, counts


@pytest.mark.parametrize("offset,limit,desc", [
    (0, 3, False), (0, 1, True), (2, 5, False), (0, -1, False),
    (1, -1, True), (5, 2, True),
])
def test_topk_matches_naive(offset, limit, desc):
    order = [(0, 8, desc)]
    spec = abi.topk_spec(abi.schema(1, 8),
                         [(o, w, d) for (o, w, d) in order],
                         offset=offset, limit=limit)
    batches, counts = churn_batches(11 + offset * 7 + (limit % 5),
                                    steps=5, n=200)
    acc, _ = run_oracle(spec, batches)
    expect = {r: c for r, c in
              naive_topk_output(counts, order, offset, limit).items()}
    assert acc == expect


def test_topk_kat_limit_boundary():
    """Multiplicity straddles the limit: key 1 has val 10 x3, val 20 x2;
    limit 4 keeps 10 x3 and 20 x1 (top_k.rs:755-758 diff clipping)."""
    order = [(0, 8, False)]
    spec = abi.topk_spec(abi.schema(1, 8), order, offset=0, limit=4)
    v10 = np.frombuffer((10).to_bytes(8, "little"), np.uint8)
    v20 = np.frombuffer((20).to_bytes(8, "little"), np.uint8)
    b = (np.array([[1], [1]], np.int64), np.stack([v10, v20]),
         np.zeros(2, np.uint64), np.array([3, 2], np.int64), 0, 1)
    acc, _ = run_oracle(spec, [b])
    assert acc == {((1,), bytes(v10)): 3, ((1,), bytes(v20)): 1}
    # retract one of val 10: window shifts, val 20 now keeps 2
    b2 = (np.array([[1]], np.int64), v10.reshape(1, -1),
          np.ones(1, np.uint64), np.array([-1], np.int64), 1, 2)
    spec2 = abi.topk_spec(abi.schema(1, 8), order, offset=0, limit=4)
    acc2, outs = run_oracle(spec2, [b, b2])
    assert acc2 == {((1,), bytes(v10)): 2, ((1,), bytes(v20)): 2}


def test_topk_negative_multiplicity_errors():
    order = [(0, 8, False)]
    spec = abi.topk_spec(abi.schema(1, 8), order, offset=0, limit=2)
    v = np.zeros((1, 8), np.uint8)
    b = abi.make_updates(np.array([[1]], np.int64), v,
                         np.zeros(1, np.uint64),
                         np.array([-1], np.int64), 0, 1)
    ctx = OracleCtx()
    op = ctx.topk_create(spec)
    with pytest.raises(RuntimeError, match="[Nn]egative multiplicities"):
        ctx.topk_push(op, b)
    ctx.close()


def test_topk_two_order_cols_multitime():
    """Secondary order column breaks primary ties; one push carries
    several timestamps processed in order."""
    order = [(0, 4, True), (4, 4, False)]
    spec = abi.topk_spec(abi.schema(1, 8), order, offset=0, limit=3)
    rng = np.random.default_rng(23)
    n, steps = 150, 3
    ks = rng.integers(0, 6, n * steps).astype(np.int64).reshape(-1, 1)
    a = rng.integers(0, 4, n * steps).astype(np.int32)
    bcol = rng.integers(-9, 9, n * steps).astype(np.int32)
    vals = np.zeros((n * steps, 8), np.uint8)
    vals[:, :4] = a.view(np.uint8).reshape(-1, 4)
    vals[:, 4:] = bcol.view(np.uint8).reshape(-1, 4)
    times = np.repeat(np.arange(steps, dtype=np.uint64), n)
    diffs = np.ones(n * steps, np.int64)
    one = [(ks, vals, times, diffs, 0, steps)]
    acc, _ = run_oracle(spec, one)
    counts = {}
    for i in range(n * steps):
        r = (tuple(ks[i]), bytes(vals[i]))
        counts[r] = counts.get(r, 0) + 1
    expect = naive_topk_output(counts, [(0, 4, True), (4, 4, False)], 0, 3)
    assert acc == expect
