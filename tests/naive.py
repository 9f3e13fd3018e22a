"""Naive Python reference implementations for property-testing the oracle.

Dictionary-based, obviously-correct semantics of an incremental equi-join
and accumulable reduce over multisets of ((key, val), time, diff).
"""
from collections import defaultdict


def consolidate(updates):
    """Sort by (key, val, time), sum diffs (wrapping i64), drop zeros."""
    acc = defaultdict(int)
    for k, v, t, d in updates:
        acc[(k, v, t)] += d
    out = []
    for (k, v, t), d in sorted(acc.items()):
        d = ((d + 2**63) % 2**64) - 2**63  # wrap to i64
        if d != 0:
            out.append((k, v, t, d))
    return out


def join_full(in1, in2):
    """Full cross-product join of two update sets sharing a key:
    for each pair, emit ((key, (v1, v2)), max(t1,t2), d1*d2)."""
    by_key2 = defaultdict(list)
    for k, v, t, d in in2:
        by_key2[k].append((v, t, d))
    out = []
    for k, v1, t1, d1 in in1:
        for v2, t2, d2 in by_key2.get(k, ()):
            out.append((k, (v1, v2), max(t1, t2), d1 * d2))
    return consolidate(out)


def reduce_snapshot(updates, time, aggs):
    """Accumulable reduce of the collection as of `time` (inclusive):
    returns {key: (finalized aggregate tuple)} for keys with any presence.

    aggs: list of ('count'|'sum',) specs where val is a tuple of datums
    (None = NULL) and agg i reads val[i].
    """
    state = defaultdict(lambda: [0] * (len(aggs) * 2 + 1))  # accum,nn pairs + total
    for k, v, t, d in updates:
        if t > time:
            continue
        st = state[k]
        for i, kind in enumerate(aggs):
            datum = v[i]
            if datum is not None:
                if kind == "sum":
                    st[2 * i] += datum * d
                st[2 * i + 1] += d  # non_nulls
        st[-1] += d  # total
    out = {}
    for k, st in state.items():
        if all(x == 0 for x in st):
            continue
        row = []
        total = st[-1]
        for i, kind in enumerate(aggs):
            accum, nn = st[2 * i], st[2 * i + 1]
            if kind == "count":
                row.append(nn)
            else:
                if total > 0 and accum == 0 and nn == 0:
                    row.append(None)  # all-null sum
                else:
                    row.append(accum)
        out[k] = tuple(row)
    return out
