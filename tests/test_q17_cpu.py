"""Q17 dataflow semantics on the CPU oracle: incremental == recomputed.

Runs the full Q17 pipeline (two linear joins + distinct + per-partkey
sum/count + correlated-average filter + global sum) over the oracle and
checks the maintained result after the snapshot and after every churn
batch against a naive numpy recomputation.
"""
import numpy as np

from materialize_amd.tpch import TpchGen
from materialize_amd.workloads import Q17Dataflow
from pyoracle import OracleCtx

BRAND, CONTAINER = 23, 10


def naive_q17(gen):
    """sum(l_extendedprice) over rows with matching part and
    l_quantity < 0.2 * avg(l_quantity per partkey); integer-exact
    comparison 5*q*count < sum."""
    match_parts = set(gen.p_partkey[(gen.p_brand == BRAND) &
                                    (gen.p_container == CONTAINER)].tolist())
    if not match_parts:
        return None
    # per-partkey sum(qty), count over ALL lineitems of partkeys present in
    # the filtered join (the Distinct limits which partkeys, but those are
    # exactly the ones probed)
    total = 0
    any_row = False
    pk = gen.l_partkey
    qty = gen.l_quantity
    ep = gen.l_extendedprice
    sums = {}
    counts = {}
    for i in range(len(pk)):
        p = int(pk[i])
        if p in match_parts:
            pass
        # per-partkey stats are over ALL lineitems of that partkey
    # vectorized per-partkey stats
    sums_v = np.zeros(gen.n_part + 1, np.int64)
    cnts_v = np.zeros(gen.n_part + 1, np.int64)
    np.add.at(sums_v, pk, qty)
    np.add.at(cnts_v, pk, 1)
    for i in range(len(pk)):
        p = int(pk[i])
        if p not in match_parts:
            continue
        any_row = True
        if 5 * int(qty[i]) * int(cnts_v[p]) < int(sums_v[p]):
            total += int(ep[i])
    return total if any_row else None


def test_q17_oracle_end_to_end():
    gen = TpchGen(sf=0.02, seed=11)  # ~4000 parts -> a few matching
    ctx = OracleCtx()
    df = Q17Dataflow(ctx)
    df.load(gen)
    want = naive_q17(gen)
    got = df.result.get(0)
    assert got == want, f"snapshot: {got} != {want}"
    for t in range(1, 5):
        churn = gen.churn(1500)
        df.step(churn, t)
        want = naive_q17(gen)
        got = df.result.get(0)
        assert got == want, f"step {t}: {got} != {want}"
