"""1-deep insert pipeline (bench path): step_dev with next_upd enqueues
batch t+1's lane consolidations under batch t's probes/reduce. A pending
batch whose lower frontier is t+1 is invisible to every le/lt probe at
time t (delta_join.rs:356-399 time tie-breaks: t2 >= t+1 > t1), so the
probes skip its flush and the engine overlaps ingest with compute.
Corrections must be bit-identical to the unpipelined sequence."""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def _run(pipelined, steps=6, batch=1200):
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import Q3Dataflow

    ctx = GpuCtx()
    df = Q3Dataflow(ctx)
    gen = TpchGen(sf=0.01, seed=11)
    df.load(gen)

    def stage(t):
        churn = gen.churn(batch)
        upd = {"lineitem": churn["lineitem"],
               "orders_by_orderkey": churn["orders"],
               "orders_by_custkey": churn["orders_by_cust"]}
        out = {}
        for name, (kk, vv, dd) in upd.items():
            out[name] = abi.make_updates(
                np.ascontiguousarray(kk, np.int64), vv,
                np.full(len(kk), t, np.uint64),
                np.ascontiguousarray(dd, np.int64), t, t + 1)
        return out

    # staged descriptors (and their numpy backing) stay alive for the
    # whole run: async inserts read them on the lane streams
    staged = [stage(t) for t in range(1, steps + 2)]
    outs = []
    for i in range(steps):
        nxt = staged[i + 1] if pipelined else None
        corr = df.step_dev(staged[i], i + 1, next_upd=nxt)
        if corr is None:
            outs.append(None)
        else:
            outs.append(tuple(np.asarray(x).copy()
                              for x in corr.to_host()))
            corr.release()
    ctx.close()
    return outs


def test_pipelined_step_dev_matches_plain():
    plain = _run(False)
    piped = _run(True)
    assert len(plain) == len(piped)
    for s, (a, b) in enumerate(zip(plain, piped)):
        assert (a is None) == (b is None), f"step {s} presence"
        if a is None:
            continue
        for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
            np.testing.assert_array_equal(
                np.asarray(x).view(np.uint8).ravel(),
                np.asarray(y).view(np.uint8).ravel(),
                err_msg=f"step {s} {what}")
