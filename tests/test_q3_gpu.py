"""Q3 dataflow parity: GPU engine vs CPU oracle, bit-exact per step."""
import numpy as np
import pytest

from materialize_amd.tpch import TpchGen
from materialize_amd.workloads import Q3Dataflow

pytestmark = pytest.mark.gpu


def _capture(df):
    holder = []
    orig = df.reduce.push

    def push(u):
        o = orig(u)
        holder.append(o.to_host())
        return o

    df.reduce.push = push
    return holder


def _norm(cols_list):
    """Concatenate correction batches into one comparable tuple."""
    if not cols_list:
        return None
    keys = np.concatenate([c[0] for c in cols_list])
    vals = np.concatenate([c[1] for c in cols_list])
    times = np.concatenate([c[2] for c in cols_list])
    diffs = np.concatenate([c[3] for c in cols_list])
    return keys, vals, times, diffs


def test_q3_gpu_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    gen_g = TpchGen(sf=0.01, seed=3)
    gen_o = TpchGen(sf=0.01, seed=3)
    df_g, df_o = Q3Dataflow(g), Q3Dataflow(o)
    cap_g, cap_o = _capture(df_g), _capture(df_o)
    df_g.load(gen_g)
    df_o.load(gen_o)
    for label in ["snapshot"]:
        a, b = _norm(cap_g), _norm(cap_o)
        assert (a is None) == (b is None)
        if a:
            for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
                np.testing.assert_array_equal(
                    x.view(np.uint8), y.view(np.uint8),
                    err_msg=f"{label}: {what}")
    for t in range(1, 6):
        cap_g.clear()
        cap_o.clear()
        rows_g, cg = df_g.step(gen_g.churn(2000), t)
        rows_o, co = df_o.step(gen_o.churn(2000), t)
        if cg is not None:
            cg.release()
        if co is not None:
            co.release()
        assert rows_g == rows_o
        a, b = _norm(cap_g), _norm(cap_o)
        assert (a is None) == (b is None), f"step {t}"
        if a:
            for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
                np.testing.assert_array_equal(
                    x.view(np.uint8), y.view(np.uint8),
                    err_msg=f"step {t}: {what}")
        # maintenance mid-stream must not change results
        if t == 3:
            df_g.maintain()
            df_o.maintain()


def test_q3_customer_churn_gpu_matches_oracle():
    """Customer churn (third delta path) under retractions: GPU engine
    bit-equal to the oracle per step."""
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    gen_g = TpchGen(sf=0.01, seed=13)
    gen_o = TpchGen(sf=0.01, seed=13)
    df_g, df_o = Q3Dataflow(g), Q3Dataflow(o)
    df_g.load(gen_g)
    df_o.load(gen_o)
    for t in range(1, 6):
        churn_g = gen_g.churn(1000)
        churn_g["customer"] = gen_g.churn_customers(30)
        churn_o = gen_o.churn(1000)
        churn_o["customer"] = gen_o.churn_customers(30)
        _, cg = df_g.step(churn_g, t)
        _, co = df_o.step(churn_o, t)
        a = cg.to_host() if cg is not None else None
        b = co.to_host() if co is not None else None
        if cg is not None:
            cg.release()
        if co is not None:
            co.release()
        assert (a is None) == (b is None)
        if a is not None:
            for x, y, what in zip(a, b, ("keys", "vals", "times", "diffs")):
                np.testing.assert_array_equal(
                    np.asarray(x).view(np.uint8),
                    np.asarray(y).view(np.uint8),
                    err_msg=f"t={t}: {what}")
    g.close()
    o.close()
