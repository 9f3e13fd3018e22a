"""Q17 dataflow parity: GPU engine vs CPU oracle per churn step."""
import numpy as np
import pytest

from materialize_amd.tpch import TpchGen
from materialize_amd.workloads import Q17Dataflow

pytestmark = pytest.mark.gpu


def test_q17_gpu_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    gen_g = TpchGen(sf=0.05, seed=19)
    gen_o = TpchGen(sf=0.05, seed=19)
    df_g, df_o = Q17Dataflow(g), Q17Dataflow(o)
    df_g.load(gen_g)
    df_o.load(gen_o)
    assert df_g.result == df_o.result, "snapshot"
    for t in range(1, 6):
        df_g.step(gen_g.churn(3000), t)
        df_o.step(gen_o.churn(3000), t)
        assert df_g.result == df_o.result, f"step {t}"
    assert df_g.avg_yearly() == df_o.avg_yearly()
