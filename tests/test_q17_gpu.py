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


def test_q17_step_dev_matches_host_path():
    """The device-resident bench step (raw interior streams, split join
    pushes, reduce_push2 concats) maintains the same result as the
    host-staged `_push` path on the oracle."""
    import torch
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    from materialize_amd import _abi as abi
    import numpy as np
    g, o = GpuCtx(), OracleCtx()
    gen_g = TpchGen(sf=0.05, seed=29)
    gen_o = TpchGen(sf=0.05, seed=29)
    df_g, df_o = Q17Dataflow(g), Q17Dataflow(o)
    df_g.load(gen_g)
    df_o.load(gen_o)
    assert df_g.result == df_o.result
    dev = "cuda:0"
    for t in range(1, 5):
        lp_k, lp_v, lp_d = gen_g.churn(2500)["lineitem_by_part"]
        n = len(lp_k)
        kt = torch.from_numpy(np.ascontiguousarray(lp_k, np.int64)).to(dev)
        vt = torch.from_numpy(
            np.ascontiguousarray(lp_v, np.uint8).reshape(-1)).to(dev)
        tt = torch.full((n,), t, dtype=torch.int64, device=dev)
        dt = torch.from_numpy(np.ascontiguousarray(lp_d, np.int64)).to(dev)
        lp_u = abi.make_updates_from_torch(kt, vt, tt, dt, t, t + 1)
        df_g.step_dev(lp_u, t)
        df_o.step(gen_o.churn(2500), t)
        assert df_g.result == df_o.result, f"step {t}"
    assert df_g.avg_yearly() == df_o.avg_yearly()
    g.close()
    o.close()
