"""Sharded Q3 on the ENGINE over gloo (world 2, both ranks on one GPU):
covers the device-resident exchange path (mz_gpu_partition + column
all_to_all) end-to-end — the union of the two ranks' maintained results
must equal the unsharded GPU run exactly. The RCCL N-GPU run is the
driver's round-end scaling job; this pins the logic it executes."""
import multiprocessing as mp
import os
import sys

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _final_state(df, gen, steps, batch_rows):
    state = {}
    holder = []
    orig = df.reduce.push

    def capture(u):
        o = orig(u)
        holder.append(o.to_host())
        return o

    df.reduce.push = capture
    df.load(gen)
    for t in range(1, steps + 1):
        df.step(gen.churn(batch_rows), t)
    for keys, vals, times, diffs in holder:
        n = len(times)
        vals = vals.reshape(n, 24) if n else vals
        for i in range(n):
            k = (int(keys[2 * i]), int(np.uint64(keys[2 * i + 1])))
            lo = int(vals[i][8:16].view(np.uint64)[0])
            hi = int(vals[i][16:24].view(np.int64)[0])
            v = hi * 2**64 + lo
            if int(diffs[i]) == 1:
                state[k] = v
            else:
                assert state.pop(k) == v
    return state


def _worker(rank, world, port, ret):
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.dist import TorchExchange
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import ShardedQ3Dataflow
    df = ShardedQ3Dataflow(GpuCtx(), TorchExchange("cpu"))
    gen = TpchGen(sf=0.01, seed=17)
    state = _final_state(df, gen, steps=3, batch_rows=800)
    ret[rank] = state
    dist.destroy_process_group()


def test_sharded_q3_engine_gloo_world2():
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import Q3Dataflow
    df = Q3Dataflow(GpuCtx())
    gen = TpchGen(sf=0.01, seed=17)
    want = _final_state(df, gen, steps=3, batch_rows=800)
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    ret = mgr.dict()
    ps = [ctx.Process(target=_worker, args=(r, 2, 29519, ret))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(timeout=300)
        assert p.exitcode == 0
    s0, s1 = ret[0], ret[1]
    assert not (set(s0) & set(s1)), "shards overlap"
    merged = {**s0, **s1}
    assert merged == want
    assert len(want) > 0


def _final_state_dev(df, gen, steps, batch_rows, device):
    """Bench-path variant: staged device updates + step_dev (flush_take
    hand-off + device exchange) — exactly what bench.py executes."""
    from bench import filter_shard, stage_churn
    state = {}
    holder = []
    orig = df.reduce.push
    orig2 = getattr(df.ctx, "reduce_push2_dev", None)

    def capture(u):
        o = orig(u)
        holder.append(o.to_host())
        return o

    def capture2(op, u1, u2):
        o = orig2(op, u1, u2)
        holder.append(o.to_host())
        return o

    df.reduce.push = capture
    if orig2 is not None:
        df.ctx.reduce_push2_dev = capture2
    df.load(gen)
    world = getattr(getattr(df, "exchange", None), "world", 1)
    rank = getattr(getattr(df, "exchange", None), "rank", 0)
    for t in range(1, steps + 1):
        churn = gen.churn(batch_rows)
        churn = filter_shard(churn, world, rank)
        corr = df.step_dev(stage_churn(churn, t, device), t)
        if corr is not None:
            corr.release()
    for keys, vals, times, diffs in holder:
        n = len(times)
        vals = vals.reshape(n, 24) if n else vals
        for i in range(n):
            k = (int(keys[2 * i]), int(np.uint64(keys[2 * i + 1])))
            lo = int(vals[i][8:16].view(np.uint64)[0])
            hi = int(vals[i][16:24].view(np.int64)[0])
            v = hi * 2**64 + lo
            if int(diffs[i]) == 1:
                state[k] = v
            else:
                assert state.pop(k) == v
    return state


def _worker_dev(rank, world, port, ret):
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.dist import TorchExchange
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import ShardedQ3Dataflow
    df = ShardedQ3Dataflow(GpuCtx(), TorchExchange("cpu"))
    gen = TpchGen(sf=0.01, seed=23)
    ret[rank] = _final_state_dev(df, gen, steps=3, batch_rows=1000,
                                 device="cuda:0")
    dist.destroy_process_group()


def test_sharded_q3_engine_gloo_world2_bench_path():
    """The exact bench flow (step_dev: lane inserts, flush_take sorted
    hand-off, device partition + all_to_all) at world 2 equals the
    unsharded engine run."""
    from materialize_amd._ffi import GpuCtx
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import Q3Dataflow
    df = Q3Dataflow(GpuCtx())
    gen = TpchGen(sf=0.01, seed=23)
    want = _final_state_dev(df, gen, steps=3, batch_rows=1000,
                            device="cuda:0")
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    ret = mgr.dict()
    ps = [ctx.Process(target=_worker_dev, args=(r, 2, 29521, ret))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(600)
        assert p.exitcode == 0
    merged = {}
    for r in range(2):
        for k, v in ret[r].items():
            assert k not in merged, f"key {k} on both shards"
            merged[k] = v
    assert merged == want
