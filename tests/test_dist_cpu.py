"""Sharded Q3 over torch.distributed gloo (world_size 2, CPU).

Covers the multi-GPU path's logic without GPUs: hash-sharded
arrangements, inter-stage all-to-all-v exchanges, sharded reduce. The
union of the two ranks' maintained results must equal the unsharded
oracle run exactly.
"""
import multiprocessing as mp
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _final_state(df, gen, steps, batch_rows):
    """Run load + churn steps, returning the maintained {key: revenue}."""
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
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from materialize_amd.dist import TorchExchange
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import ShardedQ3Dataflow
    from pyoracle import OracleCtx
    ctx = OracleCtx()
    df = ShardedQ3Dataflow(ctx, TorchExchange("cpu"))
    gen = TpchGen(sf=0.002, seed=7)
    state = _final_state(df, gen, steps=3, batch_rows=300)
    ret[rank] = state
    dist.destroy_process_group()


def test_sharded_q3_gloo_world2():
    from materialize_amd.workloads import Q3Dataflow
    from materialize_amd.tpch import TpchGen
    from pyoracle import OracleCtx
    # unsharded reference run
    df = Q3Dataflow(OracleCtx())
    gen = TpchGen(sf=0.002, seed=7)
    want = _final_state(df, gen, steps=3, batch_rows=300)
    # sharded run, world 2
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    ret = mgr.dict()
    port = 29511
    ps = [ctx.Process(target=_worker, args=(r, 2, port, ret))
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


def test_sharded_q3_threads_world2():
    """Same invariant via the in-process ThreadExchange (the CPU-baseline
    transport): union of 2 thread-shards == unsharded."""
    import threading
    from materialize_amd.thread_exchange import ThreadExchangeGroup
    from materialize_amd.workloads import Q3Dataflow, ShardedQ3Dataflow
    from materialize_amd.tpch import TpchGen
    from pyoracle import OracleCtx
    df = Q3Dataflow(OracleCtx())
    gen = TpchGen(sf=0.002, seed=7)
    want = _final_state(df, gen, steps=3, batch_rows=300)

    import copy
    gen2 = TpchGen(sf=0.002, seed=7)
    base = copy.deepcopy(gen2.__dict__)
    churns = [gen2.churn(300) for _ in range(3)]
    gen2.__dict__.update(base)
    group = ThreadExchangeGroup(2)
    states = [None, None]

    def worker(rank):
        ctx = OracleCtx()
        dfr = ShardedQ3Dataflow(ctx, group.member(rank))
        holder = []
        orig = dfr.reduce.push

        def capture(u):
            o = orig(u)
            holder.append(o.to_host())
            return o

        dfr.reduce.push = capture
        dfr.load(gen2)
        for t in range(1, 4):
            dfr.step(churns[t - 1], t)
        st = {}
        for keys, vals, times, diffs in holder:
            n = len(times)
            vals = vals.reshape(n, 24) if n else vals
            for i in range(n):
                k = (int(keys[2 * i]), int(np.uint64(keys[2 * i + 1])))
                lo = int(vals[i][8:16].view(np.uint64)[0])
                hi = int(vals[i][16:24].view(np.int64)[0])
                v = hi * 2**64 + lo
                if int(diffs[i]) == 1:
                    st[k] = v
                else:
                    assert st.pop(k) == v
        states[rank] = st

    ths = [threading.Thread(target=worker, args=(r,)) for r in range(2)]
    for t in ths:
        t.start()
    for t in ths:
        t.join(timeout=300)
    s0, s1 = states
    assert s0 is not None and s1 is not None
    assert not (set(s0) & set(s1))
    assert {**s0, **s1} == want


def _q17_worker(rank, world, port, ret):
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from materialize_amd.dist import TorchExchange
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import ShardedQ17Dataflow
    from pyoracle import OracleCtx
    df = ShardedQ17Dataflow(OracleCtx(), TorchExchange("cpu"))
    gen = TpchGen(sf=0.02, seed=13)
    df.load(gen)
    for t in range(1, 4):
        df.step(gen.churn(800), t)
    ret[rank] = dict(df.result)
    dist.destroy_process_group()


def test_sharded_q17_gloo_world2():
    """Q17 shards by partkey with only the final global-SUM exchange: the
    const-key owner rank's maintained result equals the unsharded run."""
    from materialize_amd.tpch import TpchGen
    from materialize_amd.workloads import Q17Dataflow
    from materialize_amd.dist import shard_of
    from pyoracle import OracleCtx
    df = Q17Dataflow(OracleCtx())
    gen = TpchGen(sf=0.02, seed=13)
    df.load(gen)
    for t in range(1, 4):
        df.step(gen.churn(800), t)
    want = dict(df.result)
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    ret = mgr.dict()
    ps = [ctx.Process(target=_q17_worker, args=(r, 2, 29517, ret))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(timeout=300)
        assert p.exitcode == 0
    owner = int(shard_of(np.zeros(1, np.int64), 1, 2)[0])
    assert ret[owner] == want
    assert ret[1 - owner] == {}
    assert len(want) > 0
