"""Benchmark workload dataflows (hand-written plans matching the
reference's pinned EXPLAIN output).

Q3's plan is pinned at test/sqllogictest/tpch_create_index.slt:382-386:

    %0:customer » %1:orders[o_custkey]KAif » %2:lineitem[l_orderkey]KAif
    %1:orders   » %0:customer[c_custkey]KAef » %2:lineitem[l_orderkey]KAif
    %2:lineitem » %1:orders[o_orderkey]KAif  » %0:customer[c_custkey]KAef

with filters c_mktsegment='BUILDING', o_orderdate < 1995-03-15,
l_shipdate > 1995-03-15, and a Reduce group_by=(o_orderkey, o_orderdate,
o_shippriority) aggregates=[sum(l_extendedprice * (1 - l_discount))].
Time tie-breaks: le when source relation precedes the lookup relation,
lt otherwise (delta_join.rs:356-399).
"""
import numpy as np

from . import _abi as abi
from .render import (DeltaJoinPlan, DeltaPathPlan, DeltaStagePlan, ReducePlan,
                     render_delta_join, render_reduce)
from .tpch import CUTOFF_19950315 as CUTOFF

F = abi.field
FL = abi.filt
KEY, VS, VL, CP = (abi.MZ_SRC_KEY, abi.MZ_SRC_VAL_STREAM,
                   abi.MZ_SRC_VAL_LOOKUP, abi.MZ_SRC_COMPUTE)


def _ident(keys, vals, diffs):
    return keys, vals, diffs


def q3_plan():
    """The three delta paths of Q3 (see module docstring)."""
    # stage closures ------------------------------------------------
    # customer path, stage A: lookup orders_by_custkey (le; 0 < 1)
    #   stream: key=c_custkey, val=[mkt i64]
    #   lookup val: [o_orderkey i64][o_orderdate i32][o_shippriority i32]
    cl_c_orders = abi.closure(
        [FL(VS, 0, 8, abi.MZ_CMP_EQ, 0),          # c_mktsegment = BUILDING
         FL(VL, 8, 4, abi.MZ_CMP_LT, CUTOFF)],    # o_orderdate < cutoff
        [F(VL, 0, 8)],                            # key := o_orderkey
        [F(VL, 8, 8)],                            # val := date||prio
        abi.schema(1, 8))
    # shared final stage: lookup lineitem[l_orderkey]
    #   stream: key=o_orderkey, val=[date||prio 8B]
    #   lookup val: [extprice i64][discount i64][shipdate i32][pad]
    cl_x_lineitem = abi.closure(
        [FL(VL, 16, 4, abi.MZ_CMP_GT, CUTOFF)],   # l_shipdate > cutoff
        [F(KEY, 0, 8), F(VS, 0, 8)],              # key := (orderkey, d||p)
        [F(CP, abi.MZ_COMPUTE_REVENUE, 8, arg0=0, arg1=8,
           arg0_src=VL, arg1_src=VL)],            # val := revenue
        abi.schema(2, 8))
    # orders path, stage A: lookup customer[c_custkey] (lt; 1 > 0)
    #   stream: key=o_custkey, val=[o_orderkey i64][date i32][prio i32]
    #   lookup val: [mkt i64]
    cl_o_customer = abi.closure(
        [FL(VL, 0, 8, abi.MZ_CMP_EQ, 0),
         FL(VS, 8, 4, abi.MZ_CMP_LT, CUTOFF)],
        [F(VS, 0, 8)],                            # key := o_orderkey
        [F(VS, 8, 8)],                            # val := date||prio
        abi.schema(1, 8))
    # lineitem path, stage A: lookup orders_by_orderkey (lt; 2 > 1)
    #   stream: key=l_orderkey, val=[extprice i64][disc i64][shipdate i32]
    #   lookup val: [o_custkey i64][date i32][prio i32]
    cl_l_orders = abi.closure(
        [FL(VS, 16, 4, abi.MZ_CMP_GT, CUTOFF),
         FL(VL, 8, 4, abi.MZ_CMP_LT, CUTOFF)],
        [F(VL, 0, 8)],                            # key := o_custkey
        [F(KEY, 0, 8),                            # val := orderkey
         F(VS, 0, 16),                            #        extprice, disc
         F(VL, 8, 8)],                            #        date||prio
        abi.schema(1, 32))
    # lineitem path, stage B: lookup customer[c_custkey] (lt; 2 > 0)
    #   stream: key=c_custkey, val=[orderkey][extprice][disc][date||prio]
    cl_l_customer = abi.closure(
        [FL(VL, 0, 8, abi.MZ_CMP_EQ, 0)],
        [F(VS, 0, 8), F(VS, 24, 8)],              # key := (orderkey, d||p)
        [F(CP, abi.MZ_COMPUTE_REVENUE, 8, arg0=8, arg1=16,
           arg0_src=VS, arg1_src=VS)],
        abi.schema(2, 8))

    return DeltaJoinPlan(paths=[
        DeltaPathPlan("customer", _ident, [
            DeltaStagePlan("orders_by_custkey", True, cl_c_orders, 8),
            DeltaStagePlan("lineitem", True, cl_x_lineitem, 8),
        ]),
        DeltaPathPlan("orders", _ident, [
            DeltaStagePlan("customer", False, cl_o_customer, 16),
            DeltaStagePlan("lineitem", True, cl_x_lineitem, 8),
        ]),
        DeltaPathPlan("lineitem", _ident, [
            DeltaStagePlan("orders_by_orderkey", False, cl_l_orders, 24),
            DeltaStagePlan("customer", False, cl_l_customer, 32),
        ]),
    ])


def q3_reduce_plan():
    aggs = [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                          is_float=0, nullable=0)]
    return ReducePlan(abi.reduce_spec(aggs, abi.schema(2, 8)))


class Q3Dataflow:
    """TPC-H Q3 maintained incrementally: 3-path delta join + accumulable
    SUM reduce, all arrangements engine-resident. Works over any engine
    context exposing the GpuCtx interface (the oracle wrapper does too,
    which is what the parity tests and the CPU-baseline bench leg use)."""

    SCHEMAS = {
        "customer": (1, 8),
        "orders_by_orderkey": (1, 16),
        "orders_by_custkey": (1, 16),
        "lineitem": (1, 24),
    }

    def __init__(self, ctx):
        self.ctx = ctx
        self.arrs = {name: ctx.arr_create(abi.schema(kw, vb))
                     for name, (kw, vb) in self.SCHEMAS.items()}
        self.plan = q3_plan()
        self.join = render_delta_join(ctx, self.arrs, self.plan)
        self.reduce = render_reduce(ctx, q3_reduce_plan())
        self.paths = {p.source_relation: p for p in self.plan.paths}

    def _seal_push(self, name, keys, vals, diffs, t):
        ctx = self.ctx
        kw, vb = self.SCHEMAS[name]
        u = abi.make_updates(keys, vals, np.full(len(keys), t, np.uint64),
                             diffs, t, t + 1)
        sealed = ctx.consolidate_dev(abi.schema(kw, vb), u)
        ctx.arr_push(self.arrs[name], sealed.updates(t, t + 1))
        sealed.release()

    def load(self, gen):
        """Load the base snapshot at t=0 and seed the reduce with the
        snapshot join result via path 0 (the as-of rule: only path 0 emits
        at the as-of, delta_join.rs:752-798)."""
        ones = lambda n: np.ones(n, np.int64)
        ck, cm = gen.customer_updates()
        self._seal_push("customer", ck, cm.view(np.uint8), ones(len(ck)), 0)
        oidx = np.arange(gen.n_orders)
        self._seal_push("orders_by_orderkey", gen.o_orderkey,
                        gen.orders_vals(oidx), ones(gen.n_orders), 0)
        self._seal_push("orders_by_custkey", gen.o_custkey,
                        gen.orders_bycust_vals(oidx), ones(gen.n_orders), 0)
        lk, lv = gen.lineitem_updates()
        self._seal_push("lineitem", lk, lv, ones(len(lk)), 0)
        # snapshot: path 0 with the full customer collection
        out = self.join.push_path(self.paths["customer"], ck,
                                  cm.view(np.uint8).reshape(-1, 8),
                                  ones(len(ck)), 0)
        n_corr = 0
        if out is not None:
            corr = self.reduce.push(out.updates(0, 1))
            n_corr = corr.n
            corr.release()
            out.release()
        return n_corr

    def step(self, churn, t):
        """Maintain one churn batch at time t. Returns (input_rows,
        corrections DevOut or None)."""
        ctx = self.ctx
        l_keys, l_vals, l_diffs = churn["lineitem"]
        o_keys, o_vals, o_diffs = churn["orders"]
        oc_keys, oc_vals, oc_diffs = churn["orders_by_cust"]
        rows = len(l_keys) + len(o_keys)
        # 1. arrangements first (the paths' le/lt tie-breaks then count
        #    concurrent cross-terms exactly once — DESIGN.md §5)
        self._seal_push("lineitem", l_keys, l_vals, l_diffs, t)
        self._seal_push("orders_by_orderkey", o_keys, o_vals, o_diffs, t)
        self._seal_push("orders_by_custkey", oc_keys, oc_vals, oc_diffs, t)
        # 2. delta paths (customer static in the churn workload)
        outs = []
        o = self.join.push_path(self.paths["orders"], oc_keys, oc_vals,
                                oc_diffs, t)
        if o is not None:
            outs.append(o)
        o = self.join.push_path(self.paths["lineitem"], l_keys, l_vals,
                                l_diffs, t)
        if o is not None:
            outs.append(o)
        if not outs:
            return rows, None
        # 3. concatenate path outputs -> reduce
        if len(outs) == 1:
            u = outs[0].updates(t, t + 1)
            corr = self.reduce.push(u)
        else:
            cols = [out.to_host() for out in outs]
            keys = np.concatenate([c[0] for c in cols])
            vals = np.concatenate([c[1] for c in cols])
            times = np.concatenate([c[2] for c in cols])
            diffs = np.concatenate([c[3] for c in cols])
            u = abi.make_updates(keys, vals, times, diffs, t, t + 1)
            corr = self.reduce.push(u)
        for out in outs:
            out.release()
        return rows, corr

    _final_exchange = False

    def step_dev(self, upd, t):
        """Bench path: one churn step whose update columns are ALREADY
        staged (device tensors at N GPUs — inputs resident in HBM when the
        timed region starts). `upd` maps the three updated relations to
        Updates descriptors with times == t. Returns the corrections
        DevOut (or None)."""
        ctx = self.ctx
        for name in ("lineitem", "orders_by_orderkey", "orders_by_custkey"):
            u = upd[name]
            kw, vb = self.SCHEMAS[name]
            sealed = ctx.consolidate_dev(abi.schema(kw, vb), u)
            ctx.arr_push(self.arrs[name], sealed.updates(t, t + 1))
            sealed.release()
        outs = []
        for rel, src in (("orders", "orders_by_custkey"),
                         ("lineitem", "lineitem")):
            o = self.join.push_path_updates(
                self.paths[rel], upd[src], t,
                final_exchange=self._final_exchange)
            if o is not None:
                outs.append(o)
        if not outs:
            return None
        if len(outs) == 1:
            corr = self.reduce.push(outs[0].updates(t, t + 1))
        else:
            cols = [out.to_host() for out in outs]
            keys = np.concatenate([c[0] for c in cols])
            vals = np.concatenate([c[1] for c in cols])
            times = np.concatenate([c[2] for c in cols])
            diffs = np.concatenate([c[3] for c in cols])
            u = abi.make_updates(keys, vals, times, diffs, t, t + 1)
            corr = self.reduce.push(u)
        for out in outs:
            out.release()
        return corr

    def maintain(self):
        for arr in self.arrs.values():
            self.ctx.arr_maintain(arr)

    def stats(self):
        return {name: self.ctx.arr_stats(arr)
                for name, arr in self.arrs.items()}


class ShardedQ3Dataflow(Q3Dataflow):
    """Q3 sharded across ranks: every arrangement is hash-partitioned by
    its key (shard = route_hash(key) % world); stage outputs
    are re-distributed by the next key's hash via the exchange (RCCL
    all-to-all over xGMI at N GPUs; gloo in CPU tests). Every rank
    generates the same deterministic churn and keeps its own shard."""

    _final_exchange = True

    def __init__(self, ctx, exchange):
        from .dist import shard_of
        self.shard_of = shard_of
        self.exchange = exchange
        super().__init__(ctx)
        self.join = render_delta_join(ctx, self.arrs, self.plan,
                                      exchange=exchange)

    def _filter_shard(self, keys, vals, diffs):
        W, r = self.exchange.world, self.exchange.rank
        if W == 1:
            return keys, vals, diffs
        m = self.shard_of(np.ascontiguousarray(keys, np.int64), 1, W) == r
        vals = vals.reshape(len(keys), -1)
        return keys[m], vals[m], diffs[m]

    def _seal_push(self, name, keys, vals, diffs, t):
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.uint8).reshape(len(keys), -1)
        diffs = np.ascontiguousarray(diffs, np.int64)
        keys, vals, diffs = self._filter_shard(keys, vals, diffs)
        if len(keys) == 0:
            return
        super()._seal_push(name, keys, vals, diffs, t)

    def _push_sharded_path(self, name, keys, vals, diffs, t):
        """Filter source updates to this rank's shard of the first stage's
        key, then run the path with inter-stage exchanges."""
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.uint8).reshape(len(keys), -1)
        diffs = np.ascontiguousarray(diffs, np.int64)
        keys, vals, diffs = self._filter_shard(keys, vals, diffs)
        return self.join.push_path(self.paths[name], keys, vals, diffs, t,
                                   final_exchange=True)

    def load(self, gen):
        ones = lambda n: np.ones(n, np.int64)
        ck, cm = gen.customer_updates()
        self._seal_push("customer", ck, cm.view(np.uint8), ones(len(ck)), 0)
        oidx = np.arange(gen.n_orders)
        self._seal_push("orders_by_orderkey", gen.o_orderkey,
                        gen.orders_vals(oidx), ones(gen.n_orders), 0)
        self._seal_push("orders_by_custkey", gen.o_custkey,
                        gen.orders_bycust_vals(oidx), ones(gen.n_orders), 0)
        lk, lv = gen.lineitem_updates()
        self._seal_push("lineitem", lk, lv, ones(len(lk)), 0)
        out = self._push_sharded_path("customer", ck,
                                      cm.view(np.uint8).reshape(-1, 8),
                                      ones(len(ck)), 0)
        n_corr = 0
        if out is not None:
            corr = self.reduce.push(out.updates(0, 1))
            n_corr = corr.n
            corr.release()
            out.release()
        return n_corr

    def step(self, churn, t):
        l_keys, l_vals, l_diffs = churn["lineitem"]
        o_keys, o_vals, o_diffs = churn["orders"]
        oc_keys, oc_vals, oc_diffs = churn["orders_by_cust"]
        rows = len(l_keys) + len(o_keys)  # global input rows (same on all
        # ranks; bench divides by wall time once, using the global count)
        self._seal_push("lineitem", l_keys, l_vals, l_diffs, t)
        self._seal_push("orders_by_orderkey", o_keys, o_vals, o_diffs, t)
        self._seal_push("orders_by_custkey", oc_keys, oc_vals, oc_diffs, t)
        outs = []
        o = self._push_sharded_path("orders", oc_keys, oc_vals, oc_diffs, t)
        if o is not None:
            outs.append(o)
        o = self._push_sharded_path("lineitem", l_keys, l_vals, l_diffs, t)
        if o is not None:
            outs.append(o)
        if not outs:
            return rows, None
        if len(outs) == 1:
            corr = self.reduce.push(outs[0].updates(t, t + 1))
        else:
            cols = [out.to_host() for out in outs]
            keys = np.concatenate([c[0] for c in cols])
            vals = np.concatenate([c[1] for c in cols])
            times = np.concatenate([c[2] for c in cols])
            diffs = np.concatenate([c[3] for c in cols])
            u = abi.make_updates(keys, vals, times, diffs, t, t + 1)
            corr = self.reduce.push(u)
        for out in outs:
            out.release()
        return rows, corr
