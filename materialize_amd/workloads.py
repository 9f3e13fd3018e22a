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
        u = abi.make_updates(keys, vals, np.full(len(keys), t, np.uint64),
                             diffs, t, t + 1)
        ctx.arr_insert(self.arrs[name], u)

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

    def _advance_compaction(self, t):
        """Logical compaction to one step back of the frontier
        (mz_join_core.rs:461 via the delta path's step_back held frontier,
        delta_join.rs:614-616): merges then cancel retract/insert pairs at
        compacted times, keeping steady-state arrangements bounded, while
        lt tie-breaks at data times >= t stay exact (frontier-1 < t)."""
        if t >= 1:
            for arr in self.arrs.values():
                self.ctx.arr_set_logical_compaction(arr, t - 1)

    def step(self, churn, t):
        """Maintain one churn batch at time t. Returns (input_rows,
        corrections DevOut or None)."""
        ctx = self.ctx
        self._advance_compaction(t)
        l_keys, l_vals, l_diffs = churn["lineitem"]
        o_keys, o_vals, o_diffs = churn["orders"]
        oc_keys, oc_vals, oc_diffs = churn["orders_by_cust"]
        c_churn = churn.get("customer")  # optional customer churn
        rows = len(l_keys) + len(o_keys) + \
            (len(c_churn[0]) if c_churn is not None else 0)
        # 1. arrangements first (the paths' le/lt tie-breaks then count
        #    concurrent cross-terms exactly once — DESIGN.md §5)
        self._seal_push("lineitem", l_keys, l_vals, l_diffs, t)
        self._seal_push("orders_by_orderkey", o_keys, o_vals, o_diffs, t)
        self._seal_push("orders_by_custkey", oc_keys, oc_vals, oc_diffs, t)
        if c_churn is not None:
            self._seal_push("customer", c_churn[0],
                            np.ascontiguousarray(c_churn[1],
                                                 np.int64).view(np.uint8),
                            c_churn[2], t)
        # 2. delta paths, one per updated source relation
        outs = []
        if c_churn is not None:
            o = self.join.push_path(
                self.paths["customer"], c_churn[0],
                np.ascontiguousarray(c_churn[1], np.int64).view(np.uint8)
                .reshape(-1, 8), c_churn[2], t)
            if o is not None:
                outs.append(o)
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

    _CHURNED = ("lineitem", "orders_by_orderkey", "orders_by_custkey")
    _preingested = None

    def step_dev(self, upd, t, next_upd=None):
        """Bench path: one churn step whose update columns are ALREADY
        staged (device tensors at N GPUs — inputs resident in HBM when the
        timed region starts). `upd` maps the three updated relations to
        Updates descriptors with times == t. Returns the corrections
        DevOut (or None).

        `next_upd` (batch t+1's staged columns) enables the 1-deep insert
        pipeline: batch t+1's lane consolidations are enqueued right
        after batch t's flush and run concurrently with batch t's probes
        and reduce on the main stream — legal because a pending batch
        whose lower frontier is t+1 is invisible to every le/lt probe at
        time t, so the probes skip its flush (timely's operator
        concurrency across capabilities, done with HIP streams)."""
        ctx = self.ctx
        # Each relation's churn is consolidated ONCE, on its arrangement's
        # own lane (the three sort pipelines overlap); the path probes
        # then consume the arrangements' published update streams — the
        # flush_take hand-off of the sealed sorted rows, exactly
        # mz_arrange_core feeding both the trace and downstream operators
        # (extensions/arrange.rs:69-114). No per-path re-sort, and the
        # sorted flag lets large-table probes take the merge path.
        if self._preingested is not upd:
            for name in self._CHURNED:
                ctx.arr_insert_async(self.arrs[name], upd[name])
        take = {}
        for name in ("orders_by_custkey", "lineitem"):
            take[name] = (ctx.arr_flush_take(self.arrs[name])
                          if hasattr(ctx, "arr_flush_take") else None)
        if next_upd is not None and hasattr(ctx, "arr_flush_take"):
            # 1-deep pipeline: enqueue batch t+1's lane consolidations now
            # (orders_by_orderkey's pending batch t auto-flushes first);
            # they execute under this step's probes/reduce.
            for name in self._CHURNED:
                ctx.arr_insert_async(self.arrs[name], next_upd[name])
            self._preingested = next_upd
        else:
            self._preingested = None
        outs = []
        for rel, src in (("orders", "orders_by_custkey"),
                         ("lineitem", "lineitem")):
            if take.get(src) is not None:
                u = take[src].updates(t, t + 1)
            else:  # oracle / fallback: the raw staged batch
                u = upd[src]
            o = self.join.push_path_updates(
                self.paths[rel], u, t,
                final_exchange=self._final_exchange)
            if o is not None:
                outs.append(o)
        for tk in take.values():
            if tk is not None:
                tk.release()
        if not outs:
            return None
        if len(outs) == 1:
            corr = self.reduce.push(outs[0].updates(t, t + 1))
        else:
            corr = self.ctx.reduce_push2_dev(self.reduce.op,
                                             outs[0].updates(t, t + 1),
                                             outs[1].updates(t, t + 1))
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


# ===================================================================== Q17

def q17_closures():
    """Closures of Q17's pinned plan (tpch_create_index.slt:1449-1505):
    l1 = lineitem ⋈ part (brand/container filters); per-partkey
    sum(quantity)/count over Distinct(l1.partkey) ⋈ lineitem; final
    l1 ⋈ avg with quantity < 0.2*avg (exact integer form),
    then a global SUM(extendedprice). Brand#23 = code 23, MED BOX = 10."""
    # join1: lineitem(by partkey) ⋈ part — input1 = lineitem
    #   lineitem val: [quantity i64][extprice i64]; part val: [brand][cont]
    cl_j1 = abi.closure(
        [FL(VL, 0, 8, abi.MZ_CMP_EQ, 23),        # p_brand = Brand#23
         FL(VL, 8, 8, abi.MZ_CMP_EQ, 10)],       # p_container = MED BOX
        [F(KEY, 0, 8)],                          # key := partkey
        [F(VS, 0, 16)],                          # val := (qty, extprice)
        abi.schema(1, 16))
    # join2: distinct(partkey) ⋈ lineitem(by partkey) — input1 = distinct
    cl_j2 = abi.closure(
        [],
        [F(KEY, 0, 8)],                          # key := partkey
        [F(VL, 0, 8)],                           # val := quantity
        abi.schema(1, 8))
    # join3: l1 ⋈ avg-reduce output — input1 = l1 (qty, extprice),
    #   input2 val = 48B reduce row: slot0 = SUM_I64(qty), slot1 = COUNT
    cl_j3 = abi.closure(
        [FL(CP, abi.MZ_COMPUTE_Q17_QTYLT, 8, abi.MZ_CMP_LT, 0,
            arg0=0, arg1=8, arg0_src=VS, arg1_src=VL)],
        [F(CP, abi.MZ_COMPUTE_CONST0, 8)],       # key := 0 (global sum)
        [F(VS, 8, 8)],                           # val := extprice
        abi.schema(1, 8))
    return cl_j1, cl_j2, cl_j3


class Q17Dataflow:
    """TPC-H Q17 maintained incrementally (config 5 shape): two linear
    joins, a distinct, a per-partkey SUM/COUNT reduce feeding an
    arrangement, the correlated-average filter join, and a global SUM.
    Exactly-once across the concurrent per-step deltas follows the
    mz_join_core drain discipline: for each binary join,
    one side's delta probes the other side BEFORE that side's delta is
    pushed, and vice versa."""

    def __init__(self, ctx):
        from .render import (LinearJoinPlan, LinearStagePlan, render_join)
        self.ctx = ctx
        self.arr_l0 = ctx.arr_create(abi.schema(1, 16))    # lineitem by pk
        self.arr_part = ctx.arr_create(abi.schema(1, 16))
        self.arr_l1 = ctx.arr_create(abi.schema(1, 16))    # filtered rows
        self.arr_dist = ctx.arr_create(abi.schema(1, 0))   # distinct pk
        self.arr_avg = ctx.arr_create(abi.schema(1, 48))   # sum/count rows
        self.arrs = {"lineitem_by_pk": self.arr_l0, "part": self.arr_part,
                     "l1": self.arr_l1, "distinct": self.arr_dist,
                     "avg": self.arr_avg}
        cl_j1, cl_j2, cl_j3 = q17_closures()
        # The three Plan::Join{Linear} nodes of the pinned Q17 plan
        # (tpch_create_index.slt:1449-1505), each expressed through the
        # LinearJoinPlan rendering surface; the
        # reduces between them are separate operators, as in the
        # reference's render loop.
        self.j1op = render_join(ctx, self.arrs, LinearJoinPlan(
            source_relation="lineitem_by_pk",
            stage_plans=[LinearStagePlan("part", cl_j1,
                                         stream_val_bytes=16)]))
        self.j2op = render_join(ctx, self.arrs, LinearJoinPlan(
            source_relation="distinct",
            stage_plans=[LinearStagePlan("lineitem_by_pk", cl_j2,
                                         stream_val_bytes=0)]))
        self.j3op = render_join(ctx, self.arrs, LinearJoinPlan(
            source_relation="l1",
            stage_plans=[LinearStagePlan("avg", cl_j3,
                                         stream_val_bytes=16)]))
        self.j1 = self.j1op.joins[0]
        self.j2 = self.j2op.joins[0]
        self.j3 = self.j3op.joins[0]
        self.distinct = ctx.reduce_create(
            abi.reduce_spec([], abi.schema(1, 0)))
        self.avg = ctx.reduce_create(abi.reduce_spec(
            [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                           is_float=0, nullable=0),
             abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                           is_float=0, nullable=0)],
            abi.schema(1, 8)))
        self.total = ctx.reduce_create(abi.reduce_spec(
            [abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                           is_float=0, nullable=0)],
            abi.schema(1, 8)))
        self.result = {}  # maintained {0: sum_extendedprice}

    def _apply_total(self, corr):
        keys, vals, times, diffs = corr
        n = len(times)
        vals = vals.reshape(n, 24) if n else vals
        # apply retractions before insertions (consolidated output orders
        # rows by val bytes, not by diff sign)
        order = sorted(range(n), key=lambda i: (int(times[i]),
                                                int(diffs[i])))
        for i in order:
            lo = int(vals[i][8:16].view(np.uint64)[0])
            hi = int(vals[i][16:24].view(np.int64)[0])
            v = hi * 2**64 + lo
            if int(diffs[i]) == 1:
                self.result[int(keys[i])] = v
            else:
                assert self.result.pop(int(keys[i])) == v

    def avg_yearly(self):
        """sum(l_extendedprice)/7.0 in cents (the reference's final Map)."""
        s = self.result.get(0)
        return None if s is None else s / 7.0

    def _push(self, t, lp=None, part=None):
        """One timestamp's worth of updates (lp = lineitem-by-partkey
        columns, part = part columns); pass None for an idle input."""
        ctx = self.ctx
        empty = (np.empty(0, np.int64), np.empty((0, 16), np.uint8),
                 np.empty(0, np.int64))
        lp_k, lp_v, lp_d = lp if lp is not None else empty
        p_k, p_v, p_d = part if part is not None else empty

        def updates(k, v, d, vb):
            return abi.make_updates(
                np.ascontiguousarray(k, np.int64),
                np.ascontiguousarray(v, np.uint8).reshape(-1) if vb else None,
                np.full(len(k), t, np.uint64),
                np.ascontiguousarray(d, np.int64), t, t + 1)

        def seal(out_cols, vb):
            k, v, tm, d = out_cols
            return abi.make_updates(k, v, tm, d, t, t + 1)

        lp_u = updates(lp_k, lp_v, lp_d, 16)
        p_u = updates(p_k, p_v, p_d, 16)
        times = np.full(len(lp_k), t, np.uint64)
        lp_cols = (np.ascontiguousarray(lp_k, np.int64),
                   np.ascontiguousarray(lp_v, np.uint8).reshape(-1),
                   times, np.ascontiguousarray(lp_d, np.int64))
        # --- join1 through its LinearJoinPlan: source = lineitem delta
        # (installed first), part's delta drained as the lookup side
        ctx.arr_insert(self.arr_l0, lp_u)
        l1_cols = self.j1op.step(t, lp_cols, {"part": (p_u, False)})
        l1_u = seal(l1_cols, 16)
        # --- distinct of l1's partkeys
        dk, dv, dt_, dd = l1_cols
        dist_u = abi.make_updates(dk, None, dt_, dd, t, t + 1)
        dcorr = ctx.reduce_push(self.distinct, dist_u)
        dcorr_u = abi.make_updates(*dcorr, t, t + 1)
        # --- join2 through its plan: source = distinct corrections,
        # lineitem's delta drained as the lookup side (already installed
        # by join1's phase; the op pairs probes for exactly-once)
        ctx.arr_insert(self.arr_dist, dcorr_u)
        q_cols = self.j2op.step(t, dcorr, {"lineitem_by_pk": (lp_u, True)})
        # --- per-partkey sum(quantity)/count
        q_u = abi.make_updates(*q_cols, t, t + 1)
        acorr = ctx.reduce_push(self.avg, q_u)
        acorr_u = abi.make_updates(*acorr, t, t + 1)
        # --- join3 through its plan: source = l1 delta, avg corrections
        # drained as the lookup side
        ctx.arr_insert(self.arr_l1, l1_u)
        e_cols = self.j3op.step(t, l1_cols, {"avg": (acorr_u, False)})
        e_cols = self._route_total(list(e_cols))
        e_u = abi.make_updates(*e_cols, t, t + 1)
        tcorr = ctx.reduce_push(self.total, e_u)
        self._apply_total(tcorr)
        return len(lp_k) + len(p_k)

    def _route_total(self, cols):
        """Hook: the global-SUM input is keyed by a constant, so the
        sharded variant exchanges it to the key's owner rank (timely
        Exchange for a keyed-by-unit reduce); unsharded = identity."""
        return cols

    def load(self, gen):
        lp_k, lp_v = gen.lineitem_bypart_updates()
        p_k, p_v = gen.part_updates()
        ones = lambda n: np.ones(n, np.int64)
        return self._push(0, (lp_k, lp_v, ones(len(lp_k))),
                          (p_k, p_v, ones(len(p_k))))

    def step(self, churn, t):
        lp = churn["lineitem_by_part"]
        return self._push(t, lp, None)

    def step_dev(self, lp_u, t):
        """Bench path: one churn step whose lineitem-by-partkey update
        columns are ALREADY staged (device Updates at time t; the part
        input is idle in the churn workload). Same drain order as
        `_push`, with interior streams staying device-resident: the two
        halves of each join's delta push separately (exactly-once per
        DESIGN.md §5 holds — both probe the not-yet-updated other side),
        and two-stream reduce inputs go through reduce_push2."""
        ctx = self.ctx
        # --- join1 drain: side1 (lineitem) first; part idle this step
        ctx.arr_insert(self.arr_l0, lp_u)
        l1a = ctx.join_push_dev(self.j1, 1, lp_u)   # probes part
        # --- join2 drain: side2 (lineitem) BEFORE the distinct delta
        j2b = ctx.join_push_dev(self.j2, 2, lp_u)   # probes dist (old)
        dcorr = ctx.reduce_push_dev(self.distinct, l1a.updates(t, t + 1))
        dcorr_u = dcorr.updates(t, t + 1)
        self.ctx.arr_insert(self.arr_dist, dcorr_u)
        j2a = ctx.join_push_dev(self.j2, 1, dcorr_u)  # probes l0 (new)
        # --- per-partkey sum(quantity)/count over both join2 halves
        acorr = ctx.reduce_push2_dev(self.avg, j2a.updates(t, t + 1),
                                     j2b.updates(t, t + 1))
        acorr_u = acorr.updates(t, t + 1)
        # --- join3 drain: side1 (l1 delta) BEFORE the avg corrections
        l1_u = l1a.updates(t, t + 1)
        ctx.arr_insert(self.arr_l1, l1_u)
        j3a = ctx.join_push_dev(self.j3, 1, l1_u)     # probes avg (old)
        ctx.arr_insert(self.arr_avg, acorr_u)
        j3b = ctx.join_push_dev(self.j3, 2, acorr_u)  # probes l1 (new)
        tcorr = ctx.reduce_push2_dev(self.total, j3a.updates(t, t + 1),
                                     j3b.updates(t, t + 1))
        cols = tcorr.to_host()
        self._apply_total(cols)
        for o in (l1a, j2a, j2b, dcorr, acorr, j3a, j3b, tcorr):
            o.release()


class ShardedQ17Dataflow(Q17Dataflow):
    """Q17 sharded by l_partkey hash (config 5 at N GPUs): every join
    and the per-partkey AVG key on partkey, so the dataflow is
    shard-local end-to-end; only the final global SUM's input stream is
    exchanged to the constant key's owner rank (route_hash(0) % world).
    Weak scaling with one tiny all-to-all per step."""

    def __init__(self, ctx, exchange):
        from .dist import shard_of
        self.shard_of = shard_of
        self.exchange = exchange
        super().__init__(ctx)

    def _filter_shard(self, keys, vals, diffs):
        W, r = self.exchange.world, self.exchange.rank
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.uint8).reshape(len(keys), -1)
        diffs = np.ascontiguousarray(diffs, np.int64)
        if W == 1:
            return keys, vals, diffs
        m = self.shard_of(keys, 1, W) == r
        return keys[m], vals[m], diffs[m]

    def load(self, gen):
        lp_k, lp_v = gen.lineitem_bypart_updates()
        p_k, p_v = gen.part_updates()
        lp_k, lp_v, lp_d = self._filter_shard(lp_k, lp_v,
                                              np.ones(len(lp_k), np.int64))
        p_k, p_v, p_d = self._filter_shard(p_k, p_v,
                                           np.ones(len(p_k), np.int64))
        return self._push(0, (lp_k, lp_v, lp_d), (p_k, p_v, p_d))

    def step(self, churn, t):
        lp_k, lp_v, lp_d = churn["lineitem_by_part"]
        lp = self._filter_shard(lp_k, lp_v, lp_d)
        return self._push(t, lp, None)

    def _route_total(self, cols):
        k, v, tm, d = cols
        return self.exchange.exchange(
            np.ascontiguousarray(k, np.int64),
            np.ascontiguousarray(v, np.uint8),
            np.ascontiguousarray(tm, np.uint64),
            np.ascontiguousarray(d, np.int64), 1, 8)

    def step_dev(self, lp_u, t):
        """Device bench step: identical to the base until join3's
        outputs, whose (tiny) constant-key stream leaves the device for
        the owner-rank exchange."""
        ctx = self.ctx
        ctx.arr_insert(self.arr_l0, lp_u)
        l1a = ctx.join_push_dev(self.j1, 1, lp_u)
        j2b = ctx.join_push_dev(self.j2, 2, lp_u)
        dcorr = ctx.reduce_push_dev(self.distinct, l1a.updates(t, t + 1))
        dcorr_u = dcorr.updates(t, t + 1)
        ctx.arr_insert(self.arr_dist, dcorr_u)
        j2a = ctx.join_push_dev(self.j2, 1, dcorr_u)
        acorr = ctx.reduce_push2_dev(self.avg, j2a.updates(t, t + 1),
                                     j2b.updates(t, t + 1))
        acorr_u = acorr.updates(t, t + 1)
        l1_u = l1a.updates(t, t + 1)
        ctx.arr_insert(self.arr_l1, l1_u)
        j3a = ctx.join_push_dev(self.j3, 1, l1_u)
        ctx.arr_insert(self.arr_avg, acorr_u)
        j3b = ctx.join_push_dev(self.j3, 2, acorr_u)
        ka, va, ta, da = j3a.to_host()
        kb, vb_, tb, db = j3b.to_host()
        cols = [np.concatenate([ka, kb]), np.concatenate([va, vb_]),
                np.concatenate([ta, tb]), np.concatenate([da, db])]
        cols = self._route_total(cols)
        e_u = abi.make_updates(*cols, t, t + 1)
        tcorr = ctx.reduce_push(self.total, e_u)
        self._apply_total(tcorr)
        for o in (l1a, j2a, j2b, dcorr, acorr, j3a, j3b):
            o.release()
