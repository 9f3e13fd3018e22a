"""materialize_amd.render — mirror of the reference's rendering surface.

Names and argument meanings follow src/compute/src/render/join/
delta_join.rs (render_delta_join, DeltaPathPlan/DeltaStagePlan —
src/compute-types/src/plan/join/delta_join.rs:38-78) and
src/compute/src/render/reduce.rs (render_reduce). The plan types here are
the drop-in surface: a hand-written plan (see workloads.py) plays the role
of the optimizer-produced `Plan::Join { plan: DeltaJoinPlan }`.

The engine context (`materialize_amd._ffi.GpuCtx`) executes stages on the
GPU via the C ABI; intermediate stage outputs stay device-resident.
"""
import ctypes as C
from dataclasses import dataclass, field
from typing import Callable, Optional

import numpy as np

from . import _abi as abi


class DevOut:
    """An engine out-batch kept where the engine produced it (device for
    GpuCtx, host for OracleCtx), usable as the next stage's input."""

    def __init__(self, ctx, outp, host_cols=None, sorted=False):
        self.ctx = ctx
        self.outp = outp  # POINTER(OutBatch) or None (host)
        self.host_cols = host_cols
        # canonical (key,val,time)-sorted content (consolidated producers)
        self.sorted = sorted

    @property
    def n(self):
        if self.outp is not None:
            return int(self.outp.contents.n)
        return len(self.host_cols[2])

    @property
    def schema(self):
        if self.outp is not None:
            s = self.outp.contents.schema
            return s.key_words, s.val_bytes
        return self._host_schema

    def updates(self, lower, upper):
        if self.outp is not None:
            ob = self.outp.contents
            u = abi.Updates()
            u.keys = ob.keys
            u.vals = ob.vals
            u.times = ob.times
            u.diffs = ob.diffs
            u.n = ob.n
            u.lower = lower
            u.upper = upper
            u.on_device = 1
            u.sorted = 1 if self.sorted else 0
            u._ref = self
            return u
        k, v, t, d = self.host_cols
        return abi.make_updates(k, v, t, d, lower, upper)

    def to_host(self):
        if self.outp is not None:
            return self.ctx._take_copy(self.outp)
        return self.host_cols

    def errs_to_host(self):
        """Copy the out-batch's consolidated (code, time, diff) error
        stream to host without releasing the batch."""
        import ctypes as C

        import numpy as np
        en = int(self.outp.contents.err_n) if self.outp is not None else 0
        ecodes = np.empty(en, np.uint64)
        etimes = np.empty(en, np.uint64)
        ediffs = np.empty(en, np.int64)
        if en:
            self.ctx._check(self.ctx.lib.mz_gpu_out_err_to_host(
                self.ctx.ctx, self.outp,
                ecodes.ctypes.data_as(C.POINTER(C.c_uint64)),
                etimes.ctypes.data_as(C.POINTER(C.c_uint64)),
                ediffs.ctypes.data_as(C.POINTER(C.c_int64))))
        return ecodes, etimes, ediffs

    def release(self):
        if self.outp is not None:
            self.ctx.lib.mz_gpu_out_release(self.ctx.ctx, self.outp)
            self.outp = None


@dataclass
class DeltaStagePlan:
    """One lookup stage of a delta path (delta_join.rs DeltaStagePlan:70).

    `lookup_relation` names the arrangement probed; `le` is the time
    tie-break (source_relation < lookup_relation -> le, else lt —
    delta_join.rs:356-399); `closure` is the stage's JoinClosure."""
    lookup_relation: str
    le: bool
    closure: abi.Closure
    stream_val_bytes: int


@dataclass
class DeltaPathPlan:
    """One update path (delta_join.rs DeltaPathPlan:48): reacts to
    `source_relation`'s updates; `initial_prep` is the key preparation of
    the source stream (the initial_closure analog)."""
    source_relation: str
    initial_prep: Callable  # (keys, vals, diffs) -> (keys, vals, diffs)
    stages: list


@dataclass
class DeltaJoinPlan:
    paths: list


class DeltaJoinOp:
    """render_delta_join (delta_join.rs:51-251): one operator per path,
    outputs concatenated by the caller. With an exchange (world > 1), the
    stream is re-distributed by the next stage's key hash between stages —
    the Exchange pacts of linear_join.rs:390 / delta_join.rs:442-464, one
    all-to-all-v per stage per batch."""

    def __init__(self, ctx, arrangements, plan: DeltaJoinPlan,
                 exchange=None):
        self.ctx = ctx
        self.arrangements = arrangements  # name -> arr handle
        self.plan = plan
        self.exchange = exchange  # None or dist.TorchExchange

    def _exchange_out(self, cur, t, kw, vb):
        """Re-distribute a stage output by its (new) key hash. COLLECTIVE:
        every rank calls this for every stage, with however few local rows
        it has — a rank-dependent skip would desynchronize the
        all-to-all. Device outputs take the device-resident exchange
        (partition in HBM + column all_to_all; no host round trip on
        RCCL); host outputs and the oracle keep the numpy path."""
        if (isinstance(cur, DevOut)
                and hasattr(self.exchange, "exchange_dev")
                and hasattr(self.ctx, "partition_dev")):
            u = self.exchange.exchange_dev(self.ctx, cur.updates(t, t + 1),
                                           kw, vb, t)
            cur.release()
            return u
        keys, vals, times, diffs = cur.to_host()
        cur.release()
        keys, vals, times, diffs = self.exchange.exchange(
            keys, vals, times, diffs, kw, vb)
        return abi.make_updates(keys, vals, times, diffs, t, t + 1)

    def push_path(self, path: DeltaPathPlan, keys, vals, diffs, t,
                  final_exchange=False):
        """Run one path for a batch of source updates at time t.
        Returns a DevOut (or, post-exchange, a host-resident one), or None
        when nothing is produced. Source updates must already be sharded by
        the first stage's key (the caller filters them). When an exchange
        is attached, EVERY rank runs every stage and every exchange, even
        with zero local rows (collectives must match across ranks)."""
        keys, vals, diffs = path.initial_prep(keys, vals, diffs)
        n = len(keys)
        if n == 0 and self.exchange is None:
            return None
        times = np.full(n, t, np.uint64)
        u = abi.make_updates(np.ascontiguousarray(keys, np.int64),
                             vals, times,
                             np.ascontiguousarray(diffs, np.int64),
                             t, t + 1)
        return self.push_path_updates(path, u, t,
                                      final_exchange=final_exchange)

    def push_path_updates(self, path: DeltaPathPlan, u, t,
                          final_exchange=False):
        """As push_path, but the source stream arrives as a ready Updates
        descriptor (host or device — the bench pre-stages churn batches
        into HBM and passes device columns)."""
        ctx = self.ctx
        collective = self.exchange is not None
        # Fused two-stage path (k_probe_path2): one kernel runs both
        # lookups with the intermediate in registers. Only when no
        # inter-stage exchange is needed and the intermediate shape fits
        # the kernel's register buffers (C side validates too).
        if (not collective and len(path.stages) == 2
                and hasattr(ctx, "halfjoin2_dev")):
            s1, s2 = path.stages
            o1 = s1.closure.out
            arr2 = self.arrangements[s2.lookup_relation]
            if 0 < o1.key_words <= 2 and o1.val_bytes <= 48:
                cur = ctx.halfjoin2_dev(
                    self.arrangements[s1.lookup_relation], s1.le,
                    s1.closure, arr2, s2.le, s2.closure, u,
                    s1.stream_val_bytes)
                if cur.n == 0:
                    cur.release()
                    return None
                return cur
        cur = None
        for i, st in enumerate(path.stages):
            arr = self.arrangements[st.lookup_relation]
            out = ctx.halfjoin_dev(arr, u, st.stream_val_bytes, st.le,
                                   st.closure)
            if cur is not None:
                cur.release()
            cur = out
            more = i + 1 < len(path.stages)
            if collective and (more or final_exchange):
                okw = st.closure.out.key_words
                ovb = st.closure.out.val_bytes
                u = self._exchange_out(cur, t, okw, ovb)
                cur = HostOut(u, okw, ovb)
            elif more:
                u = cur.updates(t, t + 1)
        if cur.n == 0:
            cur.release()
            return None
        return cur


class HostOut:
    """A host-resident stage result (post-exchange) presenting the DevOut
    interface."""

    def __init__(self, upd, kw, vb):
        self._upd = upd
        self._schema = (kw, vb)

    @property
    def n(self):
        return int(self._upd.n)

    @property
    def schema(self):
        return self._schema

    def updates(self, lower, upper):
        self._upd.lower = lower
        self._upd.upper = upper
        return self._upd

    def to_host(self):
        import numpy as np
        u = self._upd
        kw, vb = self._schema
        n = int(u.n)
        keys = np.ctypeslib.as_array(u.keys, shape=(n * kw,)).view(np.int64)
        vals = (np.ctypeslib.as_array(u.vals, shape=(n * vb,))
                if vb else np.empty(0, np.uint8))
        times = np.ctypeslib.as_array(u.times, shape=(n,))
        diffs = np.ctypeslib.as_array(u.diffs, shape=(n,))
        return keys.copy(), vals.copy(), times.copy(), diffs.copy()

    def release(self):
        self._upd = None


@dataclass
class LinearStagePlan:
    """One stage of a linear join (compute-types/src/plan/join/
    linear_join.rs:50-65): probe `lookup_relation` with the accumulated
    stream; `stream_key` (when set) is the key-prep closure — the
    stream_key exprs + stream_thinning re-keying of linear_join.rs:349-385,
    evaluated via mz_gpu_map; `closure` is the stage's JoinClosure over
    (key, stream val, lookup val)."""
    lookup_relation: str
    closure: abi.Closure
    stream_key: Optional[abi.Closure] = None
    stream_key_words: int = 1
    stream_val_bytes: int = 0


@dataclass
class LinearJoinPlan:
    """compute-types/src/plan/join/linear_join.rs:27-42: a source
    relation, an optional initial closure, a sequence of stages, an
    optional final closure. `source_relation` names the arrangement that
    plays arranged1 of the FIRST stage's mz_join_core (the reference's
    source_key arrangement); interior stages arrange the accumulated
    stream in op-owned "JoinStage" arrangements (linear_join.rs:461)."""
    source_relation: str
    stage_plans: list
    initial_closure: Optional[abi.Closure] = None
    final_closure: Optional[abi.Closure] = None


class LinearJoinOp:
    """render_join / differential_join (linear_join.rs:204-543): executes
    a LinearJoinPlan over the engine. Each stage is the two-sided
    mz_join_core between the stream's arrangement and the lookup
    arrangement; `step` drives one timestamp's deltas through the
    drain-input-1-first discipline (mz_join_core.rs:237-368):
    the stream delta probes the lookup BEFORE the lookup's delta is
    installed, then the lookup delta probes the updated stream side —
    each concurrent pair counted exactly once."""

    def __init__(self, ctx, arrangements, plan: LinearJoinPlan):
        self.ctx = ctx
        self.arrangements = arrangements
        self.plan = plan
        self.joins = []
        self.stage_arrs = []  # op-owned interior stream arrangements
        stream_arr = arrangements[plan.source_relation]
        for i, st in enumerate(plan.stage_plans):
            lookup = arrangements[st.lookup_relation]
            j = ctx.join_create(stream_arr, lookup, st.closure)
            self.joins.append(j)
            if i + 1 < len(plan.stage_plans):
                nxt = plan.stage_plans[i + 1]
                stream_arr = ctx.arr_create(
                    abi.schema(nxt.stream_key_words, nxt.stream_val_bytes))
                self.stage_arrs.append(stream_arr)

    def _map(self, cl, cols, t, in_kw, in_vb):
        k, v, tm, d = cols
        u = abi.make_updates(k, v, tm, d, t, t + 1)
        return self.ctx.map(abi.schema(in_kw, in_vb), u, cl)

    def step(self, t, source_cols, lookup_deltas=None):
        """One timestamp: `source_cols` = (keys, vals, times, diffs) of
        the source relation's delta (the caller has ALREADY installed it
        in the source arrangement — it is shared state, like the
        reference's CollectionBundle arrangements); `lookup_deltas` maps
        lookup relation name -> (updates_desc, already_installed: bool).
        The op installs not-yet-installed lookup deltas at the correct
        drain point. Returns the final output columns (host)."""
        ctx = self.ctx
        lookup_deltas = lookup_deltas or {}
        plan = self.plan
        cur = source_cols
        if plan.initial_closure is not None:
            st0 = plan.stage_plans[0]
            cur = self._map(plan.initial_closure, cur, t,
                            st0.stream_key_words, st0.stream_val_bytes)
        for i, st in enumerate(plan.stage_plans):
            if st.stream_key is not None:
                cur = self._map(st.stream_key, cur, t, st.stream_key_words,
                                st.stream_val_bytes)
            u = abi.make_updates(cur[0], cur[1], cur[2], cur[3], t, t + 1)
            if i > 0:
                # interior stage: arrange the accumulated stream
                ctx.arr_insert(self.stage_arrs[i - 1], u)
            out1 = ctx.join_push(self.joins[i], 1, u)  # probes lookup OLD
            ld = lookup_deltas.get(st.lookup_relation)
            if ld is not None:
                lu, installed = ld
                if not installed:
                    ctx.arr_insert(self.arrangements[st.lookup_relation],
                                   lu)
                out2 = ctx.join_push(self.joins[i], 2, lu)  # probes NEW
                cur = tuple(np.concatenate([a, b])
                            for a, b in zip(out1, out2))
            else:
                cur = out1
        if plan.final_closure is not None:
            lst = plan.stage_plans[-1]
            cur = self._map(plan.final_closure, cur, t,
                            lst.closure.out.key_words,
                            lst.closure.out.val_bytes)
        return cur


def render_join(ctx, arrangements, plan: LinearJoinPlan) -> LinearJoinOp:
    """Context::render_join dispatch for Plan::Join{Linear}
    (render.rs:1321-1350)."""
    return LinearJoinOp(ctx, arrangements, plan)


@dataclass
class ReducePlan:
    """render_reduce / AccumulablePlan surface (reduce.rs:71,
    plan/reduce.rs:233)."""
    spec: abi.ReduceSpec


class ReduceOp:
    def __init__(self, ctx, plan: ReducePlan):
        self.ctx = ctx
        self.op = ctx.reduce_create(plan.spec)

    def push(self, updates) -> DevOut:
        return self.ctx.reduce_push_dev(self.op, updates)


class TopKPlan:
    """render_topk / BasicTopKPlan surface (top_k.rs:289-311,
    plan/top_k.rs): group key = the arrangement key; order columns over
    the record val bytes; literal offset/limit."""

    def __init__(self, spec):
        self.spec = spec


class TopKOp:
    """build_topk (top_k.rs:322-418): maintains per-group kept windows."""

    def __init__(self, ctx, plan: TopKPlan):
        self.ctx = ctx
        self.op = ctx.topk_create(plan.spec)

    def push(self, updates):
        return self.ctx.topk_push(self.op, updates)


def render_topk(ctx, plan) -> "TopKOp":
    return TopKOp(ctx, plan)


class ThresholdPlan:
    """render_threshold / BasicThresholdPlan surface
    (src/compute/src/render/threshold.rs:100-116): the ensure_arrangement
    key is the full record in our fixed-width model."""

    def __init__(self, schema):
        self.schema = schema


class ThresholdOp:
    """build_threshold_basic (threshold.rs:75-97): keeps rows with a
    positive accumulated count, at that count."""

    def __init__(self, ctx, plan: ThresholdPlan):
        self.ctx = ctx
        self.op = ctx.threshold_create(plan.schema)

    def push(self, updates) -> DevOut:
        return self.ctx.threshold_push_dev(self.op, updates)


def render_delta_join(ctx, arrangements, plan, exchange=None) -> DeltaJoinOp:
    return DeltaJoinOp(ctx, arrangements, plan, exchange=exchange)


def render_reduce(ctx, plan) -> ReduceOp:
    return ReduceOp(ctx, plan)


def render_threshold(ctx, plan) -> ThresholdOp:
    return ThresholdOp(ctx, plan)
