"""Varlen fault isolation: staged scenarios from trivial to the failing
test shape; prints engine-vs-oracle row diffs at the first divergence."""
import os
import sys

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))
import numpy as np

from materialize_amd import _abi as abi
from materialize_amd._ffi import GpuCtx
from pyoracle_vl import VlOracle

VARLEN = abi.MZ_GPU_VARLEN


def mk(keys, vals_list, times, diffs, lo, hi):
    offs = np.zeros(len(vals_list) + 1, np.uint32)
    arena = bytearray()
    for i, v in enumerate(vals_list):
        offs[i] = len(arena)
        arena.extend(v)
    offs[len(vals_list)] = len(arena)
    arena_np = (np.frombuffer(bytes(arena), np.uint8).copy()
                if arena else np.empty(0, np.uint8))
    u = abi.make_updates(np.asarray(keys, np.int64), arena_np,
                         np.asarray(times, np.uint64),
                         np.asarray(diffs, np.int64), lo, hi,
                         val_offs=offs)
    return u, arena_np, offs


def rows_of(k, arena, offs, t, d):
    return [(int(k[i]), bytes(arena[offs[i]:offs[i + 1]]), int(t[i]),
             int(d[i])) for i in range(len(t))]


def scenario(name, nts, nrows, compaction, probe_t, le, maxlen, seed):
    g = GpuCtx()
    o = VlOracle()
    rng = np.random.default_rng(seed)
    ga = g.arr_create(abi.schema(1, VARLEN))
    oa = o.arr_create(1)
    for t in range(nts):
        keys = rng.integers(0, 20, nrows).astype(np.int64)
        vals = [bytes(rng.integers(0, 256, int(rng.integers(0, maxlen)),
                                   dtype=np.uint8)) for _ in range(nrows)]
        diffs = rng.choice([-1, 1, 1], nrows).astype(np.int64)
        times = np.full(nrows, t, np.uint64)
        u, arena, offs = mk(keys, vals, times, diffs, t, t + 1)
        g.arr_insert(ga, u)
        o.arr_insert(oa, keys.view(np.uint64), 1, arena, offs, times,
                     diffs)
    if compaction is not None:
        g.arr_set_logical_compaction(ga, compaction)
        o.set_logical_compaction(oa, compaction)
    cl = abi.closure(
        [], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 0)],
        abi.Schema(key_words=1, val_bytes=VARLEN))
    m = 60
    pk = rng.integers(0, 20, m).astype(np.int64)
    pt = np.full(m, probe_t, np.uint64)
    pd = rng.choice([-1, 1], m).astype(np.int64)
    pu = abi.make_updates(pk, None, pt, pd, probe_t, probe_t + 1)
    gk, garena, gt, gd = g.halfjoin(ga, pu, 0, le, cl)
    goffs = g.last_voffs
    ok, oarena, ooffs, ot, od = o.halfjoin(oa, pk.view(np.uint64), 1, pt,
                                           pd, le)
    gr = rows_of(gk, garena, goffs, gt, gd)
    orr = rows_of(ok, oarena, ooffs, ot, od)
    if gr == orr:
        print(f"{name}: OK ({len(gr)} rows)", flush=True)
        g.close()
        return True
    print(f"{name}: MISMATCH g={len(gr)} o={len(orr)}", flush=True)
    sg, so = set(gr), set(orr)
    for r in list(sorted(sg - so))[:5]:
        print("  only-gpu", r, flush=True)
    for r in list(sorted(so - sg))[:5]:
        print("  only-oracle", r, flush=True)
    g.close()
    return False


scenario("s1-onebatch-le", 1, 40, None, 2, True, 10, 1)
scenario("s2-threebatch", 3, 40, None, 5, True, 10, 2)
scenario("s3-lt", 3, 40, None, 2, False, 10, 3)
scenario("s4-manybatch-merge", 14, 60, None, 15, True, 10, 4)
scenario("s5-compaction", 6, 40, 3, 8, True, 10, 5)
scenario("s6-longvals", 3, 40, None, 5, True, 40, 6)
scenario("s7-fulltest-shape", 14, 250, 5, 9, True, 24, 43)
