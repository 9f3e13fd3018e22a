"""Variable-length vals (schema.val_bytes == MZ_GPU_VARLEN): the
reference's byte-arena row layout (row-spine/src/lib.rs:110-135) with
GENUINELY varlen byte strings — embedded NULs and prefix pairs included,
which zero-padded fixed-width encodings cannot represent. CPU tests pin
the python varlen oracle against hand-computed orderings; GPU tests hold
the engine (consolidate, arrangement spine incl. merges, halfjoin/join
probes) bit-exact to that oracle."""
import numpy as np
import pytest

from materialize_amd import _abi as abi
from pyoracle_vl import VlOracle, consolidate_rows, rows_to_cols

VARLEN = abi.MZ_GPU_VARLEN


def _mk_updates(keys, vals_list, times, diffs, lower, upper, sorted=0):
    offs = np.zeros(len(vals_list) + 1, np.uint32)
    arena = bytearray()
    for i, v in enumerate(vals_list):
        offs[i] = len(arena)
        arena.extend(v)
    offs[len(vals_list)] = len(arena)
    arena_np = (np.frombuffer(bytes(arena), np.uint8).copy()
                if arena else np.empty(0, np.uint8))
    return abi.make_updates(
        np.asarray(keys, np.int64), arena_np,
        np.asarray(times, np.uint64), np.asarray(diffs, np.int64),
        lower, upper, sorted=sorted, val_offs=offs), (arena_np, offs)


def _rand_val(rng, maxlen=40):
    n = int(rng.integers(0, maxlen))
    return bytes(rng.integers(0, 256, n, dtype=np.uint8))


def test_vl_oracle_ordering_hand_cases():
    """Canonical varlen order: lexicographic bytes, shorter-prefix-first
    — incl. embedded NULs ('a' < 'a\\0' < 'a\\0b' < 'ab')."""
    rows = [((1,), b"ab", 0, 1), ((1,), b"a", 0, 1),
            ((1,), b"a\x00b", 0, 1), ((1,), b"a\x00", 0, 1),
            ((0,), b"zzz", 0, 1), ((1,), b"", 0, 1)]
    out = consolidate_rows(rows)
    vals = [v for (k, v, t, d) in out if k == (1,)]
    assert vals == [b"", b"a", b"a\x00", b"a\x00b", b"ab"]
    assert out[0][0] == (0,)
    # cancellation
    out2 = consolidate_rows(rows + [(k, v, t, -d) for (k, v, t, d) in rows])
    assert out2 == []


def _gpu_consolidate(g, kw, keys, vals_list, times, diffs):
    u, _keep = _mk_updates(keys, vals_list, times, diffs, 0,
                           int(max(times)) + 1 if len(times) else 1)
    k, arena, t, d = g.consolidate(abi.schema(kw, VARLEN), u)
    offs = g.last_voffs
    return k, arena, offs, t, d


pytestmark_gpu = pytest.mark.gpu


@pytest.mark.gpu
def test_vl_consolidate_gpu_matches_oracle():
    from materialize_amd._ffi import GpuCtx
    g = GpuCtx()
    o = VlOracle()
    rng = np.random.default_rng(41)
    n = 3000
    keys = rng.integers(-50, 50, n).astype(np.int64)
    vals = [_rand_val(rng) for _ in range(n)]
    # force collisions: repeat some (key, val) pairs with mixed signs
    for i in range(0, n, 3):
        keys[i] = keys[(i + 1) % n]
        vals[i] = vals[(i + 1) % n]
    times = rng.integers(0, 3, n).astype(np.uint64)
    diffs = rng.choice([-1, 1, 1, 2], n).astype(np.int64)
    gk, garena, goffs, gt, gd = _gpu_consolidate(g, 1, keys, vals, times,
                                                 diffs)
    u, (arena_np, offs) = _mk_updates(keys, vals, times, diffs, 0, 3)
    ok, oarena, ooffs, ot, od = o.consolidate(
        np.asarray(keys, np.int64).view(np.uint64), 1, arena_np, offs,
        times, diffs)
    np.testing.assert_array_equal(np.asarray(gk), np.asarray(ok))
    np.testing.assert_array_equal(np.asarray(goffs), np.asarray(ooffs))
    np.testing.assert_array_equal(np.asarray(garena), np.asarray(oarena))
    np.testing.assert_array_equal(np.asarray(gt), np.asarray(ot))
    np.testing.assert_array_equal(np.asarray(gd), np.asarray(od))
    g.close()


@pytest.mark.gpu
def test_vl_arrangement_and_halfjoin_gpu_matches_oracle():
    """Spine inserts across many timestamps (forcing merge_range_vl via
    the >10-batch hard cap), logical compaction, then le and lt probes
    with a passthrough closure — engine vs varlen oracle bit-exact."""
    from materialize_amd._ffi import GpuCtx
    g = GpuCtx()
    o = VlOracle()
    rng = np.random.default_rng(43)
    garr = g.arr_create(abi.schema(1, VARLEN))
    oarr = o.arr_create(1)
    for t in range(14):  # > 10 batches -> varlen spine merges
        n = 250
        keys = rng.integers(0, 60, n).astype(np.int64)
        vals = [_rand_val(rng, 24) for _ in range(n)]
        diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
        times = np.full(n, t, np.uint64)
        u, (arena_np, offs) = _mk_updates(keys, vals, times, diffs, t,
                                          t + 1)
        g.arr_insert(garr, u)
        o.arr_insert(oarr, np.asarray(keys, np.int64).view(np.uint64), 1,
                     arena_np, offs, times, diffs)
    g.arr_set_logical_compaction(garr, 5)
    o.set_logical_compaction(oarr, 5)
    cl = abi.closure(
        [], [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 0)],  # whole-val passthrough
        abi.Schema(key_words=1, val_bytes=VARLEN))
    for le in (True, False):
        m = 500
        pk = rng.integers(0, 60, m).astype(np.int64)
        pt = np.full(m, 9, np.uint64)
        pd = rng.choice([-1, 1], m).astype(np.int64)
        pu = abi.make_updates(pk, None, pt, pd, 9, 10)
        gk, garena, gt, gd = g.halfjoin(garr, pu, 0, le, cl)
        goffs = g.last_voffs
        ok, oarena, ooffs, ot, od = o.halfjoin(
            oarr, pk.view(np.uint64), 1, pt, pd, le)
        np.testing.assert_array_equal(np.asarray(gk), np.asarray(ok),
                                      err_msg=f"le={le} keys")
        np.testing.assert_array_equal(np.asarray(goffs), np.asarray(ooffs),
                                      err_msg=f"le={le} offs")
        np.testing.assert_array_equal(np.asarray(garena),
                                      np.asarray(oarena),
                                      err_msg=f"le={le} arena")
        np.testing.assert_array_equal(np.asarray(gt), np.asarray(ot))
        np.testing.assert_array_equal(np.asarray(gd), np.asarray(od))
    g.close()


@pytest.mark.gpu
def test_vl_closure_validation_errors():
    """Closures touching the varlen side beyond the passthrough fail
    loudly instead of computing garbage."""
    from materialize_amd._ffi import GpuCtx, MzGpuError
    g = GpuCtx()
    arr = g.arr_create(abi.schema(1, VARLEN))
    u, _ = _mk_updates([1], [b"xy"], [0], [1], 0, 1)
    g.arr_insert(arr, u)
    bad = abi.closure(
        [abi.filt(abi.MZ_SRC_VAL_LOOKUP, 0, 8, abi.MZ_CMP_EQ, 0)],
        [abi.field(abi.MZ_SRC_KEY, 0, 8)],
        [abi.field(abi.MZ_SRC_VAL_LOOKUP, 0, 0)],
        abi.Schema(key_words=1, val_bytes=VARLEN))
    pu = abi.make_updates(np.array([1], np.int64), None,
                          np.ones(1, np.uint64), np.ones(1, np.int64), 1, 2)
    with pytest.raises(MzGpuError):
        g.halfjoin(arr, pu, 0, True, bad)
    g.close()
