"""Pin the virgin-context first-op scenario: a freshly created engine
context's FIRST operation being a reduce push must already be correct
(smoke() runs exactly this shape; the structured suites always warmed
the context with other ops first and missed it).

Round-2 root cause: hipMemsetAsync fills into freshly-carved
hipMallocAsync blocks were silently LOST on a process's first
operations — the reduce hash table kept its pre-memset zeros, so a
zeroed slot read as "key 0 at row 0" and distinct keys aliased onto one
accumulator row. Semantic state init now uses compute-kernel fills
(mzgpu.hip fill_u64/fill_u32/fill_u8); this test pins that path with no
warmup crutch."""
import numpy as np
import pytest

from materialize_amd import _abi as abi

pytestmark = pytest.mark.gpu


def test_first_op_reduce_parity():
    from materialize_amd._ffi import GpuCtx
    from pyoracle import OracleCtx
    g, o = GpuCtx(), OracleCtx()
    aggs = [abi.Aggregate(func=abi.MZ_AGG_COUNT, off=0, width=8,
                          is_float=0, nullable=0),
            abi.Aggregate(func=abi.MZ_AGG_SUM_I64, off=0, width=8,
                          is_float=0, nullable=0)]
    spec = abi.reduce_spec(aggs, abi.schema(1, 16))
    gop, oop = g.reduce_create(spec), o.reduce_create(spec)
    rng = np.random.default_rng(5)
    for push in range(3):
        n = 200
        keys = rng.integers(0, 40, n).astype(np.int64)
        vals = np.zeros((n, 16), np.uint8)
        vals[:, :8] = rng.integers(0, 100, n).astype(np.int64) \
            .reshape(-1, 1).view(np.uint8).reshape(n, 8)
        diffs = rng.choice([-1, 1, 1], n).astype(np.int64)
        u = abi.make_updates(keys, vals, np.zeros(n, np.uint64), diffs,
                             0, 1)
        a, b = g.reduce_push(gop, u), o.reduce_push(oop, u)
        for x, y in zip(a, b):
            np.testing.assert_array_equal(x.view(np.uint8),
                                          y.view(np.uint8),
                                          err_msg=f"push {push}")
    g.close()
    o.close()
