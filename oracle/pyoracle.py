"""oracle/pyoracle.py — ctypes wrapper over liboracle.so.

*** TEST INFRASTRUCTURE ONLY *** — only tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg may import this module. The product package
(materialize_amd/) never does.
"""
import ctypes as C
import os
import subprocess

from materialize_amd._abi import (  # type defs of the shared boundary
    Closure, OutBatch, ReduceSpec, Schema, TopKSpec, Updates, out_to_numpy,
)

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB = None


def _load():
    global _LIB
    if _LIB is not None:
        return _LIB
    path = os.path.join(_HERE, "liboracle.so")
    if not os.path.exists(path):
        subprocess.run(["make", "-C", _HERE], check=True,
                       capture_output=True)
    lib = C.CDLL(path)
    lib.orc_init.restype = C.c_void_p
    lib.orc_fini.argtypes = [C.c_void_p]
    lib.orc_last_error.restype = C.c_char_p
    lib.orc_last_error.argtypes = [C.c_void_p]
    lib.orc_arr_create.restype = C.c_void_p
    lib.orc_arr_create.argtypes = [C.c_void_p, C.POINTER(Schema)]
    lib.orc_arr_push_batch.argtypes = [C.c_void_p, C.c_void_p,
                                       C.POINTER(Updates)]
    lib.orc_arr_insert.argtypes = [C.c_void_p, C.c_void_p,
                                   C.POINTER(Updates)]
    lib.orc_arr_set_logical_compaction.argtypes = [C.c_void_p, C.c_void_p,
                                                   C.c_uint64]
    lib.orc_arr_maintain.argtypes = [C.c_void_p, C.c_void_p, C.c_uint64]
    lib.orc_arr_stats.argtypes = [C.c_void_p, C.c_void_p] + \
        [C.POINTER(C.c_uint64)] * 3
    lib.orc_join_create.restype = C.c_void_p
    lib.orc_join_create.argtypes = [C.c_void_p, C.c_void_p, C.c_void_p,
                                    C.POINTER(Closure)]
    lib.orc_join_push.argtypes = [C.c_void_p, C.c_void_p, C.c_int,
                                  C.POINTER(Updates),
                                  C.POINTER(C.POINTER(OutBatch))]
    lib.orc_halfjoin.argtypes = [C.c_void_p, C.c_void_p, C.POINTER(Updates),
                                 C.c_uint32, C.c_int, C.POINTER(Closure),
                                 C.POINTER(C.POINTER(OutBatch))]
    lib.orc_reduce_create.restype = C.c_void_p
    lib.orc_reduce_create.argtypes = [C.c_void_p, C.POINTER(ReduceSpec)]
    lib.orc_reduce_push.argtypes = [C.c_void_p, C.c_void_p,
                                    C.POINTER(Updates),
                                    C.POINTER(C.POINTER(OutBatch))]
    lib.orc_threshold_create.restype = C.c_void_p
    lib.orc_threshold_create.argtypes = [C.c_void_p, C.POINTER(Schema)]
    lib.orc_threshold_push.argtypes = [C.c_void_p, C.c_void_p,
                                       C.POINTER(Updates),
                                       C.POINTER(C.POINTER(OutBatch))]
    lib.orc_topk_create.restype = C.c_void_p
    lib.orc_topk_create.argtypes = [C.c_void_p, C.POINTER(TopKSpec)]
    lib.orc_topk_push.argtypes = [C.c_void_p, C.c_void_p,
                                  C.POINTER(Updates),
                                  C.POINTER(C.POINTER(OutBatch))]
    lib.orc_consolidate.argtypes = [C.c_void_p, C.POINTER(Schema),
                                    C.POINTER(Updates),
                                    C.POINTER(C.POINTER(OutBatch))]
    lib.orc_out_release.argtypes = [C.c_void_p, C.POINTER(OutBatch)]
    lib.orc_map.argtypes = [C.c_void_p, C.POINTER(Schema),
                            C.POINTER(Updates), C.POINTER(Closure),
                            C.POINTER(C.POINTER(OutBatch))]
    lib.orc_minmax_create.restype = C.c_void_p
    lib.orc_minmax_create.argtypes = [C.c_void_p, C.POINTER(Schema),
                                      C.c_int, C.POINTER(C.c_uint32),
                                      C.c_uint32]
    lib.orc_minmax_push.argtypes = [C.c_void_p, C.c_void_p,
                                    C.POINTER(Updates),
                                    C.POINTER(C.POINTER(OutBatch))]
    lib.orc_peek.argtypes = [C.c_void_p, C.c_void_p,
                             C.POINTER(C.c_uint64), C.c_uint64, C.c_uint64,
                             C.POINTER(C.POINTER(OutBatch))]
    lib.orc_route_hash.restype = C.c_uint64
    lib.orc_route_hash.argtypes = [C.POINTER(C.c_uint64), C.c_uint32]
    _LIB = lib
    return lib


class OracleCtx:
    def __init__(self):
        self.lib = _load()
        self.ctx = self.lib.orc_init()

    def close(self):
        if self.ctx:
            self.lib.orc_fini(self.ctx)
            self.ctx = None

    def __del__(self):
        self.close()

    def _take(self, outp):
        import numpy as np
        ob = outp.contents
        res = out_to_numpy(ob)
        en = ob.err_n
        ecodes = np.empty(en, np.uint64)
        etimes = np.empty(en, np.uint64)
        ediffs = np.empty(en, np.int64)
        for i in range(en):  # host pointers: direct reads
            ecodes[i] = ob.err_codes[i]
            etimes[i] = ob.err_times[i]
            ediffs[i] = ob.err_diffs[i]
        self.last_errs = (ecodes, etimes, ediffs)
        self.lib.orc_out_release(self.ctx, outp)
        return res

    def arr_create(self, sch):
        return self.lib.orc_arr_create(self.ctx, C.byref(sch))

    def arr_push(self, arr, upd):
        rc = self.lib.orc_arr_push_batch(self.ctx, arr, C.byref(upd))
        assert rc == 0

    def arr_insert_async(self, arr, upd):
        # the oracle has no device lanes: async == sync
        return self.arr_insert(arr, upd)

    def arr_flush(self, arr):
        return 0

    def arr_insert(self, arr, upd):
        rc = self.lib.orc_arr_insert(self.ctx, arr, C.byref(upd))
        assert rc == 0
    def arr_set_logical_compaction(self, arr, frontier):
        self.lib.orc_arr_set_logical_compaction(self.ctx, arr, frontier)

    def arr_maintain(self, arr, fuel=0):
        self.lib.orc_arr_maintain(self.ctx, arr, fuel)

    def arr_stats(self, arr):
        nb, nu, by = C.c_uint64(), C.c_uint64(), C.c_uint64()
        self.lib.orc_arr_stats(self.ctx, arr, C.byref(nb), C.byref(nu),
                               C.byref(by))
        return nb.value, nu.value, by.value

    def join_create(self, a1, a2, cl):
        return self.lib.orc_join_create(self.ctx, a1, a2, C.byref(cl))

    def join_push(self, op, side, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_join_push(self.ctx, op, side, C.byref(upd),
                                    C.byref(outp))
        assert rc == 0
        return self._take(outp)

    def halfjoin(self, lookup, upd, stream_vb, le, cl):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_halfjoin(self.ctx, lookup, C.byref(upd), stream_vb,
                                   1 if le else 0, C.byref(cl),
                                   C.byref(outp))
        assert rc == 0
        return self._take(outp)

    _red_specs = None

    def reduce_create(self, spec):
        op = self.lib.orc_reduce_create(self.ctx, C.byref(spec))
        if self._red_specs is None:
            self._red_specs = {}
        self._red_specs[op] = (spec.in_.key_words, spec.in_.val_bytes)
        return op

    def reduce_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_reduce_push(self.ctx, op, C.byref(upd),
                                      C.byref(outp))
        assert rc == 0
        return self._take(outp)

    def topk_create(self, spec):
        return self.lib.orc_topk_create(self.ctx, C.byref(spec))

    def topk_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_topk_push(self.ctx, op, C.byref(upd),
                                    C.byref(outp))
        if rc != 0:
            raise RuntimeError(self.lib.orc_last_error(self.ctx).decode())
        return self._take(outp)

    def threshold_create(self, sch):
        return self.lib.orc_threshold_create(self.ctx, C.byref(sch))

    def threshold_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_threshold_push(self.ctx, op, C.byref(upd),
                                         C.byref(outp))
        assert rc == 0
        return self._take(outp)

    def consolidate(self, sch, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_consolidate(self.ctx, C.byref(sch), C.byref(upd),
                                      C.byref(outp))
        assert rc == 0
        return self._take(outp)

    # --- DevOut-style variants mirroring GpuCtx (host-resident here) ---
    def _host_out(self, res, kw, vb):
        from materialize_amd.render import DevOut
        o = DevOut(self, None, host_cols=res)
        o._host_schema = (kw, vb)
        return o

    def consolidate_dev(self, sch, upd):
        res = self.consolidate(sch, upd)
        return self._host_out(res, sch.key_words, sch.val_bytes)

    def halfjoin_dev(self, lookup, upd, stream_vb, le, cl):
        res = self.halfjoin(lookup, upd, stream_vb, le, cl)
        return self._host_out(res, cl.out.key_words, cl.out.val_bytes)

    def reduce_push_dev(self, op, upd):
        res = self.reduce_push(op, upd)
        return self._host_out(res, 0, 0)

    def reduce_push2_dev(self, op, u1, u2):
        import numpy as np
        from materialize_amd import _abi as abi

        def cols(u, kw, vb):
            n = int(u.n)
            k = np.ctypeslib.as_array(u.keys, shape=(n * kw,)) if n else \
                np.empty(0, np.uint64)
            v = np.ctypeslib.as_array(u.vals, shape=(n * vb,)) if n and vb \
                else np.empty(0, np.uint8)
            t = np.ctypeslib.as_array(u.times, shape=(n,)) if n else \
                np.empty(0, np.uint64)
            d = np.ctypeslib.as_array(u.diffs, shape=(n,)) if n else \
                np.empty(0, np.int64)
            return k, v, t, d
        # concat on host (schema from the op's spec is not exposed here;
        # infer strides from the updates' n and array shapes via the spec
        # stored at create time)
        kw, vb = self._red_specs[op]
        a = cols(u1, kw, vb)
        b = cols(u2, kw, vb)
        cat = [np.concatenate([x, y]) for x, y in zip(a, b)]
        u = abi.make_updates(cat[0].view(np.int64), cat[1], cat[2], cat[3],
                             min(u1.lower, u2.lower),
                             max(u1.upper, u2.upper))
        return self.reduce_push_dev(op, u)

    def route_hash(self, words):
        arr = (C.c_uint64 * len(words))(*[w & 0xFFFFFFFFFFFFFFFF
                                          for w in words])
        return self.lib.orc_route_hash(arr, len(words))

    def map(self, in_schema, upd, cl):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_map(self.ctx, C.byref(in_schema), C.byref(upd),
                              C.byref(cl), C.byref(outp))
        assert rc == 0
        return self._take(outp)

    def minmax_create(self, in_schema, is_max, buckets):
        arr = (C.c_uint32 * len(buckets))(*buckets)
        return self.lib.orc_minmax_create(self.ctx, C.byref(in_schema),
                                          1 if is_max else 0, arr,
                                          len(buckets))

    def minmax_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_minmax_push(self.ctx, op, C.byref(upd),
                                      C.byref(outp))
        assert rc == 0
        return self._take(outp)

    def peek(self, arr, keys, time, kw=1):
        import numpy as np
        keys = np.ascontiguousarray(keys, np.int64).ravel()
        outp = C.POINTER(OutBatch)()
        rc = self.lib.orc_peek(
            self.ctx, arr,
            keys.view(np.uint64).ctypes.data_as(C.POINTER(C.c_uint64)),
            len(keys) // kw, time, C.byref(outp))
        assert rc == 0
        return self._take(outp)
