"""materialize_amd._ffi — ctypes bindings to libmzgpu.so (the product).

Loads the in-tree HIP library. On a machine with a GPU this module FAILS
LOUDLY if the extension is missing or fails to initialize — there is no
CPU fallback on the product path.
"""
import ctypes as C
import os
import subprocess

from ._abi import (Closure, OutBatch, ReduceSpec, Schema, TopKSpec, Updates,
                   out_to_numpy)

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "csrc", "libmzgpu.so")
_LIB = None


class MzGpuError(RuntimeError):
    pass


def _torch_has_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


def load():
    """Load and bind libmzgpu.so. Raises MzGpuError if unavailable."""
    global _LIB
    if _LIB is not None:
        return _LIB
    if not os.path.exists(_LIB_PATH):
        # attempt an in-tree build (hipcc cross-compiles without a GPU)
        try:
            subprocess.run(["make", "-C", os.path.dirname(_LIB_PATH)],
                           check=True, capture_output=True)
        except Exception as e:
            raise MzGpuError(
                f"libmzgpu.so missing at {_LIB_PATH} and in-tree build "
                f"failed: {e}. The HIP extension is required — there is no "
                f"CPU fallback.") from e
    lib = C.CDLL(_LIB_PATH)
    lib.mz_gpu_init.restype = C.c_void_p
    lib.mz_gpu_init.argtypes = [C.c_void_p]
    lib.mz_gpu_fini.argtypes = [C.c_void_p]
    lib.mz_gpu_last_error.restype = C.c_char_p
    lib.mz_gpu_last_error.argtypes = [C.c_void_p]
    lib.mz_gpu_sync.argtypes = [C.c_void_p]
    lib.mz_gpu_arr_create.restype = C.c_void_p
    lib.mz_gpu_arr_create.argtypes = [C.c_void_p, C.POINTER(Schema)]
    lib.mz_gpu_arr_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_arr_push_batch.argtypes = [C.c_void_p, C.c_void_p,
                                          C.POINTER(Updates)]
    lib.mz_gpu_arr_insert.argtypes = [C.c_void_p, C.c_void_p,
                                      C.POINTER(Updates)]
    lib.mz_gpu_arr_set_logical_compaction.argtypes = [C.c_void_p, C.c_void_p,
                                                      C.c_uint64]
    lib.mz_gpu_arr_maintain.argtypes = [C.c_void_p, C.c_void_p, C.c_uint64]
    lib.mz_gpu_arr_stats.argtypes = [C.c_void_p, C.c_void_p] + \
        [C.POINTER(C.c_uint64)] * 3
    lib.mz_gpu_out_release.argtypes = [C.c_void_p, C.POINTER(OutBatch)]
    lib.mz_gpu_out_to_host.argtypes = [
        C.c_void_p, C.POINTER(OutBatch), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.POINTER(C.c_int64)]
    lib.mz_gpu_out_err_to_host.argtypes = [
        C.c_void_p, C.POINTER(OutBatch), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint64), C.POINTER(C.c_int64)]
    lib.mz_gpu_out_voffs_to_host.argtypes = [
        C.c_void_p, C.POINTER(OutBatch), C.POINTER(C.c_uint32)]
    lib.mz_gpu_consolidate.argtypes = [C.c_void_p, C.POINTER(Schema),
                                       C.POINTER(Updates),
                                       C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_join_create.restype = C.c_void_p
    lib.mz_gpu_join_create.argtypes = [C.c_void_p, C.c_void_p, C.c_void_p,
                                       C.POINTER(Closure)]
    lib.mz_gpu_join_push.argtypes = [C.c_void_p, C.c_void_p, C.c_int,
                                     C.POINTER(Updates),
                                     C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_halfjoin.argtypes = [C.c_void_p, C.c_void_p,
                                    C.POINTER(Updates), C.c_uint32, C.c_int,
                                    C.POINTER(Closure),
                                    C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_halfjoin_raw.argtypes = lib.mz_gpu_halfjoin.argtypes
    lib.mz_gpu_halfjoin2.argtypes = [C.c_void_p, C.c_void_p, C.c_int,
                                     C.POINTER(Closure), C.c_void_p,
                                     C.c_int, C.POINTER(Closure),
                                     C.POINTER(Updates), C.c_uint32,
                                     C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_reduce_create.restype = C.c_void_p
    lib.mz_gpu_reduce_create.argtypes = [C.c_void_p, C.POINTER(ReduceSpec)]
    lib.mz_gpu_reduce_push.argtypes = [C.c_void_p, C.c_void_p,
                                       C.POINTER(Updates),
                                       C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_reduce_push2.argtypes = [C.c_void_p, C.c_void_p,
                                        C.POINTER(Updates),
                                        C.POINTER(Updates),
                                        C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_arr_insert_async.argtypes = [C.c_void_p, C.c_void_p,
                                            C.POINTER(Updates)]
    lib.mz_gpu_arr_flush.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_arr_flush_take.argtypes = [C.c_void_p, C.c_void_p,
                                          C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_prof_dump.argtypes = [C.c_void_p]
    lib.mz_gpu_topk_create.restype = C.c_void_p
    lib.mz_gpu_topk_create.argtypes = [C.c_void_p, C.POINTER(TopKSpec)]
    lib.mz_gpu_topk_push.argtypes = [C.c_void_p, C.c_void_p,
                                     C.POINTER(Updates),
                                     C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_topk_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_threshold_create.restype = C.c_void_p
    lib.mz_gpu_threshold_create.argtypes = [C.c_void_p, C.POINTER(Schema)]
    lib.mz_gpu_threshold_push.argtypes = [C.c_void_p, C.c_void_p,
                                          C.POINTER(Updates),
                                          C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_threshold_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_partition.argtypes = [
        C.c_void_p, C.POINTER(Schema), C.POINTER(Updates), C.c_uint32,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_int64), C.POINTER(C.c_uint64)]
    lib.mz_gpu_map.argtypes = [C.c_void_p, C.POINTER(Schema),
                               C.POINTER(Updates), C.POINTER(Closure),
                               C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_minmax_create.restype = C.c_void_p
    lib.mz_gpu_minmax_create.argtypes = [C.c_void_p, C.POINTER(Schema),
                                         C.c_int, C.POINTER(C.c_uint32),
                                         C.c_uint32]
    lib.mz_gpu_minmax_push.argtypes = [C.c_void_p, C.c_void_p,
                                       C.POINTER(Updates),
                                       C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_minmax_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_join_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_reduce_drop.argtypes = [C.c_void_p, C.c_void_p]
    lib.mz_gpu_peek.argtypes = [C.c_void_p, C.c_void_p,
                                C.POINTER(C.c_uint64), C.c_uint64,
                                C.c_uint64, C.POINTER(C.POINTER(OutBatch))]
    lib.mz_gpu_route_hash.restype = C.c_uint64
    lib.mz_gpu_route_hash.argtypes = [C.POINTER(C.c_uint64), C.c_uint32]
    lib.mz_gpu_set_kernel_timing.argtypes = [C.c_void_p, C.c_int]
    lib.mz_gpu_get_probe_stats.argtypes = [C.c_void_p, C.POINTER(C.c_double),
                                           C.POINTER(C.c_uint64),
                                           C.POINTER(C.c_uint64)]
    _LIB = lib
    return lib


class GpuCtx:
    """One engine context on one GPU (one driver thread per GPU —
    mirrors a timely worker, server.rs:327-377)."""

    def __init__(self, device=0):
        self.lib = load()
        from ._abi import Cfg
        cfg = Cfg(hbm_pool_bytes=0, device_index=device)
        self.ctx = self.lib.mz_gpu_init(C.byref(cfg))
        if not self.ctx:
            raise MzGpuError(
                "mz_gpu_init failed: no HIP device available. The product "
                "path requires an MI355X; there is no CPU fallback.")
        # (The round-1 "virgin context" warmup is gone: the fault was
        # hipMemsetAsync fills being lost on a fresh process's first ops —
        # all semantic state init now uses compute-kernel fills in the C
        # layer, so every ABI consumer is covered. DESIGN.md §9.)

    def close(self):
        if getattr(self, "ctx", None):
            self.lib.mz_gpu_fini(self.ctx)
            self.ctx = None

    def __del__(self):
        # contexts own multi-GB device state; dropping one without
        # close() must not leak it for the rest of the process
        if os.environ.get("MZ_NO_GCCLOSE"):
            return
        try:
            self.close()
        except Exception:
            pass

    def _check(self, rc):
        if rc != 0:
            raise MzGpuError(self.lib.mz_gpu_last_error(self.ctx).decode())

    def _take(self, outp):
        """Copy an out-batch to host numpy arrays and release it. Any
        error rows (ok/err split) land in self.last_errs as
        (codes, times, diffs)."""
        import numpy as np
        from ._abi import MZ_GPU_VARLEN
        ob = outp.contents
        n = ob.n
        kw, vb = ob.schema.key_words, ob.schema.val_bytes
        keys = np.empty(n * kw, np.uint64)
        if vb == MZ_GPU_VARLEN:
            vals = np.empty(int(ob.val_arena_bytes), np.uint8)
            voffs = np.zeros(n + 1, np.uint32)
            if n:
                self._check(self.lib.mz_gpu_out_voffs_to_host(
                    self.ctx, outp,
                    voffs.ctypes.data_as(C.POINTER(C.c_uint32))))
            self.last_voffs = voffs
        else:
            vals = np.empty(n * vb, np.uint8)
            self.last_voffs = None
        times = np.empty(n, np.uint64)
        diffs = np.empty(n, np.int64)
        if n:
            self._check(self.lib.mz_gpu_out_to_host(
                self.ctx, outp,
                keys.ctypes.data_as(C.POINTER(C.c_uint64)),
                vals.ctypes.data_as(C.POINTER(C.c_uint8)),
                times.ctypes.data_as(C.POINTER(C.c_uint64)),
                diffs.ctypes.data_as(C.POINTER(C.c_int64))))
        en = ob.err_n
        ecodes = np.empty(en, np.uint64)
        etimes = np.empty(en, np.uint64)
        ediffs = np.empty(en, np.int64)
        if en:
            self._check(self.lib.mz_gpu_out_err_to_host(
                self.ctx, outp,
                ecodes.ctypes.data_as(C.POINTER(C.c_uint64)),
                etimes.ctypes.data_as(C.POINTER(C.c_uint64)),
                ediffs.ctypes.data_as(C.POINTER(C.c_int64))))
        self.last_errs = (ecodes, etimes, ediffs)
        self.lib.mz_gpu_out_release(self.ctx, outp)
        return keys.view("int64"), vals, times, diffs

    # --- mirrors of the OracleCtx interface (parity rig symmetry) ---
    def arr_create(self, sch):
        return self.lib.mz_gpu_arr_create(self.ctx, C.byref(sch))

    def arr_push(self, arr, upd):
        self._check(self.lib.mz_gpu_arr_push_batch(self.ctx, arr,
                                                   C.byref(upd)))

    def arr_insert(self, arr, upd):
        """Consolidate raw updates + push, fused (one call)."""
        self._check(self.lib.mz_gpu_arr_insert(self.ctx, arr,
                                               C.byref(upd)))

    def arr_set_logical_compaction(self, arr, frontier):
        self.lib.mz_gpu_arr_set_logical_compaction(self.ctx, arr, frontier)

    def arr_maintain(self, arr, fuel=0):
        self._check(self.lib.mz_gpu_arr_maintain(self.ctx, arr, fuel))

    def arr_stats(self, arr):
        nb, nu, by = C.c_uint64(), C.c_uint64(), C.c_uint64()
        self.lib.mz_gpu_arr_stats(self.ctx, arr, C.byref(nb), C.byref(nu),
                                  C.byref(by))
        return nb.value, nu.value, by.value

    def join_create(self, a1, a2, cl):
        return self.lib.mz_gpu_join_create(self.ctx, a1, a2, C.byref(cl))

    def join_push(self, op, side, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_join_push(self.ctx, op, side,
                                              C.byref(upd), C.byref(outp)))
        return self._take(outp)

    def halfjoin(self, lookup, upd, stream_vb, le, cl):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_halfjoin(
            self.ctx, lookup, C.byref(upd), stream_vb, 1 if le else 0,
            C.byref(cl), C.byref(outp)))
        return self._take(outp)

    def reduce_create(self, spec):
        return self.lib.mz_gpu_reduce_create(self.ctx, C.byref(spec))

    def reduce_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_reduce_push(self.ctx, op, C.byref(upd),
                                                C.byref(outp)))
        return self._take(outp)

    def arr_insert_async(self, arr, upd):
        self._check(self.lib.mz_gpu_arr_insert_async(self.ctx, arr,
                                                     C.byref(upd)))

    def arr_flush(self, arr):
        self._check(self.lib.mz_gpu_arr_flush(self.ctx, arr))

    def arr_flush_take(self, arr):
        """Flush, returning the pending insert's consolidated rows as a
        sorted DevOut (None when nothing was pending/empty) — the
        arrangement's update stream (mz_arrange_core's published sealed
        rows, extensions/arrange.rs:69-114)."""
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_arr_flush_take(self.ctx, arr,
                                                   C.byref(outp)))
        if not outp:
            return None
        return self._dev_out(outp, sorted=True)

    def prof_dump(self):
        self.lib.mz_gpu_prof_dump(self.ctx)

    def topk_create(self, spec):
        return self.lib.mz_gpu_topk_create(self.ctx, C.byref(spec))

    def topk_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_topk_push(self.ctx, op, C.byref(upd),
                                              C.byref(outp)))
        return self._take(outp)

    def threshold_create(self, sch):
        return self.lib.mz_gpu_threshold_create(self.ctx, C.byref(sch))

    def threshold_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_threshold_push(self.ctx, op,
                                                   C.byref(upd),
                                                   C.byref(outp)))
        return self._take(outp)

    def consolidate(self, sch, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_consolidate(self.ctx, C.byref(sch),
                                                C.byref(upd), C.byref(outp)))
        return self._take(outp)

    def route_hash(self, words):
        arr = (C.c_uint64 * len(words))(*[w & 0xFFFFFFFFFFFFFFFF
                                          for w in words])
        return self.lib.mz_gpu_route_hash(arr, len(words))

    def map(self, in_schema, upd, cl):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_map(self.ctx, C.byref(in_schema),
                                        C.byref(upd), C.byref(cl),
                                        C.byref(outp)))
        return self._take(outp)

    def minmax_create(self, in_schema, is_max, buckets):
        arr = (C.c_uint32 * len(buckets))(*buckets)
        return self.lib.mz_gpu_minmax_create(self.ctx, C.byref(in_schema),
                                             1 if is_max else 0, arr,
                                             len(buckets))

    def minmax_push(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_minmax_push(self.ctx, op, C.byref(upd),
                                                C.byref(outp)))
        return self._take(outp)

    def peek(self, arr, keys, time, kw=1):
        """Read (val, summed diff) per requested key as of `time`."""
        import numpy as np
        keys = np.ascontiguousarray(keys, np.int64).ravel()
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_peek(
            self.ctx, arr,
            keys.view(np.uint64).ctypes.data_as(C.POINTER(C.c_uint64)),
            len(keys) // kw, time, C.byref(outp)))
        return self._take(outp)

    # --- device-resident variants (outputs stay on the GPU; used by the
    # render layer to chain stages without host round trips) ---
    def _dev_out(self, outp, sorted=False):
        from .render import DevOut
        return DevOut(self, outp, sorted=sorted)

    def _take_copy(self, outp):
        """Copy an OutBatch to host WITHOUT releasing it."""
        import numpy as np
        ob = outp.contents
        n = ob.n
        kw, vb = ob.schema.key_words, ob.schema.val_bytes
        keys = np.empty(n * kw, np.uint64)
        vals = np.empty(n * vb, np.uint8)
        times = np.empty(n, np.uint64)
        diffs = np.empty(n, np.int64)
        if n:
            self._check(self.lib.mz_gpu_out_to_host(
                self.ctx, outp,
                keys.ctypes.data_as(C.POINTER(C.c_uint64)),
                vals.ctypes.data_as(C.POINTER(C.c_uint8)),
                times.ctypes.data_as(C.POINTER(C.c_uint64)),
                diffs.ctypes.data_as(C.POINTER(C.c_int64))))
        return keys.view("int64"), vals, times, diffs

    def consolidate_dev(self, sch, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_consolidate(self.ctx, C.byref(sch),
                                                C.byref(upd), C.byref(outp)))
        return self._dev_out(outp, sorted=True)

    def halfjoin_dev(self, lookup, upd, stream_vb, le, cl):
        # raw (unconsolidated) output: the render layer's consumers —
        # the next probe stage and the reduce — consolidate themselves
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_halfjoin_raw(
            self.ctx, lookup, C.byref(upd), stream_vb, 1 if le else 0,
            C.byref(cl), C.byref(outp)))
        return self._dev_out(outp)

    def halfjoin2_dev(self, lk1, le1, cl1, lk2, le2, cl2, upd, stream_vb):
        # fused two-stage delta path (k_probe_path2); raw output
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_halfjoin2(
            self.ctx, lk1, 1 if le1 else 0, C.byref(cl1), lk2,
            1 if le2 else 0, C.byref(cl2), C.byref(upd), stream_vb,
            C.byref(outp)))
        return self._dev_out(outp)

    def partition_dev(self, sch, upd, nshards, out_tensors):
        """mz_gpu_partition into caller-provided torch device tensors
        (keys, vals-or-None, times, diffs); returns per-shard counts.
        The call synchronizes: the tensors are fully written on return."""
        counts = (C.c_uint64 * nshards)()
        kt, vt, tt, dt = out_tensors
        self._check(self.lib.mz_gpu_partition(
            self.ctx, C.byref(sch), C.byref(upd), nshards,
            C.cast(kt.data_ptr(), C.POINTER(C.c_uint64)),
            (C.cast(vt.data_ptr(), C.POINTER(C.c_uint8))
             if vt is not None else None),
            C.cast(tt.data_ptr(), C.POINTER(C.c_uint64)),
            C.cast(dt.data_ptr(), C.POINTER(C.c_int64)),
            counts))
        return list(counts)

    def join_push_dev(self, op, side, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_join_push(self.ctx, op, side,
                                              C.byref(upd), C.byref(outp)))
        return self._dev_out(outp, sorted=True)  # consolidated output

    def threshold_push_dev(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_threshold_push(self.ctx, op,
                                                   C.byref(upd),
                                                   C.byref(outp)))
        return self._dev_out(outp, sorted=True)  # consolidated output

    def reduce_push_dev(self, op, upd):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_reduce_push(self.ctx, op, C.byref(upd),
                                                C.byref(outp)))
        return self._dev_out(outp, sorted=True)  # consolidated output

    def reduce_push2_dev(self, op, u1, u2):
        outp = C.POINTER(OutBatch)()
        self._check(self.lib.mz_gpu_reduce_push2(
            self.ctx, op, C.byref(u1), C.byref(u2), C.byref(outp)))
        return self._dev_out(outp, sorted=True)  # consolidated output

    def set_kernel_timing(self, on):
        self.lib.mz_gpu_set_kernel_timing(self.ctx, 1 if on else 0)

    def probe_stats(self):
        ms, rows, launches = C.c_double(), C.c_uint64(), C.c_uint64()
        self.lib.mz_gpu_get_probe_stats(self.ctx, C.byref(ms), C.byref(rows),
                                        C.byref(launches))
        return ms.value, rows.value, launches.value
