"""ctypes mirror of include/mz_gpu.h (the C-ABI drop-in boundary).

These are pure type definitions of the boundary; both the product bindings
(materialize_amd._ffi) and the test-side oracle wrapper (oracle/pyoracle.py)
use them so that parity tests drive both implementations through identical
descriptors.
"""
import ctypes as C

import numpy as np

MZ_SRC_KEY = 0
MZ_SRC_VAL_STREAM = 1
MZ_SRC_VAL_LOOKUP = 2
MZ_SRC_COMPUTE = 3

MZ_CMP_LT, MZ_CMP_LE, MZ_CMP_GT, MZ_CMP_GE, MZ_CMP_EQ, MZ_CMP_NE = range(6)
MZ_COMPUTE_REVENUE = 0
MZ_COMPUTE_CONST0 = 1
MZ_COMPUTE_Q17_QTYLT = 2
MZ_COMPUTE_DIV_I64 = 3
MZ_COMPUTE_MUL_I64 = 4
MZ_COMPUTE_CMP_FIELDS = 5
MZ_GPU_VARLEN = 0xFFFFFFFF
MZ_AGG_COUNT, MZ_AGG_SUM_I64, MZ_AGG_SUM_F64 = range(3)

MZ_GPU_MAX_FILTERS = 6
MZ_GPU_MAX_FIELDS = 8
MZ_GPU_MAX_AGGS = 4

# Finalized aggregate output slot: 24 bytes {u8 null; pad[7]; 16B value}
AGG_SLOT_BYTES = 24


class Schema(C.Structure):
    _fields_ = [("key_words", C.c_uint32), ("val_bytes", C.c_uint32)]


class Updates(C.Structure):
    _fields_ = [
        ("keys", C.POINTER(C.c_uint64)),
        ("vals", C.POINTER(C.c_uint8)),
        ("times", C.POINTER(C.c_uint64)),
        ("diffs", C.POINTER(C.c_int64)),
        ("n", C.c_uint64),
        ("lower", C.c_uint64),
        ("upper", C.c_uint64),
        ("on_device", C.c_int32),
        # sorted=1: rows already in canonical (key, val, time) order (the
        # consolidate output form) -> large-table probes take the
        # streaming merge path instead of hash lookups. 0 always safe.
        ("sorted", C.c_int32),
        # VARLEN schemas: [n+1] offsets into the vals byte arena
        ("val_offs", C.POINTER(C.c_uint32)),
    ]


class Filter(C.Structure):
    _fields_ = [
        ("src", C.c_uint8),
        ("off", C.c_uint16),
        ("width", C.c_uint8),
        ("cmp", C.c_uint8),
        ("imm", C.c_int64),
        ("arg0", C.c_uint16),
        ("arg1", C.c_uint16),
        ("arg0_src", C.c_uint8),
        ("arg1_src", C.c_uint8),
    ]


class Field(C.Structure):
    _fields_ = [
        ("src", C.c_uint8),
        ("off", C.c_uint16),
        ("width", C.c_uint8),
        ("arg0", C.c_uint16),
        ("arg1", C.c_uint16),
        ("arg0_src", C.c_uint8),
        ("arg1_src", C.c_uint8),
    ]


class Closure(C.Structure):
    _fields_ = [
        ("n_filters", C.c_uint32),
        ("filters", Filter * MZ_GPU_MAX_FILTERS),
        ("n_key_fields", C.c_uint32),
        ("key_fields", Field * MZ_GPU_MAX_FIELDS),
        ("n_val_fields", C.c_uint32),
        ("val_fields", Field * MZ_GPU_MAX_FIELDS),
        ("out", Schema),
    ]


class Aggregate(C.Structure):
    _fields_ = [
        ("func", C.c_uint8),
        ("off", C.c_uint16),
        ("width", C.c_uint8),
        ("is_float", C.c_uint8),
        ("nullable", C.c_uint8),
    ]


class ReduceSpec(C.Structure):
    _fields_ = [
        ("n_aggs", C.c_uint32),
        ("aggs", Aggregate * MZ_GPU_MAX_AGGS),
        ("in_", Schema),
        ("out", Schema),
    ]


MZ_GPU_MAX_ORDER = 4


class OrderCol(C.Structure):
    _fields_ = [
        ("off", C.c_uint16),
        ("width", C.c_uint8),
        ("desc", C.c_uint8),
    ]


class TopKSpec(C.Structure):
    _fields_ = [
        ("in_", Schema),
        ("offset", C.c_uint64),
        ("limit", C.c_int64),
        ("n_order", C.c_uint32),
        ("order", OrderCol * MZ_GPU_MAX_ORDER),
    ]


MZ_ERR_DIVISION_BY_ZERO = 1


class OutBatch(C.Structure):
    _fields_ = [
        ("keys", C.POINTER(C.c_uint64)),
        ("vals", C.POINTER(C.c_uint8)),
        ("times", C.POINTER(C.c_uint64)),
        ("diffs", C.POINTER(C.c_int64)),
        ("n", C.c_uint64),
        ("on_device", C.c_int32),
        ("schema", Schema),
        # error-row stream (the could_error ok/err split,
        # linear_join.rs:495-541): (code, time, diff) rows
        ("err_n", C.c_uint64),
        ("err_codes", C.POINTER(C.c_uint64)),
        ("err_times", C.POINTER(C.c_uint64)),
        ("err_diffs", C.POINTER(C.c_int64)),
        # VARLEN: [n+1] offsets into vals (the byte arena)
        ("val_offs", C.POINTER(C.c_uint32)),
        ("val_arena_bytes", C.c_uint64),
    ]


class Cfg(C.Structure):
    _fields_ = [("hbm_pool_bytes", C.c_uint64), ("device_index", C.c_uint32)]


def schema(kw, vb):
    return Schema(key_words=kw, val_bytes=vb)


def _as_u64(a):
    return np.ascontiguousarray(a, dtype=np.uint64)


def make_updates(keys, vals, times, diffs, lower, upper, on_device=0,
                 sorted=0, val_offs=None):
    """Build an Updates descriptor over numpy arrays (host memory).

    keys: int64/uint64 array of n*key_words; vals: uint8 array of
    n*val_bytes (or None); times: uint64[n]; diffs: int64[n].
    Keeps references to the arrays on the returned struct (._refs).
    """
    keys = np.ascontiguousarray(keys).view(np.uint64).ravel()
    times = _as_u64(times).ravel()
    diffs = np.ascontiguousarray(diffs, dtype=np.int64).ravel()
    n = len(times)
    u = Updates()
    u.keys = keys.ctypes.data_as(C.POINTER(C.c_uint64))
    if vals is not None and len(vals):
        vals = np.ascontiguousarray(vals, dtype=np.uint8).ravel()
        u.vals = vals.ctypes.data_as(C.POINTER(C.c_uint8))
    else:
        vals = None
        u.vals = None
    u.times = times.ctypes.data_as(C.POINTER(C.c_uint64))
    u.diffs = diffs.ctypes.data_as(C.POINTER(C.c_int64))
    u.n = n
    u.lower = lower
    u.upper = upper
    u.on_device = on_device
    u.sorted = sorted
    if val_offs is not None:
        val_offs = np.ascontiguousarray(val_offs, np.uint32).ravel()
        u.val_offs = val_offs.ctypes.data_as(C.POINTER(C.c_uint32))
    u._refs = (keys, vals, times, diffs, val_offs)
    return u


def make_updates_from_torch(keys_t, vals_t, times_t, diffs_t, lower, upper):
    """Updates descriptor over torch CUDA tensors (on_device=1). Tensors:
    keys int64 [n*kw], vals uint8 [n*vb] or None, times int64 (bit-pattern
    u64) [n], diffs int64 [n]. Keeps tensor refs alive on the struct."""
    u = Updates()
    n = times_t.numel()
    u.keys = C.cast(keys_t.data_ptr(), C.POINTER(C.c_uint64))
    u.vals = (C.cast(vals_t.data_ptr(), C.POINTER(C.c_uint8))
              if vals_t is not None and vals_t.numel() else None)
    u.times = C.cast(times_t.data_ptr(), C.POINTER(C.c_uint64))
    u.diffs = C.cast(diffs_t.data_ptr(), C.POINTER(C.c_int64))
    u.n = n
    u.lower = lower
    u.upper = upper
    u.on_device = 1
    u._refs = (keys_t, vals_t, times_t, diffs_t)
    return u


def out_to_numpy(out, copy=True):
    """Read a host OutBatch into numpy arrays (keys, vals, times, diffs)."""
    n = out.n
    kw = out.schema.key_words
    vb = out.schema.val_bytes
    if n == 0:
        return (np.empty(0, np.int64), np.empty(0, np.uint8),
                np.empty(0, np.uint64), np.empty(0, np.int64))
    keys = np.ctypeslib.as_array(out.keys, shape=(n * kw,)).view(np.int64)
    vals = (np.ctypeslib.as_array(out.vals, shape=(n * vb,))
            if vb else np.empty(0, np.uint8))
    times = np.ctypeslib.as_array(out.times, shape=(n,))
    diffs = np.ctypeslib.as_array(out.diffs, shape=(n,))
    if copy:
        return keys.copy(), vals.copy(), times.copy(), diffs.copy()
    return keys, vals, times, diffs


def filt(src, off, width, cmp, imm, arg0=0, arg1=0, arg0_src=0,
         arg1_src=0):
    return Filter(src=src, off=off, width=width, cmp=cmp, imm=imm,
                  arg0=arg0, arg1=arg1, arg0_src=arg0_src,
                  arg1_src=arg1_src)


def field(src, off, width=8, arg0=0, arg1=0, arg0_src=0, arg1_src=0):
    return Field(src=src, off=off, width=width, arg0=arg0, arg1=arg1,
                 arg0_src=arg0_src, arg1_src=arg1_src)


def closure(filters, key_fields, val_fields, out_schema):
    cl = Closure()
    cl.n_filters = len(filters)
    for i, f in enumerate(filters):
        cl.filters[i] = f
    cl.n_key_fields = len(key_fields)
    for i, f in enumerate(key_fields):
        cl.key_fields[i] = f
    cl.n_val_fields = len(val_fields)
    for i, f in enumerate(val_fields):
        cl.val_fields[i] = f
    cl.out = out_schema
    return cl


def topk_spec(in_schema, order, offset=0, limit=-1):
    """order: list of (off, width, desc) over the val bytes."""
    sp = TopKSpec()
    sp.in_ = in_schema
    sp.offset = offset
    sp.limit = limit
    sp.n_order = len(order)
    for i, (off, width, desc) in enumerate(order):
        sp.order[i] = OrderCol(off=off, width=width, desc=1 if desc else 0)
    return sp


def reduce_spec(aggs, in_schema):
    sp = ReduceSpec()
    sp.n_aggs = len(aggs)
    for i, a in enumerate(aggs):
        sp.aggs[i] = a
    sp.in_ = in_schema
    sp.out = Schema(key_words=in_schema.key_words,
                    val_bytes=AGG_SLOT_BYTES * len(aggs))
    return sp
