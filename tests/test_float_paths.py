"""Device float fixed-point encode/decode vs host-exact references.

Isolates the two float conversions of the SUM(float) path
(reduce.rs:1663-1697 encode; :1952 decode) on the GPU against
arbitrary-precision Python references.
"""
import ctypes as C
import struct

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

M128 = 1 << 128


def py_float_to_fixed_point(n: float) -> int:
    """Exact restatement of reduce.rs:1663-1697 in Python ints."""
    bits = struct.unpack("<Q", struct.pack("<d", n))[0]
    mantissa = bits & ((1 << 52) - 1)
    exp_bits = (bits >> 52) & 0x7FF
    if exp_bits == 0:
        exponent = -1074
    else:
        mantissa |= 1 << 52
        exponent = exp_bits - 1075
    sign = -1 if bits >> 63 else 1
    e = exponent + 24
    if e >= 0:
        mag = (mantissa << e) % M128 if e < 128 else 0
    else:
        mag = mantissa >> (-e) if -e < 128 else 0
    v = mag if sign > 0 else (M128 - mag) % M128
    return v  # as unsigned 128-bit representation


def py_decode(u: int) -> float:
    """i128 (as unsigned) -> double (Python float() is round-nearest-even)
    then / 2^24."""
    s = u - M128 if u >= (1 << 127) else u
    return float(s) / 16777216.0


def test_device_float_paths():
    from materialize_amd._ffi import GpuCtx, load
    lib = load()
    lib.mz_gpu_debug_float_paths.argtypes = [
        C.c_void_p, C.POINTER(C.c_double), C.c_uint64,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint64), C.POINTER(C.c_double)]
    g = GpuCtx()
    rng = np.random.default_rng(5)
    xs = np.concatenate([
        rng.uniform(-1e6, 1e6, 3000),
        rng.uniform(-1e-8, 1e-8, 1000),
        rng.uniform(-1e30, 1e30, 1000),
        np.array([0.0, -0.0, 1.5, -1.5, 2.0**-25, 5e-324, 1e308]),
    ])
    n = len(xs)
    fp_out = np.zeros(2 * n, np.uint64)
    # decode inputs: random 128-bit ints with varied magnitudes
    dec_in_py = []
    for _ in range(n):
        b = int(rng.integers(1, 128))
        v = int(rng.integers(0, 2**63)) | (int(rng.integers(0, 2**63)) << 63)
        v &= (1 << b) - 1
        if rng.random() < 0.5:
            v = (M128 - v) % M128
        dec_in_py.append(v)
    dec_in = np.zeros(2 * n, np.uint64)
    for i, v in enumerate(dec_in_py):
        dec_in[2 * i] = v & 0xFFFFFFFFFFFFFFFF
        dec_in[2 * i + 1] = v >> 64
    dec_out = np.zeros(n, np.float64)
    lib.mz_gpu_debug_float_paths(
        g.ctx, xs.ctypes.data_as(C.POINTER(C.c_double)), n,
        fp_out.ctypes.data_as(C.POINTER(C.c_uint64)),
        dec_in.ctypes.data_as(C.POINTER(C.c_uint64)),
        dec_out.ctypes.data_as(C.POINTER(C.c_double)))
    # encode parity
    bad = []
    for i, x in enumerate(xs):
        want = py_float_to_fixed_point(float(x))
        got = int(fp_out[2 * i]) | (int(fp_out[2 * i + 1]) << 64)
        if got != want:
            bad.append((float(x), got, want))
    assert not bad, f"encode mismatches: {bad[:5]} ({len(bad)} total)"
    # decode parity (bitwise)
    badd = []
    for i, v in enumerate(dec_in_py):
        want = py_decode(v)
        got = float(dec_out[i])
        if struct.pack("<d", want) != struct.pack("<d", got):
            badd.append((v, got, want))
    assert not badd, f"decode mismatches: {badd[:5]} ({len(badd)} total)"
