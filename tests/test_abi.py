"""CPU checks of the C-ABI boundary: the product library loads and
exports every symbol include/mz_gpu.h declares (no compute without a
GPU), and init fails loudly rather than falling back."""
import ctypes as C
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "mz_gpu.h")
LIB = os.path.join(REPO, "materialize_amd", "csrc", "libmzgpu.so")


def declared_symbols():
    src = open(HEADER).read()
    # function declarations: return type then mz_gpu_xxx(
    syms = set(re.findall(r"\b(mz_gpu_[a-z0-9_]+)\s*\(", src))
    return syms


def test_library_exports_every_declared_symbol():
    if not os.path.exists(LIB):
        import subprocess
        subprocess.run(["make", "-C", os.path.dirname(LIB)], check=True,
                       capture_output=True)
    lib = C.CDLL(LIB)
    missing = [s for s in sorted(declared_symbols())
               if not hasattr(lib, s)]
    assert not missing, f"undefined ABI symbols: {missing}"


def test_init_fails_loudly_without_gpu():
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from materialize_amd._ffi import GpuCtx, MzGpuError
    with pytest.raises(MzGpuError):
        GpuCtx()


def test_oracle_is_not_imported_by_product():
    """The product package must never import the oracle (no CPU fallback
    through test infrastructure)."""
    import subprocess
    import sys
    code = (
        "import sys; sys.path.insert(0, %r); "
        "import materialize_amd, materialize_amd.render, "
        "materialize_amd.workloads, materialize_amd.tpch, "
        "materialize_amd.dist, materialize_amd._abi; "
        "bad = [m for m in sys.modules if 'oracle' in m.lower()]; "
        "assert not bad, bad" % REPO)
    subprocess.run([sys.executable, "-c", code], check=True)
