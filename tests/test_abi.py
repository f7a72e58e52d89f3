# CPU-side checks of the C-ABI shared library: it loads and exports every
# symbol include/bkgpu.h declares (no compute calls — no GPU here).
import ctypes as C
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "baikaldb_amd", "libbkgpu.so")
HDR = os.path.join(REPO, "include", "bkgpu.h")


def _build_if_needed():
    src = os.path.join(REPO, "baikaldb_amd", "csrc", "bkgpu.hip")
    if not os.path.exists(LIB) or os.path.getmtime(LIB) < os.path.getmtime(src):
        subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
                        "-munsafe-fp-atomics", "-fPIC", "-shared", src, "-o", LIB],
                       check=True, capture_output=True)


def _declared_symbols():
    with open(HDR) as f:
        text = f.read()
    return sorted(set(re.findall(r"\b(bkgpu_\w+)\s*\(", text)))


def test_library_loads_and_exports_header_symbols():
    _build_if_needed()
    lib = C.CDLL(LIB)
    syms = _declared_symbols()
    assert len(syms) >= 15
    for s in syms:
        assert hasattr(lib, s), f"missing export: {s}"


def test_engine_refuses_without_gpu():
    """On a machine with no GPU the engine must fail loudly, not fall back."""
    _build_if_needed()
    import sys
    sys.path.insert(0, REPO)
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from baikaldb_amd import GpuEngine, NativeEngineMissing
    with pytest.raises((NativeEngineMissing, RuntimeError)):
        GpuEngine()
