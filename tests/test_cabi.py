"""CPU-side checks of the product library: it builds for gfx950, loads, and
exports every symbol include/arroyo_amd.h declares.  No compute calls (no
GPU here); creation failure without a device must be loud."""
import ctypes
import os

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(ROOT, "arroyo_amd", "libarroyo_amd.so")

def declared_symbols():
    """Every arroyo_amd_* function include/arroyo_amd.h declares."""
    import re
    hdr = open(os.path.join(ROOT, "include", "arroyo_amd.h")).read()
    syms = sorted(set(re.findall(r"\b(arroyo_amd_[a-z0-9_]+)\s*\(", hdr)))
    assert len(syms) > 40, syms  # all seven families + shared helpers
    return syms


SYMBOLS = declared_symbols()


def _build():
    from arroyo_amd import gpu
    return gpu.build()


def test_library_builds_and_exports_all_symbols():
    so = _build()
    lib = ctypes.CDLL(so)
    for sym in SYMBOLS:
        assert getattr(lib, sym, None) is not None, f"missing symbol {sym}"


def test_device_code_is_gfx950():
    import subprocess
    so = _build()
    out = subprocess.run(
        ["/opt/rocm/lib/llvm/bin/llvm-objdump", "--offloading", so],
        capture_output=True, text=True)
    assert "gfx950" in out.stdout + out.stderr


def test_create_without_gpu_fails_loudly():
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by -m gpu tests")
    from arroyo_amd import cabi, gpu
    lib = gpu.lib()
    cfg = cabi.make_config(width_ns=10**9, slide_ns=10**9,
                           aggs=[(cabi.COUNT, -1)])
    fn = cabi.bind(lib, "arroyo_amd_")
    h = fn["create"](ctypes.byref(cfg))
    assert not h, "create must fail without a HIP device (no CPU fallback)"
    msg = fn["last_error"](None).decode()
    assert "hip" in msg.lower() or "device" in msg.lower()
