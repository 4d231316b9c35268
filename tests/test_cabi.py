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


def test_oracle_handle_watermarks_fallback():
    """WindowOp.handle_watermarks falls back to a sequential Python loop
    on libraries without the batched export (the oracle): emissions must
    equal per-watermark calls, concatenated."""
    import numpy as np

    import oracle
    from arroyo_amd import cabi as c

    def stream(op, batched):
        rng = np.random.default_rng(11)
        outs = []
        for i in range(6):
            ts = (1_600_000_000 * 10**9 +
                  np.arange(2000, dtype=np.int64) * 10**6 +
                  i * 2 * 10**9)
            key = rng.integers(0, 50, size=2000).astype(np.int64)
            op.process_batch([key, ts])
            wms = [int(ts[-1]) - 10**9]
            if batched:
                out = op.handle_watermarks(wms)
                if out and len(out[0]):
                    outs.append(out)
            else:
                for w in wms:
                    out = op.handle_watermark(w)
                    if out and len(out[0]):
                        outs.append(out)
        return outs

    cfg = lambda: c.make_config(width_ns=4 * 10**9, slide_ns=2 * 10**9,
                                n_keys=1, n_value_cols=0,
                                aggs=[(c.COUNT, -1)], log2_capacity=12)
    a = oracle.make_op(cfg())
    b = oracle.make_op(cfg())
    sa = stream(a, False)
    sb = stream(b, True)
    a.close()
    b.close()
    assert len(sa) == len(sb)
    for x, y in zip(sa, sb):
        for cx, cy in zip(x, y):
            assert np.array_equal(cx, cy)
