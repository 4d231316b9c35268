"""Supporting throughput datapoints for the non-headline operators
(profiles/): TTL join probe+insert rate and updating-aggregate update rate
on synthetic streams.  Run on a GPU box: python scripts/bench_ops.py"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

NS = 10**9
T0 = 1_600_000_000 * NS


def bench_expjoin():
    from arroyo_amd import cabi, gpu
    rng = np.random.default_rng(1)
    n = 1 << 20
    batches = 48
    op = gpu.make_expjoin_op(cabi.make_expjoin_config(
        3600 * NS, n_left_vals=1, n_right_vals=1, log2_capacity=25,
        log2_rows_cap=26, log2_out_cap=24, emit_to_host=False))
    # ~1M keys so matches stay ~1 per probe (updating-join regime)
    def batch(i, side):
        k = rng.integers(0, 1 << 24, size=n).astype(np.int64)
        v = rng.integers(0, 100, size=n).astype(np.int64)
        ts = T0 + np.full(n, i, dtype=np.int64) * NS
        return [k, v, ts]
    data = [batch(i, i % 2) for i in range(batches)]
    t0 = time.perf_counter()
    rows = 0
    for i, cols in enumerate(data):
        out = op.process_batch(i % 2, cols)
        rows += n
    dt = time.perf_counter() - t0
    op.close()
    print(f"expjoin: {rows/dt/1e9:.3f} Grows/s ingest+probe "
          f"({dt*1e6/batches:.0f} us per 1M-row batch, ~1 match/row "
          f"by the last batches)")


def bench_updagg():
    from arroyo_amd import cabi, gpu
    rng = np.random.default_rng(2)
    n = 1 << 20
    batches = 48
    op = gpu.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1, n_value_cols=1,
        log2_capacity=22, log2_out_cap=24))
    key = rng.integers(0, 1 << 21, size=n * 2).astype(np.int64)
    val = rng.integers(0, 1000, size=n * 2).astype(np.int64)
    retract = np.zeros(n * 2, dtype=np.int64)
    t0 = time.perf_counter()
    rows = 0
    for i in range(batches):
        lo = (i % 2) * n
        op.process_batch([key[lo:lo + n], val[lo:lo + n],
                          retract[lo:lo + n]])
        rows += n
        if i % 8 == 7:
            op.flush()
    dt = time.perf_counter() - t0
    op.close()
    print(f"updagg:  {rows/dt/1e9:.3f} Grows/s update (COUNT+SUM, 2M keys, "
          f"flush every 8 batches; {dt*1e6/batches:.0f} us per 1M-row "
          f"batch incl. H2D staging)")


def bench_expjoin_device():
    import torch
    from arroyo_amd import cabi, gpu
    rng = np.random.default_rng(3)
    n = 1 << 20
    batches = 48
    dev = torch.device("cuda", 0)
    op = gpu.make_expjoin_op(cabi.make_expjoin_config(
        3600 * NS, n_left_vals=1, n_right_vals=1, log2_capacity=25,
        log2_rows_cap=26, log2_out_cap=24, emit_to_host=False))
    tens = []
    for i in range(batches):
        k = rng.integers(0, 1 << 24, size=n).astype(np.int64)
        v = rng.integers(0, 100, size=n).astype(np.int64)
        ts = T0 + np.full(n, i, dtype=np.int64) * NS
        tens.append([torch.from_numpy(c).to(dev) for c in (k, v, ts)])
    import ctypes
    lib = gpu.lib()
    lib.arroyo_amd_expjoin_match_count.restype = ctypes.c_int
    lib.arroyo_amd_expjoin_match_count.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64)]

    def consume():
        c = ctypes.c_int64()
        rc = lib.arroyo_amd_expjoin_match_count(op._h, ctypes.byref(c))
        if rc != 0:
            raise RuntimeError(op._fn["last_error"](op._h).decode())
        return int(c.value)

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    matches = 0
    for i, cols in enumerate(tens):
        op.process_batch_device(i % 2, [c.data_ptr() for c in cols], n)
        if i % 8 == 7:  # consume in place before the match buffer fills
            matches += consume()
    matches += consume()  # final count syncs the op stream
    dt = time.perf_counter() - t0
    op.close()
    print(f"expjoin (device-resident): {batches*n/dt/1e9:.3f} Grows/s "
          f"ingest+probe ({dt*1e6/batches:.0f} us per 1M-row batch, "
          f"{matches} matches consumed in place)")


def bench_updagg_device():
    import torch
    from arroyo_amd import cabi, gpu
    rng = np.random.default_rng(4)
    n = 1 << 20
    batches = 48
    dev = torch.device("cuda", 0)
    op = gpu.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1, n_value_cols=1,
        log2_capacity=22, log2_out_cap=24))
    key = torch.from_numpy(
        rng.integers(0, 1 << 21, size=n * 2).astype(np.int64)).to(dev)
    val = torch.from_numpy(
        rng.integers(0, 1000, size=n * 2).astype(np.int64)).to(dev)
    ret = torch.zeros(n * 2, dtype=torch.int64, device=dev)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(batches):
        lo = (i % 2) * n * 8
        op.process_batch_device([key.data_ptr() + lo, val.data_ptr() + lo,
                                 ret.data_ptr() + lo], n)
        if i % 8 == 7:
            op.flush()
    op.flush()  # drains the op stream (no-op emission if nothing changed)
    dt = time.perf_counter() - t0
    op.close()
    print(f"updagg (device-resident):  {batches*n/dt/1e9:.3f} Grows/s "
          f"update (COUNT+SUM, 2M keys, flush every 8 batches, "
          f"{dt*1e6/batches:.0f} us per 1M-row batch)")


if __name__ == "__main__":
    # torch's bundled HIP runtime must initialize before the op library's
    # (system-ROCm) runtime touches the device, or torch.cuda breaks
    import torch
    torch.zeros(1, device="cuda")
    bench_expjoin()
    bench_updagg()
    bench_expjoin_device()
    bench_updagg_device()
