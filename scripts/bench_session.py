"""Session-operator throughput note (BASELINE config 5 shape, 1 GPU):
Zipf-keyed stream from a 10M-key space through the HIP session operator,
device-resident batches, 1/s event-time watermark cadence.  Not the
headline bench (configs[1] is); a supporting datapoint for profiles/.
Run on a GPU box: python scripts/bench_session.py [n_rows]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from arroyo_amd import cabi, gpu

NS = 10**9
BATCH = 65536


def zipf_keys(rng, n, key_space=10_000_000, s=1.0):
    ranks = np.arange(1, key_space + 1, dtype=np.float64)
    p = 1.0 / ranks ** s
    p /= p.sum()
    return rng.choice(key_space, size=n, p=p).astype(np.int64)


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8 << 20
    rng = np.random.default_rng(5)
    t0 = 1_600_000_000 * NS
    key = zipf_keys(rng, n)
    ts = t0 + (np.arange(n, dtype=np.int64) * NS) // (16 * BATCH)
    dev = torch.device("cuda", 0)
    d_key = torch.from_numpy(key).to(dev)
    d_ts = torch.from_numpy(ts).to(dev)

    op = gpu.make_session_op(cabi.make_session_config(
        5 * NS, [(cabi.COUNT, -1)], n_keys=1, n_value_cols=0,
        log2_capacity=24, max_sessions=8, log2_batch_capacity=17,
        log2_out_cap=24, emit_to_host=False))

    import ctypes
    lib = gpu.lib()
    lib.arroyo_amd_session_process_batches_device.restype = ctypes.c_int
    lib.arroyo_amd_session_process_batches_device.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
        ctypes.c_int64, ctypes.c_int32, ctypes.c_uint64]

    def submit(b, reps=1):
        arr = (ctypes.c_void_p * 2)(d_key.data_ptr() + b * BATCH * 8,
                                    d_ts.data_ptr() + b * BATCH * 8)
        rc = lib.arroyo_amd_session_process_batches_device(
            op._h, arr, 2, BATCH, reps, 0)
        if rc != 0:
            raise RuntimeError(op._fn["last_error"](op._h).decode())

    # one C call per watermark interval (16 batches): the per-batch
    # update+merge rounds enqueue C-side, python cost amortizes
    n_batches = n // BATCH
    warm = n_batches // 8
    for b in range(0, warm - warm % 16, 16):
        submit(b, 16)
        op.handle_watermark(int(ts[(b + 16) * BATCH - 1]) - NS)
    torch.cuda.synchronize()
    t_start = time.perf_counter()
    start = warm - warm % 16
    for b in range(start, n_batches - 15, 16):
        submit(b, 16)
        op.handle_watermark(int(ts[(b + 16) * BATCH - 1]) - NS)
    n_batches = (n_batches - start) - (n_batches - start) % 16 + start
    torch.cuda.synchronize()
    dt = time.perf_counter() - t_start
    rows = (n_batches - start) * BATCH
    print(f"session op (Zipf s=1.0, 10M-key space, gap 5s): "
          f"{rows / dt / 1e9:.3f} Grows/s over {rows} rows "
          f"({dt*1e6/ max(n_batches - start, 1):.1f} us/64K batch)")
    op.close()


if __name__ == "__main__":
    main()
