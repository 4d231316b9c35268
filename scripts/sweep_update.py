"""Grid/LDS/batch-size sweep of the fused update kernel (k_update[_lds]) on
the q5 bench workload, to locate where the 64K-row launch time goes.
Run on a GPU box:  python scripts/sweep_update.py
"""
import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from arroyo_amd import cabi, nexmark  # noqa: E402
from arroyo_amd.pipeline import NS  # noqa: E402

BATCH_ROWS_LIST = [65536, 262144, 1048576]
BLOCKS_LIST = [0, 32, 64, 128, 256, 512, 1024, 2048]
STEPS = 120
EVENTS_PER_SEC = 1_000_000


def run_one(rows, blocks, lds):
    os.environ["ARROYO_AMD_LDS"] = str(lds)
    os.environ["ARROYO_AMD_BLOCKS"] = str(blocks)
    from arroyo_amd import gpu
    n = rows * 4
    key, ts = nexmark.bids(n, events_per_sec=EVENTS_PER_SEC, seed=1)
    span = int(((n * nexmark.TOTAL_PROPORTION) // nexmark.BID_PROPORTION)
               * NS // EVENTS_PER_SEC)
    d_key = torch.from_numpy(key).cuda()
    d_ts = torch.from_numpy(ts).cuda()
    views = [(d_key[i * rows:(i + 1) * rows], d_ts[i * rows:(i + 1) * rows])
             for i in range(4)]
    cfg = cabi.make_config(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1,
                           n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                           log2_capacity=19, ring_panes=16,
                           emit_to_host=False)
    op = gpu.make_op(cfg)
    lib = gpu.lib()
    torch.cuda.synchronize()

    last_wm_step = 0

    def step(s):
        nonlocal last_wm_step
        b = s % 4
        off = (s // 4) * span
        bk, bt = views[b]
        op.process_batch_device([bk.data_ptr(), bt.data_ptr()], rows, off)
        # retire panes ~ once per 1s event time so the 16-slot ring never
        # conflicts, matching the bench cadence
        batch_span = span // 4
        if (s - last_wm_step) * batch_span > NS:
            last_wm_step = s
            wm = int(ts[(b + 1) * rows - 1]) + off - NS
            rc = lib.arroyo_amd_handle_watermark(op._h, ctypes.c_uint64(wm),
                                                 None)
            if rc:
                raise RuntimeError(op._fn["last_error"](op._h).decode())

    for s in range(20):
        step(s)
    torch.cuda.synchronize()
    op.perf()
    t0 = time.perf_counter()
    for s in range(20, 20 + STEPS):
        step(s)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    p = op.perf()
    op.close()
    del d_key, d_ts, views
    kms = p["update_ms"] / max(p["launches"], 1)
    gbps = rows * 16 / (kms * 1e-3) / 1e9 if kms else 0
    wall_gbps = STEPS * rows * 16 / (t1 - t0) / 1e9
    print(f"rows={rows:8d} blocks={blocks:5d} lds={lds} "
          f"kernel={kms * 1000:9.2f}us ({gbps:7.1f} GB/s alg) "
          f"wall/step={(t1 - t0) / STEPS * 1e6:9.2f}us ({wall_gbps:7.1f} GB/s)",
          flush=True)


if __name__ == "__main__":
    for rows in BATCH_ROWS_LIST:
        for lds in (1, 0):
            for blocks in BLOCKS_LIST:
                run_one(rows, blocks, lds)
