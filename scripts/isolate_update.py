"""Stage isolation for the fused update kernel: time the kernel with later
stages compiled out (ARROYO_AMD_KMODE) to find where the launch time goes,
plus a pure streaming-read calibration of the box.
Run on a GPU box:  python scripts/isolate_update.py
"""
import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from arroyo_amd import cabi, nexmark  # noqa: E402
from arroyo_amd.pipeline import NS  # noqa: E402

EVENTS_PER_SEC = 1_000_000
STEPS = 100

MODES = {1: "read+bin only", 2: "+tag protocol", 3: "+table upsert",
         0: "full"}


def main():
    from arroyo_amd import gpu
    lib = gpu.lib()

    # streaming-read ceiling
    n = 32 * 1024 * 1024
    a = torch.randint(0, 1 << 40, (n,), dtype=torch.int64, device="cuda")
    b = torch.randint(0, 1 << 40, (n,), dtype=torch.int64, device="cuda")
    torch.cuda.synchronize()
    gbps = lib.arroyo_amd_stream_gbps(a.data_ptr(), b.data_ptr(), n, 20)
    print(f"stream-read calibration (512 MiB, 2 cols): {gbps:.0f} GB/s",
          flush=True)
    del a, b

    for rows in (65536, 1048576):
        nall = rows * 4
        key, ts = nexmark.bids(nall, events_per_sec=EVENTS_PER_SEC, seed=1)
        span = int(((nall * nexmark.TOTAL_PROPORTION)
                    // nexmark.BID_PROPORTION) * NS // EVENTS_PER_SEC)
        d_key = torch.from_numpy(key).cuda()
        d_ts = torch.from_numpy(ts).cuda()
        views = [(d_key[i * rows:(i + 1) * rows],
                  d_ts[i * rows:(i + 1) * rows]) for i in range(4)]
        for lds in (1, 0):
            for mode in (1, 2, 3, 0):
                if lds and mode == 3:
                    continue
                os.environ["ARROYO_AMD_LDS"] = str(lds)
                os.environ["ARROYO_AMD_KMODE"] = str(mode)
                cfg = cabi.make_config(
                    width_ns=10 * NS, slide_ns=2 * NS, n_keys=1,
                    n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                    log2_capacity=19, ring_panes=16, emit_to_host=False)
                op = gpu.make_op(cfg)
                torch.cuda.synchronize()
                last_wm = [0]
                batch_span = span // 4

                def step(s):
                    bidx = s % 4
                    off = (s // 4) * span
                    bk, bt = views[bidx]
                    op.process_batch_device(
                        [bk.data_ptr(), bt.data_ptr()], rows, off)
                    if (s - last_wm[0]) * batch_span > NS:
                        last_wm[0] = s
                        wm = int(ts[(bidx + 1) * rows - 1]) + off - NS
                        lib.arroyo_amd_handle_watermark(
                            op._h, ctypes.c_uint64(wm), None)

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
                kus = p["update_ms"] / max(p["launches"], 1) * 1000
                gb = rows * 16 / (kus * 1e-6) / 1e9
                print(f"rows={rows:8d} lds={lds} mode={mode} "
                      f"({MODES[mode]:15s}) kernel={kus:9.2f}us "
                      f"({gb:7.1f} GB/s alg)", flush=True)
        del d_key, d_ts, views


if __name__ == "__main__":
    main()
