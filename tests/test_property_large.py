"""Size-independent properties at BASELINE.json full sizes (64K-row batches,
nexmark-shaped stream): every fired window's aggregate must reconcile with a
direct numpy count over the raw input — catches row loss, double counting,
and window-boundary errors at scales where the row-by-row oracle is too slow.

Property (q5 shape, COUNT GROUP BY auction over hop(2s, 10s)):
  for each emitted window [ws, we):
    sum of emitted counts == #rows with bin in [ws, we)
    #emitted rows        == #distinct keys among those rows
    per-key counts match numpy exactly for a sample of keys
(bin = ts - ts % slide; the stream is event-time monotone so no late drops.)
"""
import numpy as np
import pytest

from arroyo_amd import cabi, nexmark
from arroyo_amd.pipeline import NS, U64MAX, batches_from_columns, run_stream

pytestmark = pytest.mark.gpu

WIDTH, SLIDE = 10 * NS, 2 * NS


def collect_windows(outputs):
    """outputs: list of [key, count, ws, we, _ts] column sets -> {ws: cols}"""
    wins = {}
    for cols in outputs:
        if cols is None or len(cols[0]) == 0:
            continue
        key, cnt, ws, we = cols[0], cols[1], cols[2], cols[3]
        for w in np.unique(ws):
            m = ws == w
            assert int(w) not in wins, "window fired twice"
            wins[int(w)] = (key[m], cnt[m], int(we[m][0]))
    return wins


def test_q5_counts_reconcile_at_full_size():
    from arroyo_amd import gpu

    n = 8_000_000
    key, ts = nexmark.bids(n, events_per_sec=1_000_000, seed=123)
    op = gpu.make_op(cabi.make_config(
        width_ns=WIDTH, slide_ns=SLIDE, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=19, ring_panes=16))
    outputs = run_stream(op, batches_from_columns([key, ts], 65536), NS)
    op.close()

    wins = collect_windows(outputs)
    assert len(wins) >= 3
    bins = ts - ts % SLIDE
    rng = np.random.default_rng(7)
    for ws, (wkey, wcnt, we) in wins.items():
        m = (bins >= ws) & (bins < we)
        total = int(m.sum())
        assert int(wcnt.sum()) == total, f"window {ws}: row count mismatch"
        distinct = np.unique(key[m])
        assert len(wkey) == len(distinct), f"window {ws}: key count mismatch"
        # exact per-key counts for a sample
        for k in rng.choice(distinct, size=min(16, len(distinct)),
                            replace=False):
            want = int(((key == k) & m).sum())
            got = int(wcnt[wkey == k][0])
            assert got == want, f"window {ws} key {k}: {got} != {want}"


def test_q5_device_resident_path_reconciles():
    """Same property through the device-resident multi-batch submit path the
    bench uses (process_batches_device + emitted-rows accounting)."""
    import torch

    from arroyo_amd import gpu

    n = 4_194_304
    key, ts = nexmark.bids(n, events_per_sec=1_000_000, seed=99)
    dev = torch.device("cuda", 0)
    d_key = torch.from_numpy(key).to(dev)
    d_ts = torch.from_numpy(ts).to(dev)
    op = gpu.make_op(cabi.make_config(
        width_ns=WIDTH, slide_ns=SLIDE, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=19, ring_panes=16,
        emit_to_host=True))
    batch = 65536
    outputs = []
    wm_last = 0
    for b in range(n // batch):
        op.process_batches_device(
            [d_key.data_ptr() + b * batch * 8, d_ts.data_ptr() + b * batch * 8],
            batch, 1, contiguous=True)
        mx = int(ts[(b + 1) * batch - 1])
        if mx - wm_last > NS:
            wm_last = mx
            outputs.append(op.handle_watermark(mx - NS))
    outputs.append(op.handle_watermark(U64MAX))
    perf = op.perf()
    op.close()
    assert perf["rows"] == n

    wins = collect_windows(outputs)
    bins = ts - ts % SLIDE
    for ws, (wkey, wcnt, we) in wins.items():
        m = (bins >= ws) & (bins < we)
        assert int(wcnt.sum()) == int(m.sum())
        assert len(wkey) == len(np.unique(key[m]))


def test_q5_fused_watermark_submission_identical():
    """Submitting two watermark periods of rows in one pass and then the two
    watermarks (the bench's BENCH_WM_FUSE batching) must produce the same
    windows as strict per-period submission: a post-watermark row can never
    land in a pane that watermark's windows read."""
    import torch

    from arroyo_amd import gpu

    n = 2_097_152
    key, ts = nexmark.bids(n, events_per_sec=1_000_000, seed=77)
    dev = torch.device("cuda", 0)
    d_key = torch.from_numpy(key).to(dev)
    d_ts = torch.from_numpy(ts).to(dev)
    batch = 65536

    def run(fuse):
        op = gpu.make_op(cabi.make_config(
            width_ns=WIDTH, slide_ns=SLIDE, n_keys=1, n_value_cols=0,
            aggs=[(cabi.COUNT, -1)], log2_capacity=19, ring_panes=16,
            emit_to_host=True))
        outs = []
        pending = []
        wm_last = 0
        for b in range(n // batch):
            op.process_batches_device(
                [d_key.data_ptr() + b * batch * 8,
                 d_ts.data_ptr() + b * batch * 8], batch, 1,
                contiguous=True)
            mx = int(ts[(b + 1) * batch - 1])
            if mx - wm_last > NS:
                wm_last = mx
                pending.append(mx - NS)
            if len(pending) >= fuse:
                for wm in pending:
                    outs.append(op.handle_watermark(wm))
                pending = []
        for wm in pending:
            outs.append(op.handle_watermark(wm))
        outs.append(op.handle_watermark(U64MAX))
        op.close()
        rows = []
        for cols in outs:
            if cols is None or len(cols) == 0 or len(cols[0]) == 0:
                continue
            rows += [tuple(int(c[i]) for c in cols)
                     for i in range(len(cols[0]))]
        return sorted(rows)

    def run_epoch(fuse):
        """the bench's actual submission shape: fused periods, epoch
        marked after the period's rows, next period submitted BEFORE the
        previous epoch's watermarks fold"""
        op = gpu.make_op(cabi.make_config(
            width_ns=WIDTH, slide_ns=SLIDE, n_keys=1, n_value_cols=0,
            aggs=[(cabi.COUNT, -1)], log2_capacity=19, ring_panes=16,
            emit_to_host=True))
        outs = []
        pending = []
        deferred = None
        wm_last = 0
        for b in range(n // batch):
            op.process_batches_device(
                [d_key.data_ptr() + b * batch * 8,
                 d_ts.data_ptr() + b * batch * 8], batch, 1,
                contiguous=True)
            mx = int(ts[(b + 1) * batch - 1])
            if mx - wm_last > NS:
                wm_last = mx
                pending.append(mx - NS)
            if len(pending) >= fuse:
                op.mark_epoch()
                op.set_filter_watermark(pending[-1])
                if deferred is not None:
                    outs.append(op.handle_watermarks_epoch(deferred))
                deferred = pending
                pending = []
        if deferred is not None:
            outs.append(op.handle_watermarks_epoch(deferred))
        for wm in pending:
            outs.append(op.handle_watermark(wm))
        outs.append(op.handle_watermark(U64MAX))
        op.close()
        rows = []
        for cols in outs:
            if cols is None or len(cols) == 0 or len(cols[0]) == 0:
                continue
            rows += [tuple(int(c[i]) for c in cols)
                     for i in range(len(cols[0]))]
        return sorted(rows)

    assert run(1) == run(2) == run(4) == run_epoch(4)
