"""ROW_NUMBER window-function tests: the reference's
most_active_driver_last_hour golden vector through the real two-operator
pipeline (sliding window aggregate -> window function with the downstream
row_number=1 filter fused), plus fuzz against a numpy restatement.

Reference semantics: crates/arroyo-worker/src/arrow/window_fn.rs (per-instant
BoundedWindowAggExec: ROW_NUMBER() OVER (PARTITION BY window ORDER BY ...),
instants fired in timestamp order at the watermark, late rows filtered)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, fmt_ts, load_golden, load_inputs
from arroyo_amd.pipeline import U64MAX, WatermarkGen, batches_from_columns

HOUR = 3600 * NS


def rows_of(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return sorted(tuple(int(c[r]) for c in cols) for r in range(len(cols[0])))


def run_most_active_driver(make_window_op, make_windowfn_op):
    """most_active_driver_last_hour.sql: hop(1min, 1h) COUNT GROUP BY
    driver_id -> ROW_NUMBER() OVER (PARTITION BY window ORDER BY count DESC,
    driver_id DESC) -> filter row_number = 1."""
    inp = load_inputs()["cars"]
    ts = np.array(inp["ts"], dtype=np.int64)
    key = np.array(inp["driver_id"], dtype=np.int64)

    win = make_window_op(cabi.make_config(
        width_ns=HOUR, slide_ns=60 * NS, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=14, ring_panes=256))
    # window output columns: [driver, count, ws, we, _ts]
    wf = make_windowfn_op(cabi.make_windowfn_config(
        n_cols=5, part_col=2, order=[(1, True), (0, True)], limit=1,
        log2_rows_cap=12, instants=256))

    wg = WatermarkGen(lateness_ns=HOUR)
    final = []
    for cols in batches_from_columns([key, ts], 32):
        win.process_batch(cols)
        wm = wg.on_batch(cols[-1])
        if wm is not None:
            out = win.handle_watermark(wm)
            if out and len(out[0]):
                wf.process_batch(out)
            final.append(wf.handle_watermark(wm))
    out = win.handle_watermark(U64MAX)
    if out and len(out[0]):
        wf.process_batch(out)
    final.append(wf.handle_watermark(U64MAX))
    win.close()
    wf.close()

    got = []
    for cols in final:
        if cols is None or len(cols) == 0 or len(cols[0]) == 0:
            continue
        for d, c, s, e, _t, rn in zip(*cols):
            got.append({"driver_id": int(d), "count": int(c),
                        "start": fmt_ts(int(s)), "end": fmt_ts(int(e)),
                        "row_number": int(rn)})
    assert_rows_match(got, load_golden("most_active_driver_last_hour"))


def test_most_active_driver_pipeline_oracle():
    run_most_active_driver(oracle.make_op, oracle.make_windowfn_op)


def np_row_numbers(cols, part_col, order, limit):
    """Independent restatement over one instant's rows."""
    n = len(cols[0])
    idx = list(range(n))

    def sort_key(i):
        k = [cols[part_col][i]] if part_col >= 0 else []
        for col, desc in order:
            v = int(cols[col][i])
            k.append(-v if desc else v)
        k.append(i)
        return tuple(k)

    idx.sort(key=sort_key)
    out = []
    rn, prev = 0, None
    for i in idx:
        p = int(cols[part_col][i]) if part_col >= 0 else 0
        rn = 1 if p != prev else rn + 1
        prev = p
        if limit and rn > limit:
            continue
        out.append(tuple(int(c[i]) for c in cols) + (rn,))
    return sorted(out)


def wf_fuzz(make_op, seed=13, n=4000, limit=0):
    rng = np.random.default_rng(seed)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(30, dtype=np.int64) * NS
    part = rng.integers(0, 6, size=n).astype(np.int64)
    v1 = rng.integers(0, 50, size=n).astype(np.int64)
    v2 = rng.integers(0, 1000, size=n).astype(np.int64)
    ts = np.sort(rng.choice(instants, size=n)).astype(np.int64)
    cols = [part, v1, v2, ts]
    op = make_op(cabi.make_windowfn_config(
        n_cols=4, part_col=0, order=[(1, True), (2, False)], limit=limit,
        log2_rows_cap=12, instants=128))
    got = []
    mid = int(t0 + 15 * NS)
    m = ts < mid
    op.process_batch([c[m] for c in cols])
    got += rows_of(op.handle_watermark(mid))
    op.process_batch([c[~m] for c in cols])
    got += rows_of(op.handle_watermark(U64MAX))
    op.close()

    want = []
    for t in np.unique(ts):
        im = ts == t
        want += np_row_numbers([c[im] for c in cols], 0,
                               [(1, True), (2, False)], limit)
    return sorted(got), sorted(want)


@pytest.mark.parametrize("seed", [13, 29])
@pytest.mark.parametrize("limit", [0, 3])
def test_windowfn_oracle_vs_numpy_fuzz(limit, seed):
    got, want = wf_fuzz(oracle.make_windowfn_op, limit=limit, seed=seed)
    assert got == want
    assert len(want) > 100


def test_windowfn_oracle_drops_late_rows():
    op = oracle.make_windowfn_op(cabi.make_windowfn_config(
        n_cols=2, part_col=-1, order=[(0, False)], limit=0))
    t0 = 1_600_000_000 * NS
    op.process_batch([np.array([7], dtype=np.int64),
                      np.array([t0 + 5 * NS], dtype=np.int64)])
    op.handle_watermark(t0 + 3 * NS)
    # late: silently filtered (window_fn.rs filter_and_split_batches)
    op.process_batch([np.array([9], dtype=np.int64),
                      np.array([t0], dtype=np.int64)])
    out = rows_of(op.handle_watermark(U64MAX))
    assert out == [(7, t0 + 5 * NS, 1)]
    op.close()


# ---------------------------------------------------------------- GPU parity


@pytest.mark.gpu
def test_most_active_driver_pipeline_gpu():
    from arroyo_amd import gpu
    run_most_active_driver(gpu.make_op, gpu.make_windowfn_op)


@pytest.mark.gpu
@pytest.mark.parametrize("limit", [0, 3])
def test_windowfn_gpu_vs_numpy_fuzz(limit):
    from arroyo_amd import gpu
    got, want = wf_fuzz(gpu.make_windowfn_op, limit=limit, n=20000)
    assert got == want


@pytest.mark.gpu
def test_windowfn_gpu_checkpoint_roundtrip():
    from arroyo_amd import gpu
    rng = np.random.default_rng(3)
    t0 = 1_600_000_000 * NS
    n = 2000
    part = rng.integers(0, 4, size=n).astype(np.int64)
    v = rng.integers(0, 30, size=n).astype(np.int64)
    ts = t0 + rng.integers(0, 10, size=n).astype(np.int64) * NS
    cfg = lambda: cabi.make_windowfn_config(
        n_cols=3, part_col=0, order=[(1, True)], limit=2,
        log2_rows_cap=12, instants=64)

    a = gpu.make_windowfn_op(cfg())
    a.process_batch([part, v, ts])
    drained = a.checkpoint_drain()
    a.close()
    assert len(drained[0]) == n

    b = gpu.make_windowfn_op(cfg())
    b.restore(drained)
    got = rows_of(b.handle_watermark(U64MAX))
    b.close()

    o = oracle.make_windowfn_op(cfg())
    o.process_batch([part, v, ts])
    want = rows_of(o.handle_watermark(U64MAX))
    o.close()
    assert got == want


@pytest.mark.gpu
def test_windowfn_gpu_large_instant():
    """One 500K-row instant: exercises the hipCUB sort pipeline at scale."""
    from arroyo_amd import gpu
    rng = np.random.default_rng(6)
    t0 = 1_600_000_000 * NS
    n = 500_000
    part = rng.integers(0, 1000, size=n).astype(np.int64)
    v = rng.integers(0, 10**6, size=n).astype(np.int64)
    ts = np.full(n, t0, dtype=np.int64)
    op = gpu.make_windowfn_op(cabi.make_windowfn_config(
        n_cols=3, part_col=0, order=[(1, True)], limit=3,
        log2_rows_cap=19, instants=64, log2_out_cap=20))
    op.process_batch([part, v, ts])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    want = np_row_numbers([part, v, ts], 0, [(1, True)], 3)
    assert got == want


@pytest.mark.gpu
def test_windowfn_gpu_device_resident_matches_host():
    """process_batch_device (device-resident ingest surface) must leave the
    instant table in the same state as the host path: identical fired rows,
    including the arrival-sequence tiebreak (sequence stamping is shared)."""
    import torch
    from arroyo_amd import gpu
    rng = np.random.default_rng(61)
    t0 = 1_600_000_000 * NS
    n = 3000
    instants = t0 + np.arange(20, dtype=np.int64) * NS
    part = rng.integers(0, 5, size=n).astype(np.int64)
    v1 = rng.integers(0, 40, size=n).astype(np.int64)
    ts = np.sort(rng.choice(instants, size=n)).astype(np.int64)
    cols = [part, v1, ts]
    kw = dict(n_cols=3, part_col=0, order=[(1, False)], log2_rows_cap=12,
              instants=64)
    h = gpu.make_windowfn_op(cabi.make_windowfn_config(**kw))
    d = gpu.make_windowfn_op(cabi.make_windowfn_config(**kw))
    dev = torch.device("cuda", 0)
    h.process_batch(cols)
    tens = [torch.from_numpy(c.copy()).to(dev) for c in cols]
    torch.cuda.synchronize()
    d.process_batch_device([t.data_ptr() for t in tens], n)
    got = sorted(rows_of(d.handle_watermark(U64MAX)))
    want = sorted(rows_of(h.handle_watermark(U64MAX)))
    h.close()
    d.close()
    assert got == want
    assert len(want) == n


def _wf_roundtrip(make_op):
    """Checkpoint/restore roundtrip: drain a half-ingested op, restore into
    a fresh one, finish the stream, and match the uninterrupted run
    (including arrival-order ROW_NUMBER ties: drain preserves per-instant
    arrival order)."""
    rng = np.random.default_rng(71)
    t0 = 1_600_000_000 * NS
    n = 3000
    instants = t0 + np.arange(16, dtype=np.int64) * NS
    part = rng.integers(0, 5, size=n).astype(np.int64)
    v1 = rng.integers(0, 8, size=n).astype(np.int64)  # many ties
    ts = np.sort(rng.choice(instants, size=n)).astype(np.int64)
    cols = [part, v1, ts]
    kw = dict(n_cols=3, part_col=0, order=[(1, False)], log2_rows_cap=12,
              instants=64)

    a = make_op(cabi.make_windowfn_config(**kw))
    half = n // 2
    a.process_batch([c[:half] for c in cols])
    drained = a.checkpoint_drain()
    a.close()
    assert len(drained[0]) == half

    b = make_op(cabi.make_windowfn_config(**kw))
    b.restore(drained)
    b.process_batch([c[half:] for c in cols])
    got = sorted(rows_of(b.handle_watermark(U64MAX)))
    b.close()

    c_ = make_op(cabi.make_windowfn_config(**kw))
    c_.process_batch(cols)
    want = sorted(rows_of(c_.handle_watermark(U64MAX)))
    c_.close()
    assert got == want
    assert len(want) == n


def test_windowfn_oracle_checkpoint_restore_roundtrip():
    _wf_roundtrip(oracle.make_windowfn_op)


@pytest.mark.gpu
def test_windowfn_gpu_checkpoint_restore_roundtrip():
    from arroyo_amd import gpu
    _wf_roundtrip(gpu.make_windowfn_op)
