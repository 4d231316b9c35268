"""GPU parity: the HIP product path (arroyo_amd/csrc/arroyo_amd.hip) against
the CPU oracle on identical batch streams, bit-exact for counts / keys /
timestamps / integer aggregates, 1e-9 relative for AVG (BASELINE.md
tolerance).  All tests here require an MI355X (-m gpu)."""
import os

import numpy as np
import pytest

import oracle
from arroyo_amd import cabi, nexmark
from arroyo_amd.pipeline import (NS, U64MAX, batches_from_columns,
                                 concat_outputs, run_stream)

pytestmark = pytest.mark.gpu


def gpu_op(**kw):
    from arroyo_amd import gpu
    return gpu.make_op(cabi.make_config(**kw))


def oracle_op(**kw):
    return oracle.make_op(cabi.make_config(**kw))


def sorted_rows(cols, f64_idx=()):
    """Rows sorted by the integer columns; returns (int_matrix, f64_cols)."""
    n = len(cols[0])
    int_cols = [c for i, c in enumerate(cols) if i not in f64_idx]
    order = np.lexsort(tuple(int_cols[::-1]))
    ints = np.stack([c[order] for c in int_cols]) if int_cols else None
    floats = [cols[i][order] for i in f64_idx]
    return ints, floats


def assert_parity(got, want, f64_idx=()):
    assert (got is None) == (want is None)
    if got is None:
        return
    assert len(got[0]) == len(want[0]), \
        f"row count {len(got[0])} != {len(want[0])}"
    gi, gf = sorted_rows(got, f64_idx)
    wi, wf = sorted_rows(want, f64_idx)
    assert np.array_equal(gi, wi)
    for a, b in zip(gf, wf):
        np.testing.assert_allclose(a, b, rtol=1e-9)


def run_both(cols, lateness=NS, batch=65536, **kw):
    g = gpu_op(**kw)
    o = oracle_op(**kw)
    batches = batches_from_columns(cols, batch)
    got = concat_outputs(run_stream(g, batches, lateness))
    want = concat_outputs(run_stream(o, batches, lateness))
    g.close()
    o.close()
    return got, want


@pytest.mark.parametrize("use_lds", ["0", "1"])
def test_q5_sliding_count_by_auction(use_lds, monkeypatch):
    monkeypatch.setenv("ARROYO_AMD_LDS", use_lds)
    cols = nexmark.bids(500_000, events_per_sec=50_000)
    got, want = run_both(cols, width_ns=10 * NS, slide_ns=2 * NS, n_keys=1,
                         n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                         log2_capacity=16)
    assert_parity(got, want)


def test_sliding_multi_agg_with_avg():
    cols = nexmark.bids(300_000, events_per_sec=30_000, with_price=True)
    got, want = run_both(
        cols, width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=1,
        aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MIN, 0), (cabi.MAX, 0),
              (cabi.AVG, 0)],
        log2_capacity=16)
    # AVG output column index = n_keys + 4
    assert_parity(got, want, f64_idx=(5,))


def test_tumbling_keyed():
    cols = nexmark.bids(200_000, events_per_sec=20_000, with_price=True)
    got, want = run_both(cols, width_ns=5 * NS, slide_ns=0, is_tumbling=True,
                         n_keys=1, n_value_cols=1,
                         aggs=[(cabi.COUNT, -1), (cabi.MAX, 0)],
                         log2_capacity=16)
    assert_parity(got, want)


def test_unkeyed_sliding():
    cols = nexmark.bids(100_000, events_per_sec=20_000)
    got, want = run_both([cols[-1]], width_ns=10 * NS, slide_ns=2 * NS,
                         n_keys=0, n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                         log2_capacity=12)
    assert_parity(got, want)


def test_out_of_order_late_drops_and_gaps():
    """Non-monotone timestamps (late drops depend on watermark timing) and
    gaps larger than the window, small batches."""
    rng = np.random.default_rng(7)
    n = 50_000
    base = np.cumsum(rng.integers(0, 2_000_000, size=n))
    jitter = rng.integers(-3 * NS, 3 * NS, size=n)
    gaps = (rng.random(n) < 0.001) * rng.integers(0, 90 * NS, size=n)
    ts = (1_600_000_000 * NS + base + jitter + np.cumsum(gaps)).astype(np.int64)
    key = rng.integers(0, 211, size=n).astype(np.int64)
    got, want = run_both([key, ts], batch=997, width_ns=10 * NS,
                         slide_ns=2 * NS, n_keys=1, n_value_cols=0,
                         aggs=[(cabi.COUNT, -1)], log2_capacity=12,
                         ring_panes=256)
    assert_parity(got, want)


def test_key_minus_one_special_slot():
    """key == -1 collides with the empty-slot sentinel; must still aggregate
    (dedicated per-pane special entry)."""
    n = 10_000
    rng = np.random.default_rng(3)
    key = rng.choice(np.array([-1, 5, 7], dtype=np.int64), size=n)
    ts = (1_600_000_000 * NS +
          np.arange(n, dtype=np.int64) * 1_000_000).astype(np.int64)
    got, want = run_both([key, ts], batch=1024, width_ns=4 * NS,
                         slide_ns=2 * NS, n_keys=1, n_value_cols=0,
                         aggs=[(cabi.COUNT, -1)], log2_capacity=10)
    assert_parity(got, want)


def test_golden_vectors_on_gpu():
    """The reference's own golden scenarios through the HIP path (batch 32,
    as in the reference smoke tests)."""
    from tests.golden_util import (assert_rows_match, fmt_ts, load_golden,
                                   load_inputs)
    inp = load_inputs()["impulse"]
    ts = np.array(inp["ts"], dtype=np.int64)
    counter = np.array(inp["counter"], dtype=np.int64)
    g = gpu_op(width_ns=10 * NS, slide_ns=2 * NS, n_keys=0, n_value_cols=1,
               aggs=[(cabi.COUNT, -1), (cabi.MIN, 0), (cabi.MAX, 0)],
               log2_capacity=10)
    outs = run_stream(g, batches_from_columns([counter, ts], 32), NS)
    cnt, mn, mx, ws, we, _ = concat_outputs(outs)
    got = [{"count": int(c), "min": int(a), "max": int(b),
            "start": fmt_ts(s), "end": fmt_ts(e)}
           for c, a, b, s, e in zip(cnt, mn, mx, ws, we)]
    assert_rows_match(got, load_golden("sliding_window_end"))
    g.close()

    cars = load_inputs()["cars"]
    ts = np.array(cars["ts"], dtype=np.int64)
    key = np.array(cars["event_type_id"], dtype=np.int64)
    names = cars["event_type_dict"]
    g = gpu_op(width_ns=3600 * NS, slide_ns=0, is_tumbling=True, n_keys=1,
               n_value_cols=0, aggs=[(cabi.COUNT, -1)], log2_capacity=10)
    outs = run_stream(g, batches_from_columns([key, ts], 32), NS)
    k, cnt, ws, we, _ = concat_outputs(outs)
    got = [{"event_type": names[int(kk)], "hour": fmt_ts(s), "count": int(c)}
           for kk, c, s in zip(k, cnt, ws)]
    assert_rows_match(got, load_golden("hourly_by_event_type"))
    g.close()

    # month_loose_watermark.sql: tumble(30 days) unkeyed COUNT, 1-minute
    # watermark lateness
    g = gpu_op(width_ns=30 * 86400 * NS, slide_ns=0, is_tumbling=True,
               n_keys=0, n_value_cols=0, aggs=[(cabi.COUNT, -1)],
               log2_capacity=10)
    outs = run_stream(g, batches_from_columns([ts], 32), 60 * NS)
    cnt, ws, we, _ = concat_outputs(outs)
    got = [{"month": fmt_ts(s), "count": int(c)} for c, s in zip(cnt, ws)]
    assert_rows_match(got, load_golden("month_loose_watermark"))
    g.close()


def test_checkpoint_drain_restore_roundtrip():
    """Drain mid-stream, restore into a fresh operator, finish the stream;
    outputs must equal the uninterrupted oracle run."""
    cols = nexmark.bids(120_000, events_per_sec=20_000)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
              aggs=[(cabi.COUNT, -1)], log2_capacity=14)
    batches = batches_from_columns(cols, 8192)
    half = len(batches) // 2

    from arroyo_amd.pipeline import WatermarkGen
    wg = WatermarkGen(NS)
    g = gpu_op(**kw)
    outs = []
    wm = None
    for b in batches[:half]:
        g.process_batch(b)
        w = wg.on_batch(b[-1])
        if w is not None:
            wm = w
            out = g.handle_watermark(w)
            if out and len(out[0]):
                outs.append(out)
    state = g.checkpoint_drain()
    g.close()

    g2 = gpu_op(**kw)
    g2.restore(state, watermark=wm)
    for b in batches[half:]:
        g2.process_batch(b)
        w = wg.on_batch(b[-1])
        if w is not None:
            out = g2.handle_watermark(w)
            if out and len(out[0]):
                outs.append(out)
    out = g2.handle_watermark(U64MAX)
    if out and len(out[0]):
        outs.append(out)
    g2.close()
    got = concat_outputs(outs)

    o = oracle_op(**kw)
    want = concat_outputs(run_stream(o, batches, NS))
    o.close()
    assert_parity(got, want)


def test_table_full_fails_loudly():
    n = 20_000
    key = np.arange(n, dtype=np.int64)
    ts = np.full(n, 1_600_000_000 * NS, dtype=np.int64)
    g = gpu_op(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
               aggs=[(cabi.COUNT, -1)], log2_capacity=8)
    g.process_batch([key, ts])
    with pytest.raises(RuntimeError, match="full"):
        g.handle_watermark(U64MAX)
    g.close()


def test_partition_kernel_matches_host():
    """K8: device partition ids must match the host restatement of
    server_for_hash (splitmix64 over contiguous u64 ranges)."""
    import ctypes

    from arroyo_amd import gpu

    def splitmix64(x):
        x = (x + 0x9E3779B97F4A7C15) & (2**64 - 1)
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & (2**64 - 1)
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & (2**64 - 1)
        return x ^ (x >> 31)

    n, parts = 100_000, 8
    rng = np.random.default_rng(11)
    keys = rng.integers(-10**12, 10**12, size=n).astype(np.int64)
    vals = rng.integers(0, 100, size=n).astype(np.int64)
    ts = np.arange(n, dtype=np.int64)

    lib = gpu.lib()
    lib.arroyo_amd_create.restype = ctypes.c_void_p  # ensure loaded
    import torch
    dk = torch.from_numpy(keys).cuda()
    dv = torch.from_numpy(vals).cuda()
    dt = torch.from_numpy(ts).cuda()
    ok = torch.empty_like(dk)
    ov = torch.empty_like(dv)
    ot = torch.empty_like(dt)
    counts = gpu.partition_device(dk.data_ptr(), dv.data_ptr(), dt.data_ptr(),
                                  n, parts, ok.data_ptr(), ov.data_ptr(),
                                  ot.data_ptr())
    torch.cuda.synchronize()
    range_size = (2**64 // parts)
    host_pid = np.array([splitmix64(int(np.uint64(k))) // range_size
                         for k in keys], dtype=np.int64)
    host_counts = np.bincount(host_pid, minlength=parts)
    assert list(host_counts) == [int(c) for c in counts]
    # every output segment holds exactly the rows whose pid == segment index
    ok_h = ok.cpu().numpy()
    ot_h = ot.cpu().numpy()
    off = 0
    for p in range(parts):
        seg = set(zip(ok_h[off:off + counts[p]].tolist(),
                      ot_h[off:off + counts[p]].tolist()))
        wantseg = set(zip(keys[host_pid == p].tolist(),
                          ts[host_pid == p].tolist()))
        assert seg == wantseg
        off += counts[p]


def test_nondivisible_width_slide_rejected():
    """hop(3s, 10s): the reference's planner rejects width not a multiple
    of slide (arroyo-planner/src/lib.rs:644) — and the operator's
    earliest-bin state machine would not terminate there — so creation
    fails loudly on both paths."""
    with pytest.raises(Exception):
        gpu_op(width_ns=10 * NS, slide_ns=3 * NS, n_keys=1, n_value_cols=0,
               aggs=[(cabi.COUNT, -1)], log2_capacity=14, ring_panes=32)
    with pytest.raises(Exception):
        oracle_op(width_ns=10 * NS, slide_ns=3 * NS, n_keys=1,
                  n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                  log2_capacity=14)


def test_width_equals_slide():
    """hop(10s, 10s) behaves like tumbling through the sliding machinery."""
    cols = nexmark.bids(200_000, events_per_sec=50_000, seed=19)
    got, want = run_both(cols, width_ns=10 * NS, slide_ns=10 * NS, n_keys=1,
                         n_value_cols=0, aggs=[(cabi.COUNT, -1)],
                         log2_capacity=14, ring_panes=16)
    assert_parity(got, want)


def test_batched_watermarks_equal_sequential():
    """arroyo_amd_handle_watermarks (one device-status read per group of
    row-free watermarks, the WatermarkGenerator's idle-source cadence)
    must be emission-identical to per-watermark handle_watermark calls."""
    cols = nexmark.bids(400_000, events_per_sec=100_000, seed=23,
                        with_price=True)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=1,
              aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MAX, 0)],
              log2_capacity=15, ring_panes=32)
    a, b = gpu_op(**kw), gpu_op(**kw)
    batches = batches_from_columns(cols, 50_000)
    seq, bat = [], []
    pend = []
    for i, cb in enumerate(batches):
        a.process_batch(cb)
        b.process_batch(cb)
        wm = int(cb[-1][-1]) - NS
        pend.append(wm)
        if len(pend) == 3 or i == len(batches) - 1:
            for w in pend:
                out = a.handle_watermark(w)
                if out and len(out[0]):
                    seq.append(out)
            out = b.handle_watermarks(pend)
            if out and len(out[0]):
                bat.append(out)
            pend = []
    a.close()
    b.close()
    got = concat_outputs(bat)
    want = concat_outputs(seq)
    assert_parity(got, want)


def test_epoch_pipelined_equals_sequential():
    """mark_epoch / set_filter_watermark / handle_watermarks_epoch with the
    next period's rows submitted BEFORE the fold must be emission-identical
    to the sequential order (include/arroyo_amd.h epoch-pipelining doc)."""
    cols = nexmark.bids(600_000, events_per_sec=100_000, seed=29,
                        with_price=True)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=1,
              aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.AVG, 0)],
              log2_capacity=15, ring_panes=32)
    a, b = gpu_op(**kw), gpu_op(**kw)
    batches = batches_from_columns(cols, 60_000)
    seq, pip = [], []
    pending = None
    for i, cb in enumerate(batches):
        wm = int(cb[-1][-1]) - NS
        # sequential reference
        a.process_batch(cb)
        out = a.handle_watermark(wm)
        if out and len(out[0]):
            seq.append(out)
        # pipelined: submit THEN fold the previous epoch
        b.process_batch(cb)
        b.mark_epoch()
        b.set_filter_watermark(wm)
        if pending is not None:
            out = b.handle_watermarks_epoch([pending])
            if out and len(out[0]):
                pip.append(out)
        pending = wm
    if pending is not None:
        out = b.handle_watermarks_epoch([pending])
        if out and len(out[0]):
            pip.append(out)
    a.close()
    b.close()
    assert_parity(concat_outputs(pip), concat_outputs(seq), f64_idx=(3,))


def test_small_ring_fused_groups_loud_or_correct():
    """ring_panes=8 under batched 4-watermark groups exercises the slot
    reuse margin of the concurrent fire path: the result must be either
    exact parity with the oracle or the documented loud ring-conflict
    error -- never silently wrong output."""
    cols = nexmark.bids(400_000, events_per_sec=100_000, seed=31)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
              aggs=[(cabi.COUNT, -1)], log2_capacity=15, ring_panes=8)
    g = gpu_op(**kw)
    o = oracle_op(**kw)
    batches = batches_from_columns(cols, 40_000)
    got, want = [], []
    pend = []
    try:
        for i, cb in enumerate(batches):
            g.process_batch(cb)
            o.process_batch(cb)
            pend.append(int(cb[-1][-1]) - NS)
            if len(pend) == 4 or i == len(batches) - 1:
                out = g.handle_watermarks(pend)
                if out and len(out[0]):
                    got.append(out)
                for w in pend:
                    ow = o.handle_watermark(w)
                    if ow and len(ow[0]):
                        want.append(ow)
                pend = []
    except RuntimeError as e:
        assert "ring" in str(e), e   # the loud error is acceptable
        g.close()
        o.close()
        return
    g.close()
    o.close()
    assert_parity(concat_outputs(got), concat_outputs(want))


def test_dual_window_merge_matches_sequential(monkeypatch):
    """The gated dual-window merge (ARROYO_AMD_DUAL=1: one kernel for two
    consecutive fires in a batched watermark call) must be
    emission-identical to the default single-window path."""
    monkeypatch.setenv("ARROYO_AMD_DUAL", "1")
    cols = nexmark.bids(500_000, events_per_sec=100_000, seed=37)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
              aggs=[(cabi.COUNT, -1)], log2_capacity=15, ring_panes=32)
    a = gpu_op(**kw)
    monkeypatch.setenv("ARROYO_AMD_DUAL", "0")
    b = gpu_op(**kw)
    batches = batches_from_columns(cols, 50_000)
    outs = {0: [], 1: []}
    pend = []
    for i, cb in enumerate(batches):
        a.process_batch(cb)
        b.process_batch(cb)
        pend.append(int(cb[-1][-1]) - NS)
        if len(pend) == 4 or i == len(batches) - 1:
            monkeypatch.setenv("ARROYO_AMD_DUAL", "1")
            out = a.handle_watermarks(pend)
            if out and len(out[0]):
                outs[0].append(out)
            monkeypatch.setenv("ARROYO_AMD_DUAL", "0")
            out = b.handle_watermarks(pend)
            if out and len(out[0]):
                outs[1].append(out)
            pend = []
    a.close()
    b.close()
    assert_parity(concat_outputs(outs[0]), concat_outputs(outs[1]))
