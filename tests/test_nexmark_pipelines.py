"""BASELINE.json configs 3 and 4 as operator-composition parity tests:

  q7 (config 3): tumbling MAX(price) per 10s window, self-joined back with
      the bids to recover the max bid's auction — window aggregate + map
      (bid -> window instant) + instant join on price.
  q8 (config 4): person ⋈ auction within a 10s tumbling window — two map
      ops (side -> window instant) + keyed instant join.

Each pipeline runs end to end through the product operators (GPU) and the
oracle operators, compared bit-exact; a numpy restatement checks q7's
semantics independently.  The multi-GPU legs of these configs are the keyed
shuffle covered by tests/test_multigpu_cpu.py and bench.py's N>1 path.
"""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi, nexmark

NS = 10**9
U64MAX = 2**64 - 1
W = 10 * NS


def rows_of(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return sorted(tuple(int(c[r]) for c in cols) for r in range(len(cols[0])))


def window_end_map_config(n_vals):
    """[vals..., ts] -> [vals..., window_end_as_ts]: we = ts - ts%W + W.
    Registers: r0..r(n_vals-1) vals, r(n_vals) ts."""
    t = n_vals
    prog = [
        (cabi.MOP_CONST, 0, 0, t + 1, W),        # r = W
        (cabi.MOP_MOD, t, t + 1, t + 2),         # ts % W
        (cabi.MOP_SUB, t, t + 2, t + 3),         # bin
        (cabi.MOP_ADD, t + 3, t + 1, t + 4),     # we
    ]
    return cabi.make_map_config(
        n_in_cols=n_vals + 1, prog=prog,
        out_reg=list(range(n_vals)) + [t + 4])


def q7_pipeline(mk_win, mk_join, mk_map, price, auction, ts):
    """tumbling MAX(price); join each window's bids on price == max."""
    win = mk_win(cabi.make_config(
        width_ns=W, slide_ns=0, is_tumbling=True, n_keys=0, n_value_cols=1,
        aggs=[(cabi.MAX, 0)], log2_capacity=14))
    win.process_batch([price, ts])
    wout = win.handle_watermark(U64MAX)  # [maxprice, ws, we, _ts]
    win.close()

    mp = mk_map(window_end_map_config(n_vals=2))
    bids_mapped = mp.process_batch([price, auction, ts])  # [price, auction, we]
    mp.close()

    j = mk_join(cabi.make_join_config(
        n_keys=1, n_left_vals=0, n_right_vals=1, log2_rows_cap=14,
        instants=64, log2_out_cap=18))
    # left: (key=maxprice, ts=we); right: (key=price, auction, ts=we)
    j.process_batch(j.LEFT, [wout[0].astype(np.int64),
                             wout[2].astype(np.int64)])
    j.process_batch(j.RIGHT, [c.astype(np.int64) for c in bids_mapped])
    out = j.handle_watermark(U64MAX)  # [price, auction, _ts=we]
    j.close()
    return rows_of(out)


def np_q7(price, auction, ts):
    rows = []
    bins = ts - ts % W
    for b in np.unique(bins):
        m = bins == b
        mx = price[m].max()
        for a in auction[m][price[m] == mx]:
            rows.append((int(mx), int(a), int(b + W)))
    return sorted(rows)


def q7_data(n=20_000, seed=3):
    cols = nexmark.bids(n, events_per_sec=1_000, seed=seed, with_price=True)
    auction, price, ts = cols
    return price, auction, ts


def test_q7_pipeline_oracle_vs_numpy():
    price, auction, ts = q7_data()
    got = q7_pipeline(oracle.make_op, oracle.make_join_op,
                      oracle.make_map_op, price, auction, ts)
    assert got == np_q7(price, auction, ts)
    assert len(got) >= 3  # >= one max-price bid per fired window


@pytest.mark.gpu
def test_q7_pipeline_gpu_vs_oracle():
    from arroyo_amd import gpu
    price, auction, ts = q7_data(n=200_000)
    got = q7_pipeline(gpu.make_op, gpu.make_join_op, gpu.make_map_op,
                      price, auction, ts)
    want = q7_pipeline(oracle.make_op, oracle.make_join_op,
                       oracle.make_map_op, price, auction, ts)
    assert got == want


def q8_pipeline(mk_join, mk_map, pid, pts, aid, aseller, ats):
    """person ⋈ auction on person id == seller within the same window."""
    mp1 = mk_map(window_end_map_config(n_vals=1))
    persons = mp1.process_batch([pid, pts])            # [id, we]
    mp1.close()
    mp2 = mk_map(window_end_map_config(n_vals=2))
    auctions = mp2.process_batch([aseller, aid, ats])  # [seller, id, we]
    mp2.close()
    j = mk_join(cabi.make_join_config(
        n_keys=1, n_left_vals=0, n_right_vals=1, log2_rows_cap=14,
        instants=64, log2_out_cap=20))
    j.process_batch(j.LEFT, [c.astype(np.int64) for c in persons])
    j.process_batch(j.RIGHT, [c.astype(np.int64) for c in auctions])
    out = j.handle_watermark(U64MAX)  # [person, auction_id, _ts=we]
    j.close()
    return rows_of(out)


def q8_data(n=30_000, seed=11):
    rng = np.random.default_rng(seed)
    t0 = 1_600_000_000 * NS
    pid = rng.integers(0, 500, size=n).astype(np.int64)
    pts = t0 + np.sort(rng.integers(0, 60 * NS, size=n)).astype(np.int64)
    aid = np.arange(n, dtype=np.int64)
    aseller = rng.integers(0, 500, size=n).astype(np.int64)
    ats = t0 + np.sort(rng.integers(0, 60 * NS, size=n)).astype(np.int64)
    return pid, pts, aid, aseller, ats


def np_q8(pid, pts, aid, aseller, ats):
    rows = []
    pwe = pts - pts % W + W
    awe = ats - ats % W + W
    for w in np.unique(pwe):
        people = pid[pwe == w]
        am = awe == w
        for p in np.unique(people):
            cnt = int((people == p).sum())
            for a in aid[am][aseller[am] == p]:
                rows.extend([(int(p), int(a), int(w))] * cnt)
    return sorted(rows)


def test_q8_pipeline_oracle_vs_numpy():
    data = q8_data(n=3000)
    got = q8_pipeline(oracle.make_join_op, oracle.make_map_op, *data)
    assert got == np_q8(*data)
    assert len(got) > 50


@pytest.mark.gpu
def test_q8_pipeline_gpu_vs_oracle():
    from arroyo_amd import gpu
    data = q8_data(n=60_000)
    got = q8_pipeline(gpu.make_join_op, gpu.make_map_op, *data)
    want = q8_pipeline(oracle.make_join_op, oracle.make_map_op, *data)
    assert got == want
    assert len(got) > 1000


def q5_full_pipeline(mod, auction, ts):
    """The COMPLETE nexmark q5 query (nexmark_q5.sql): hop(2s,10s) COUNT
    GROUP BY auction, per-window MAX of the counts, window-joined back on
    AuctionBids.num >= MaxBids.maxn.  The bench measures the first stage
    (BASELINE configs[1]); this composes all three through the product
    operators.  Stage-1 outputs carry _timestamp = window_end - 1, so the
    per-window MAX is a tumble(slide) over them and the window join is an
    exact-_timestamp instant join (n_keys=0 cross product) with the
    num >= maxn filter fused behind it."""
    SL = 2 * NS
    agg = mod.make_op(cabi.make_config(
        width_ns=10 * NS, slide_ns=SL, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=14))
    agg.process_batch([auction, ts])
    a_out = agg.handle_watermark(U64MAX)  # [auction, num, ws, we, _ts]
    agg.close()

    mx = mod.make_op(cabi.make_config(
        width_ns=SL, slide_ns=0, is_tumbling=True, n_keys=0, n_value_cols=1,
        aggs=[(cabi.MAX, 0)], log2_capacity=10, ring_panes=256))
    mx.process_batch([a_out[1].astype(np.int64),
                      a_out[4].astype(np.int64)])
    m_out = mx.handle_watermark(U64MAX)  # [maxn, ws, we, _ts]
    mx.close()

    j = mod.make_join_op(cabi.make_join_config(
        n_keys=0, n_left_vals=2, n_right_vals=1, log2_rows_cap=14,
        instants=256, log2_out_cap=20))
    j.process_batch(j.LEFT, [a_out[0].astype(np.int64),
                             a_out[1].astype(np.int64),
                             a_out[4].astype(np.int64)])
    j.process_batch(j.RIGHT, [m_out[0].astype(np.int64),
                              m_out[3].astype(np.int64)])
    j_out = j.handle_watermark(U64MAX)  # [auction, num, maxn, _ts]
    j.close()

    f = mod.make_map_op(cabi.make_map_config(
        n_in_cols=4, prog=[(cabi.MOP_GE, 1, 2, 4)], out_reg=[0, 1, 3],
        filter_reg=4))
    out = f.process_batch(list(j_out))  # [auction, num, _ts]
    f.close()
    return rows_of(out)


def np_q5_full(auction, ts):
    rows = {}
    SL, Wd = 2 * NS, 10 * NS
    for a, t in zip(auction, ts):
        b = (t // SL) * SL
        for k in range(5):
            ws = b - k * SL
            rows.setdefault(ws, {})
            rows[ws][a] = rows[ws].get(a, 0) + 1
    out = []
    for ws, counts in rows.items():
        mx = max(counts.values())
        for a, n in counts.items():
            if n >= mx:
                out.append((int(a), int(n), int(ws + Wd - 1)))
    return sorted(out)


def q5_data(n=30_000, seed=17):
    rng = np.random.default_rng(seed)
    t0 = 1_600_000_000 * NS
    # hot-auction skew as nexmark sends it
    hot = rng.random(n) < 0.5
    auction = np.where(hot, 1007,
                       1000 + rng.integers(0, 500, size=n)).astype(np.int64)
    ts = t0 + np.sort(rng.integers(0, 60 * NS, size=n)).astype(np.int64)
    return auction, ts


def test_q5_full_pipeline_oracle_vs_numpy():
    auction, ts = q5_data()
    got = q5_full_pipeline(oracle, auction, ts)
    assert got == np_q5_full(auction, ts)
    assert len(got) > 20


@pytest.mark.gpu
def test_q5_full_pipeline_gpu_vs_oracle():
    auction, ts = q5_data(seed=19)
    from arroyo_amd import gpu
    got = q5_full_pipeline(gpu, auction, ts)
    want = q5_full_pipeline(oracle, auction, ts)
    assert got == want
    assert got == np_q5_full(auction, ts)
