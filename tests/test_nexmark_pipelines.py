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
