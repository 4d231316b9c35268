"""windowed_outer_join.sql as an operator pipeline, pinned to the
reference's golden (crates/arroyo-sql-testing/golden_outputs/
windowed_outer_join.json):

    tumble(1h) COUNT DISTINCT driver over dropoffs
      FULL OUTER JOIN
    tumble(1h) COUNT DISTINCT driver over pickups
      ON dropoffs.window.start = pickups.window.start

Each side composes COUNT DISTINCT as keyed-count -> unkeyed-count (stage A
emits one row per (driver, window)); the FULL OUTER instant join is keyed
on the window start and pads the absent side, mapping its presence flags
to the golden's SQL nulls (the last golden row is a dropoffs-only hour
with pickups = null)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, fmt_ts, load_golden, load_inputs

U64MAX = 2**64 - 1
HOUR = 3600 * NS


def run_pipeline(mod):
    inp = load_inputs()["cars"]
    ts = np.array(inp["ts"], dtype=np.int64)
    driver = np.array(inp["driver_id"], dtype=np.int64)
    etype = np.array(inp["event_type_id"], dtype=np.int64)
    dropoff = inp["event_type_dict"].index("dropoff")
    pickup = inp["event_type_dict"].index("pickup")

    def win_config(keyed):
        return cabi.make_config(
            width_ns=HOUR, slide_ns=0, is_tumbling=True,
            n_keys=1 if keyed else 0, n_value_cols=0,
            aggs=[(cabi.COUNT, -1)], log2_capacity=12 if keyed else 10)

    sides = {}
    for name, want_et in (("dropoffs", dropoff), ("pickups", pickup)):
        sides[name] = {
            # source filter on event_type, projecting [driver, ts]
            "f": mod.make_map_op(cabi.make_map_config(
                n_in_cols=3,
                prog=[(cabi.MOP_CONST, 0, 0, 3, want_et),
                      (cabi.MOP_EQ, 1, 3, 4)],
                out_reg=[0, 2], filter_reg=4)),
            "wa": mod.make_op(win_config(keyed=True)),
            # stage A out [driver, cnt, ws, we, ts] -> [ts] for stage B
            "fp": mod.make_map_op(cabi.make_map_config(
                n_in_cols=5, prog=[], out_reg=[4])),
            "wb": mod.make_op(win_config(keyed=False)),
            # stage B out [cnt, ws, we, ts] -> join input [ws, cnt, ts]
            "fj": mod.make_map_op(cabi.make_map_config(
                n_in_cols=4, prog=[], out_reg=[1, 0, 3])),
        }
    join = mod.make_join_op(cabi.make_join_config(
        n_keys=1, n_left_vals=1, n_right_vals=1,
        join_type=cabi.JOIN_FULL))

    from arroyo_amd.pipeline import WatermarkGen
    wg = WatermarkGen(NS)
    outs = []

    def advance(wm):
        for i, name in enumerate(("dropoffs", "pickups")):
            s = sides[name]
            a_out = s["wa"].handle_watermark(wm)
            if a_out is not None and len(a_out[0]):
                s["wb"].process_batch([s["fp"].process_batch(list(a_out))[0]])
            b_out = s["wb"].handle_watermark(wm)
            if b_out is not None and len(b_out[0]):
                join.process_batch(i, list(s["fj"].process_batch(list(b_out))))
        j = join.handle_watermark(wm)
        if j is not None and len(j[0]):
            outs.append(j)

    n = len(ts)
    for lo in range(0, n, 32):
        sl = slice(lo, min(lo + 32, n))
        for name in ("dropoffs", "pickups"):
            s = sides[name]
            cols = s["f"].process_batch([driver[sl], etype[sl], ts[sl]])
            if len(cols[0]):
                s["wa"].process_batch(list(cols))
        wm = wg.on_batch(ts[sl])
        if wm is not None:
            advance(wm)
    advance(U64MAX)
    for s in sides.values():
        for op in s.values():
            op.close()
    join.close()

    got = []
    for out in outs:
        ws, d, p, _ts, lp, rp = out
        got += [{"hour": fmt_ts(w),
                 "drivers": int(dd) if int(l) else None,
                 "pickups": int(pp) if int(r) else None}
                for w, dd, pp, l, r in zip(ws, d, p, lp, rp)]
    assert_rows_match(got, load_golden("windowed_outer_join"))


def test_windowed_outer_join_oracle():
    run_pipeline(oracle)


@pytest.mark.gpu
def test_windowed_outer_join_gpu():
    from arroyo_amd import gpu
    run_pipeline(gpu)
