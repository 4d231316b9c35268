"""offset_impulse_join.sql as an operator pipeline, pinned to the
reference's golden: the same impulse source read twice with ASYMMETRIC
watermark strategies (plain `WATERMARK FOR timestamp` vs a 10-minute
delay), each through tumble(1s) COUNT GROUP BY counter, instant-joined on
counter.  The join's watermark is the MIN across its inputs
(WatermarkHolder, crates/arroyo-operator/src/context.rs:63-86), so the
delayed side holds every instant back until the final watermark — the
semantics this pin exercises."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, fmt_ts, load_golden, load_inputs

U64MAX = 2**64 - 1
DELAY = 600 * NS


def run_pipeline(mod):
    inp = load_inputs()["impulse"]
    ts = np.array(inp["ts"], dtype=np.int64)
    counter = np.array(inp["counter"], dtype=np.int64)

    def win():
        return mod.make_op(cabi.make_config(
            width_ns=NS, slide_ns=0, is_tumbling=True, n_keys=1,
            n_value_cols=0, aggs=[(cabi.COUNT, -1)], log2_capacity=10,
            ring_panes=256))

    wa, wb = win(), win()
    join = mod.make_join_op(cabi.make_join_config(
        n_keys=1, n_left_vals=1, n_right_vals=1, instants=256))

    outs = []

    def feed_join(side, w_out):
        # window rows [counter, cnt, ws, we, ts] -> join [counter, ws, ts]
        if w_out is None or len(w_out[0]) == 0:
            return
        join.process_batch(side, [w_out[0], w_out[2], w_out[4]])

    def advance(wm_a, wm_b):
        feed_join(0, wa.handle_watermark(wm_a))
        feed_join(1, wb.handle_watermark(wm_b))
        out = join.handle_watermark(min(wm_a, wm_b))
        if out is not None and len(out[0]):
            outs.append(out)

    n = len(ts)
    for lo in range(0, n, 32):
        sl = slice(lo, min(lo + 32, n))
        wa.process_batch([counter[sl], ts[sl]])
        wb.process_batch([counter[sl], ts[sl]])
        hi = int(ts[sl].max())
        # side A: watermark = timestamp; side B: timestamp - 10 minutes
        advance(hi, max(hi - DELAY, 0))
    advance(U64MAX, U64MAX)
    wa.close()
    wb.close()
    join.close()

    got = []
    for out in outs:
        key, ws_l, ws_r, _ts = out
        got += [{"counter": int(k), "start": fmt_ts(s)}
                for k, s in zip(key, ws_l)]
    assert_rows_match(got, load_golden("offset_impulse_join"))


def test_offset_impulse_join_oracle():
    run_pipeline(oracle)


@pytest.mark.gpu
def test_offset_impulse_join_gpu():
    from arroyo_amd import gpu
    run_pipeline(gpu)
