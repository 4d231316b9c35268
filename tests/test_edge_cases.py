"""Edge cases across every operator family: empty batches, single rows,
watermarks with no data, and empty-output fires — mirroring the reference's
own defensive paths (empty-batch warnings, zero-row filters)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi

NS = 10**9
U64MAX = 2**64 - 1
T0 = 1_600_000_000 * NS


def e(): return np.empty(0, dtype=np.int64)


def one(v): return np.array([v], dtype=np.int64)


def factories(gpu_mode):
    if gpu_mode:
        from arroyo_amd import gpu as m
    else:
        m = oracle
    return m


@pytest.fixture(params=[False])
def mod(request):
    return factories(request.param)


def gpu_mod():
    return factories(True)


def check_window_edges(m):
    op = m.make_op(cabi.make_config(width_ns=10 * NS, slide_ns=2 * NS,
                                    aggs=[(cabi.COUNT, -1)], n_keys=1,
                                    log2_capacity=12))
    op.process_batch([e(), e()])                     # empty batch
    out = op.handle_watermark(T0)                    # watermark, no data
    assert out is None or len(out[0]) == 0
    op.process_batch([one(5), one(T0 + 20 * NS)])    # single row
    out = op.handle_watermark(U64MAX)
    rows = [tuple(int(c[i]) for c in out) for i in range(len(out[0]))]
    # 5 sliding windows cover the single row's bin
    assert len(rows) == 5 and all(r[0] == 5 and r[1] == 1 for r in rows)
    drained = op.checkpoint_drain()
    assert len(drained[0]) == 0                      # everything fired
    op.close()


def check_join_edges(m):
    j = m.make_join_op(cabi.make_join_config(n_keys=1, n_left_vals=0,
                                             n_right_vals=0))
    j.process_batch(j.LEFT, [e(), e()])
    out = j.handle_watermark(T0)
    assert out is None or len(out[0]) == 0
    j.process_batch(j.LEFT, [one(1), one(T0 + NS)])  # left only: no match
    out = j.handle_watermark(U64MAX)
    assert len(out[0]) == 0
    j.close()


def check_session_edges(m):
    sp = m.make_session_op(cabi.make_session_config(
        2 * NS, [(cabi.COUNT, -1)], n_keys=1))
    sp.process_batch([e(), e()])
    out = sp.handle_watermark(T0)
    assert len(out[0]) == 0
    sp.process_batch([one(9), one(T0 + NS)])
    out = sp.handle_watermark(U64MAX)
    assert [int(c[0]) for c in out] == [9, 1, T0 + NS, T0 + 3 * NS,
                                        T0 + 3 * NS - 1]
    drained = sp.checkpoint_drain()
    assert len(drained[0]) == 0
    sp.close()


def check_expjoin_edges(m):
    ej = m.make_expjoin_op(cabi.make_expjoin_config(NS))
    out = ej.process_batch(ej.LEFT, [e(), e()])
    assert len(out[0]) == 0
    out = ej.process_batch(ej.RIGHT, [one(3), one(T0)])  # nothing stored
    assert len(out[0]) == 0
    ej.close()


def check_updagg_edges(m):
    ua = m.make_updagg_op(cabi.make_updagg_config([(cabi.COUNT, -1)],
                                                  n_keys=1))
    ua.process_batch([e(), e()])
    assert len(ua.flush()[0]) == 0                   # nothing changed
    assert len(ua.flush()[0]) == 0                   # idempotent
    ua.close()


def check_windowfn_edges(m):
    wf = m.make_windowfn_op(cabi.make_windowfn_config(
        n_cols=2, part_col=-1, order=[(0, False)], limit=1))
    wf.process_batch([e(), e()])
    out = wf.handle_watermark(U64MAX)
    assert len(out[0]) == 0
    wf.close()


def check_map_edges(m):
    mp = m.make_map_op(cabi.make_map_config(
        n_in_cols=2, prog=[(cabi.MOP_ADD, 0, 1, 2)], out_reg=[2]))
    out = mp.process_batch([e(), e()])
    assert len(out[0]) == 0
    mp.close()


def check_watermark_idempotence(m):
    """Re-delivering the same watermark (WatermarkHolder can forward the
    min repeatedly) fires nothing new, and a REGRESSED watermark is a
    no-op: already-fired panes never re-fire or error."""
    op = m.make_op(cabi.make_config(width_ns=10 * NS, slide_ns=2 * NS,
                                    aggs=[(cabi.COUNT, -1)], n_keys=1,
                                    log2_capacity=12))
    op.process_batch([one(7), one(T0 + 20 * NS)])
    out = op.handle_watermark(T0 + 40 * NS)
    assert len(out[0]) == 5
    for wm in (T0 + 40 * NS, T0 + 10 * NS):  # repeat, then regress
        out = op.handle_watermark(wm)
        assert out is None or len(out[0]) == 0
    op.close()

    sp = m.make_session_op(cabi.make_session_config(
        2 * NS, [(cabi.COUNT, -1)], n_keys=1))
    sp.process_batch([one(9), one(T0 + NS)])
    out = sp.handle_watermark(T0 + 60 * NS)
    assert len(out[0]) == 1
    for wm in (T0 + 60 * NS, T0 + 5 * NS):
        out = sp.handle_watermark(wm)
        assert out is None or len(out[0]) == 0
    sp.close()


ALL = [check_window_edges, check_join_edges, check_session_edges,
       check_expjoin_edges, check_updagg_edges, check_windowfn_edges,
       check_map_edges, check_watermark_idempotence]


@pytest.mark.parametrize("check", ALL, ids=lambda f: f.__name__)
def test_oracle_edges(check):
    check(oracle)


@pytest.mark.gpu
@pytest.mark.parametrize("check", ALL, ids=lambda f: f.__name__)
def test_gpu_edges(check):
    from arroyo_amd import gpu
    check(gpu)
