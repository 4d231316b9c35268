"""f64 value columns through the window path (VERDICT round-2 item: at
least one non-i64 value type).  Arrow Float64 columns travel as bit
patterns in the i64 planes (AmdWindowConfig.val_is_f64); SUM/MIN/MAX/AVG
aggregate as doubles and mark their output column f64.

Pinned two ways: the oracle against a direct numpy groupby restatement
(CPU), and the HIP path against the oracle (-m gpu).  MIN/MAX are
bit-exact (order-free comparisons); SUM/AVG carry the documented 1e-9
relative tolerance (float summation order)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from arroyo_amd.pipeline import (NS, batches_from_columns, concat_outputs,
                                 run_stream)

T0 = 1_600_000_000 * NS


def gen(n=200_000, n_keys=40, seed=9):
    rng = np.random.default_rng(seed)
    key = rng.integers(0, n_keys, size=n).astype(np.int64)
    val = rng.normal(100.0, 37.0, size=n)
    ts = T0 + (np.arange(n, dtype=np.int64) * NS) // 20_000
    return key, val, ts


def cfg(**kw):
    base = dict(width_ns=4 * NS, slide_ns=4 * NS, is_tumbling=True,
                n_keys=1, n_value_cols=1,
                aggs=[(cabi.SUM, 0), (cabi.MIN, 0), (cabi.MAX, 0),
                      (cabi.AVG, 0), (cabi.COUNT, -1)],
                val_is_f64=(0,), log2_capacity=14, ring_panes=16)
    base.update(kw)
    return cabi.make_config(**base)


def run_op(op, key, val, ts):
    cols = [key, val.view(np.int64), ts]
    outs = run_stream(op, batches_from_columns(cols, 8192), NS)
    got = concat_outputs(outs)
    op.close()
    return got


def test_oracle_f64_aggs_match_numpy():
    key, val, ts = gen()
    got = run_op(oracle.make_op(cfg()), key, val, ts)
    assert got is not None
    # got: [key, sum(f64), min(f64), max(f64), avg(f64), count, ws, we, _ts]
    assert got[1].dtype == np.float64 and got[4].dtype == np.float64
    bins = (ts // (4 * NS)) * 4 * NS
    rows = {}
    for k in np.unique(key):
        for b in np.unique(bins):
            m = (key == k) & (bins == b)
            if not m.any():
                continue
            rows[(int(k), int(b))] = (val[m].sum(), val[m].min(),
                                      val[m].max(), val[m].mean(),
                                      int(m.sum()))
    got_rows = {}
    for r in range(len(got[0])):
        got_rows[(int(got[0][r]), int(got[6][r]))] = (
            got[1][r], got[2][r], got[3][r], got[4][r], int(got[5][r]))
    # the final watermark flush emits every window
    assert set(got_rows) == set(rows)
    for kb, (s, lo, hi, avg, cnt) in rows.items():
        gs, glo, ghi, gavg, gcnt = got_rows[kb]
        np.testing.assert_allclose(gs, s, rtol=1e-9)
        assert glo == lo and ghi == hi     # min/max exact
        np.testing.assert_allclose(gavg, avg, rtol=1e-9)
        assert gcnt == cnt


@pytest.mark.gpu
def test_gpu_f64_aggs_match_oracle():
    from arroyo_amd import gpu
    key, val, ts = gen(seed=11)
    got = run_op(gpu.make_op(cfg()), key, val, ts)
    want = run_op(oracle.make_op(cfg()), key, val, ts)
    assert got is not None and len(got[0]) == len(want[0])
    def keyed(cols):
        return {(int(cols[0][r]), int(cols[6][r])):
                tuple(cols[c][r] for c in (1, 2, 3, 4, 5))
                for r in range(len(cols[0]))}
    g, w = keyed(got), keyed(want)
    assert set(g) == set(w)
    for kb in w:
        np.testing.assert_allclose(g[kb][0], w[kb][0], rtol=1e-9)  # sum
        assert g[kb][1] == w[kb][1] and g[kb][2] == w[kb][2]  # min/max
        np.testing.assert_allclose(g[kb][3], w[kb][3], rtol=1e-9)  # avg
        assert g[kb][4] == w[kb][4]


@pytest.mark.gpu
def test_gpu_f64_sliding_with_negative_values():
    """Sliding (pane-merge) path with negatives and exact zero: the f64
    order-preserving encodes must survive merge + retirement."""
    from arroyo_amd import gpu
    rng = np.random.default_rng(3)
    n = 120_000
    key = rng.integers(0, 25, size=n).astype(np.int64)
    val = np.where(rng.random(n) < 0.1, 0.0, rng.normal(0.0, 1e6, size=n))
    ts = T0 + (np.arange(n, dtype=np.int64) * NS) // 15_000
    c = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=1,
             aggs=[(cabi.MIN, 0), (cabi.MAX, 0)], val_is_f64=(0,),
             log2_capacity=14, ring_panes=16)
    got = run_op(gpu.make_op(cabi.make_config(**c)), key, val, ts)
    want = run_op(oracle.make_op(cabi.make_config(**c)), key, val, ts)
    def keyed(cols):
        return {(int(cols[0][r]), int(cols[3][r])):
                (cols[1][r], cols[2][r]) for r in range(len(cols[0]))}
    g, w = keyed(got), keyed(want)
    assert g == w   # min/max bit-exact
