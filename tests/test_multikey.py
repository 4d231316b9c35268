"""Multi-column keys (n_keys 2..4) through the window path (ArroyoSchema
key_indices, crates/arroyo-rpc/src/df.rs:24-30: key columns first).

Oracle keys on exact tuples; the HIP path normalizes each composite key to
a dictionary id on device (k_dict_encode — the analogue of the reference's
Arrow-Row key conversion, expiring_time_key_map.rs:1008-1049) and expands
ids back to the original columns at emission.  Pinned against a numpy
groupby restatement and GPU-vs-oracle."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from arroyo_amd.pipeline import (NS, batches_from_columns, concat_outputs,
                                 run_stream)

T0 = 1_600_000_000 * NS


def gen(n=150_000, seed=5, nk=2):
    rng = np.random.default_rng(seed)
    keys = [rng.integers(0, c, size=n).astype(np.int64)
            for c in (7, 5, 3, 11)[:nk]]
    # include negative key values and the -1 sentinel in one column
    keys[0] = keys[0] - 3
    val = rng.integers(0, 1000, size=n).astype(np.int64)
    ts = T0 + (np.arange(n, dtype=np.int64) * NS) // 25_000
    return keys, val, ts


def cfg(nk, **kw):
    base = dict(width_ns=4 * NS, slide_ns=2 * NS, n_keys=nk, n_value_cols=1,
                aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MIN, 0)],
                log2_capacity=13, ring_panes=16)
    base.update(kw)
    return cabi.make_config(**base)


def run_op(op, keys, val, ts):
    outs = run_stream(op, batches_from_columns([*keys, val, ts], 8192), NS)
    got = concat_outputs(outs)
    op.close()
    return got


def rows_of(cols, nk):
    # [k0..k(nk-1), count, sum, min, ws, we, _ts]
    return {tuple(int(c[r]) for c in cols): None or
            tuple(int(c[r]) for c in cols)
            for r in range(len(cols[0]))}


def numpy_reference(keys, val, ts, nk):
    """Sliding hop(2s,4s): each row lands in ceil(width/slide)=2 windows."""
    want = {}
    bins = (ts // (2 * NS)) * 2 * NS
    import collections
    acc = collections.defaultdict(lambda: [0, 0, None])
    for r in range(len(val)):
        kt = tuple(int(k[r]) for k in keys)
        for ws in (int(bins[r]) - 2 * NS, int(bins[r])):
            # window [ws, ws+4s) contains the row iff ws <= ts < ws+4s
            if not (ws <= ts[r] < ws + 4 * NS):
                continue
            e = acc[kt + (ws,)]
            e[0] += 1
            e[1] += int(val[r])
            e[2] = int(val[r]) if e[2] is None else min(e[2], int(val[r]))
    for k, (cnt, sm, mn) in acc.items():
        want[k[:-1] + (cnt, sm, mn, k[-1], k[-1] + 4 * NS,
                       k[-1] + 4 * NS - 1)] = 1
    return set(want)


@pytest.mark.parametrize("nk", [2, 3, 4])
def test_oracle_multikey_matches_numpy(nk):
    keys, val, ts = gen(nk=nk)
    got = run_op(oracle.make_op(cfg(nk)), keys, val, ts)
    assert got is not None
    got_rows = {tuple(int(c[r]) for c in got) for r in range(len(got[0]))}
    assert got_rows == numpy_reference(keys, val, ts, nk)


@pytest.mark.gpu
@pytest.mark.parametrize("nk", [2, 4])
def test_gpu_multikey_matches_oracle(nk):
    from arroyo_amd import gpu
    keys, val, ts = gen(seed=6, nk=nk)
    got = run_op(gpu.make_op(cfg(nk)), keys, val, ts)
    want = run_op(oracle.make_op(cfg(nk)), keys, val, ts)
    g = {tuple(int(c[r]) for c in got) for r in range(len(got[0]))}
    w = {tuple(int(c[r]) for c in want) for r in range(len(want[0]))}
    assert g == w


@pytest.mark.gpu
def test_gpu_multikey_checkpoint_roundtrip():
    """Drain raw states (k key columns first), restore into a fresh op,
    finish the stream: outputs equal the uninterrupted run."""
    from arroyo_amd import gpu
    keys, val, ts = gen(seed=8, nk=2)
    half = len(val) // 2
    c = dict(nk=2)

    op1 = gpu.make_op(cfg(**c))
    outs1 = run_stream(op1, batches_from_columns(
        [*(k[:half] for k in keys), val[:half], ts[:half]], 8192), NS,
        final_watermark=False)
    drained = op1.checkpoint_drain()
    wm1 = int(ts[half - 1]) - NS
    op1.close()

    op2 = gpu.make_op(cfg(**c))
    op2.restore(drained, watermark=wm1)
    outs2 = run_stream(op2, batches_from_columns(
        [*(k[half:] for k in keys), val[half:], ts[half:]], 8192), NS)
    got = concat_outputs(outs1 + outs2)
    op2.close()

    want = run_op(oracle.make_op(cfg(**c)), keys, val, ts)
    g = {tuple(int(col[r]) for col in got) for r in range(len(got[0]))}
    w = {tuple(int(col[r]) for col in want) for r in range(len(want[0]))}
    assert g == w


@pytest.mark.gpu
def test_gpu_multikey_f64_values_match_oracle():
    """Composite keys AND f64 value columns together: the device key
    dictionary and the f64 state encodings interact nowhere, which this
    pins (SUM/MIN/AVG over f64 bit-pattern values, 2 key columns)."""
    from arroyo_amd import gpu

    rng = np.random.default_rng(23)
    n = 120_000
    keys = [rng.integers(0, 7, size=n).astype(np.int64) - 3,
            rng.integers(0, 5, size=n).astype(np.int64)]
    fval = rng.normal(0.0, 100.0, size=n)
    val = fval.view(np.int64).copy()
    ts = T0 + (np.arange(n, dtype=np.int64) * NS) // 25_000
    kw = dict(width_ns=4 * NS, slide_ns=2 * NS, n_keys=2, n_value_cols=1,
              aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MIN, 0),
                    (cabi.AVG, 0)],
              val_is_f64=(0,), log2_capacity=13, ring_panes=16)
    g = gpu.make_op(cabi.make_config(**kw))
    got = run_op(g, keys, val, ts)
    o = oracle.make_op(cabi.make_config(**kw))
    want = run_op(o, keys, val, ts)
    assert got is not None and want is not None
    assert len(got[0]) == len(want[0])
    # [k0, k1, count, sum(f64), min(f64), avg(f64), ws, we, ts]
    ints = [0, 1, 2, 6, 7, 8]
    gi = np.lexsort(tuple(got[i] for i in reversed(ints)))
    wi = np.lexsort(tuple(want[i] for i in reversed(ints)))
    for i in ints:
        assert np.array_equal(got[i][gi], want[i][wi]), i
    for i in (3, 4, 5):
        np.testing.assert_allclose(got[i][gi], want[i][wi], rtol=1e-9)


@pytest.mark.gpu
def test_checkpoint_under_armed_epoch():
    """checkpoint_drain while an epoch is armed must drain the correct
    open-pane state; the armed epoch folds afterwards with unchanged
    emissions (the snapshot predates the checkpoint, which mutates no
    pane)."""
    from arroyo_amd import gpu

    keys, val, ts = gen(n=60_000, nk=2)
    kw = cfg(2)
    g = gpu.make_op(kw)
    batches = batches_from_columns([*keys, val, ts], 8192)
    half = len(batches) // 2
    outs = []
    for b in batches[:half]:
        g.process_batch(b)
    wm = int(batches[half - 1][-1][-1]) - NS
    g.mark_epoch()
    g.set_filter_watermark(wm)
    drained = g.checkpoint_drain()          # with the epoch still armed
    assert drained is not None
    out = g.handle_watermarks_epoch([wm])   # fold after the drain
    if out and len(out[0]):
        outs.append(out)
    for b in batches[half:]:
        g.process_batch(b)
        w2 = int(b[-1][-1]) - NS
        if w2 > wm:
            out = g.handle_watermark(w2)
            wm = w2
            if out and len(out[0]):
                outs.append(out)
    got = concat_outputs(outs)
    g.close()

    o = oracle.make_op(cfg(2))
    from arroyo_amd.pipeline import WatermarkGen
    outs_o = []
    wm = None
    wmo = int(batches[half - 1][-1][-1]) - NS
    for i, b in enumerate(batches):
        o.process_batch(b)
        if i == half - 1:
            out = o.handle_watermark(wmo)
            wm = wmo
            if out and len(out[0]):
                outs_o.append(out)
        elif i >= half:
            w2 = int(b[-1][-1]) - NS
            if w2 > wm:
                out = o.handle_watermark(w2)
                wm = w2
                if out and len(out[0]):
                    outs_o.append(out)
    want = concat_outputs(outs_o)
    o.close()
    assert (got is None) == (want is None)
    if got is None:
        return
    assert len(got[0]) == len(want[0])
    gi = np.lexsort(tuple(got[::-1]))
    wi = np.lexsort(tuple(want[::-1]))
    for gc, wc in zip(got, want):
        assert np.array_equal(gc[gi], wc[wi])
