"""Session (gap) window tests: the reference's own golden vectors
(session_window.sql / global_session_window.sql over inputs/impulse.json) and
fuzz against an independent numpy restatement.

Reference semantics: crates/arroyo-worker/src/arrow/session_aggregating_window.rs
(per-key maximal runs where each next ts is strictly within prev-max + gap;
a session fires when data_end + gap < watermark; window = [min_ts,
max_ts + gap), _timestamp = end - 1; late rows ts < watermark dropped)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, fmt_ts, load_golden, load_inputs

U64MAX = 2**64 - 1
GAP20 = 20 * NS


def np_sessions(key, ts, gap):
    """Independent restatement: per key, sort ts, split where the next ts is
    >= running-max + gap; emit (key, count, start, end, end-1)."""
    rows = []
    for k in np.unique(key):
        t = np.sort(ts[key == k])
        lo = 0
        for i in range(1, len(t) + 1):
            if i == len(t) or t[i] >= t[i - 1] + gap:
                rows.append((int(k), i - lo, int(t[lo]),
                             int(t[i - 1] + gap), int(t[i - 1] + gap - 1)))
                lo = i
    return sorted(rows)


def rows_of(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return sorted(tuple(int(c[r]) for c in cols) for r in range(len(cols[0])))


def impulse_cols():
    d = load_inputs()["impulse"]
    return (np.array(d["counter"], dtype=np.int64),
            np.array(d["ts"], dtype=np.int64))


def run_session_window_golden(make_op):
    """session_window.sql: SESSION(20s) COUNT(*) GROUP BY user_id where
    user_id = 0 if counter % 10 == 0 else counter."""
    counter, ts = impulse_cols()
    user = np.where(counter % 10 == 0, 0, counter).astype(np.int64)
    op = make_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1, n_value_cols=0))
    op.process_batch([user, ts])
    out = op.handle_watermark(U64MAX)
    op.close()
    got = [{"user_id": int(k), "rows": int(n), "start": fmt_ts(int(s)),
            "end": fmt_ts(int(e))}
           for k, n, s, e, _t in zip(*out)]
    assert_rows_match(got, load_golden("session_window"))


def run_global_session_window_golden(make_op):
    """global_session_window.sql: unkeyed SESSION(20s) COUNT(*)."""
    counter, ts = impulse_cols()
    op = make_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=0, n_value_cols=0))
    op.process_batch([ts])
    out = op.handle_watermark(U64MAX)
    op.close()
    got = [{"rows": int(n), "start": fmt_ts(int(s)), "end": fmt_ts(int(e))}
           for n, s, e, _t in zip(*out)]
    assert_rows_match(got, load_golden("global_session_window"))


def test_session_window_golden_oracle():
    run_session_window_golden(oracle.make_session_op)


def test_global_session_window_golden_oracle():
    run_global_session_window_golden(oracle.make_session_op)


def stream_fuzz(make_op, n=4000, seed=3, gap_s=5, aggs=None, n_value_cols=0,
                **cfg_kw):
    """Stream a key/ts stream through the op with periodic watermarks;
    returns all emitted rows.  Input is watermark-clean (each batch only
    has ts >= previous watermark) so results equal the batch restatement."""
    rng = np.random.default_rng(seed)
    t0 = 1_600_000_000 * NS
    gap = gap_s * NS
    # ragged event times: bursts separated by quiet periods > gap
    ts = t0 + np.cumsum(rng.choice(
        [NS // 10, NS // 2, 2 * NS, 7 * NS], size=n,
        p=[0.55, 0.3, 0.1, 0.05]).astype(np.int64))
    key = rng.integers(0, 37, size=n).astype(np.int64)
    vals = [rng.integers(-50, 50, size=n).astype(np.int64)
            for _ in range(n_value_cols)]
    aggs = aggs or [(cabi.COUNT, -1)]
    # sparse watermark cadence (one per ~n/7 rows) leaves many unfired
    # sessions per key live at once: size the inline session store for it
    cfg_kw.setdefault("max_sessions", 256)
    cfg_kw.setdefault("log2_capacity", 10)
    op = make_op(cabi.make_session_config(
        gap, aggs, n_keys=1, n_value_cols=n_value_cols, **cfg_kw))
    got = []
    step = n // 7
    for b in range(0, n, step):
        sl = slice(b, min(b + step, n))
        cols = [key[sl]] + [v[sl] for v in vals] + [ts[sl]]
        op.process_batch(cols)
        wm = int(ts[sl][len(ts[sl]) // 2])  # mid-batch watermark, no late rows
        got += rows_of(op.handle_watermark(wm))
    got += rows_of(op.handle_watermark(U64MAX))
    op.close()
    return key, ts, vals, got


@pytest.mark.parametrize("seed", [3, 17, 71])
def test_session_oracle_vs_numpy_fuzz(seed):
    key, ts, _vals, got = stream_fuzz(oracle.make_session_op, seed=seed)
    want = [(k, n, s, e, t)
            for k, n, s, e, t in np_sessions(key, ts, 5 * NS)]
    assert sorted(got) == sorted(want)


def test_session_oracle_multi_agg():
    """SUM/MIN/MAX over a value column alongside COUNT."""
    key, ts, vals, got = stream_fuzz(
        oracle.make_session_op, n_value_cols=1,
        aggs=[(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MIN, 0), (cabi.MAX, 0)])
    v = vals[0]
    want = []
    for k, n, s, e, t in np_sessions(key, ts, 5 * NS):
        m = (key == k) & (ts >= s) & (ts < e)
        want.append((k, n, int(v[m].sum()), int(v[m].min()), int(v[m].max()),
                     s, e, t))
    assert sorted(got) == sorted(want)


def test_session_oracle_drops_late_rows():
    op = oracle.make_session_op(cabi.make_session_config(
        2 * NS, [(cabi.COUNT, -1)], n_keys=1))
    t0 = 1_600_000_000 * NS
    op.process_batch([np.array([7], dtype=np.int64),
                      np.array([t0 + 10 * NS], dtype=np.int64)])
    op.handle_watermark(t0 + 10 * NS)
    # ts < watermark: silently dropped (session_aggregating_window.rs:849-874)
    op.process_batch([np.array([7], dtype=np.int64),
                      np.array([t0 + 9 * NS], dtype=np.int64)])
    out = op.handle_watermark(U64MAX)
    rows = rows_of(out)
    assert rows == [(7, 1, t0 + 10 * NS, t0 + 12 * NS, t0 + 12 * NS - 1)]
    op.close()


def test_session_oracle_checkpoint_roundtrip():
    counter, ts = impulse_cols()
    user = np.where(counter % 10 == 0, 0, counter).astype(np.int64)
    mid = len(user) // 2

    a = oracle.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    a.process_batch([user[:mid], ts[:mid]])
    drained = a.checkpoint_drain()
    a.close()
    assert len(drained[0]) == mid

    b = oracle.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    b.restore(drained)
    b.process_batch([user[mid:], ts[mid:]])
    got = rows_of(b.handle_watermark(U64MAX))
    b.close()

    c = oracle.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    c.process_batch([user, ts])
    want = rows_of(c.handle_watermark(U64MAX))
    c.close()
    assert got == want


# ---------------------------------------------------------------- GPU parity


@pytest.mark.gpu
def test_session_window_golden_gpu():
    from arroyo_amd import gpu
    run_session_window_golden(gpu.make_session_op)


@pytest.mark.gpu
def test_global_session_window_golden_gpu():
    from arroyo_amd import gpu
    run_global_session_window_golden(gpu.make_session_op)


@pytest.mark.gpu
def test_session_gpu_vs_oracle_fuzz():
    """Streamed fuzz: HIP session path vs the CPU oracle, bit-exact."""
    from arroyo_amd import gpu
    _k, _t, _v, got = stream_fuzz(gpu.make_session_op, n=20000, seed=9)
    _k2, _t2, _v2, want = stream_fuzz(oracle.make_session_op, n=20000,
                                      seed=9)
    assert sorted(got) == sorted(want)
    assert len(want) > 50


@pytest.mark.gpu
def test_session_gpu_multi_agg_vs_oracle():
    from arroyo_amd import gpu
    aggs = [(cabi.COUNT, -1), (cabi.SUM, 0), (cabi.MIN, 0), (cabi.MAX, 0)]
    _k, _t, _v, got = stream_fuzz(gpu.make_session_op, n=8000, seed=4,
                                  aggs=aggs, n_value_cols=1)
    _k2, _t2, _v2, want = stream_fuzz(oracle.make_session_op, n=8000, seed=4,
                                      aggs=aggs, n_value_cols=1)
    assert sorted(got) == sorted(want)


@pytest.mark.gpu
def test_session_gpu_wide_batch_split():
    """A single batch spanning many gaps must still sessionize correctly
    (the host splits it into gap/2 buckets)."""
    from arroyo_amd import gpu
    rng = np.random.default_rng(31)
    t0 = 1_600_000_000 * NS
    gap = 2 * NS
    n = 3000
    ts = t0 + np.sort(rng.integers(0, 600 * NS, size=n)).astype(np.int64)
    key = rng.integers(0, 5, size=n).astype(np.int64)
    op = gpu.make_session_op(cabi.make_session_config(
        gap, [(cabi.COUNT, -1)], n_keys=1, max_sessions=512,
        log2_capacity=8))
    op.process_batch([key, ts])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    want = [(k, c, s, e, t) for k, c, s, e, t in np_sessions(key, ts, gap)]
    assert got == sorted(want)


@pytest.mark.gpu
def test_session_gpu_checkpoint_roundtrip():
    from arroyo_amd import gpu
    counter, ts = impulse_cols()
    user = np.where(counter % 10 == 0, 0, counter).astype(np.int64)
    mid = len(user) // 2

    a = gpu.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    a.process_batch([user[:mid], ts[:mid]])
    drained = a.checkpoint_drain()
    a.close()
    assert len(drained[0]) > 0

    b = gpu.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    b.restore(drained)
    b.process_batch([user[mid:], ts[mid:]])
    got = rows_of(b.handle_watermark(U64MAX))
    b.close()

    c = oracle.make_session_op(cabi.make_session_config(
        GAP20, [(cabi.COUNT, -1)], n_keys=1))
    c.process_batch([user, ts])
    want = rows_of(c.handle_watermark(U64MAX))
    c.close()
    assert got == want


@pytest.mark.gpu
def test_session_gpu_key_minus_one():
    """key == -1 collides with the empty-slot sentinel: special slot."""
    from arroyo_amd import gpu
    t0 = 1_600_000_000 * NS
    key = np.array([-1, -1, 3, -1], dtype=np.int64)
    ts = np.array([t0, t0 + NS, t0 + NS, t0 + 10 * NS], dtype=np.int64)
    op = gpu.make_session_op(cabi.make_session_config(
        2 * NS, [(cabi.COUNT, -1)], n_keys=1))
    op.process_batch([key, ts])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    want = [(-1, 2, t0, t0 + 3 * NS, t0 + 3 * NS - 1),
            (-1, 1, t0 + 10 * NS, t0 + 12 * NS, t0 + 12 * NS - 1),
            (3, 1, t0 + NS, t0 + 3 * NS, t0 + 3 * NS - 1)]
    assert got == sorted(want)


def zipf_keys(rng, n, key_space=10_000_000, s=1.0):
    """Bounded Zipf(s) over `key_space` ranks (BASELINE.json config 5)."""
    ranks = np.arange(1, key_space + 1, dtype=np.float64)
    p = 1.0 / ranks ** s
    p /= p.sum()
    return rng.choice(key_space, size=n, p=p).astype(np.int64)


@pytest.mark.gpu
def test_session_config5_zipf_checkpoint_under_load():
    """BASELINE.json config 5 verbatim at single-GPU scale: session-window
    COUNT DISTINCT over a Zipf stream drawn from a 10M-key space, with a
    checkpoint (sessions + distinct-value sets) drained and restored
    mid-stream while data keeps flowing; results must equal the oracle's
    uninterrupted run."""
    from arroyo_amd import gpu

    rng = np.random.default_rng(55)
    n = 1_000_000
    gap = 5 * NS
    t0 = 1_600_000_000 * NS
    key = zipf_keys(rng, n)
    item = rng.integers(0, 1000, size=n).astype(np.int64)
    # ~64K rows per second of event time
    ts = t0 + (np.arange(n, dtype=np.int64) * NS) // 65536
    batch = 65536

    def cfg():
        return cabi.make_session_config(
            gap, [(cabi.COUNT_DISTINCT, 0)], n_keys=1, n_value_cols=1,
            log2_capacity=21, max_sessions=8, log2_batch_capacity=17,
            log2_out_cap=22, log2_distinct=11, log2_cd_regions=21)

    # GPU run with a mid-stream checkpoint: drain into a fresh op
    op = gpu.make_session_op(cfg())
    got = []
    wm = 0
    for b in range(0, n, batch):
        cols = [key[b:b + batch], item[b:b + batch], ts[b:b + batch]]
        op.process_batch(cols)
        wm = int(ts[min(b + batch, n) - 1]) - NS
        got += rows_of(op.handle_watermark(wm))
        if b == (n // batch // 2) * batch:
            drained = op.checkpoint_drain()
            values = op.drain_values()
            op.close()
            op = gpu.make_session_op(cfg())
            op.restore(drained)
            op.restore_values(values)
    got += rows_of(op.handle_watermark(U64MAX))
    op.close()

    # oracle, uninterrupted
    o = oracle.make_session_op(cfg())
    want = []
    for b in range(0, n, batch):
        o.process_batch([key[b:b + batch], item[b:b + batch],
                         ts[b:b + batch]])
        want += rows_of(o.handle_watermark(
            int(ts[min(b + batch, n) - 1]) - NS))
    want += rows_of(o.handle_watermark(U64MAX))
    o.close()

    assert len(got) == len(want)
    assert sorted(got) == sorted(want)
    assert len(want) > 100_000  # cold Zipf keys produce many 1-row sessions


def np_session_count_distinct(key, val, ts, gap):
    rows = []
    for k in np.unique(key):
        m = key == k
        t = ts[m]
        v = val[m]
        o = np.argsort(t, kind="stable")
        t, v = t[o], v[o]
        lo = 0
        for i in range(1, len(t) + 1):
            if i == len(t) or t[i] >= t[i - 1] + gap:
                rows.append((int(k), len(np.unique(v[lo:i])), int(t[lo]),
                             int(t[i - 1] + gap), int(t[i - 1] + gap - 1)))
                lo = i
    return sorted(rows)


def cd_stream(n=20_000, seed=21, gap_s=5):
    rng = np.random.default_rng(seed)
    t0 = 1_600_000_000 * NS
    ts = t0 + np.cumsum(rng.choice(
        [NS // 10, NS // 2, 2 * NS, 7 * NS], size=n,
        p=[0.55, 0.3, 0.1, 0.05]).astype(np.int64))
    key = rng.integers(0, 29, size=n).astype(np.int64)
    val = rng.integers(0, 200, size=n).astype(np.int64)
    return key, val, ts, gap_s * NS


def run_cd(make_op, key, val, ts, gap, checkpoint_at=None, **cfg_kw):
    # sparse watermarks (one per ~n/7 rows) leave many live sessions per key
    cfg_kw.setdefault("max_sessions", 1024)
    cfg_kw.setdefault("log2_capacity", 10)
    op = make_op(cabi.make_session_config(
        gap, [(cabi.COUNT_DISTINCT, 0)], n_keys=1, n_value_cols=1, **cfg_kw))
    got = []
    n = len(ts)
    step = n // 7
    for i, b in enumerate(range(0, n, step)):
        sl = slice(b, min(b + step, n))
        op.process_batch([key[sl], val[sl], ts[sl]])
        got += rows_of(op.handle_watermark(int(ts[sl][len(ts[sl]) // 2])))
        if checkpoint_at is not None and i == checkpoint_at:
            drained = op.checkpoint_drain()
            values = op.drain_values()
            op.close()
            op = make_op(cabi.make_session_config(
                gap, [(cabi.COUNT_DISTINCT, 0)], n_keys=1, n_value_cols=1,
                **cfg_kw))
            op.restore(drained)
            op.restore_values(values)
    got += rows_of(op.handle_watermark(U64MAX))
    op.close()
    return got


def test_session_count_distinct_oracle_vs_numpy():
    key, val, ts, gap = cd_stream()
    op = oracle.make_session_op(cabi.make_session_config(
        gap, [(cabi.COUNT_DISTINCT, 0)], n_keys=1, n_value_cols=1))
    got = []
    n = len(ts)
    step = n // 7
    for b in range(0, n, step):
        sl = slice(b, min(b + step, n))
        op.process_batch([key[sl], val[sl], ts[sl]])
        got += rows_of(op.handle_watermark(int(ts[sl][len(ts[sl]) // 2])))
    got += rows_of(op.handle_watermark(U64MAX))
    op.close()
    assert sorted(got) == np_session_count_distinct(key, val, ts, gap)


@pytest.mark.gpu
def test_session_count_distinct_gpu_vs_numpy():
    from arroyo_amd import gpu
    key, val, ts, gap = cd_stream(n=60_000)
    got = run_cd(gpu.make_session_op, key, val, ts, gap)
    assert sorted(got) == np_session_count_distinct(key, val, ts, gap)


@pytest.mark.gpu
def test_session_count_distinct_gpu_checkpoint_roundtrip():
    from arroyo_amd import gpu
    key, val, ts, gap = cd_stream(n=30_000, seed=31)
    got = run_cd(gpu.make_session_op, key, val, ts, gap, checkpoint_at=3)
    assert sorted(got) == np_session_count_distinct(key, val, ts, gap)
