"""Instant-join oracle tests (CPU): fuzz against an independent numpy
restatement, and the reference's own windowed_inner_join golden vector
reproduced by composing window-aggregate ops with the join op.

Reference semantics: crates/arroyo-worker/src/arrow/instant_join.rs
(per-exact-timestamp hash join, fired in timestamp order when the watermark
passes the instant)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, fmt_ts, load_golden, load_inputs

HOUR = 3600 * NS
U64MAX = 2**64 - 1


def np_instant_inner_join(l_cols, r_cols, n_keys):
    """Independent restatement: inner equi-join on (instant, key)."""
    lk = l_cols[0] if n_keys else None
    lt = l_cols[-1]
    rk = r_cols[0] if n_keys else None
    rt = r_cols[-1]
    rows = []
    for li in range(len(lt)):
        for ri in range(len(rt)):
            if lt[li] != rt[ri]:
                continue
            if n_keys and lk[li] != rk[ri]:
                continue
            row = []
            if n_keys:
                row.append(int(lk[li]))
            for c in l_cols[n_keys:-1]:
                row.append(int(c[li]))
            for c in r_cols[n_keys:-1]:
                row.append(int(c[ri]))
            row.append(int(lt[li]))
            rows.append(tuple(row))
    return sorted(rows)


def rows_of(cols):
    if cols is None or len(cols[0]) == 0:
        return []
    return sorted(tuple(int(c[r]) for c in cols) for r in range(len(cols[0])))


@pytest.mark.parametrize("n_keys", [0, 1])
def test_join_oracle_vs_numpy_fuzz(n_keys):
    rng = np.random.default_rng(5)
    cfg = cabi.make_join_config(n_keys=n_keys, n_left_vals=1, n_right_vals=2)
    op = oracle.make_join_op(cfg)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(20, dtype=np.int64) * NS

    def gen(n, nv):
        cols = []
        if n_keys:
            cols.append(rng.integers(0, 8, size=n).astype(np.int64))
        for _ in range(nv):
            cols.append(rng.integers(0, 100, size=n).astype(np.int64))
        cols.append(rng.choice(instants, size=n).astype(np.int64))
        return cols

    n_l = 40 if n_keys else 12   # cross product stays small when unkeyed
    left = gen(n_l, 1)
    right = gen(n_l, 2)
    # stream in two chunks with a mid-stream watermark
    mid_wm = int(t0 + 10 * NS)
    lmask = left[-1] >= mid_wm
    rmask = right[-1] >= mid_wm
    op.process_batch(op.LEFT, [c[~lmask] for c in left])
    op.process_batch(op.RIGHT, [c[~rmask] for c in right])
    out1 = op.handle_watermark(mid_wm)
    op.process_batch(op.LEFT, [c[lmask] for c in left])
    op.process_batch(op.RIGHT, [c[rmask] for c in right])
    out2 = op.handle_watermark(U64MAX)
    op.close()

    got = rows_of(out1) + rows_of(out2)
    want = np_instant_inner_join(left, right, n_keys)
    assert sorted(got) == want


def test_join_oracle_rejects_late_rows():
    cfg = cabi.make_join_config(n_keys=1, n_left_vals=0, n_right_vals=0)
    op = oracle.make_join_op(cfg)
    t0 = 1_600_000_000 * NS
    op.process_batch(op.LEFT, [np.array([1], dtype=np.int64),
                               np.array([t0 + 5 * NS], dtype=np.int64)])
    op.handle_watermark(t0 + 3 * NS)
    with pytest.raises(RuntimeError, match="before the watermark"):
        # instant_join.rs:129-139 panics on pre-watermark data
        op.process_batch(op.LEFT, [np.array([1], dtype=np.int64),
                                   np.array([t0], dtype=np.int64)])
    op.close()


def count_distinct_per_hour(key, ts, make_window_op):
    """COUNT(DISTINCT key) per tumbling hour via composition: keyed tumbling
    COUNT GROUP BY key, then unkeyed tumbling COUNT over the output rows —
    the same partial/final split DataFusion plans for distinct aggregates."""
    op1 = make_window_op(cabi.make_config(
        width_ns=HOUR, slide_ns=0, is_tumbling=True, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=12))
    op1.process_batch([key, ts])
    o1 = op1.handle_watermark(U64MAX)
    op1.close()
    # o1: [driver, count, ws, we, _ts]
    op2 = make_window_op(cabi.make_config(
        width_ns=HOUR, slide_ns=0, is_tumbling=True, n_keys=0, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=12))
    op2.process_batch([o1[-1].astype(np.int64)])
    o2 = op2.handle_watermark(U64MAX)
    op2.close()
    return o2  # [count_distinct, ws, we, _ts]


def run_windowed_inner_join(make_window_op, make_join_op):
    cars = load_inputs()["cars"]
    ts = np.array(cars["ts"], dtype=np.int64)
    driver = np.array(cars["driver_id"], dtype=np.int64)
    etype = np.array(cars["event_type_id"], dtype=np.int64)
    names = cars["event_type_dict"]
    dropoff_id = names.index("dropoff")
    pickup_id = names.index("pickup")

    sides = {}
    for name, eid in (("dropoff", dropoff_id), ("pickup", pickup_id)):
        m = etype == eid
        sides[name] = count_distinct_per_hour(driver[m], ts[m],
                                              make_window_op)

    j = make_join_op(cabi.make_join_config(n_keys=0, n_left_vals=1,
                                           n_right_vals=1,
                                           log2_rows_cap=10, instants=64,
                                           log2_out_cap=12))
    d, p = sides["dropoff"], sides["pickup"]
    j.process_batch(j.LEFT, [d[0].astype(np.int64), d[-1].astype(np.int64)])
    j.process_batch(j.RIGHT, [p[0].astype(np.int64), p[-1].astype(np.int64)])
    out = j.handle_watermark(U64MAX)
    j.close()
    # out: [dropoff_drivers, pickup_drivers, _ts]; hour = window start
    got = [{"hour": fmt_ts(int(t) + 1 - HOUR), "drivers": int(a),
            "pickups": int(b)}
           for a, b, t in zip(out[0], out[1], out[2])]
    assert_rows_match(got, load_golden("windowed_inner_join"))


def test_windowed_inner_join_golden_oracle():
    """The reference's windowed_inner_join golden vector (tumbling COUNT
    DISTINCT per side, inner join on the window) through the oracle ops."""
    run_windowed_inner_join(oracle.make_op, oracle.make_join_op)


# ---------------------------------------------------------------- GPU parity


def gpu_join(cfg):
    from arroyo_amd import gpu
    return gpu.make_join_op(cfg)


@pytest.mark.gpu
@pytest.mark.parametrize("n_keys", [0, 1])
def test_join_gpu_vs_oracle_fuzz(n_keys):
    """HIP instant-join vs the CPU oracle on identical streamed batches with
    mid-stream watermarks; bit-exact row sets."""
    rng = np.random.default_rng(11)
    kw = dict(n_keys=n_keys, n_left_vals=1, n_right_vals=2,
              log2_rows_cap=12, instants=256, log2_out_cap=22)
    g = gpu_join(cabi.make_join_config(**kw))
    o = oracle.make_join_op(cabi.make_join_config(**kw))
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(50, dtype=np.int64) * NS

    def gen(n, nv):
        cols = []
        if n_keys:
            cols.append(rng.integers(0, 64, size=n).astype(np.int64))
        for _ in range(nv):
            cols.append(rng.integers(0, 10**6, size=n).astype(np.int64))
        cols.append(np.sort(rng.choice(instants, size=n)).astype(np.int64))
        return cols

    n = 5000 if n_keys else 300
    got, want = [], []
    wm = 0
    for step in range(5):
        lo = t0 + step * 10 * NS
        left = gen(n, 1)
        right = gen(n, 2)
        lm = left[-1] >= wm
        rm = right[-1] >= wm
        left = [c[lm] for c in left]
        right = [c[rm] for c in right]
        for op, acc in ((g, got), (o, want)):
            op.process_batch(op.LEFT, left)
            op.process_batch(op.RIGHT, right)
        wm = int(lo + 10 * NS)
        got_o = g.handle_watermark(wm)
        want_o = o.handle_watermark(wm)
        got += rows_of(got_o)
        want += rows_of(want_o)
    got += rows_of(g.handle_watermark(U64MAX))
    want += rows_of(o.handle_watermark(U64MAX))
    g.close()
    o.close()
    assert sorted(got) == sorted(want)
    assert len(want) > 0


@pytest.mark.gpu
def test_windowed_inner_join_golden_gpu():
    """The reference's windowed_inner_join golden vector through the HIP
    window + join ops end to end."""
    from arroyo_amd import gpu
    run_windowed_inner_join(gpu.make_op, gpu.make_join_op)


@pytest.mark.gpu
def test_join_gpu_checkpoint_drain_restore():
    """checkpoint_drain of each side's buffered rows, restored into a fresh
    op, matches an uninterrupted run (instant_join.rs:205-230, :285-303)."""
    rng = np.random.default_rng(7)
    kw = dict(n_keys=1, n_left_vals=1, n_right_vals=1,
              log2_rows_cap=10, instants=64, log2_out_cap=18)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(10, dtype=np.int64) * NS

    def gen(n):
        return [rng.integers(0, 16, size=n).astype(np.int64),
                rng.integers(0, 100, size=n).astype(np.int64),
                rng.choice(instants, size=n).astype(np.int64)]

    left, right = gen(500), gen(500)

    base = gpu_join(cabi.make_join_config(**kw))
    base.process_batch(base.LEFT, left)
    base.process_batch(base.RIGHT, right)
    want = rows_of(base.handle_watermark(U64MAX))
    base.close()

    a = gpu_join(cabi.make_join_config(**kw))
    a.process_batch(a.LEFT, left)
    a.process_batch(a.RIGHT, right)
    ld = a.checkpoint_drain(a.LEFT)
    rd = a.checkpoint_drain(a.RIGHT)
    a.close()
    assert len(ld[0]) == 500 and len(rd[0]) == 500

    b = gpu_join(cabi.make_join_config(**kw))
    b.restore(b.LEFT, ld)
    b.restore(b.RIGHT, rd)
    got = rows_of(b.handle_watermark(U64MAX))
    b.close()
    assert got == want


@pytest.mark.gpu
def test_join_gpu_rejects_late_rows():
    cfg = cabi.make_join_config(n_keys=1, n_left_vals=0, n_right_vals=0)
    op = gpu_join(cfg)
    t0 = 1_600_000_000 * NS
    op.process_batch(op.LEFT, [np.array([1], dtype=np.int64),
                               np.array([t0 + 5 * NS], dtype=np.int64)])
    op.handle_watermark(t0 + 3 * NS)
    op.process_batch(op.LEFT, [np.array([1], dtype=np.int64),
                               np.array([t0], dtype=np.int64)])
    with pytest.raises(RuntimeError, match="before .*watermark"):
        # device-side error is surfaced at the next watermark
        op.handle_watermark(t0 + 4 * NS)
    op.close()


def np_outer_join(lk, lv, lt, rk, rv, rt, join_type):
    """numpy/python restatement of per-instant LEFT/RIGHT/FULL join with
    presence flags (missing side zero-filled)."""
    rows = []
    matched_l = set()
    for j in range(len(rk)):
        hit = False
        for i in range(len(lk)):
            if lt[i] == rt[j] and lk[i] == rk[j]:
                rows.append((int(lk[i]), int(lv[i]), int(rv[j]),
                             int(lt[i]), 1, 1))
                matched_l.add(i)
                hit = True
        if not hit and join_type in (2, 3):
            rows.append((int(rk[j]), 0, int(rv[j]), int(rt[j]), 0, 1))
    if join_type in (1, 3):
        for i in range(len(lk)):
            if i not in matched_l:
                rows.append((int(lk[i]), int(lv[i]), 0, int(lt[i]), 1, 0))
    return sorted(rows)


def outer_join_scenario(make_op, join_type):
    rng = np.random.default_rng(83 + join_type)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(8, dtype=np.int64) * NS
    def side(seed, n):
        r = np.random.default_rng(seed)
        k = r.integers(0, 24, size=n).astype(np.int64)
        v = r.integers(0, 10**6, size=n).astype(np.int64)
        t = np.sort(r.choice(instants, size=n)).astype(np.int64)
        return k, v, t
    lk, lv, lt = side(7 + join_type, 400)
    rk, rv, rt = side(8 + join_type, 350)
    op = make_op(cabi.make_join_config(n_keys=1, n_left_vals=1,
                                       n_right_vals=1, join_type=join_type))
    op.process_batch(op.LEFT, [lk, lv, lt])
    op.process_batch(op.RIGHT, [rk, rv, rt])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    return got, np_outer_join(lk, lv, lt, rk, rv, rt, join_type)


@pytest.mark.parametrize("join_type", [1, 2, 3],
                         ids=["left", "right", "full"])
def test_outer_join_oracle_vs_numpy(join_type):
    """JoinType Left/Right/Full on the instant join (planner plan/join.rs
    passes the SQL join type through to the per-instant HashJoinExec):
    unmatched rows emit once at instant fire with zero-filled values and
    presence flags for the absent side."""
    got, want = outer_join_scenario(oracle.make_join_op, join_type)
    assert got == want
    assert any(r[4] == 0 or r[5] == 0 for r in want)  # pads exercised


def test_outer_join_oracle_one_sided_instants():
    """Instants where one side is entirely empty: FULL emits every row of
    the present side padded; INNER emits nothing."""
    t0 = 1_600_000_000 * NS
    for join_type, expect in ((0, []), (3, [(5, 11, 0, t0, 1, 0)])):
        op = oracle.make_join_op(cabi.make_join_config(
            n_keys=1, n_left_vals=1, n_right_vals=1, join_type=join_type))
        op.process_batch(op.LEFT,
                         [np.array([5], dtype=np.int64),
                          np.array([11], dtype=np.int64),
                          np.array([t0], dtype=np.int64)])
        got = rows_of(op.handle_watermark(U64MAX))
        op.close()
        assert got == expect


def test_outer_join_oracle_window_condition():
    """n_keys=0 (join ON the window itself): cross product when both sides
    present; FULL pads the present side when the other is empty."""
    t0 = 1_600_000_000 * NS
    op = oracle.make_join_op(cabi.make_join_config(
        n_keys=0, n_left_vals=1, n_right_vals=1, join_type=3))
    # instant t0: both sides; instant t0+1s: left only; t0+2s: right only
    op.process_batch(op.LEFT,
                     [np.array([10, 20], dtype=np.int64),
                      np.array([t0, t0 + NS], dtype=np.int64)])
    op.process_batch(op.RIGHT,
                     [np.array([30, 40], dtype=np.int64),
                      np.array([t0, t0 + 2 * NS], dtype=np.int64)])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    assert got == sorted([(10, 30, t0, 1, 1),
                          (20, 0, t0 + NS, 1, 0),
                          (0, 40, t0 + 2 * NS, 0, 1)])


@pytest.mark.gpu
@pytest.mark.parametrize("join_type", [1, 2, 3],
                         ids=["left", "right", "full"])
def test_outer_join_gpu_vs_numpy(join_type):
    from arroyo_amd import gpu
    got, want = outer_join_scenario(gpu.make_join_op, join_type)
    assert got == want


@pytest.mark.gpu
def test_outer_join_gpu_window_condition_and_one_sided():
    from arroyo_amd import gpu
    t0 = 1_600_000_000 * NS
    op = gpu.make_join_op(cabi.make_join_config(
        n_keys=0, n_left_vals=1, n_right_vals=1, join_type=3))
    op.process_batch(op.LEFT,
                     [np.array([10, 20], dtype=np.int64),
                      np.array([t0, t0 + NS], dtype=np.int64)])
    op.process_batch(op.RIGHT,
                     [np.array([30, 40], dtype=np.int64),
                      np.array([t0, t0 + 2 * NS], dtype=np.int64)])
    got = rows_of(op.handle_watermark(U64MAX))
    op.close()
    assert got == sorted([(10, 30, t0, 1, 1),
                          (20, 0, t0 + NS, 1, 0),
                          (0, 40, t0 + 2 * NS, 0, 1)])


@pytest.mark.parametrize("join_type", [1, 2, 3],
                         ids=["left", "right", "full"])
def test_outer_join_oracle_multiwatermark_fuzz(join_type):
    """Outer joins across several watermarks with skewed sides (some
    instants left-only or right-only): each instant fires exactly once,
    with pads decided by that instant's final contents."""
    rng = np.random.default_rng(91 + join_type)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(30, dtype=np.int64) * NS
    # left skews early instants, right skews late: many one-sided instants
    lk = rng.integers(0, 12, size=300).astype(np.int64)
    lv = rng.integers(0, 10**6, size=300).astype(np.int64)
    lt = rng.choice(instants[:20], size=300).astype(np.int64)
    rk = rng.integers(0, 12, size=300).astype(np.int64)
    rv = rng.integers(0, 10**6, size=300).astype(np.int64)
    rt = rng.choice(instants[10:], size=300).astype(np.int64)
    # three rounds, each feeding the rows in [prev_wm, wm)
    op = oracle.make_join_op(cabi.make_join_config(
        n_keys=1, n_left_vals=1, n_right_vals=1, join_type=join_type))
    got = []
    prev = 0
    for wm in (int(t0 + 10 * NS), int(t0 + 20 * NS), U64MAX):
        lm = (lt >= prev) & (lt < wm)
        rm = (rt >= prev) & (rt < wm)
        op.process_batch(op.LEFT, [lk[lm], lv[lm], lt[lm]])
        op.process_batch(op.RIGHT, [rk[rm], rv[rm], rt[rm]])
        got += rows_of(op.handle_watermark(wm))
        prev = wm
    op.close()
    want = np_outer_join(lk, lv, lt, rk, rv, rt, join_type)
    assert sorted(got) == want
    assert any(r[4] == 0 for r in want) or any(r[5] == 0 for r in want)


def test_outer_join_oracle_checkpoint_roundtrip():
    """Drain both sides of a half-fed FULL join, restore into a fresh op,
    finish, and match the uninterrupted run (join_type orthogonal to the
    raw-row drain format)."""
    rng = np.random.default_rng(97)
    t0 = 1_600_000_000 * NS
    instants = t0 + np.arange(10, dtype=np.int64) * NS
    def gen(seed, n):
        r = np.random.default_rng(seed)
        return (r.integers(0, 10, size=n).astype(np.int64),
                r.integers(0, 1000, size=n).astype(np.int64),
                np.sort(r.choice(instants, size=n)).astype(np.int64))
    lk, lv, lt = gen(1, 200)
    rk, rv, rt = gen(2, 150)
    cfgk = dict(n_keys=1, n_left_vals=1, n_right_vals=1,
                join_type=cabi.JOIN_FULL)

    a = oracle.make_join_op(cabi.make_join_config(**cfgk))
    a.process_batch(a.LEFT, [lk, lv, lt])
    ld = a.checkpoint_drain(a.LEFT)
    rd = a.checkpoint_drain(a.RIGHT)
    a.close()

    b = oracle.make_join_op(cabi.make_join_config(**cfgk))
    b.restore(b.LEFT, ld)
    b.restore(b.RIGHT, rd)
    b.process_batch(b.RIGHT, [rk, rv, rt])
    got = rows_of(b.handle_watermark(U64MAX))
    b.close()

    c = oracle.make_join_op(cabi.make_join_config(**cfgk))
    c.process_batch(c.LEFT, [lk, lv, lt])
    c.process_batch(c.RIGHT, [rk, rv, rt])
    want = rows_of(c.handle_watermark(U64MAX))
    c.close()
    assert got == want
