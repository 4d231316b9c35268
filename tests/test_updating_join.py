"""Updating (retraction-carrying) and outer TTL joins, pinned to the
reference's updating_left/right/full_join goldens.

Reference semantics: the planner passes the SQL join type into the
DataFusion join inside JoinWithExpiration (plan/join.rs:326-379); with an
updating input the join is maintained incrementally — a new match retracts
the earlier null-padded row, a retraction that removes a key's last match
brings the null-padded rows back.  Round 2: the HIP path (k_ej_upd,
arroyo_amd/csrc/expjoin.hip — per-key ordered replay after a
radix-grouped batch sort) runs the same tests under -m gpu via the
`ej_factory` fixture.
"""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, load_golden, load_inputs


@pytest.fixture(params=["oracle",
                        pytest.param("gpu", marks=pytest.mark.gpu)])
def ej_factory(request):
    """Runs each test against the CPU oracle and (under -m gpu) the HIP
    library through the same C ABI."""
    if request.param == "oracle":
        return oracle.make_expjoin_op
    from arroyo_amd import gpu
    return gpu.make_expjoin_op

HOUR = 3600 * NS
T0 = 1_600_000_000 * NS


def rows_of(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return [tuple(int(c[r]) for c in cols) for r in range(len(cols[0]))]


def fold(emissions):
    """Fold (cols..., is_retract) emission rows into a live multiset."""
    live = {}
    for row in emissions:
        body, retr = row[:-1], row[-1]
        if retr:
            assert live.get(body, 0) > 0, f"retract of absent row {body}"
            live[body] -= 1
            if not live[body]:
                del live[body]
        else:
            live[body] = live.get(body, 0) + 1
    return live


def test_updating_left_join_scenario(ej_factory):
    """Hand-checked LEFT sequence: null row -> match retracts it -> right
    retraction brings it back."""
    op = ej_factory(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=0, n_right_vals=1,
        join_type=cabi.JOIN_LEFT, updating=True))
    a = np.array
    # left 7 arrives, no right: [7, null]
    out = rows_of(op.process_batch(op.LEFT, [a([7]), a([0]), a([T0])]))
    # (key, rval, ts, lp, rp, retract)
    assert out == [(7, 0, T0, 1, 0, 0)]
    # right (7, 42) arrives: retract [7, null], append [7, 42]
    out = rows_of(op.process_batch(
        op.RIGHT, [a([7]), a([42]), a([0]), a([T0 + NS])]))
    assert sorted(out) == sorted([(7, 0, T0, 1, 0, 1),
                                  (7, 42, T0 + NS, 1, 1, 0)])
    # right (7, 42) retracts: retract [7, 42], [7, null] comes back
    out = rows_of(op.process_batch(
        op.RIGHT, [a([7]), a([42]), a([1]), a([T0 + 2 * NS])]))
    assert sorted(out) == sorted([(7, 42, T0 + NS, 1, 1, 1),
                                  (7, 0, T0, 1, 0, 0)])
    op.close()


def test_updating_full_join_scenario(ej_factory):
    """FULL: both sides pad; a match retracts both null rows."""
    op = ej_factory(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=1, n_right_vals=1,
        join_type=cabi.JOIN_FULL, updating=True))
    a = np.array
    ems = []
    ems += rows_of(op.process_batch(op.LEFT,
                                    [a([3]), a([10]), a([0]), a([T0])]))
    ems += rows_of(op.process_batch(op.RIGHT,
                                    [a([4]), a([20]), a([0]), a([T0])]))
    # different keys: two null rows live
    assert fold(ems) == {(3, 10, 0, T0, 1, 0): 1, (4, 0, 20, T0, 0, 1): 1}
    ems += rows_of(op.process_batch(op.RIGHT,
                                    [a([3]), a([30]), a([0]), a([T0 + NS])]))
    # key 3 matched: its null row retracted, pair appended
    assert fold(ems) == {(3, 10, 30, T0 + NS, 1, 1): 1,
                         (4, 0, 20, T0, 0, 1): 1}
    op.close()


def test_updating_inner_retraction_propagates(ej_factory):
    """INNER with updating inputs: retracting a stored right row retracts
    the pairs it participated in (no null rows)."""
    op = ej_factory(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=0, n_right_vals=1, updating=True))
    a = np.array
    assert rows_of(op.process_batch(op.LEFT,
                                    [a([5]), a([0]), a([T0])])) == []
    out = rows_of(op.process_batch(op.RIGHT,
                                   [a([5]), a([9]), a([0]), a([T0])]))
    assert out == [(5, 9, T0, 0)]
    out = rows_of(op.process_batch(op.RIGHT,
                                   [a([5]), a([9]), a([1]), a([T0])]))
    assert out == [(5, 9, T0, 1)]
    op.close()


def run_updating_join_golden(ej_factory, join_type, golden):
    """updating_{left,right,full}_join.sql: impulse JOIN (counter % 2,
    count(*) WHERE counter < 3 GROUP BY 1) ON counter = right_count WHERE
    counter < 3, debezium sink.  The right side is an updating aggregate
    stream driven through our updagg operator; the emission cadence
    differs from the reference's, so the comparison folds both streams to
    their final live multiset (cadence-independent)."""
    d = load_inputs()["impulse"]
    counter = np.array(d["counter"], dtype=np.int64)
    ts = np.array(d["ts"], dtype=np.int64)

    agg = oracle.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1)], n_keys=1, n_value_cols=0))
    join = ej_factory(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=0, n_right_vals=1, join_type=join_type,
        updating=True))
    ems = []
    step = 32
    for b in range(0, len(counter), step):
        sl = slice(b, min(b + step, len(counter)))
        c, t = counter[sl], ts[sl]
        ems += rows_of(join.process_batch(
            join.LEFT, [c, np.zeros(len(c), dtype=np.int64), t]))
        m = c < 3
        if m.any():
            agg.process_batch([c[m] % 2, np.zeros(int(m.sum()),
                                                  dtype=np.int64)])
        for mod2, cnt, retr in rows_of(agg.flush()):
            a = np.array
            ems += rows_of(join.process_batch(
                join.RIGHT, [a([cnt]), a([mod2]), a([retr]),
                             a([int(t[-1])])]))
    agg.close()
    join.close()

    # WHERE counter < 3: filter each emission independently (nulls drop)
    # emission: (key, mod2, ts, lp, rp, retract); left_counter = key if lp
    kept = [r for r in ems if r[3] == 1 and r[0] < 3]
    got = fold([(r[0], r[1] if r[4] else None, r[0] if r[4] else None,
                 r[5]) for r in kept])

    want = {}
    for g in load_golden(golden):
        def row(d):
            return (d["left_counter"], d["counter_mod_2"], d["right_count"])
        if g["op"] == "c":
            want[row(g["after"])] = want.get(row(g["after"]), 0) + 1
        elif g["op"] == "d":
            want[row(g["before"])] -= 1
            if not want[row(g["before"])]:
                del want[row(g["before"])]
        elif g["op"] == "u":
            want[row(g["before"])] -= 1
            if not want[row(g["before"])]:
                del want[row(g["before"])]
            want[row(g["after"])] = want.get(row(g["after"]), 0) + 1
    assert got == want


def test_updating_left_join_golden(ej_factory):
    run_updating_join_golden(ej_factory, cabi.JOIN_LEFT,
                             "updating_left_join")


def test_updating_right_join_golden(ej_factory):
    run_updating_join_golden(ej_factory, cabi.JOIN_RIGHT,
                             "updating_right_join")


def test_updating_full_join_golden(ej_factory):
    run_updating_join_golden(ej_factory, cabi.JOIN_FULL,
                             "updating_full_join")


@pytest.mark.gpu
@pytest.mark.parametrize("join_type", [0, 1, 2, 3],
                         ids=["inner", "left", "right", "full"])
def test_gpu_matches_oracle_updating_batches(join_type):
    """Multi-row mixed append/retract batches: the HIP path's emission
    multiset per batch equals the oracle's (within-batch emission order
    is unspecified across keys)."""
    from arroyo_amd import gpu
    rng = np.random.default_rng(7 + join_type)
    cfg = dict(n_left_vals=1, n_right_vals=1, join_type=join_type,
               updating=True)
    o1 = oracle.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfg))
    o2 = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfg))
    live = [{}, {}]   # (key,val) -> count, to generate valid retracts
    for step in range(30):
        side = int(rng.integers(0, 2))
        n = int(rng.integers(1, 64))
        keys, vals, retr, tss = [], [], [], []
        for _ in range(n):
            stored = [kv for kv, c in live[side].items() if c > 0]
            if stored and rng.random() < 0.35:
                k, v = stored[int(rng.integers(0, len(stored)))]
                live[side][(k, v)] -= 1
                keys.append(k); vals.append(v); retr.append(1)
            else:
                k = int(rng.integers(0, 8)); v = int(rng.integers(0, 4))
                live[side][(k, v)] = live[side].get((k, v), 0) + 1
                keys.append(k); vals.append(v); retr.append(0)
            tss.append(T0 + (k * 7 + v) * NS)
        cols = [np.array(keys, dtype=np.int64),
                np.array(vals, dtype=np.int64),
                np.array(retr, dtype=np.int64),
                np.array(tss, dtype=np.int64)]
        e1 = sorted(rows_of(o1.process_batch(side, cols)))
        e2 = sorted(rows_of(o2.process_batch(side, cols)))
        assert e1 == e2, f"step {step} side {side}"
    o1.close()
    o2.close()


def np_join_of_multisets(left, right, join_type, nlv, nrv):
    """Brute-force join of two final live multisets (list of (key, vals,
    ts) tuples) with null padding -- what the folded emission stream must
    equal regardless of arrival/retraction order."""
    out = {}

    def add(row):
        out[row] = out.get(row, 0) + 1

    r_by_key = {}
    for rk, rv, rt in right:
        r_by_key.setdefault(rk, []).append((rv, rt))
    l_by_key = {}
    for lk, lv, lt in left:
        l_by_key.setdefault(lk, []).append((lv, lt))
    for lk, lv, lt in left:
        matches = r_by_key.get(lk, [])
        if matches:
            for rv, rt in matches:
                add((lk, *lv, *rv, max(lt, rt), 1, 1))
        elif join_type in (cabi.JOIN_LEFT, cabi.JOIN_FULL):
            add((lk, *lv, *((0,) * nrv), lt, 1, 0))
    if join_type in (cabi.JOIN_RIGHT, cabi.JOIN_FULL):
        for rk, rv, rt in right:
            if rk not in l_by_key:
                add((rk, *((0,) * nlv), *rv, rt, 0, 1))
    return out


@pytest.mark.parametrize("seed", [103, 211, 307])
@pytest.mark.parametrize("join_type", [1, 2, 3],
                         ids=["left", "right", "full"])
def test_updating_join_fuzz_fold_invariant(ej_factory, join_type, seed):
    """Random append/retract interleavings on both sides: the folded
    emission stream must equal the brute-force join of the two final live
    multisets, for every prefix cadence."""
    rng = np.random.default_rng(seed + join_type)
    op = ej_factory(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=1, n_right_vals=1, join_type=join_type,
        updating=True))
    live = [[], []]  # per-side live rows (key, (val,), ts)
    ems = []
    # retraction removes the first VALUE-equal stored row; give equal-value
    # rows equal timestamps so that policy is observationally unambiguous
    def ts_of(key, val):
        return T0 + (key * 5 + val) * NS
    for step in range(500):
        side = int(rng.integers(0, 2))
        do_retract = live[side] and rng.random() < 0.4
        if do_retract:
            key, vals, ts = live[side].pop(
                int(rng.integers(0, len(live[side]))))
            cols = [np.array([key], dtype=np.int64),
                    np.array([vals[0]], dtype=np.int64),
                    np.array([1], dtype=np.int64),
                    np.array([ts], dtype=np.int64)]
        else:
            key = int(rng.integers(0, 10))
            val = int(rng.integers(0, 5))
            ts = ts_of(key, val)
            live[side].append((key, (val,), ts))
            cols = [np.array([key], dtype=np.int64),
                    np.array([val], dtype=np.int64),
                    np.array([0], dtype=np.int64),
                    np.array([ts], dtype=np.int64)]
        ems += rows_of(op.process_batch(side, cols))
        if step % 100 == 99:
            want = np_join_of_multisets(live[0], live[1], join_type, 1, 1)
            assert fold(ems) == want
    op.close()


def test_updating_join_checkpoint_roundtrip(ej_factory):
    """Drain/restore carries the net live multiset: a restored LEFT join
    continues emitting correct retract/append pairs for rows stored before
    the checkpoint."""
    a = np.array
    cfgk = dict(n_left_vals=0, n_right_vals=1, join_type=cabi.JOIN_LEFT,
                updating=True)
    op1 = ej_factory(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    # left 7 live and unmatched at checkpoint time
    ems = rows_of(op1.process_batch(op1.LEFT, [a([7]), a([0]), a([T0])]))
    assert ems == [(7, 0, T0, 1, 0, 0)]
    ld = op1.checkpoint_drain(op1.LEFT)
    rd = op1.checkpoint_drain(op1.RIGHT)
    op1.close()
    assert len(ld[0]) == 1 and len(rd[0]) == 0

    op2 = ej_factory(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    op2.restore(op2.LEFT, ld)
    op2.restore(op2.RIGHT, rd)
    # first match after restore retracts the pre-checkpoint null row
    out = rows_of(op2.process_batch(
        op2.RIGHT, [a([7]), a([42]), a([0]), a([T0 + NS])]))
    op2.close()
    assert sorted(out) == sorted([(7, 0, T0, 1, 0, 1),
                                  (7, 42, T0 + NS, 1, 1, 0)])


@pytest.mark.gpu
def test_gpu_outer_join_validity_bitmaps():
    """The C ABI carries Arrow validity bitmaps (LSB bit order) for the
    null-padded value columns; the bitmap must equal the presence flags,
    which are themselves oracle-pinned (df.rs:24-30 RecordBatch
    contract)."""
    import ctypes

    from arroyo_amd import gpu
    from arroyo_amd.cabi import AmdOutBatch, _out_to_numpy, out_validity

    a = np.array
    cfg = cabi.make_expjoin_config(24 * HOUR, n_left_vals=1, n_right_vals=1,
                                   join_type=cabi.JOIN_FULL, updating=True)
    op = gpu.make_expjoin_op(cfg)
    emissions = []
    batches = [
        (op.LEFT, [a([3]), a([10]), a([0]), a([T0])]),
        (op.RIGHT, [a([4]), a([20]), a([0]), a([T0])]),
        (op.RIGHT, [a([3]), a([30]), a([0]), a([T0 + NS])]),
        (op.RIGHT, [a([3]), a([30]), a([1]), a([T0 + NS])]),
    ]
    for side, cols in batches:
        cols = [np.ascontiguousarray(c, dtype=np.int64) for c in cols]
        n = len(cols[0])
        arr = (ctypes.c_void_p * len(cols))(
            *[c.ctypes.data_as(ctypes.c_void_p).value for c in cols])
        out = AmdOutBatch()
        rc = op._fn["process_batch"](op._h, side, arr, len(cols), n,
                                     ctypes.byref(out))
        assert rc == 0, op._fn["last_error"](op._h).decode()
        got = _out_to_numpy(out)
        masks = out_validity(out)
        op._fn["free_out"](ctypes.byref(out))
        if not len(got[0]):
            continue
        # layout: [key, lval, rval, ts, lp, rp, retract]
        lp, rp = got[4].astype(bool), got[5].astype(bool)
        assert masks[1] is not None and masks[2] is not None
        assert (masks[1] == lp).all(), "left-value validity != presence"
        assert (masks[2] == rp).all(), "right-value validity != presence"
        for c in (0, 3, 4, 5, 6):
            assert masks[c] is None  # key/ts/flags always valid
        emissions.extend(
            tuple(int(c[r]) for c in got) for r in range(len(got[0])))
    op.close()
    # sanity: the emission fold matches the oracle for the same sequence
    o2 = oracle.make_expjoin_op(cfg)
    ems2 = []
    for side, cols in batches:
        ems2 += rows_of(o2.process_batch(side, cols))
    o2.close()
    assert fold(emissions) == fold(ems2)
