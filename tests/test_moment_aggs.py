"""Co-moment aggregates (stddev/stddev_pop/var/var_pop) + bit_xor in the
updating aggregate — the retractable rows of every_aggregate.sql's
"unsupported" list (COVERAGE.md).  State = (Σx, Σx²) f64 + the key's live
row count; sample variants emit NaN when n < 2 (the reference emits SQL
NULL).  Pinned against a numpy restatement and GPU-vs-oracle."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi

AGGS = [(cabi.STDDEV, 0), (cabi.STDDEV_POP, 0), (cabi.VAR, 0),
        (cabi.VAR_POP, 0), (cabi.BIT_XOR, 0), (cabi.COUNT, -1)]


def cfg():
    return cabi.make_updagg_config(AGGS, n_keys=1, n_value_cols=1)


def drive(op, steps=40, seed=3):
    """Random append/retract stream; returns (emissions, live) where live
    maps key -> list of values."""
    rng = np.random.default_rng(seed)
    live = {}
    ems = []
    for _ in range(steps):
        n = int(rng.integers(1, 50))
        keys, vals, retr = [], [], []
        for _ in range(n):
            k = int(rng.integers(0, 6))
            stored = live.get(k, [])
            if stored and rng.random() < 0.35:
                v = stored.pop(int(rng.integers(0, len(stored))))
                keys.append(k); vals.append(v); retr.append(1)
            else:
                v = int(rng.integers(-50, 50))
                live.setdefault(k, []).append(v)
                keys.append(k); vals.append(v); retr.append(0)
        op.process_batch([np.array(keys, dtype=np.int64),
                          np.array(vals, dtype=np.int64),
                          np.array(retr, dtype=np.int64)])
        out = op.flush()
        if out and len(out[0]):
            ems.append(out)
    return ems, live


def final_state(ems):
    """Fold retract/append emissions to the final per-key values."""
    state = {}
    for out in ems:
        # [key, stddev, stddev_pop, var, var_pop, bit_xor, count, retract]
        for r in range(len(out[0])):
            k = int(out[0][r])
            row = tuple(out[c][r] for c in range(1, 7))
            if int(out[7][r]):
                state.pop(k, None)
            else:
                state[k] = row
    return state


def expect(live):
    want = {}
    for k, vals in live.items():
        if not vals:
            continue
        a = np.array(vals, dtype=np.float64)
        n = len(a)
        xor = 0
        for v in vals:
            xor ^= int(v)
        want[k] = (
            a.std(ddof=1) if n > 1 else np.nan,
            a.std(ddof=0),
            a.var(ddof=1) if n > 1 else np.nan,
            a.var(ddof=0),
            xor,
            n,
        )
    return want


def check(state, want):
    assert set(state) == set(want)
    for k, w in want.items():
        g = state[k]
        for i in range(4):
            if np.isnan(w[i]):
                assert np.isnan(g[i]), (k, i, g[i])
            else:
                np.testing.assert_allclose(g[i], w[i], rtol=1e-9, atol=1e-9)
        assert int(g[4]) == w[4] and int(g[5]) == w[5]


def test_oracle_moment_aggs_match_numpy():
    op = oracle.make_updagg_op(cfg())
    ems, live = drive(op)
    op.close()
    check(final_state(ems), expect(live))


@pytest.mark.gpu
def test_gpu_moment_aggs_match_numpy():
    from arroyo_amd import gpu
    op = gpu.make_updagg_op(cfg())
    ems, live = drive(op, seed=4)
    op.close()
    check(final_state(ems), expect(live))
