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


# ---- two-argument co-moment family + bit aggregates ----

# 14 aggregates split across two op instances (AMD_MAX_AGGS = 8 per op),
# driven with the same stream; finals are merged by key
COV_AGGS_A = [(cabi.COVAR_POP, 0, 1), (cabi.COVAR_SAMP, 0, 1),
              (cabi.CORR, 0, 1), (cabi.REGR_SLOPE, 0, 1),
              (cabi.REGR_INTERCEPT, 0, 1), (cabi.REGR_R2, 0, 1)]
COV_AGGS_B = [(cabi.REGR_AVGX, 0, 1), (cabi.REGR_AVGY, 0, 1),
              (cabi.REGR_COUNT, 0, 1), (cabi.REGR_SXX, 0, 1),
              (cabi.REGR_SYY, 0, 1), (cabi.REGR_SXY, 0, 1),
              (cabi.BIT_AND, 0), (cabi.BIT_OR, 0)]


def cov_cfgs():
    return (cabi.make_updagg_config(COV_AGGS_A, n_keys=1, n_value_cols=2),
            cabi.make_updagg_config(COV_AGGS_B, n_keys=1, n_value_cols=2))


def cov_drive(ops, steps=30, seed=9):
    rng = np.random.default_rng(seed)
    live = {}
    ems = []
    for _ in range(steps):
        n = int(rng.integers(1, 40))
        keys, ys, xs, retr = [], [], [], []
        for _ in range(n):
            k = int(rng.integers(0, 5))
            stored = live.get(k, [])
            if stored and rng.random() < 0.3:
                y, x = stored.pop(int(rng.integers(0, len(stored))))
                keys.append(k); ys.append(y); xs.append(x); retr.append(1)
            else:
                y = int(rng.integers(1, 40))
                x = int(rng.integers(1, 40))
                live.setdefault(k, []).append((y, x))
                keys.append(k); ys.append(y); xs.append(x); retr.append(0)
        cols = [np.array(keys, dtype=np.int64),
                np.array(ys, dtype=np.int64),
                np.array(xs, dtype=np.int64),
                np.array(retr, dtype=np.int64)]
        outs = []
        for op in ops:
            op.process_batch(cols)
            outs.append(op.flush())
        ems.append(outs)
    return ems, live


def cov_final(ems):
    # merge the two ops' per-key final values into 14-wide rows
    state = [{}, {}]
    for outs in ems:
        for which, out in enumerate(outs):
            if not out or not len(out[0]):
                continue
            na = len(out) - 2
            for r in range(len(out[0])):
                k = int(out[0][r])
                row = tuple(out[c][r] for c in range(1, 1 + na))
                if int(out[1 + na][r]):
                    state[which].pop(k, None)
                else:
                    state[which][k] = row
    merged = {}
    for k in set(state[0]) | set(state[1]):
        merged[k] = state[0].get(k, ()) + state[1].get(k, ())
    return merged


def cov_expect(live):
    want = {}
    for k, pairs in live.items():
        if not pairs:
            continue
        y = np.array([p[0] for p in pairs], dtype=np.float64)
        x = np.array([p[1] for p in pairs], dtype=np.float64)
        n = len(y)
        my, mx = y.mean(), x.mean()
        Sxy = ((x - mx) * (y - my)).sum()
        Sxx = ((x - mx) ** 2).sum()
        Syy = ((y - my) ** 2).sum()
        corr = (Sxy / np.sqrt(Sxx * Syy)
                if n > 1 and Sxx > 0 and Syy > 0 else np.nan)
        slope = Sxy / Sxx if Sxx > 0 else np.nan
        xor_and = -1  # all-ones start for AND
        bor = 0
        band = ~np.uint64(0)
        for yy, _ in pairs:
            bor |= int(yy)
            band &= np.uint64(yy)
        want[k] = (
            Sxy / n,
            Sxy / (n - 1) if n > 1 else np.nan,
            corr,
            slope,
            my - slope * mx if Sxx > 0 else np.nan,
            (np.nan if Sxx <= 0 else
             (1.0 if Syy <= 0 else Sxy * Sxy / (Sxx * Syy))),
            mx, my, n, Sxx, Syy, Sxy,
            int(band), bor,
        )
    return want


def cov_check(state, want):
    assert set(state) == set(want)
    for k, w in want.items():
        g = state[k]
        for i in range(12):
            if i == 8:               # regr_count: i64
                assert int(g[i]) == w[i], (k, i)
            elif isinstance(w[i], float) and np.isnan(w[i]):
                assert np.isnan(g[i]), (k, i, g[i])
            else:
                np.testing.assert_allclose(g[i], w[i], rtol=1e-9, atol=1e-9)
        assert int(g[12]) == int(w[12]) and int(g[13]) == int(w[13])


def test_oracle_cov_family_matches_numpy():
    ca, cb = cov_cfgs()
    ops = [oracle.make_updagg_op(ca), oracle.make_updagg_op(cb)]
    ems, live = cov_drive(ops)
    for op in ops:
        op.close()
    cov_check(cov_final(ems), cov_expect(live))


@pytest.mark.gpu
def test_gpu_cov_family_matches_numpy():
    from arroyo_amd import gpu
    ca, cb = cov_cfgs()
    ops = [gpu.make_updagg_op(ca), gpu.make_updagg_op(cb)]
    ems, live = cov_drive(ops, seed=10)
    for op in ops:
        op.close()
    cov_check(cov_final(ems), cov_expect(live))
