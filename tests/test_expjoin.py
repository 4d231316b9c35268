"""Non-windowed (TTL'd) join tests: the reference's updating_inner_join
golden vector and fuzz against an independent numpy restatement.

Reference semantics: crates/arroyo-worker/src/arrow/join_with_expiration.rs
(each incoming batch inserted into its side's per-key state and joined
against the other side's stored rows; each pair emitted once, when the later
row arrives; output _timestamp = max of the two sides')."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi
from tests.golden_util import NS, assert_rows_match, load_golden, load_inputs

U64MAX = 2**64 - 1
HOUR = 3600 * NS


def rows_of(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return sorted(tuple(int(c[r]) for c in cols) for r in range(len(cols[0])))


def run_updating_inner_join_golden(make_op):
    """updating_inner_join.sql: impulse JOIN (impulse WHERE counter % 2 = 1)
    ON counter = counter -> (c, c) for every odd counter."""
    d = load_inputs()["impulse"]
    counter = np.array(d["counter"], dtype=np.int64)
    ts = np.array(d["ts"], dtype=np.int64)
    odd = counter % 2 == 1
    op = make_op(cabi.make_expjoin_config(24 * HOUR))
    got = []
    # interleave sides batch-by-batch as the engine does (both sides come
    # from the same source stream)
    step = 32  # reference smoke tests force source batch size 32
    for b in range(0, len(counter), step):
        sl = slice(b, min(b + step, len(counter)))
        got += rows_of(op.process_batch(op.LEFT, [counter[sl], ts[sl]]))
        m = odd[sl]
        got += rows_of(op.process_batch(op.RIGHT,
                                        [counter[sl][m], ts[sl][m]]))
    op.close()
    # the sink wraps non-windowed join output in Debezium create envelopes
    # (format = 'debezium_json'; appends are op "c")
    rows = [{"before": None,
             "after": {"left_count": int(k), "right_count": int(k)},
             "op": "c"} for k, _t in got]
    assert_rows_match(rows, load_golden("updating_inner_join"))


def test_updating_inner_join_golden_oracle():
    run_updating_inner_join_golden(oracle.make_expjoin_op)


def np_join(lk, lv, lt, rk, rv, rt):
    """All (l, r) key-equal pairs with max timestamp."""
    rows = []
    for i in range(len(lk)):
        for j in range(len(rk)):
            if lk[i] == rk[j]:
                rows.append((int(lk[i]), int(lv[i]), int(rv[j]),
                             max(int(lt[i]), int(rt[j]))))
    return sorted(rows)


def gen_side(rng, n, t0):
    return (rng.integers(0, 40, size=n).astype(np.int64),
            rng.integers(0, 1000, size=n).astype(np.int64),
            t0 + np.sort(rng.integers(0, 60 * NS, size=n)).astype(np.int64))


def test_expjoin_oracle_vs_numpy_fuzz():
    rng = np.random.default_rng(17)
    t0 = 1_600_000_000 * NS
    lk, lv, lt = gen_side(rng, 300, t0)
    rk, rv, rt = gen_side(rng, 300, t0)
    op = oracle.make_expjoin_op(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=1, n_right_vals=1))
    got = []
    step = 50
    for b in range(0, 300, step):
        sl = slice(b, b + step)
        got += rows_of(op.process_batch(op.LEFT, [lk[sl], lv[sl], lt[sl]]))
        got += rows_of(op.process_batch(op.RIGHT, [rk[sl], rv[sl], rt[sl]]))
    op.close()
    want = np_join(lk, lv, lt, rk, rv, rt)
    assert sorted(got) == want
    assert len(want) > 100


def test_expjoin_oracle_expire():
    """expire() drops rows with ts < watermark - ttl; expired rows no longer
    match (the cutoff the reference applies across checkpoint/restore)."""
    t0 = 1_600_000_000 * NS
    ttl = 10 * NS
    op = oracle.make_expjoin_op(cabi.make_expjoin_config(ttl))
    op.process_batch(op.LEFT, [np.array([5, 6], dtype=np.int64),
                               np.array([t0, t0 + 30 * NS], dtype=np.int64)])
    op.handle_watermark(t0 + 25 * NS)
    op.expire()  # key 5 (ts=t0) is older than wm - ttl = t0+15s: dropped
    out = op.process_batch(op.RIGHT,
                           [np.array([5, 6], dtype=np.int64),
                            np.array([t0 + 31 * NS, t0 + 31 * NS],
                                     dtype=np.int64)])
    rows = rows_of(out)
    assert rows == [(6, t0 + 31 * NS)]
    op.close()


def test_expjoin_oracle_checkpoint_restore():
    rng = np.random.default_rng(23)
    t0 = 1_600_000_000 * NS
    lk, lv, lt = gen_side(rng, 100, t0)
    rk, rv, rt = gen_side(rng, 100, t0)
    cfg = lambda: cabi.make_expjoin_config(24 * HOUR, n_left_vals=1,
                                           n_right_vals=1)
    a = oracle.make_expjoin_op(cfg())
    a.process_batch(a.LEFT, [lk, lv, lt])
    ld = a.checkpoint_drain(a.LEFT)
    rd = a.checkpoint_drain(a.RIGHT)
    a.close()
    assert len(ld[0]) == 100 and len(rd[0]) == 0

    b = oracle.make_expjoin_op(cfg())
    b.restore(b.LEFT, ld)
    got = rows_of(b.process_batch(b.RIGHT, [rk, rv, rt]))
    b.close()
    assert got == np_join(lk, lv, lt, rk, rv, rt)


def test_expjoin_oracle_restore_ttl_filter():
    t0 = 1_600_000_000 * NS
    ttl = 10 * NS
    op = oracle.make_expjoin_op(cabi.make_expjoin_config(ttl))
    # restore with watermark: rows older than wm - ttl are filtered
    op.restore(op.LEFT, [np.array([1, 2], dtype=np.int64),
                         np.array([t0, t0 + 20 * NS], dtype=np.int64)],
               watermark=t0 + 25 * NS)
    out = op.process_batch(op.RIGHT,
                           [np.array([1, 2], dtype=np.int64),
                            np.array([t0 + 26 * NS, t0 + 26 * NS],
                                     dtype=np.int64)])
    assert rows_of(out) == [(2, t0 + 26 * NS)]
    op.close()


# ---------------------------------------------------------------- GPU parity


@pytest.mark.gpu
def test_updating_inner_join_golden_gpu():
    from arroyo_amd import gpu
    run_updating_inner_join_golden(gpu.make_expjoin_op)


@pytest.mark.gpu
def test_expjoin_gpu_vs_oracle_fuzz():
    from arroyo_amd import gpu
    rng = np.random.default_rng(41)
    t0 = 1_600_000_000 * NS
    cfgk = dict(n_left_vals=1, n_right_vals=2)
    g = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    o = oracle.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    got, want = [], []
    for step in range(6):
        n = 400
        k = rng.integers(0, 64, size=n).astype(np.int64)
        v1 = rng.integers(0, 10**6, size=n).astype(np.int64)
        v2 = rng.integers(0, 10**6, size=n).astype(np.int64)
        ts = t0 + (step * 60 + np.sort(
            rng.integers(0, 60, size=n))).astype(np.int64) * NS
        side = step % 2
        cols = [k, v1, ts] if side == 0 else [k, v1, v2, ts]
        got += rows_of(g.process_batch(side, cols))
        want += rows_of(o.process_batch(side, cols))
    g.close()
    o.close()
    assert sorted(got) == sorted(want)
    assert len(want) > 1000


@pytest.mark.gpu
def test_expjoin_gpu_expire_matches_oracle():
    from arroyo_amd import gpu
    rng = np.random.default_rng(43)
    t0 = 1_600_000_000 * NS
    ttl = 30 * NS
    g = gpu.make_expjoin_op(cabi.make_expjoin_config(ttl))
    o = oracle.make_expjoin_op(cabi.make_expjoin_config(ttl))
    got, want = [], []
    for step in range(5):
        n = 200
        k = rng.integers(0, 32, size=n).astype(np.int64)
        ts = t0 + (step * 20 + np.sort(
            rng.integers(0, 20, size=n))).astype(np.int64) * NS
        side = step % 2
        for op, acc in ((g, got), (o, want)):
            acc += rows_of(op.process_batch(side, [k, ts]))
            op.handle_watermark(int(ts[-1]))
            op.expire()
    g.close()
    o.close()
    assert sorted(got) == sorted(want)


@pytest.mark.gpu
def test_expjoin_gpu_checkpoint_restore():
    from arroyo_amd import gpu
    rng = np.random.default_rng(47)
    t0 = 1_600_000_000 * NS
    lk, lv, lt = gen_side(rng, 500, t0)
    rk, rv, rt = gen_side(rng, 500, t0)
    cfgk = dict(n_left_vals=1, n_right_vals=1)

    a = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    a.process_batch(a.LEFT, [lk, lv, lt])
    ld = a.checkpoint_drain(a.LEFT)
    rd = a.checkpoint_drain(a.RIGHT)
    a.close()
    assert len(ld[0]) == 500 and len(rd[0]) == 0

    b = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    b.restore(b.LEFT, ld)
    got = rows_of(b.process_batch(b.RIGHT, [rk, rv, rt]))
    b.close()
    assert got == np_join(lk, lv, lt, rk, rv, rt)


@pytest.mark.gpu
def test_expjoin_gpu_key_minus_one():
    from arroyo_amd import gpu
    t0 = 1_600_000_000 * NS
    op = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR))
    op.process_batch(op.LEFT, [np.array([-1, 2], dtype=np.int64),
                               np.array([t0, t0], dtype=np.int64)])
    out = op.process_batch(op.RIGHT,
                           [np.array([-1, -1, 3], dtype=np.int64),
                            np.array([t0 + NS] * 3, dtype=np.int64)])
    op.close()
    assert rows_of(out) == [(-1, t0 + NS), (-1, t0 + NS)]


@pytest.mark.gpu
def test_expjoin_gpu_device_resident_matches_host():
    """process_batch_device + collect (device-resident pipeline surface,
    no host staging) must produce the same match set as the host path."""
    import torch
    from arroyo_amd import gpu
    rng = np.random.default_rng(53)
    t0 = 1_600_000_000 * NS
    cfgk = dict(n_left_vals=1, n_right_vals=1)
    d = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    h = gpu.make_expjoin_op(cabi.make_expjoin_config(24 * HOUR, **cfgk))
    dev = torch.device("cuda", 0)
    want = []
    keep = []  # keep device tensors alive until collect
    for step in range(6):
        n = 300
        k = rng.integers(0, 48, size=n).astype(np.int64)
        v = rng.integers(0, 10**6, size=n).astype(np.int64)
        ts = t0 + (step * 60 + np.sort(
            rng.integers(0, 60, size=n))).astype(np.int64) * NS
        side = step % 2
        want += rows_of(h.process_batch(side, [k, v, ts]))
        tk = torch.from_numpy(k).to(dev)
        tv = torch.from_numpy(v).to(dev)
        tt = torch.from_numpy(ts).to(dev)
        keep += [tk, tv, tt]
        # H2D copies run on torch's stream, the op on its own: fence before
        # handing the pointers over
        torch.cuda.synchronize()
        d.process_batch_device(side, [tk.data_ptr(), tv.data_ptr(),
                                      tt.data_ptr()], n)
    got = rows_of(d.collect())
    # a second collect after draining must return nothing
    assert rows_of(d.collect()) == []
    d.close()
    h.close()
    assert sorted(got) == sorted(want)
    assert len(want) > 300
