"""Stateless map/filter/projection tests: numpy fuzz, nexmark-q1-shaped
pipeline, and bit-exact GPU-vs-oracle parity (including f64 bit patterns).

Reference semantics: crates/arroyo-worker/src/arrow/mod.rs (expression
operators evaluate a plan per batch and emit immediately; filters preserve
row order; integer division by zero is an error)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi


def q1_config():
    """nexmark q1 shape: SELECT auction, price * 0.908 AS price_eur, bidder,
    _timestamp FROM bids WHERE price > 1000.
    Input cols: r0 auction, r1 price, r2 bidder, r3 _timestamp."""
    prog = [
        (cabi.MOP_I2F, 1, 0, 4),                 # r4 = (f64) price
        (cabi.MOP_CONST, 0, 0, 5, 0.908),        # r5 = 0.908
        (cabi.MOP_FMUL, 4, 5, 6),                # r6 = price * 0.908
        (cabi.MOP_CONST, 0, 0, 7, 1000),         # r7 = 1000
        (cabi.MOP_GT, 1, 7, 8),                  # r8 = price > 1000
    ]
    return cabi.make_map_config(
        n_in_cols=4, prog=prog, out_reg=[0, 6, 2, 3],
        out_is_f64=[0, 1, 0, 0], filter_reg=8)


def q1_inputs(n=50_000, seed=7):
    rng = np.random.default_rng(seed)
    return [rng.integers(1000, 2000, size=n).astype(np.int64),
            rng.integers(1, 5000, size=n).astype(np.int64),
            rng.integers(0, 100, size=n).astype(np.int64),
            np.arange(n, dtype=np.int64)]


def np_q1(cols):
    auction, price, bidder, ts = cols
    m = price > 1000
    return [auction[m], price[m].astype(np.float64) * 0.908, bidder[m],
            ts[m]]


def test_map_q1_oracle():
    cols = q1_inputs()
    op = oracle.make_map_op(q1_config())
    out = op.process_batch(cols)
    op.close()
    want = np_q1(cols)
    assert len(out[0]) == len(want[0])
    for got, exp in zip(out, want):
        assert np.array_equal(got, exp)  # row order preserved, bit-exact


def test_map_oracle_div_zero_errors():
    cfg = cabi.make_map_config(
        n_in_cols=2, prog=[(cabi.MOP_DIV, 0, 1, 2)], out_reg=[2])
    op = oracle.make_map_op(cfg)
    with pytest.raises(RuntimeError, match="division by zero"):
        op.process_batch([np.array([1], dtype=np.int64),
                          np.array([0], dtype=np.int64)])
    op.close()


def fuzz_cols(seed, n=20_000):
    rng = np.random.default_rng(seed)
    return [rng.integers(-1000, 1000, size=n).astype(np.int64)
            for _ in range(3)]


def fuzz_cfg():
    """mixed int/f64 expression with a filter:
    out = [a+b, (a*b) % 97, f64(a)/f64(b or 1), ts] where (a-b) % 3 != 0."""
    prog = [
        (cabi.MOP_ADD, 0, 1, 3),                 # r3 = a + b
        (cabi.MOP_MUL, 0, 1, 4),                 # r4 = a * b
        (cabi.MOP_CONST, 0, 0, 5, 97),
        (cabi.MOP_MOD, 4, 5, 6),                 # r6 = (a*b) % 97
        (cabi.MOP_CONST, 0, 0, 7, 0),
        (cabi.MOP_EQ, 1, 7, 8),                  # r8 = b == 0
        (cabi.MOP_ADD, 1, 8, 9),                 # r9 = b or 1 (b + (b==0))
        (cabi.MOP_I2F, 0, 0, 10),
        (cabi.MOP_I2F, 9, 0, 11),
        (cabi.MOP_FDIV, 10, 11, 12),             # r12 = f64 a / (b or 1)
        (cabi.MOP_SUB, 0, 1, 13),
        (cabi.MOP_CONST, 0, 0, 14, 3),
        (cabi.MOP_MOD, 13, 14, 15),
        (cabi.MOP_NE, 15, 7, 16),                # r16 = (a-b)%3 != 0
    ]
    return cabi.make_map_config(
        n_in_cols=3, prog=prog, out_reg=[3, 6, 12, 2],
        out_is_f64=[0, 0, 1, 0], filter_reg=16)


def np_fuzz(cols):
    a, b, ts = cols
    keep = (a - b) % 3 != 0
    # C semantics: % truncates toward zero (numpy % floors) — emulate
    trunc_mod = np.fmod(a * b, 97)
    trunc_mod3 = np.fmod(a - b, 3)
    keep = trunc_mod3 != 0
    bz = b + (b == 0)
    return [c[keep] for c in
            (a + b, trunc_mod, a.astype(np.float64) / bz, ts)]


@pytest.mark.parametrize("seed", [1, 2])
def test_map_oracle_vs_numpy_fuzz(seed):
    cols = fuzz_cols(seed)
    op = oracle.make_map_op(fuzz_cfg())
    out = op.process_batch(cols)
    op.close()
    want = np_fuzz(cols)
    assert len(out) == len(want)
    for got, exp in zip(out, want):
        assert got.dtype == exp.dtype or got.dtype == np.float64
        assert np.array_equal(got, exp)


# ---------------------------------------------------------------- GPU parity


@pytest.mark.gpu
def test_map_q1_gpu():
    from arroyo_amd import gpu
    cols = q1_inputs(n=500_000)
    g = gpu.make_map_op(q1_config())
    got = g.process_batch(cols)
    g.close()
    want = np_q1(cols)
    assert len(got[0]) == len(want[0])
    for a, b in zip(got, want):
        assert np.array_equal(a, b)  # order-preserving, bit-exact


@pytest.mark.gpu
@pytest.mark.parametrize("seed", [1, 2])
def test_map_gpu_vs_oracle_fuzz(seed):
    from arroyo_amd import gpu
    cols = fuzz_cols(seed, n=300_000)
    g = gpu.make_map_op(fuzz_cfg())
    o = oracle.make_map_op(fuzz_cfg())
    got = g.process_batch(cols)
    want = o.process_batch(cols)
    g.close()
    o.close()
    for a, b in zip(got, want):
        assert np.array_equal(a, b)


@pytest.mark.gpu
def test_map_gpu_div_zero_errors():
    from arroyo_amd import gpu
    cfg = cabi.make_map_config(
        n_in_cols=2, prog=[(cabi.MOP_DIV, 0, 1, 2)], out_reg=[2])
    op = gpu.make_map_op(cfg)
    with pytest.raises(RuntimeError, match="division by zero"):
        op.process_batch([np.array([1, 2], dtype=np.int64),
                          np.array([1, 0], dtype=np.int64)])
    op.close()


@pytest.mark.gpu
def test_mapop_gpu_device_pipeline_into_window():
    """On-GPU composition: map/filter device output feeds the window
    operator's device ingest with no host bounce (the q7-style chained
    layout).  Must equal numpy-filter + oracle-window on the host."""
    import torch
    from arroyo_amd import gpu
    from arroyo_amd.pipeline import U64MAX
    NS = 10**9
    rng = np.random.default_rng(67)
    t0 = 1_600_000_000 * NS
    n = 50_000
    key = rng.integers(0, 100, size=n).astype(np.int64)
    ts = t0 + np.sort(rng.integers(0, 120, size=n)).astype(np.int64) * NS
    # keep keys divisible by 3; project [key, ts] through the register VM
    mcfg = cabi.make_map_config(
        n_in_cols=2,
        prog=[(cabi.MOP_CONST, 0, 0, 2, 3),
              (cabi.MOP_MOD, 0, 2, 3),          # r3 = key % 3
              (cabi.MOP_CONST, 0, 0, 4, 0),
              (cabi.MOP_EQ, 3, 4, 5)],          # r5 = (key % 3 == 0)
        out_reg=[0, 1], filter_reg=5)
    mop = gpu.make_map_op(mcfg)
    wop = gpu.make_op(cabi.make_config(
        width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)]))
    dev = torch.device("cuda", 0)
    tk = torch.from_numpy(key).to(dev)
    tt = torch.from_numpy(ts).to(dev)
    torch.cuda.synchronize()
    dptrs, n_keep = mop.process_batch_device([tk.data_ptr(), tt.data_ptr()],
                                             n)
    assert n_keep == int((key % 3 == 0).sum())
    # map output columns live in the op's device buffers; the window op
    # ingests them directly (both ops sync their own streams at the API
    # boundary, and mapop's device surface syncs before returning)
    wop.process_batch_device(dptrs, n_keep)
    out = wop.handle_watermark(U64MAX)
    got = set()
    if out is not None and len(out) and len(out[0]):
        got = {tuple(int(c[r]) for c in out) for r in range(len(out[0]))}

    keep = key % 3 == 0
    oop = oracle.make_op(cabi.make_config(
        width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)]))
    oop.process_batch([key[keep], ts[keep]])
    wout = oop.handle_watermark(U64MAX)
    want = {tuple(int(c[r]) for c in wout) for r in range(len(wout[0]))}
    oop.close()
    mop.close()
    wop.close()
    assert got == want
    assert len(want) > 100
