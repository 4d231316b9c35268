"""API error paths: invalid configs are rejected at create and argument
mismatches return errors through last_error — never crashes or silent
acceptance.  Mirrors the reference's planner/operator validation behavior
(e.g. arroyo-planner/src/lib.rs:644 rejects hop widths not divisible by the
slide).  CPU (oracle) versions here; the GPU library shares the same
validation code paths (create-time checks run before any HIP call)."""
import numpy as np
import pytest

import oracle
from arroyo_amd import cabi

NS = 10**9


def test_window_create_rejects_zero_width():
    with pytest.raises(RuntimeError):
        oracle.make_op(cabi.make_config(
            width_ns=0, slide_ns=NS, n_keys=1, aggs=[(cabi.COUNT, -1)]))


def test_window_create_rejects_nondivisible_hop():
    # planner: "hop() width must be evenly divisible by the slide"
    # (arroyo-planner/src/lib.rs:642-648; the reference pins this itself:
    # most_active_driver_last_hour_unaligned.sql is a --fail test expecting
    # "hop() width 3600s currently must be a multiple of slide 2400s")
    with pytest.raises(RuntimeError):
        oracle.make_op(cabi.make_config(
            width_ns=10 * NS, slide_ns=3 * NS, n_keys=1,
            aggs=[(cabi.COUNT, -1)]))
    with pytest.raises(RuntimeError):  # the reference's exact fail case
        oracle.make_op(cabi.make_config(
            width_ns=3600 * NS, slide_ns=2400 * NS, n_keys=1,
            aggs=[(cabi.COUNT, -1)]))


def test_window_create_rejects_zero_slide():
    with pytest.raises(RuntimeError):
        oracle.make_op(cabi.make_config(
            width_ns=10 * NS, slide_ns=0, n_keys=1,
            aggs=[(cabi.COUNT, -1)]))


def test_window_process_batch_wrong_cols():
    op = oracle.make_op(cabi.make_config(
        width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=1,
        aggs=[(cabi.SUM, 0)]))
    with pytest.raises(RuntimeError):
        # expects [key, value, ts]; hand only [key, ts]
        op.process_batch([np.zeros(4, dtype=np.int64),
                          np.zeros(4, dtype=np.int64)])
    op.close()


def test_session_create_rejects_zero_gap():
    with pytest.raises(RuntimeError):
        oracle.make_session_op(cabi.make_session_config(
            0, [(cabi.COUNT, -1)], n_keys=1))


def test_session_create_rejects_no_aggs():
    with pytest.raises(RuntimeError):
        oracle.make_session_op(cabi.make_session_config(
            5 * NS, [], n_keys=1))


def test_expjoin_create_rejects_zero_ttl():
    with pytest.raises(RuntimeError):
        oracle.make_expjoin_op(cabi.make_expjoin_config(0))


def test_expjoin_wrong_side_cols():
    op = oracle.make_expjoin_op(cabi.make_expjoin_config(
        3600 * NS, n_left_vals=2, n_right_vals=0))
    with pytest.raises(RuntimeError):
        # left expects [key, v0, v1, ts]
        op.process_batch(op.LEFT, [np.zeros(2, dtype=np.int64),
                                   np.zeros(2, dtype=np.int64)])
    op.close()


def test_mapop_create_rejects_bad_register():
    with pytest.raises(RuntimeError):
        oracle.make_map_op(cabi.make_map_config(
            n_in_cols=1, prog=[(cabi.MOP_ADD, 0, 31, 40)], out_reg=[0]))


def test_mapop_div_by_zero_is_error():
    op = oracle.make_map_op(cabi.make_map_config(
        n_in_cols=2, prog=[(cabi.MOP_DIV, 0, 1, 2)], out_reg=[2]))
    with pytest.raises(RuntimeError):
        op.process_batch([np.array([4], dtype=np.int64),
                          np.array([0], dtype=np.int64)])
    op.close()


def test_windowfn_wrong_cols():
    op = oracle.make_windowfn_op(cabi.make_windowfn_config(
        n_cols=3, part_col=0, order=[(1, False)]))
    with pytest.raises(RuntimeError):
        op.process_batch([np.zeros(2, dtype=np.int64),
                          np.zeros(2, dtype=np.int64)])
    op.close()


def test_updagg_wrong_cols():
    op = oracle.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1)], n_keys=1, n_value_cols=1))
    with pytest.raises(RuntimeError):
        # expects [key, value, is_retract]
        op.process_batch([np.zeros(3, dtype=np.int64),
                          np.zeros(3, dtype=np.int64)])
    op.close()


def test_updagg_state_width_cap_is_loud():
    """> UAGG_MAX_SW state words must fail at create with a clear message
    (oracle mirrors the cap)."""
    import oracle
    from arroyo_amd import cabi

    # 8 bit aggregates = 256 words > both caps
    aggs = [(cabi.BIT_AND, 0)] * 8
    cfg = cabi.make_updagg_config(aggs, n_keys=1, n_value_cols=1)
    with pytest.raises(RuntimeError):
        oracle.make_updagg_op(cfg)


def test_window_rejects_more_than_four_keys():
    import oracle
    from arroyo_amd import cabi
    from arroyo_amd.pipeline import NS

    with pytest.raises(RuntimeError):
        oracle.make_op(cabi.make_config(
            width_ns=4 * NS, slide_ns=2 * NS, n_keys=5, n_value_cols=0,
            aggs=[(cabi.COUNT, -1)]))


@pytest.mark.gpu
def test_create_destroy_leak_free():
    """50 create/destroy cycles of the window operator (which now owns
    three streams, seven events, epoch buffers and flip cursors) must not
    leak device memory beyond allocator noise."""
    import torch

    from arroyo_amd import cabi, gpu

    def mk():
        op = gpu.make_op(cabi.make_config(
            width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
            aggs=[(cabi.COUNT, -1)], log2_capacity=14, ring_panes=16))
        op.process_batch([
            np.arange(1000, dtype=np.int64),
            (1_600_000_000 * NS +
             np.arange(1000, dtype=np.int64) * 10**6)])
        op.handle_watermark(1_600_000_000 * NS + 30 * NS)
        op.close()

    mk()  # warm allocator/pools
    torch.cuda.synchronize()
    free0, _ = torch.cuda.mem_get_info()
    for _ in range(50):
        mk()
    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    leaked = free0 - free1
    assert leaked < 64 * 1024 * 1024, f"leaked ~{leaked/1e6:.1f} MB over 50 cycles"


@pytest.mark.gpu
def test_epoch_api_misuse_is_loud():
    """Epoch-pipeline misuse must fail loudly: folding with no armed epoch,
    and arming more than depth 2."""
    from arroyo_amd import cabi, gpu

    op = gpu.make_op(cabi.make_config(
        width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
        aggs=[(cabi.COUNT, -1)], log2_capacity=12, ring_panes=16))
    with pytest.raises(RuntimeError, match="no armed epoch"):
        op.handle_watermarks_epoch([1_600_000_000 * NS])
    op.mark_epoch()
    op.mark_epoch()
    with pytest.raises(RuntimeError, match="epoch queue full"):
        op.mark_epoch()
    # folding drains the queue in order and recovers the operator
    out = op.handle_watermarks_epoch([1_600_000_000 * NS])
    assert out == [] or len(out[0]) == 0
    op.mark_epoch()   # queue has room again
    op.handle_watermarks_epoch([1_600_000_001 * NS])
    op.handle_watermarks_epoch([1_600_000_002 * NS])
    op.close()
