"""Multi-rank sharding semantics on CPU (gloo, world_size=2): the keyed
shuffle + per-shard window aggregation must reproduce the single-instance
result exactly.

Models the reference's parallel subtask layout: every windowed aggregate sits
behind a Shuffle edge partitioned by key hash over contiguous u64 ranges
(crates/arroyo-operator/src/context.rs:506-560 repartition;
crates/arroyo-types/src/lib.rs:640-647 server_for_hash), so each key lives
wholly on one shard and window firing needs no further exchange.  On GPU the
exchange is RCCL all-to-all over xGMI with the same partitioning (K8 kernel,
arroyo_amd_partition); here the identical logic runs over torch.distributed
gloo with the C oracle as the per-shard operator.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from arroyo_amd import cabi, nexmark
from arroyo_amd.pipeline import NS, batches_from_columns, concat_outputs, run_stream
from arroyo_amd.shuffle import shuffle_columns

WORLD = 2


def _rank_main(rank, world, port, result_q, op_kind="oracle",
               fused=False):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    np.seterr(over="ignore")

    cols = nexmark.bids(100_000, events_per_sec=20_000)
    key, ts = cols
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
              aggs=[(cabi.COUNT, -1)])
    if op_kind == "gpu":
        # every rank's shard op runs on the same physical GPU (device 0):
        # legal — handles are independent — and exactly the composed
        # N-rank dataflow (gloo exchange + HIP operator per shard)
        from arroyo_amd import gpu
        op = gpu.make_op(cabi.make_config(**kw, emit_to_host=True))
    else:
        op = oracle.make_op(cabi.make_config(**kw))

    outs = []
    from arroyo_amd.pipeline import U64MAX, WatermarkGen
    wg = WatermarkGen(NS)
    n = len(ts)
    bsz = 8192
    if fused:
        # the bench's period-fused cadence: K batches' local slices are
        # concatenated, exchanged in ONE shuffle round and submitted as
        # one batch, then the period's watermarks emit batched.  Output
        # equality with the per-batch cadence is the same late-filter
        # argument as watermark fusion (the stream is time-monotone).
        FUSE = 4
        acc_k, acc_t, pend_wm = [], [], []
        for i, lo in enumerate(range(0, n, bsz)):
            bk, bt = key[lo:lo + bsz], ts[lo:lo + bsz]
            mine = np.arange(len(bk)) % world == rank
            acc_k.append(bk[mine])
            acc_t.append(bt[mine])
            wm = wg.on_batch(ts[lo:lo + bsz])
            if wm is not None:
                pend_wm.append(wm)
            last = lo + bsz >= n
            if len(acc_k) == FUSE or last:
                mk, mt = shuffle_columns([np.concatenate(acc_k),
                                          np.concatenate(acc_t)], world)
                if len(mk):
                    op.process_batch([mk, mt])
                acc_k, acc_t = [], []
                if pend_wm:
                    out = op.handle_watermarks(pend_wm)
                    pend_wm = []
                    if out and len(out[0]):
                        outs.append(out)
    else:
        for lo in range(0, n, bsz):
            bk, bt = key[lo:lo + bsz], ts[lo:lo + bsz]
            # this rank's slice of the upstream stream (round-robin split,
            # mirroring parallel source subtasks)
            mine = np.arange(len(bk)) % world == rank
            bk, bt = bk[mine], bt[mine]
            # keyed shuffle: all-to-all by key-hash range owner
            mk, mt = shuffle_columns([bk, bt], world)
            if len(mk):
                op.process_batch([mk, mt])
            # watermark: min across upstream partitions (WatermarkHolder,
            # context.rs:63-86) -- here all ranks see the same source clock
            wm = wg.on_batch(ts[lo:lo + bsz])
            if wm is not None:
                out = op.handle_watermark(wm)
                if out and len(out[0]):
                    outs.append(out)
    out = op.handle_watermark(U64MAX)
    if out and len(out[0]):
        outs.append(out)
    op.close()
    got = concat_outputs(outs)
    rows = set()
    if got is not None:
        for r in range(len(got[0])):
            rows.add(tuple(int(c[r]) for c in got))
    result_q.put((rank, rows))
    dist.destroy_process_group()


def _run_world(world, port, op_kind="oracle", fused=False):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main,
                         args=(r, world, port, q, op_kind, fused))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, rows = q.get(timeout=300)
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # key-hash range partitioning: no key may appear on two ranks
    for a in range(world):
        for b in range(a + 1, world):
            assert not ({r[0] for r in results[a]} &
                        {r[0] for r in results[b]})

    merged = set()
    for rows in results.values():
        merged |= rows

    cols = nexmark.bids(100_000, events_per_sec=20_000)
    kw = dict(width_ns=10 * NS, slide_ns=2 * NS, n_keys=1, n_value_cols=0,
              aggs=[(cabi.COUNT, -1)])
    op = oracle.make_op(cabi.make_config(**kw))
    outs = run_stream(op, batches_from_columns(cols, 8192), NS)
    want_cols = concat_outputs(outs)
    want = set()
    for r in range(len(want_cols[0])):
        want.add(tuple(int(c[r]) for c in want_cols))
    op.close()
    assert merged == want


def test_two_rank_shuffle_matches_single():
    _run_world(2, 29371)


def test_shuffle_window_parity_world4():
    """Same sharded-vs-single parity at world_size=4 (the q7 config's 1->8
    GPU scaling path; gloo here, RCCL on the GPU boxes)."""
    _run_world(4, 29377)


@pytest.mark.gpu
def test_two_rank_shuffle_hip_ops_match_single():
    """The composed N>1 path with the HIP operator per shard: two gloo
    ranks exchange by key-hash range and each feeds its shard through the
    GPU window op on one device; the merged emissions equal the
    single-instance oracle run.  (An 8-GPU RCCL run swaps gloo for
    all_to_all_single over xGMI — same partitioning, same operator.)"""
    _run_world(2, 29383, op_kind="gpu")


@pytest.mark.gpu
def test_four_rank_shuffle_hip_ops_match_single():
    _run_world(4, 29389, op_kind="gpu")


def test_two_rank_period_fused_exchange_matches_single():
    """bench.py's N>1 cadence (one exchange + one submission + batched
    watermarks per fused period) through the composed dataflow: merged
    emissions still equal the single-instance run."""
    _run_world(2, 29389, fused=True)


@pytest.mark.gpu
def test_two_rank_period_fused_hip_ops_match_single():
    _run_world(2, 29393, op_kind="gpu", fused=True)
