"""Multi-rank sharding semantics for the NON-window stateful operators
(gloo, CPU): TTL join and updating aggregate behind the same keyed shuffle
as the window path (tests/test_multigpu_cpu.py).

Both operators partition by key exactly like the windowed aggregates: every
key's state lives wholly on one shard (contiguous key-hash ranges,
server_for_hash, crates/arroyo-types/src/lib.rs:640-647), joins only match
equal keys and updating aggregates group by key, so the sharded run needs no
data-path collective and must reproduce the single-instance result exactly
(SURVEY.md §8e).  The per-shard operator here is the C oracle; on GPU the
same layout runs one HIP operator per rank with the exchange over RCCL.
"""
import os

import numpy as np
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle
from arroyo_amd import cabi
from arroyo_amd.shuffle import shuffle_columns

NS = 10**9
HOUR = 3600 * NS
T0 = 1_600_000_000 * NS
WORLD = 2


def _rows(cols):
    if cols is None or len(cols) == 0 or len(cols[0]) == 0:
        return []
    return [tuple(int(c[r]) for c in cols) for r in range(len(cols[0]))]


def _gen_stream(seed, n):
    rng = np.random.default_rng(seed)
    key = rng.integers(0, 40, size=n).astype(np.int64)
    val = rng.integers(0, 10**6, size=n).astype(np.int64)
    ts = T0 + np.sort(rng.integers(0, 600, size=n)).astype(np.int64) * NS
    return key, val, ts


def _expjoin_run(op, batches):
    got = []
    for side, cols in batches:
        got += _rows(op.process_batch(side, cols))
    op.close()
    return sorted(got)


def _expjoin_rank_main(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    op = oracle.make_expjoin_op(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=1, n_right_vals=1))
    got = []
    for step in range(6):
        key, val, ts = _gen_stream(100 + step, 600)
        side = step % 2
        # this rank's slice of the source stream (parallel source subtasks)
        mine = np.arange(len(key)) % world == rank
        cols = shuffle_columns([key[mine], val[mine], ts[mine]], world)
        if len(cols[0]):
            got += _rows(op.process_batch(side, cols))
    op.close()
    result_q.put((rank, sorted(got)))
    dist.destroy_process_group()


def test_expjoin_two_rank_shuffle_matches_single():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_expjoin_rank_main,
                         args=(r, WORLD, 29381, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, rows = q.get(timeout=300)
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # key-disjointness: no join key may appear on two ranks
    assert not ({r[0] for r in results[0]} & {r[0] for r in results[1]})

    single = oracle.make_expjoin_op(cabi.make_expjoin_config(
        24 * HOUR, n_left_vals=1, n_right_vals=1))
    batches = []
    for step in range(6):
        key, val, ts = _gen_stream(100 + step, 600)
        batches.append((step % 2, [key, val, ts]))
    want = _expjoin_run(single, batches)
    assert sorted(results[0] + results[1]) == want
    assert len(want) > 1000


def _updagg_rank_main(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    op = oracle.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1, n_value_cols=1))
    got = []
    for step in range(5):
        key, val, _ts = _gen_stream(200 + step, 800)
        retract = np.zeros(len(key), dtype=np.int64)
        mine = np.arange(len(key)) % world == rank
        cols = shuffle_columns([key[mine], val[mine], retract[mine]], world)
        if len(cols[0]):
            op.process_batch(cols)
        got += _rows(op.flush())
    op.close()
    result_q.put((rank, sorted(got)))
    dist.destroy_process_group()


def test_updagg_two_rank_shuffle_matches_single():
    """Per-flush emission streams merge to the single-instance stream: each
    key's retract/append history lives wholly on its owner rank, and flush
    cadence is global (every batch), so the merged per-key final rows and
    the full emission multiset must both match."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_updagg_rank_main,
                         args=(r, WORLD, 29383, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, rows = q.get(timeout=300)
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    assert not ({r[0] for r in results[0]} & {r[0] for r in results[1]})

    single = oracle.make_updagg_op(cabi.make_updagg_config(
        [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1, n_value_cols=1))
    want = []
    for step in range(5):
        key, val, _ts = _gen_stream(200 + step, 800)
        retract = np.zeros(len(key), dtype=np.int64)
        single.process_batch([key, val, retract])
        want += _rows(single.flush())
    single.close()
    assert sorted(results[0] + results[1]) == sorted(want)
    assert len(want) > 100


def _session_rank_main(rank, world, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    op = oracle.make_session_op(cabi.make_session_config(
        5 * NS, [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1,
        n_value_cols=1))
    got = []
    for step in range(6):
        key, val, ts = _gen_stream(300 + step, 700)
        # bursty: cluster timestamps so sessions form and close
        ts = T0 + (step * 120 + (ts - T0) // (20 * NS)) * NS
        mine = np.arange(len(key)) % world == rank
        cols = shuffle_columns([key[mine], val[mine], ts[mine]], world)
        if len(cols[0]):
            op.process_batch(cols)
        got += _rows(op.handle_watermark(int(ts.max()) - NS))
    got += _rows(op.handle_watermark(2**64 - 1))
    op.close()
    result_q.put((rank, sorted(got)))
    dist.destroy_process_group()


def test_session_two_rank_shuffle_matches_single():
    """Session windows shard by key like everything else: all of a key's
    rows land on its owner rank, so session extents and aggregates are
    computed whole there and the merged firings equal the single run."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_session_rank_main,
                         args=(r, WORLD, 29387, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, rows = q.get(timeout=300)
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    assert not ({r[0] for r in results[0]} & {r[0] for r in results[1]})

    single = oracle.make_session_op(cabi.make_session_config(
        5 * NS, [(cabi.COUNT, -1), (cabi.SUM, 0)], n_keys=1,
        n_value_cols=1))
    want = []
    for step in range(6):
        key, val, ts = _gen_stream(300 + step, 700)
        ts = T0 + (step * 120 + (ts - T0) // (20 * NS)) * NS
        single.process_batch([key, val, ts])
        want += _rows(single.handle_watermark(int(ts.max()) - NS))
    want += _rows(single.handle_watermark(2**64 - 1))
    single.close()
    assert sorted(results[0] + results[1]) == sorted(want)
    assert len(want) > 50


def _join_quantize(step, ts):
    """30s instants within a step, steps in disjoint 900s blocks so no
    batch falls behind the previous step's watermark (the instant join
    errors on pre-watermark rows, matching the reference's buffering
    contract)."""
    return T0 + (step * 900 + ((ts - T0) // (30 * NS)) * 30) * NS


def _join_rank_main(rank, world, port, result_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        op = oracle.make_join_op(cabi.make_join_config(
            n_keys=1, n_left_vals=1, n_right_vals=1))
        got = []
        for step in range(4):
            lk, lv, lt = _gen_stream(400 + step, 400)
            rk, rv, rt = _gen_stream(500 + step, 400)
            lt = _join_quantize(step, lt)
            rt = _join_quantize(step, rt)
            for side, (k, v, t) in ((0, (lk, lv, lt)), (1, (rk, rv, rt))):
                mine = np.arange(len(k)) % world == rank
                cols = shuffle_columns([k[mine], v[mine], t[mine]], world)
                if len(cols[0]):
                    op.process_batch(side, cols)
            got += _rows(op.handle_watermark(
                int(max(lt.max(), rt.max())) + NS))
        op.close()
        result_q.put((rank, sorted(got)))
        dist.destroy_process_group()
    except Exception as e:  # surface child failures instead of hanging
        result_q.put((rank, e))
        raise


def test_instant_join_two_rank_shuffle_matches_single():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_join_rank_main,
                         args=(r, WORLD, 29389, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, rows = q.get(timeout=120)
        assert not isinstance(rows, Exception), rows
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    assert not ({r[0] for r in results[0]} & {r[0] for r in results[1]})

    single = oracle.make_join_op(cabi.make_join_config(
        n_keys=1, n_left_vals=1, n_right_vals=1))
    want = []
    for step in range(4):
        lk, lv, lt = _gen_stream(400 + step, 400)
        rk, rv, rt = _gen_stream(500 + step, 400)
        lt = _join_quantize(step, lt)
        rt = _join_quantize(step, rt)
        single.process_batch(0, [lk, lv, lt])
        single.process_batch(1, [rk, rv, rt])
        want += _rows(single.handle_watermark(
            int(max(lt.max(), rt.max())) + NS))
    single.close()
    assert sorted(results[0] + results[1]) == sorted(want)
    assert len(want) > 500


def _windowfn_rank_main(rank, world, port, result_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        op = oracle.make_windowfn_op(cabi.make_windowfn_config(
            n_cols=3, part_col=0, order=[(1, False)]))
        got = []
        for step in range(4):
            part, val, ts = _gen_stream(600 + step, 500)
            ts = T0 + (step * 30 + ((ts - T0) // (60 * NS)) * 3) * NS
            mine = np.arange(len(part)) % world == rank
            cols = shuffle_columns([part[mine], val[mine], ts[mine]], world)
            if len(cols[0]):
                op.process_batch(cols)
            got += _rows(op.handle_watermark(int(ts.max()) + NS))
        op.close()
        result_q.put((rank, sorted(got)))
        dist.destroy_process_group()
    except Exception as e:
        result_q.put((rank, e))
        raise


def test_windowfn_two_rank_shuffle_matches_single():
    """ROW_NUMBER shards by its PARTITION BY column: a partition's rows all
    land on one rank, so per-partition ranking is computed whole there.
    Tie order inside a partition follows arrival sequence, which the keyed
    shuffle preserves per source (rows of one partition arrive from one
    logical upstream order here), so the merged rows equal the single
    run's."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_windowfn_rank_main,
                         args=(r, WORLD, 29393, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, rows = q.get(timeout=120)
        assert not isinstance(rows, Exception), rows
        results[rank] = rows
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    assert not ({r[0] for r in results[0]} & {r[0] for r in results[1]})

    single = oracle.make_windowfn_op(cabi.make_windowfn_config(
        n_cols=3, part_col=0, order=[(1, False)]))
    want = []
    for step in range(4):
        part, val, ts = _gen_stream(600 + step, 500)
        ts = T0 + (step * 30 + ((ts - T0) // (60 * NS)) * 3) * NS
        single.process_batch([part, val, ts])
        want += _rows(single.handle_watermark(int(ts.max()) + NS))
    single.close()
    assert sorted(results[0] + results[1]) == sorted(want)
    assert len(want) > 1000


def test_exchange_split_bookkeeping_simulation():
    """Single-process simulation of the N>1 bench's split-size math: for a
    random [world x world] send matrix (row = sender, col = destination),
    every rank's decoded receive splits must match the senders' columns,
    and totals must conserve rows."""
    import numpy as np

    from arroyo_amd.shuffle import recv_splits_of

    rng = np.random.default_rng(5)
    for world in (2, 4, 8):
        sends = rng.integers(0, 70_000, size=(world, world))
        gathered = [int(x) for x in sends.reshape(-1)]  # all_gather order
        total_recv = 0
        for rank in range(world):
            rs = recv_splits_of(gathered, world, rank)
            assert rs == [int(sends[src][rank]) for src in range(world)]
            total_recv += sum(rs)
        assert total_recv == int(sends.sum())


def test_period_fused_exchange_equals_per_batch():
    """The N>1 bench exchanges once per fused watermark period (one
    partition + one all_to_all over the period's contiguous slice) instead
    of per 64K batch.  Simulation: for each rank, the multiset of rows it
    receives over a period must be identical either way, and within-period
    arrival order cannot change window results (the same late-filter
    argument as watermark fusion, tests/test_property_large.py)."""
    import numpy as np

    def splitmix64(x):
        x = (x + 0x9E3779B97F4A7C15) & (2**64 - 1)
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & (2**64 - 1)
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & (2**64 - 1)
        return x ^ (x >> 31)

    rng = np.random.default_rng(17)
    world, batches, rows = 3, 6, 500
    streams = [rng.integers(0, 10_000, size=batches * rows).astype(np.int64)
               for _ in range(world)]
    rng_size = 2**64 // world

    def owner(k):
        return int(splitmix64(int(np.uint64(k))) // rng_size)

    # per-batch exchange: each destination's received keys, batch by batch
    per_batch = [[] for _ in range(world)]
    for b in range(batches):
        for r in range(world):
            seg = streams[r][b * rows:(b + 1) * rows]
            for k in seg:
                per_batch[owner(k)].append(int(k))
    # period-fused: one partition of each rank's whole period slice
    fused = [[] for _ in range(world)]
    for r in range(world):
        for k in streams[r]:
            fused[owner(k)].append(int(k))
    for d in range(world):
        assert sorted(per_batch[d]) == sorted(fused[d])
