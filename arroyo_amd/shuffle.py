"""Keyed shuffle exchange between ranks (the MI355X-native equivalent of the
reference's cross-subtask repartition: crates/arroyo-operator/src/context.rs:
506-560 `repartition` + crates/arroyo-worker/src/network_manager.rs Arrow-IPC
transport).

Partitioning matches the reference's scheme: 64-bit hash of the key column,
owner = contiguous u64 range (server_for_hash,
crates/arroyo-types/src/lib.rs:640-647).  The hash itself is splitmix64, not
ahash — allowed because both ends of our shuffle use it consistently and
result parity does not depend on partition assignment.

Exchange primitive:
  - RCCL (backend "nccl" on ROCm): one fused `all_to_all_single` per column
    over xGMI — every GPU pair has a direct link, so all-to-all is the natural
    collective.
  - gloo (CPU tests): gloo supports neither all_to_all nor all_to_all_single,
    so the same exchange runs as point-to-point isend/irecv pairs.
"""
import numpy as np
import torch
import torch.distributed as dist

SPLITMIX_C1 = 0x9E3779B97F4A7C15
SPLITMIX_C2 = 0xBF58476D1CE4E5B9
SPLITMIX_C3 = 0x94D049BB133111EB
U64 = np.uint64


def splitmix64_np(x):
    x = x.astype(U64) + U64(SPLITMIX_C1)
    x = (x ^ (x >> U64(30))) * U64(SPLITMIX_C2)
    x = (x ^ (x >> U64(27))) * U64(SPLITMIX_C3)
    return x ^ (x >> U64(31))


def partition_ids(keys, n_parts):
    """Owner rank per row: hash / (2^64 / n) — contiguous hash ranges, same
    scheme as server_for_hash (arroyo-types:640-647)."""
    with np.errstate(over="ignore"):
        h = splitmix64_np(np.asarray(keys).astype(np.int64).view(np.uint64)
                          if np.asarray(keys).dtype != np.uint64
                          else np.asarray(keys))
    range_size = U64(2**64 // n_parts)
    return (h // range_size).astype(np.int64)


def recv_splits_of(gathered_flat, world, rank):
    """Decode an all-gathered [world x world] send-split matrix (row =
    sender, column = destination) into this rank's receive splits, indexed
    by source rank.  Shared by bench.py's RCCL exchange and the
    single-process bookkeeping test."""
    return [int(gathered_flat[src * world + rank]) for src in range(world)]


def exchange_sizes(send_counts, group=None):
    """All-gather the per-destination row counts; returns recv_counts[src]."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    t = torch.tensor(send_counts, dtype=torch.int64)
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    return [int(gathered[src][rank]) for src in range(world)]


def all_to_all_tensors(send, group=None):
    """Exchange a list of per-destination 1-D tensors; returns the received
    list indexed by source rank.  Works on gloo (isend/irecv) and nccl/RCCL
    (all_to_all_single over xGMI)."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    assert len(send) == world
    recv_counts = exchange_sizes([s.numel() for s in send], group)
    backend = dist.get_backend(group)
    if backend == "nccl":
        send_flat = torch.cat([s for s in send])
        recv_flat = send_flat.new_empty(sum(recv_counts))
        dist.all_to_all_single(
            recv_flat, send_flat,
            output_split_sizes=recv_counts,
            input_split_sizes=[s.numel() for s in send], group=group)
        out, off = [], 0
        for c in recv_counts:
            out.append(recv_flat[off:off + c])
            off += c
        return out
    # gloo: point-to-point. Self-row is a local copy; peers paired isend/irecv.
    out = [None] * world
    out[rank] = send[rank].clone()
    reqs = []
    recvs = {}
    for peer in range(world):
        if peer == rank:
            continue
        if send[peer].numel():
            reqs.append(dist.isend(send[peer].contiguous(), dst=peer,
                                   group=group, tag=0))
        if recv_counts[peer]:
            buf = send[rank].new_empty(recv_counts[peer])
            recvs[peer] = buf
            reqs.append(dist.irecv(buf, src=peer, group=group, tag=0))
    for r in reqs:
        r.wait()
    for peer in range(world):
        if peer == rank:
            continue
        out[peer] = recvs.get(peer, send[rank].new_empty(0))
    return out


def shuffle_columns(cols, n_parts, group=None):
    """Repartition a batch (list of equal-length int64 numpy columns, key col
    first) across ranks by key-hash range; returns the merged received
    columns.  CPU/gloo path; the GPU path does the partition+scatter on
    device (arroyo_amd_partition) and exchanges device buffers over RCCL."""
    world = dist.get_world_size(group)
    assert world == n_parts
    pid = partition_ids(cols[0], n_parts)
    order = np.argsort(pid, kind="stable")
    counts = np.bincount(pid, minlength=n_parts)
    merged = []
    for c in cols:
        sorted_c = np.ascontiguousarray(np.asarray(c)[order])
        send, off = [], 0
        for p in range(n_parts):
            send.append(torch.from_numpy(
                sorted_c[off:off + counts[p]].copy()))
            off += counts[p]
        recv = all_to_all_tensors(send, group)
        merged.append(torch.cat(recv).numpy())
    return merged
