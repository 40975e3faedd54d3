"""Multi-GPU shuffle exchange — RCCL all-to-all over xGMI.

Replaces the reference's byte movement between co-located executors
(ShuffleReaderExec remote fetch over Flight/raw-block transport,
shuffle_reader.rs:704-716, 1049-1118) with `torch.distributed`
all_to_all_single on the nccl backend (= RCCL on ROCm), one rank per
GPU-backed executor.  The file/Flight path stays the cross-node/parity
fallback (RCCL replaces only the byte movement between
co-located executors).

Partitioning contract: the writer hash-partitions into K = k_local x world
global partitions (`hash % K`, writer.rs:1274-1276); partition p is owned
by rank p // k_local — so the all-to-all send split for rank r is the byte
range [offsets[r*k_local], offsets[(r+1)*k_local]) of the partition-major
buffer that bg_hash_repartition materialises.  Backend-agnostic (gloo for
the CPU tests, nccl/RCCL on the 8-GPU node).
"""

import torch
import torch.distributed as dist


def owner_of(partition: int, k_local: int) -> int:
    return partition // k_local


def send_splits(offsets, world: int, elem_size: int = 1):
    """offsets: (K+1,) int64 row offsets of the partition-major buffer
    (K = world * k_local).  Returns per-rank element counts (len `world`)."""
    k = len(offsets) - 1
    assert k % world == 0
    k_local = k // world
    return [int(offsets[(r + 1) * k_local] - offsets[r * k_local]) * elem_size
            for r in range(world)]


def all_to_all_rows(t: torch.Tensor, offsets, world: int):
    """Exchange a partition-major row buffer: rank r keeps the rows of its
    own k_local partitions from every peer.  `t` is 1-D with one element
    per row unit (any dtype); returns (received tensor, recv_splits)."""
    in_splits = send_splits(offsets, world)
    counts = torch.tensor(in_splits, dtype=torch.int64, device=t.device)
    all_counts = torch.empty(world * world, dtype=torch.int64,
                             device=t.device)
    dist.all_gather_into_tensor(all_counts, counts)
    # all_counts[s * world + r] = what rank s sends to rank r
    me = dist.get_rank()
    out_splits = [int(all_counts[s * world + me]) for s in range(world)]
    out = torch.empty(sum(out_splits), dtype=t.dtype, device=t.device)
    if dist.get_backend() == "gloo" and \
            in_splits[me] != out_splits[me]:
        # gloo's alltoallv requires equal self send/recv lengths; emulate
        # with point-to-point sends (CPU test path only — the GPU path is
        # RCCL, whose all_to_all_single handles ragged self splits)
        src = t.contiguous()
        in_off = [0]
        for c in in_splits:
            in_off.append(in_off[-1] + c)
        out_off = [0]
        for c in out_splits:
            out_off.append(out_off[-1] + c)
        reqs = []
        for peer in range(world):
            if peer == me:
                out[out_off[me]:out_off[me + 1]] = \
                    src[in_off[me]:in_off[me + 1]]
                continue
            if in_splits[peer]:
                reqs.append(dist.isend(
                    src[in_off[peer]:in_off[peer + 1]], peer))
            if out_splits[peer]:
                reqs.append(dist.irecv(
                    out[out_off[peer]:out_off[peer + 1]], peer))
        for r in reqs:
            r.wait()
        return out, out_splits
    dist.all_to_all_single(out, t.contiguous(), out_splits, in_splits)
    return out, out_splits


def broadcast_build_side(t: torch.Tensor, src_rank: int = 0):
    """Broadcast-join build-side replication: RCCL
    broadcast instead of N Flight fetches of the broadcast stage's output
    (try_new_broadcast, shuffle_reader.rs:163-187; planner.rs:154-195
    lowers CollectLeft joins to a broadcast stage)."""
    shape = torch.tensor([t.shape[0] if dist.get_rank() == src_rank else 0],
                         dtype=torch.int64, device=t.device)
    dist.broadcast(shape, src_rank)
    if dist.get_rank() != src_rank:
        t = torch.empty(int(shape.item()), dtype=t.dtype, device=t.device)
    dist.broadcast(t, src_rank)
    return t
