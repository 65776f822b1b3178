"""GpuShuffleExchange — the cross-GPU repartition step.

Replaces ShuffleExchangeExec's write/fetch path
(sql/core/.../exchange/ShuffleExchangeExec.scala:357-470 +
core/.../shuffle/sort/UnsafeShuffleWriter.java + Netty block fetch): rows are
radix-partitioned on-device by pid = Pmod(Murmur3Hash(key,42), n)
(partitioning.scala:328-330, computed by gpuq_partition_perm bit-exactly),
then the per-partition runs are exchanged with ONE RCCL all-to-all(v) per
column over xGMI (torch.distributed backend "nccl" IS RCCL on ROCm; each
MI355X has 7 p2p links, so all-to-all uses all links concurrently — the
right primitive for repartition, unlike ring collectives which are
per-link-bound). CPU tests run the same code over the gloo backend with
oracle-partitioned inputs (world_size 2).
"""
from typing import Dict, List, Tuple

import torch
import torch.distributed as dist


def exchange_columns(cols: Dict[str, torch.Tensor], in_splits: List[int],
                     group=None) -> Tuple[Dict[str, torch.Tensor], List[int]]:
    """All-to-all(v) of already-partition-contiguous columns.

    cols: column name -> 1-D tensor laid out partition-contiguous
          (partition i occupies in_splits[i] rows), one partition per rank.
    Returns (received columns, out_splits). Mirrors the reducer side's
    concat-of-mapper-streams (ShuffledRowRDD.compute, ShuffledRowRDD.scala:188):
    received rows are ordered by source rank, preserving source order.
    """
    world = dist.get_world_size(group)
    device = next(iter(cols.values())).device
    counts_in = torch.tensor(in_splits, dtype=torch.int64, device=device)
    counts_out = torch.empty(world, dtype=torch.int64, device=device)
    dist.all_to_all_single(counts_out, counts_in, group=group)
    out_splits = counts_out.cpu().tolist()
    total = sum(out_splits)
    out = {}
    for name, t in cols.items():
        assert t.dim() == 1 and t.numel() == sum(in_splits)
        recv = torch.empty(total, dtype=t.dtype, device=t.device)
        dist.all_to_all_single(recv, t.contiguous(), out_splits, list(in_splits),
                               group=group)
        out[name] = recv
    return out, out_splits


def shuffle_exchange_gpu(key: torch.Tensor, payload: Dict[str, torch.Tensor],
                         group=None):
    """Full GPU exchange: partition on-device (gpuq) + RCCL all-to-all.

    num_partitions = world size (one partition per GPU, SURVEY §2 analog of
    one task per partition). Returns (key, payload) columns now holding this
    rank's partition. Fails loudly if the HIP engine is missing."""
    from . import gpuq
    world = dist.get_world_size(group)
    perm, counts = gpuq.partition_perm(key, world)
    cols = {"__key__": gpuq.gather(key, perm)}
    for name, t in payload.items():
        cols[name] = gpuq.gather(t, perm)
    in_splits = counts.cpu().tolist()
    out, _ = exchange_columns(cols, in_splits, group=group)
    k = out.pop("__key__")
    return k, out


def broadcast_gather(cols: Dict[str, torch.Tensor], group=None) -> Dict[str, torch.Tensor]:
    """BroadcastExchangeExec analog (SURVEY §8(f).3,
    joins/BroadcastHashJoinExec.scala:40): every rank contributes its slice
    of the (small) build side and receives the whole relation — an RCCL
    all-gather(v) over xGMI instead of Spark's driver-collect + Netty
    torrent broadcast. Received rows are source-rank-major, source order
    preserved (deterministic build order on every rank)."""
    world = dist.get_world_size(group)
    device = next(iter(cols.values())).device
    n_local = torch.tensor([next(iter(cols.values())).numel()],
                           dtype=torch.int64, device=device)
    sizes = torch.empty(world, dtype=torch.int64, device=device)
    dist.all_gather_into_tensor(sizes, n_local, group=group)
    counts = sizes.cpu().tolist()
    mx = max(counts)
    out = {}
    for name, t in cols.items():
        # all_gather needs equal sizes: pad to the max, slice after
        padded = torch.empty(mx, dtype=t.dtype, device=device)
        padded[:t.numel()] = t
        parts = [torch.empty(mx, dtype=t.dtype, device=device) for _ in counts]
        dist.all_gather(parts, padded, group=group)
        out[name] = torch.cat([p[:c] for p, c in zip(parts, counts)])
    return out


def range_exchange(key: torch.Tensor, payload: Dict[str, torch.Tensor],
                   desc=False, samples_per_rank: int = 4096, group=None):
    """Global ORDER BY exchange (the RangePartitioning path,
    ShuffleExchangeExec.scala:379-400): sample keys, all-gather the samples,
    take quantile bounds, range-partition on device, all-to-all. After a
    local sort on each rank, rank-major order is the global sort order.
    Returns (key, payload) holding this rank's range."""
    from . import gpuq
    world = dist.get_world_size(group)
    n = key.numel()
    # evenly-strided local sample, sorted on device with the GPU sort
    stride = max(1, n // samples_per_rank)
    idx = torch.arange(0, n, stride, dtype=torch.int32, device=key.device)
    local_sample = gpuq.gather(key, idx)
    gathered = broadcast_gather({"s": local_sample}, group=group)["s"]
    _, sorted_samples = gpuq.sort_perm(gathered, desc=desc)
    m = sorted_samples.numel()
    bidx = torch.arange(1, world, dtype=torch.int32, device=key.device) * (m // world)
    bounds = gpuq.gather(sorted_samples, bidx.clamp(max=m - 1))
    perm, counts = gpuq.range_partition_perm(key, bounds, desc=desc)
    cols = {"__key__": gpuq.gather(key, perm)}
    for name, t in payload.items():
        cols[name] = gpuq.gather(t, perm)
    out, _ = exchange_columns(cols, counts.cpu().tolist(), group=group)
    k = out.pop("__key__")
    return k, out
