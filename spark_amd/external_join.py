"""Out-of-core (spill-to-host) hash join — grace-hash-join shaped, the
spill analog for ShuffledHashJoinExec when a partition exceeds device
memory (the reference's CPU sorter/relation spill machinery,
UnsafeExternalSorter.java / HashedRelation's off-heap modes, reshaped for
the device):

  1. both sides stream through the GPU in chunks; each chunk is
     radix-partitioned by pid = Pmod(Murmur3(key,42), K) — the SAME
     partitioning the shuffle uses (partitioning.scala:328), so bucket b
     of the build side can only match bucket b of the probe side — and
     the per-bucket runs spill to pinned host memory;
  2. per bucket: its build slices gather H2D, build the hash table once,
     then its probe slices stream through the probe kernel; matched
     payload rows yield as device batches.

Traffic: one D2H + one H2D per row over PCIe + the on-device partition
and join kernels. K is sized so the largest bucket's build side fits the
row budget. NULL keys never match (inner join) and are dropped at the
partition spill.
"""
from typing import Dict, Iterator, List, Tuple

import torch


def _spill_partitioned(gq, keys, payload, nbuckets, runs):
    """partition one device chunk and append per-bucket host slices."""
    perm, counts = gq.partition_perm(keys, nbuckets)
    pk = gq.gather(keys, perm)
    pcols = {n: gq.gather(t, perm) for n, t in payload.items()}
    splits = counts.cpu().tolist()
    off = 0
    for b, c in enumerate(splits):
        if c == 0:
            off += c
            continue
        hk = torch.empty(c, dtype=pk.dtype, pin_memory=True)
        hk.copy_(pk[off:off + c], non_blocking=True)
        hp = {}
        for n, t in pcols.items():
            h = torch.empty(c, dtype=t.dtype, pin_memory=True)
            h.copy_(t[off:off + c], non_blocking=True)
            hp[n] = h
        runs[b].append((hk, hp))
        off += c
    torch.cuda.synchronize()


def external_hash_join(build_batches: Iterator[Tuple[torch.Tensor, Dict[str, torch.Tensor]]],
                       probe_batches: Iterator[Tuple[torch.Tensor, Dict[str, torch.Tensor]]],
                       budget_rows: int, nbuckets: int = 16):
    """Inner equi-join of two streams too large to co-reside on device.
    Each element: (int64 key tensor, payload dict), non-null keys.
    Yields (build_key, build_payload, probe_payload) device column dicts
    per (bucket, probe-slice) — concatenation = the full inner join
    (row order nondeterministic, as the in-core join's is)."""
    from . import gpuq as gq
    bruns: List[list] = [[] for _ in range(nbuckets)]
    pruns: List[list] = [[] for _ in range(nbuckets)]
    for keys, payload in build_batches:
        n = keys.numel()
        for lo in range(0, n, budget_rows):
            hi = min(lo + budget_rows, n)
            _spill_partitioned(gq, keys[lo:hi],
                               {k: t[lo:hi] for k, t in payload.items()},
                               nbuckets, bruns)
        del keys, payload
    for keys, payload in probe_batches:
        n = keys.numel()
        for lo in range(0, n, budget_rows):
            hi = min(lo + budget_rows, n)
            _spill_partitioned(gq, keys[lo:hi],
                               {k: t[lo:hi] for k, t in payload.items()},
                               nbuckets, pruns)
        del keys, payload

    for b in range(nbuckets):
        if not bruns[b] or not pruns[b]:
            continue
        bk = torch.cat([r[0] for r in bruns[b]]).cuda(non_blocking=True)
        bpay = {n: torch.cat([r[1][n] for r in bruns[b]])
                .cuda(non_blocking=True) for n in bruns[b][0][1]}
        torch.cuda.synchronize()
        bn = bk.numel()
        assert bn <= budget_rows * max(1, len(bruns[b])), "bucket skew"
        cap = 1 << max(4, int(bn * 2 - 1).bit_length() if bn else 4)
        ws = gq.join_build(bk, cap)
        for hk, hp in pruns[b]:
            pk = hk.cuda(non_blocking=True)
            ppay = {n: t.cuda(non_blocking=True) for n, t in hp.items()}
            torch.cuda.synchronize()
            out_cap = max(int(pk.numel() * 2) + 64, 64)
            while True:
                op, ob, nm = gq.join_probe(pk, ws, cap, bn, out_cap)
                if op is not None:
                    break
                out_cap = nm + 64
            if nm == 0:
                continue
            yield (gq.gather(bk, ob),
                   {n: gq.gather(t, ob) for n, t in bpay.items()},
                   {n: gq.gather(t, op) for n, t in ppay.items()})
