"""Single-rank RCCL validation of the tiled overlapped exchange.

Two ranks cannot share one GPU under RCCL, so the N>1 exchange code is
validated here as a world-size-1 NCCL process group: all_to_all_single
degenerates to a self-copy, but the FULL code path runs — side comm
stream, cross-stream events, record_stream allocator pinning, per-tile
partition + gather, count exchange, concat. The world-2 semantics are
covered by the gloo tests; the 8-GPU measurement is the driver's SCALE
run."""
import os

import numpy as np
import pytest

import oracle

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu


def test_tiled_overlap_exchange_self():
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29613")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    assert torch.cuda.is_available()
    torch.cuda.set_device(0)
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1,
                                device_id=torch.device("cuda", 0))
    import bench
    w = bench.JoinWorkload.__new__(bench.JoinWorkload)
    from spark_amd import gpuq as gq
    w.gq, w.world, w.rank = gq, 1, 0
    n = 2_000_000
    bk = gq.gen_i64(seed=1, n=n, range_=1_000_000)
    bp = gq.gen_i64(seed=2, n=n)
    pk = gq.gen_i64(seed=3, n=n, range_=1_000_000)
    pp = gq.gen_i64(seed=4, n=n)
    (rbk, rbp), (rpk, rpp) = w._exchange_tiled([(bk, bp), (pk, pp)],
                                               ntiles=4)
    # world 1: the exchange is a (partitioned, self-routed) permutation —
    # same multiset per side, payload stays aligned with its key
    for k0, p0, k1, p1 in ((bk, bp, rbk, rbp), (pk, pp, rpk, rpp)):
        assert k1.numel() == n
        o0 = np.lexsort((p0.cpu().numpy(), k0.cpu().numpy()))
        o1 = np.lexsort((p1.cpu().numpy(), k1.cpu().numpy()))
        assert (k0.cpu().numpy()[o0] == k1.cpu().numpy()[o1]).all()
        assert (p0.cpu().numpy()[o0] == p1.cpu().numpy()[o1]).all()
    dist.destroy_process_group()


def test_exchange_node_multikey_validity_self():
    """GpuShuffleExchangeExec end-to-end on device (world-size-1 NCCL):
    composite partition keys (seed-chained Murmur3) + validity transport +
    the ShuffleExchangeLike stats + coalesced-partition serving."""
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29614")
    torch.cuda.set_device(0)
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1,
                                device_id=torch.device("cuda", 0))
    from spark_amd import exec as gx
    n = 500_000
    k1 = oracle.gen_i64(seed=601, n=n, range_=1000)
    k2 = oracle.gen_i64(seed=602, n=n, range_=1000)
    pay = oracle.gen_i64(seed=603, n=n)
    valid = oracle.gen_i64(seed=604, n=n, range_=5) != 0
    batch = gx.ColumnarBatch(
        {"k1": torch.from_numpy(k1).cuda(),
         "k2": torch.from_numpy(k2).cuda(),
         "p": torch.from_numpy(pay).cuda()},
        validity={"p": torch.from_numpy(
            np.packbits(valid, bitorder="little")).cuda()})
    node = gx.GpuShuffleExchangeExec(("k1", "k2"), gx.InputBatches([batch]))
    out = list(node.execute_columnar())[0]
    assert out.num_rows() == n
    gk1 = out.column("k1").cpu().numpy()
    gk2 = out.column("k2").cpu().numpy()
    gp = out.column("p").cpu().numpy()
    gv = np.unpackbits(out.validity("p").cpu().numpy(), count=n,
                       bitorder="little").astype(bool)
    o0 = np.lexsort((pay, k2, k1))
    o1 = np.lexsort((gp, gk2, gk1))
    assert (gk1[o1] == k1[o0]).all() and (gk2[o1] == k2[o0]).all()
    assert (gp[o1] == pay[o0]).all() and (gv[o1] == valid[o0]).all()
    stats = node.runtime_statistics()
    assert stats["rows_written"] == n
    assert sum(stats["bytes_by_partition"]) == n * (24 + 1)  # 3 cols + u8
    # AQE coalesce serving over the single map partition
    parts = node.get_shuffle_partitions([(0, 1)])
    assert parts[0].num_rows() == n
    dist.destroy_process_group()
