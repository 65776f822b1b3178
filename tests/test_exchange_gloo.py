"""Multi-process CPU tests of the exchange host logic (world_size 2, gloo):
the same exchange_columns() code the RCCL path runs, driven with
oracle-partitioned inputs (the oracle is the test driver/checker here; on
GPU the partitioning itself is gpuq_partition_perm, covered by
tests/test_gpu_parity.py)."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

WORLD = 2


def _worker(rank, port, fail_q):
    try:
        import torch
        import torch.distributed as dist
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(WORLD))
        dist.init_process_group("gloo")
        import oracle
        from spark_amd.exchange import exchange_columns

        # each rank generates its slice and partitions it with the oracle
        n = 10_000
        keys = oracle.gen_i64(seed=100 + rank, n=n)
        pay = oracle.gen_i64(seed=200 + rank, n=n)
        pids = oracle.partition_ids(keys, WORLD)
        perm = np.argsort(pids, kind="stable")
        counts = np.bincount(pids, minlength=WORLD).tolist()
        cols = {"k": torch.from_numpy(keys[perm]),
                "p": torch.from_numpy(pay[perm])}

        out, out_splits = exchange_columns(cols, counts)

        # every received key must hash to MY partition
        got_k = out["k"].numpy()
        got_p = out["p"].numpy()
        assert (oracle.partition_ids(got_k, WORLD) == rank).all()
        # rows arrive source-rank-major with source order preserved:
        # reconstruct the expected stream from both ranks' generators
        exp_k, exp_p = [], []
        for src in range(WORLD):
            sk = oracle.gen_i64(seed=100 + src, n=n)
            sp = oracle.gen_i64(seed=200 + src, n=n)
            sel = oracle.partition_ids(sk, WORLD) == rank
            exp_k.append(sk[sel])
            exp_p.append(sp[sel])
        exp_k = np.concatenate(exp_k)
        exp_p = np.concatenate(exp_p)
        assert (got_k == exp_k).all() and (got_p == exp_p).all()
        exp_splits = [np.count_nonzero(
            oracle.partition_ids(oracle.gen_i64(seed=100 + src, n=n), WORLD) == rank)
            for src in range(WORLD)]
        assert out_splits == exp_splits
        dist.destroy_process_group()
    except Exception as e:  # surface failures to the parent
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_exchange_columns_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_worker, args=(29531, q), nprocs=WORLD, join=True,
                       start_method="spawn")
    assert q.empty(), q.get()


def _worker_empty_split(rank, port, fail_q):
    try:
        import torch
        import torch.distributed as dist
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(WORLD))
        dist.init_process_group("gloo")
        from spark_amd.exchange import exchange_columns
        # rank 0 sends everything to rank 1; rank 1 sends nothing
        if rank == 0:
            cols = {"k": torch.arange(5, dtype=torch.int64)}
            splits = [0, 5]
        else:
            cols = {"k": torch.empty(0, dtype=torch.int64)}
            splits = [0, 0]
        out, out_splits = exchange_columns(cols, splits)
        if rank == 0:
            assert out["k"].numel() == 0
        else:
            assert out["k"].tolist() == [0, 1, 2, 3, 4]
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_exchange_empty_partitions():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_worker_empty_split, args=(29532, q), nprocs=WORLD,
                       join=True, start_method="spawn")
    assert q.empty(), q.get()


def _worker_broadcast(rank, port, fail_q):
    try:
        import torch
        import torch.distributed as dist
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(WORLD))
        dist.init_process_group("gloo")
        from spark_amd.exchange import broadcast_gather
        # unequal slice sizes per rank
        n = 5 + rank * 3
        cols = {"k": torch.arange(n, dtype=torch.int64) + rank * 100,
                "p": torch.arange(n, dtype=torch.float64) * (rank + 1)}
        out = broadcast_gather(cols)
        exp_k = torch.cat([torch.arange(5 + r * 3, dtype=torch.int64) + r * 100
                           for r in range(WORLD)])
        exp_p = torch.cat([torch.arange(5 + r * 3, dtype=torch.float64) * (r + 1)
                           for r in range(WORLD)])
        assert (out["k"] == exp_k).all() and (out["p"] == exp_p).all()
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_broadcast_gather_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_worker_broadcast, args=(29533, q), nprocs=WORLD,
                       join=True, start_method="spawn")
    assert q.empty(), q.get()


def _worker_range(rank, port, fail_q):
    try:
        import numpy as np
        import torch
        import torch.distributed as dist
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(WORLD))
        dist.init_process_group("gloo")
        import oracle
        from spark_amd.exchange import exchange_columns, broadcast_gather
        # host-logic shape of range_exchange with oracle/numpy as the test
        # driver (the device kernels are covered by -m gpu parity tests):
        n = 20_000
        keys = oracle.gen_i64(seed=300 + rank, n=n)
        sample = keys[::max(1, n // 1024)]
        gathered = broadcast_gather({"s": torch.from_numpy(sample.copy())})["s"].numpy()
        ss = np.sort(gathered)
        m = len(ss)
        bounds = ss[[(j + 1) * (m // WORLD) for j in range(WORLD - 1)]]
        pids = np.searchsorted(bounds, keys, side="left")
        perm = np.argsort(pids, kind="stable")
        counts = np.bincount(pids, minlength=WORLD).tolist()
        out, _ = exchange_columns({"k": torch.from_numpy(keys[perm])}, counts)
        got = np.sort(out["k"].numpy())
        # ranges are disjoint and ordered across ranks
        lo = -2**63 if rank == 0 else int(bounds[rank - 1])
        hi = 2**63 - 1 if rank == WORLD - 1 else int(bounds[rank])
        assert (got > lo).all() if rank > 0 else True
        assert (got <= hi).all()
        # global order: concatenating rank 0's sorted range then rank 1's is sorted
        mx = torch.tensor([int(got.max()) if len(got) else -2**63])
        mn = torch.tensor([int(got.min()) if len(got) else 2**63 - 1])
        maxs = [torch.zeros(1, dtype=torch.int64) for _ in range(WORLD)]
        mins = [torch.zeros(1, dtype=torch.int64) for _ in range(WORLD)]
        dist.all_gather(maxs, mx)
        dist.all_gather(mins, mn)
        for r in range(WORLD - 1):
            assert int(maxs[r]) <= int(mins[r + 1])
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_range_exchange_logic_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_worker_range, args=(29534, q), nprocs=WORLD,
                       join=True, start_method="spawn")
    assert q.empty(), q.get()


def _worker_validity(rank, port, fail_q):
    try:
        import torch
        import torch.distributed as dist
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(WORLD))
        dist.init_process_group("gloo")
        import oracle
        from spark_amd.exchange import exchange_columns

        # NULL payloads travel as u8 wire columns alongside the data (the
        # exec node's transport form: split points aren't byte-aligned, so
        # bitmaps are unpacked for the all-to-all and repacked on receive —
        # ColumnVector null contract, ColumnVector.java:58-366)
        n = 8_000
        keys = oracle.gen_i64(seed=300 + rank, n=n)
        pay = oracle.gen_i64(seed=400 + rank, n=n)
        null_mask = (oracle.gen_i64(seed=500 + rank, n=n, range_=4) == 0)
        pids = oracle.partition_ids(keys, WORLD)
        perm = np.argsort(pids, kind="stable")
        counts = np.bincount(pids, minlength=WORLD).tolist()
        cols = {"k": torch.from_numpy(keys[perm]),
                "p": torch.from_numpy(pay[perm]),
                "__valid__p": torch.from_numpy(
                    (~null_mask[perm]).astype(np.uint8))}
        out, _ = exchange_columns(cols, counts)
        got_k = out["k"].numpy()
        got_p = out["p"].numpy()
        got_v = out["__valid__p"].numpy().astype(bool)
        exp_k, exp_p, exp_v = [], [], []
        for src in range(WORLD):
            sk = oracle.gen_i64(seed=300 + src, n=n)
            sp = oracle.gen_i64(seed=400 + src, n=n)
            sm = (oracle.gen_i64(seed=500 + src, n=n, range_=4) == 0)
            sel = oracle.partition_ids(sk, WORLD) == rank
            exp_k.append(sk[sel]); exp_p.append(sp[sel]); exp_v.append(~sm[sel])
        exp_k = np.concatenate(exp_k)
        exp_p = np.concatenate(exp_p)
        exp_v = np.concatenate(exp_v)
        assert (got_k == exp_k).all()
        assert (got_v == exp_v).all()
        # data under NULL positions is garbage by contract; compare valid only
        assert (got_p[got_v] == exp_p[exp_v]).all()
        dist.destroy_process_group()
    except Exception as e:
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


def test_exchange_validity_wire_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    mp.start_processes(_worker_validity, args=(29534, q), nprocs=WORLD,
                       join=True, start_method="spawn")
    assert q.empty(), q.get()
