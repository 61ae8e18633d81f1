"""Exchange edge cases on CPU (gloo): empty ranks / empty buckets through
the grouped-P2P all_to_all_kv (the self bucket moves by copy, peers by
batch_isend_irecv), and a world_size=4 routing check — the shapes the
8-GPU RCCL exchange must survive (skewed partitions can empty whole
buckets; a rank can own zero input rows)."""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _worker_empty(rank, port, fail_q):
    try:
        import torch.distributed as dist
        from vega_amd import datagen, shuffle
        import pyref

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=2)

        # rank 1 holds NO rows at all; rank 0's keys all hash to ONE owner
        if rank == 0:
            k = np.full(1000, 42, dtype=np.int64)  # single key -> one bucket
            v = np.arange(1000, dtype=np.int64)
        else:
            k = np.empty(0, dtype=np.int64)
            v = np.empty(0, dtype=np.int64)
        pk, pv, counts = shuffle.partition_cpu(k, v, 2)
        rk, rv = shuffle.all_to_all_kv(torch.from_numpy(pk.copy()),
                                       torch.from_numpy(pv.copy()),
                                       counts.tolist())
        owner = int(shuffle.bucket_of_np(np.array([42]), 2)[0])
        if rank == owner:
            assert rk.numel() == 1000
            assert (rk.numpy() == 42).all()
            assert sorted(rv.tolist()) == list(range(1000))
        else:
            assert rk.numel() == 0
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")


def _worker_w4(rank, port, fail_q):
    try:
        import torch.distributed as dist
        from vega_amd import datagen, shuffle
        import pyref

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=4)

        n_per = 3000
        k, v = datagen.uniform_pairs(99, n_per, key_bits=12, start=rank * n_per)
        pk, pv, counts = shuffle.partition_cpu(k, v, 4)
        rk, rv = shuffle.all_to_all_kv(torch.from_numpy(pk.copy()),
                                       torch.from_numpy(pv.copy()),
                                       counts.tolist())
        rk = rk.numpy(); rv = rv.numpy()
        assert (shuffle.bucket_of_np(rk, 4) == rank).all()
        got = pyref.reduce_by_key(rk, rv)
        obj = [None] * 4
        dist.all_gather_object(obj, sorted(got.items()))
        if rank == 0:
            allg = [p for part in obj for p in part]
            gk, gv = datagen.uniform_pairs(99, n_per * 4, key_bits=12)
            assert sorted(allg) == sorted(pyref.reduce_by_key(gk, gv).items())
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")


def _run(fn, ws, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=fn, args=(r, port, q)) for r in range(ws)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(120)
    fails = []
    while not q.empty():
        fails.append(q.get())
    assert fails == []
    assert all(p.exitcode == 0 for p in ps)


def test_exchange_empty_rank_and_bucket():
    _run(_worker_empty, 2, 29661)


def test_exchange_world4():
    _run(_worker_w4, 4, 29662)
