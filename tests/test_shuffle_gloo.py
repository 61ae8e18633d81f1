"""Multi-process exchange-plan tests on CPU (gloo, world_size=2): the same
shuffle.all_to_all_kv code path the GPU ranks use over RCCL.

Checks: counts exchange + all-to-all-v routing + per-rank local reduce
(oracle as checker) reproduces the global oracle result exactly.
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

WS = 2


def _worker(rank, port, fail_q):
    try:
        import torch.distributed as dist
        from vega_amd import datagen, shuffle
        import oracle_ctypes as oc
        import pyref

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=WS)

        n_per = 5000
        seed = 777
        # rank generates its shard of the global stream (a9 slicing at rank
        # granularity: contiguous chunks)
        k, v = datagen.uniform_pairs(seed, n_per, key_bits=10, start=rank * n_per)
        pk, pv, counts = shuffle.partition_cpu(k, v, WS)
        tk = torch.from_numpy(pk.copy())
        tv = torch.from_numpy(pv.copy())
        rk, rv = shuffle.all_to_all_kv(tk, tv, counts.tolist())
        rk = rk.numpy(); rv = rv.numpy()

        # ownership: every received key hashes to this rank
        assert (shuffle.bucket_of_np(rk, WS) == rank).all()

        # local reduce (oracle as checker)
        lk, lv = oc.reduce_by_key_i64(rk, rv, 1, 1)

        # gather all ranks' results and compare against global oracle
        obj = [None, None]
        dist.all_gather_object(obj, (lk.tolist(), lv.tolist()))
        if rank == 0:
            got = []
            for lk_i, lv_i in obj:
                got += list(zip(lk_i, lv_i))
            gk, gv = datagen.uniform_pairs(seed, n_per * WS, key_bits=10)
            ref = pyref.reduce_by_key(gk, gv)
            assert sorted(got) == sorted(ref.items())
            # no key owned by two ranks
            keys = [k for k, _ in got]
            assert len(keys) == len(set(keys))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # propagate to parent
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")
        raise


def test_exchange_reduce_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29531
    procs = [ctx.Process(target=_worker, args=(r, port, q)) for r in range(WS)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)


def test_partition_cpu_matches_gpu_semantics():
    # stable bucket-contiguous reorder; counts; hash ownership
    from vega_amd import datagen, shuffle
    k, v = datagen.uniform_pairs(3, 10_000, key_bits=12)
    pk, pv, counts = shuffle.partition_cpu(k, v, 8)
    assert counts.sum() == len(k)
    off = 0
    for b, c in enumerate(counts.tolist()):
        seg = pk[off:off + c]
        assert (shuffle.bucket_of_np(seg, 8) == b).all()
        off += c
    # multiset preserved
    import oracle_ctypes as oc
    assert oc.checksum_pairs(pk, pv) == oc.checksum_pairs(k, v)
    # stability: rows of one bucket keep row order -> (k,v) pairs in the
    # bucket appear in the same relative order as in the input
    b = shuffle.bucket_of_np(k, 8)
    for bb in range(8):
        sel = np.where(b == bb)[0]
        lo = counts[:bb].sum()
        assert (pk[lo:lo + len(sel)] == k[sel]).all()
        assert (pv[lo:lo + len(sel)] == v[sel]).all()
