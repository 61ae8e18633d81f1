"""CPU (gloo, world_size=2) coverage of the C2/C3 multi-GPU plans:
  - group_count with map-side pre-combine (skew-safe exchange)
  - sort_by_key via sampled range partition + exchange + local sort
Same shuffle.py code path the GPU ranks use over RCCL.
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

        n_per = 20000
        # ---------- C2 plan: Zipf group_count with pre-combine ----------
        k, v = datagen.zipf_pairs(555, n_per, s=1.1, keyspace=500, start=rank * n_per)
        # map-side pre-combine: local (key, count) via the oracle (checker)
        gk, gc = oc.group_count_i64(k, v, 1, 1)
        pk, pv, counts = shuffle.partition_cpu(gk, gc, WS)
        rk, rv = shuffle.all_to_all_kv(torch.from_numpy(pk.copy()),
                                       torch.from_numpy(pv.copy()), counts.tolist())
        fk, fv = oc.reduce_by_key_i64(rk.numpy(), rv.numpy(), 1, 1)
        obj = [None] * WS
        dist.all_gather_object(obj, (fk.tolist(), fv.tolist()))
        if rank == 0:
            got = []
            for a, b in obj:
                got += list(zip(a, b))
            allk, allv = datagen.zipf_pairs(555, n_per * WS, s=1.1, keyspace=500)
            ref = pyref.group_count(allk)
            assert sorted(got) == sorted(ref.items()), "C2 plan mismatch"
            # skew safety: hot key crossed the wire as <= WS rows
            assert len(pk) <= 500 + 1, "pre-combine did not shrink the exchange"

        # ---------- C3 plan: range-partitioned sort ----------
        k, v = datagen.uniform_pairs(666, n_per, key_bits=64, start=rank * n_per)
        tk = torch.from_numpy(k.copy())
        spl = shuffle.choose_splitters(tk, WS, samples=512)
        pk, pv, counts = shuffle.partition_range_cpu(k, v, spl.numpy())
        rk, rv = shuffle.all_to_all_kv(torch.from_numpy(pk.copy()),
                                       torch.from_numpy(pv.copy()), counts.tolist())
        sk, sv = oc.sort_by_key_i64(rk.numpy(), rv.numpy())
        obj = [None] * WS
        dist.all_gather_object(obj, (sk.tolist(), sv.tolist()))
        if rank == 0:
            cat_k, cat_v = [], []
            for a, b in obj:  # rank order = global order
                cat_k += a
                cat_v += b
            allk, allv = datagen.uniform_pairs(666, n_per * WS, key_bits=64)
            ok, ov = oc.sort_by_key_i64(allk, allv)
            assert cat_k == ok.tolist(), "C3 global key order mismatch"
            assert sorted(cat_v) == sorted(ov.tolist())
        # ---------- C4 plan: hash-partitioned inner join ----------
        na = nb = 8000
        ak, av = datagen.uniform_range_pairs(777, na, 4000, start=rank * na)
        bk, bv = datagen.uniform_range_pairs(888, nb, 4000, start=rank * nb)
        pak, pav, ac = shuffle.partition_cpu(ak, av, WS)
        pbk, pbv, bc = shuffle.partition_cpu(bk, bv, WS)
        rak, rav = shuffle.all_to_all_kv(torch.from_numpy(pak.copy()),
                                         torch.from_numpy(pav.copy()), ac.tolist())
        rbk, rbv = shuffle.all_to_all_kv(torch.from_numpy(pbk.copy()),
                                         torch.from_numpy(pbv.copy()), bc.tolist())
        jk, jva, jvb = oc.join_i64(rak.numpy(), rav.numpy(), rbk.numpy(), rbv.numpy(), 1, 1)
        obj = [None] * WS
        dist.all_gather_object(obj, (jk.tolist(), jva.tolist(), jvb.tolist()))
        if rank == 0:
            got = []
            for a, b, c in obj:
                got += list(zip(a, b, c))
            gak, gav = datagen.uniform_range_pairs(777, na * WS, 4000)
            gbk, gbv = datagen.uniform_range_pairs(888, nb * WS, 4000)
            assert sorted(got) == pyref.join(gak, gav, gbk, gbv), "C4 plan mismatch"

        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:
        import traceback
        fail_q.put(f"rank {rank}: {e}\n{traceback.format_exc()}")
        raise


def test_c2_c3_plans_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 29541, q)) for r in range(WS)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, "\n".join(errs)
    assert all(p.exitcode == 0 for p in procs)
