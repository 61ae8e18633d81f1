#!/usr/bin/env python3
"""bench.py — the driver-contract benchmark for the MI355X vega engine.

Measures BASELINE.json's metric: rows/sec (whole node) reduce_by_key i64->i64
on the C1 workload (1e9 (i64,i64) rows, uniform keys in [0,2^63), 256 logical
partitions). A "step" is one reduce_by_key pass over the batch already
resident in HBM (H2D/generation is untimed; collect is untimed — the
PCIe-inclusive rate is discussed in DESIGN.md).

  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched by the driver via torch.distributed.run, one rank per GPU;
   ranks read RANK/LOCAL_RANK/WORLD_SIZE from the env. Scaling is WEAK: each
   rank processes its own --rows shard of the global stream; the exchange is
   the counts-all-to-all + all-to-all-v over RCCL/xGMI.)

Rank 0 prints ONE JSON line per the contract, plus:
  roofline     — dominant kernel (radix_scatter) achieved GB/s vs the 8 TB/s
                 HBM peak, measured with HIP events on the launch stream
                 during the timed region (algorithmic bytes: DESIGN.md §Roofline)
  cpu_baseline — the CPU oracle (vega's own bucketed-HashMap algorithm,
                 oracle/oracle.c, OpenMP over partitions) timed on this box's
                 host cores on a bounded sample (N=1 rank 0 only)
"""
import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

METRIC = ("rows/sec (whole node) reduce_by_key i64→i64, 1e9 rows, "
          "1/2/4/8 GPU")
WORKLOAD = ("C1: 1e9 (i64,i64) uniform keys in [0,2^63), reduce_by_key(sum), "
            "256 partitions (BASELINE.json configs[1]); weak scaling: "
            "--rows per GPU")
HBM_PEAK_GBS = 8000.0          # spec peak, MI355X_MICROARCH.md
SCATTER_ALGO_BYTES_PER_ROW = 32  # read (k,v) 16 B + write (k,v) 16 B per pass


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(rows, key_bits, seed):
    """Time the CPU oracle (kind=port: restatement of dependency.rs:176-223 +
    shuffled_rdd.rs:153-169) on a bounded sample of the same workload."""
    sys.path.insert(0, os.path.join(ROOT, "tests"))
    import oracle_ctypes as oc
    from vega_amd import datagen
    cores = os.cpu_count()
    pilot = min(rows, 2_000_000)
    k, v = datagen.uniform_pairs(seed, pilot, key_bits=key_bits)
    t0 = time.perf_counter()
    oc.reduce_by_key_i64(k, v, 256, 256)
    dt = time.perf_counter() - t0
    rate = pilot / dt
    sample = int(min(rows, max(pilot, min(rate * 15.0, 400_000_000))))
    k, v = datagen.uniform_pairs(seed, sample, key_bits=key_bits)
    t0 = time.perf_counter()
    oc.reduce_by_key_i64(k, v, 256, 256)
    dt = time.perf_counter() - t0
    return {
        "value": sample / dt,
        "unit": "rows/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{sample} rows of the same uniform-key stream "
                  f"({dt:.1f}s on {cores} host cores, OpenMP)",
    }


def read_traffic_calibration(rows):
    """Measured HBM bytes per launch for the dominant kernel, scaled from the
    committed rocprofv3 PMC calibration (profiles/traffic_calib.json holds
    bytes/row measured via FETCH_SIZE[x2 gfx950 correction]+WRITE_SIZE)."""
    p = os.path.join(ROOT, "profiles", "traffic_calib.json")
    if os.path.exists(p):
        try:
            bpr = json.load(open(p)).get("radix_scatter_hbm_bytes_per_row")
            return round(bpr * rows / 1e9, 2) if bpr else None  # GB per launch
        except Exception:
            return None
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows", type=int, default=1_000_000_000,
                    help="rows per GPU (weak scaling)")
    ap.add_argument("--key-bits", type=int, default=63)
    ap.add_argument("--seed", type=int, default=0xC0FFEE + 1)  # config C1
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--op", choices=["reduce", "group_count", "sort", "join"],
                    default="reduce",
                    help="reduce = C1 (driver default); group_count = C2 "
                         "semantics (with map-side pre-combine before the "
                         "exchange); sort = C3 sort_by_key")
    ap.add_argument("--dist", choices=["uniform", "zipf"], default="uniform")
    ap.add_argument("--dtype", choices=["i64", "f64"], default="i64",
                    help="value type for --op reduce (f64 = SURVEY C1's f64-sum variant)")
    ap.add_argument("--zipf-keyspace", type=int, default=100_000_000)
    ap.add_argument("--force-dist", action="store_true",
                    help="run the distributed exchange path (RCCL init, "
                    "counts all-to-all, all-to-all-v) even at WORLD_SIZE=1 — "
                    "hardware coverage of the collective plumbing on a "
                    "1-GPU box (launch via torch.distributed.run)")
    args = ap.parse_args()

    import torch
    from vega_amd import gpu, shuffle

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist_on = world > 1 or (args.force_dist and "MASTER_ADDR" in os.environ)
    if dist_on:
        import torch.distributed as dist
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        dist.init_process_group("nccl")
    else:
        torch.cuda.set_device(0)
    n_gpus = world if world > 1 else args.gpus
    assert n_gpus == world or world == 1, "launch N>1 via torch.distributed.run"

    rows = args.rows
    dev = torch.device("cuda")

    # ---- setup (untimed): generate the shard resident in HBM ----
    if args.dist == "zipf":
        from vega_amd import datagen
        hk, hv = datagen.zipf_pairs(args.seed, rows, s=1.1,
                                    keyspace=args.zipf_keyspace, start=rank * rows)
        k = torch.from_numpy(hk).to(dev)
        v = torch.from_numpy(hv).to(dev)
    elif args.dtype == "f64":
        k = torch.empty(rows, dtype=torch.int64, device=dev)
        v = torch.empty(rows, dtype=torch.float64, device=dev)
        gpu.dev_gen_uniform_f64(k, v, seed=args.seed, key_bits=args.key_bits,
                                start=rank * rows)
    else:
        k = torch.empty(rows, dtype=torch.int64, device=dev)
        v = torch.empty(rows, dtype=torch.int64, device=dev)
        gpu.dev_gen_uniform(k, v, seed=args.seed, key_bits=args.key_bits,
                            start=rank * rows)
    slack = 1.10 if dist_on else 1.0
    cap = int(rows * slack) + 1024
    ws = gpu.alloc_ws(cap)
    out_k = torch.empty(cap, dtype=torch.int64, device=dev)
    out_v = torch.empty(cap, dtype=torch.int64, device=dev)
    if dist_on or args.op in ("sort", "join"):
        pk = torch.empty(rows, dtype=torch.int64, device=dev)
        pv = torch.empty(rows, dtype=torch.int64, device=dev)

    nout = None

    def step_reduce(op):
        nonlocal nout
        if not dist_on:
            nout = gpu.dev_sort_reduce(k, v, op, out_k, out_v, ws)
        else:
            counts = gpu.dev_partition(k, v, world, pk, pv, ws)
            rk, rv = shuffle.all_to_all_kv(pk, pv, counts.astype(np.int64).tolist())
            if rk.numel() > cap:
                raise RuntimeError(f"recv shard {rk.numel()} > cap {cap}")
            nout = gpu.dev_sort_reduce(rk, rv, op, out_k, out_v, ws)

    def step_group_count():
        # C2 with map-side pre-combine (vega's own map-side combine,
        # dependency.rs:191-210): local (key,count) aggregate BEFORE the
        # exchange, so Zipf hot keys cross xGMI as one row per rank
        nonlocal nout
        if not dist_on:
            nout = gpu.dev_sort_reduce(k, v, gpu.OP_COUNT, out_k, out_v, ws)
            return
        nagg = gpu.dev_sort_reduce(k, v, gpu.OP_COUNT, out_k, out_v, ws)
        counts = gpu.dev_partition(out_k[:nagg], out_v[:nagg], world, pk, pv, ws)
        rk, rv = shuffle.all_to_all_kv(pk[:nagg], pv[:nagg],
                                       counts.astype(np.int64).tolist())
        nout = gpu.dev_sort_reduce(rk, rv, gpu.OP_SUM_I64, out_k, out_v, ws)

    def step_sort():
        nonlocal nout
        if not dist_on:
            out_k[:rows].copy_(k)
            out_v[:rows].copy_(v)
            gpu.dev_sort_pairs(out_k[:rows], out_v[:rows], ws)
            nout = rows
        else:
            spl = shuffle.choose_splitters(k, world)
            counts = gpu.dev_partition_range(k, v, spl, pk, pv, ws)
            rk, rv = shuffle.all_to_all_kv(pk, pv, counts.astype(np.int64).tolist())
            if rk.numel() > cap:
                raise RuntimeError(f"recv shard {rk.numel()} > cap {cap}")
            gpu.dev_sort_pairs(rk, rv, ws)
            nout = rk.numel()

    if args.op == "join":
        # C4: second side, keys uniform in [0, rows) both sides
        bk_h, bv_h = None, None
        from vega_amd import datagen as _dg
        ak_h, av_h = _dg.uniform_range_pairs(args.seed + 3, rows, rows, start=rank * rows)
        bk_h, bv_h = _dg.uniform_range_pairs(args.seed + 30, rows, rows, start=rank * rows)
        k = torch.from_numpy(ak_h).to(dev); v = torch.from_numpy(av_h).to(dev)
        kb = torch.from_numpy(bk_h).to(dev); vb = torch.from_numpy(bv_h).to(dev)
        jk = torch.empty(cap * 3, dtype=torch.int64, device=dev)
        jva = torch.empty(cap * 3, dtype=torch.int64, device=dev)
        jvb = torch.empty(cap * 3, dtype=torch.int64, device=dev)
        sb_k = torch.empty(cap, dtype=torch.int64, device=dev)
        sb_v = torch.empty(cap, dtype=torch.int64, device=dev)

    def _join_grouped_sides(ak_t, av_t, bk_t, bv_t):
        # group both sides; tag 4 = (h32,key) lex, tag 0 = full unsigned-key
        # order (narrow keys / fallback — ALREADY sorted, no extra pass).
        # Mixed tags: only the tag-4 side needs harmonizing to key order.
        ta = gpu.dev_group_pairs(ak_t, av_t, ws)
        tb = gpu.dev_group_pairs(bk_t, bv_t, ws)
        if ta == 4 and tb == 4:
            return gpu.dev_join_grouped(ak_t, av_t, bk_t, bv_t, 2, jk, jva, jvb, ws)
        if ta == 0 and tb == 0:  # both already in full unsigned-key order
            return gpu.dev_join_grouped(ak_t, av_t, bk_t, bv_t, 1, jk, jva, jvb, ws)
        # mixed tags (rare: one narrow-key side): harmonize BOTH to the
        # signed order dev_sort_pairs produces and merge with mode 0
        gpu.dev_sort_pairs(ak_t, av_t, ws)
        gpu.dev_sort_pairs(bk_t, bv_t, ws)
        return gpu.dev_join_grouped(ak_t, av_t, bk_t, bv_t, 0, jk, jva, jvb, ws)

    def step_join():
        nonlocal nout
        if not dist_on:
            out_k[:rows].copy_(k); out_v[:rows].copy_(v)
            sb_k[:rows].copy_(kb); sb_v[:rows].copy_(vb)
            nout = _join_grouped_sides(out_k[:rows], out_v[:rows],
                                       sb_k[:rows], sb_v[:rows])
        else:
            ca = gpu.dev_partition(k, v, world, pk, pv, ws)
            rak, rav = shuffle.all_to_all_kv(pk, pv, ca.astype(np.int64).tolist())
            cb = gpu.dev_partition(kb, vb, world, pk, pv, ws)
            rbk, rbv = shuffle.all_to_all_kv(pk, pv, cb.astype(np.int64).tolist())
            nout = _join_grouped_sides(rak, rav, rbk, rbv)

    step = {"reduce": lambda: step_reduce(
                gpu.OP_SUM_F64 if args.dtype == "f64" else gpu.OP_SUM_I64),
            "group_count": step_group_count,
            "sort": step_sort,
            "join": step_join}[args.op]

    def barrier_sync():
        torch.cuda.synchronize()
        if dist_on:
            import torch.distributed as dist
            dist.barrier()
            torch.cuda.synchronize()

    log(f"[rank {rank}] warmup x{args.warmup} (rows={rows})")
    for _ in range(args.warmup):
        step()
    barrier_sync()

    profiling = (rank == 0 and not dist_on)
    if profiling:
        gpu.prof_enable(True)

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if dist_on:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_rows = rows * n_gpus
    value = total_rows / (elapsed / args.steps)
    ms_per_step = elapsed / args.steps * 1000.0

    roofline = None
    if profiling:
        stats = gpu.prof_stats()
        sc = stats.get("radix_scatter")
        if sc and sc["n"] > 0:
            avg_ms = sc["ms"] / sc["n"]
            algo_gb = SCATTER_ALGO_BYTES_PER_ROW * rows / 1e9
            achieved = algo_gb / (avg_ms / 1000.0)
            roofline = {
                "bound": "hbm",
                "achieved": round(achieved, 1),
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": round(achieved / HBM_PEAK_GBS, 4),
                "traffic": read_traffic_calibration(rows),
                "kernel": "radix_scatter",
                "launches": sc["n"],
                "avg_launch_ms": round(avg_ms, 3),
            }
        log(f"[rank 0] kernel stats: {json.dumps(stats)}")

    if rank == 0:
        out = {
            "metric": METRIC,
            "value": round(value, 1),
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "float64" if args.dtype == "f64" else "int64",
            "data": "synthetic",
            "config": {
                "workload": WORKLOAD if args.op == "reduce" and args.dist == "uniform"
                else f"op={args.op} dist={args.dist} rows/GPU={rows} "
                     f"(C2/C3-style; the driver-default line is C1)",
                "rows_per_gpu": rows,
                "key_bits": args.key_bits,
                "nparts": 256,
                "distinct_keys_out": int(nout) if world == 1 else None,
            },
            "roofline": roofline,
            "cpu_baseline": (cpu_baseline_leg(rows, args.key_bits, args.seed)
                             if (world == 1 and not args.no_cpu_baseline) else None),
        }
        print(json.dumps(out), flush=True)

    if dist_on:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
