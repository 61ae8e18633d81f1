#!/usr/bin/env python3
"""Per-phase cycle breakdown of the onesweep scatter (VEGA_PHASE_PROF=1).

Runs dev_sort_reduce at --rows and prints the per-phase shader-cycle sums
accumulated across all scatter launches: 0 prefetch, 1 rank, 2 publish+
starts, 3 lookback, 4 reorder, 5 writeout. Localizes where the ~67% parked
wave cycles go before touching the kernel.

Usage (GPU box): VEGA_PHASE_PROF=1 python tools/phase_prof.py --rows 200000000
"""
import argparse
import ctypes
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

PHASES = ["prefetch", "rank", "publish+starts", "lookback", "reorder", "writeout"]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000_000)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--op", default="reduce", choices=["reduce", "sort"])
    args = ap.parse_args()
    if os.environ.get("VEGA_PHASE_PROF") not in ("1", "2"):
        print("set VEGA_PHASE_PROF=1 (phases) or 2 (+walk counters)", file=sys.stderr)
        sys.exit(2)
    import torch
    from vega_amd import gpu
    n = args.rows
    dev = torch.device("cuda")
    k = torch.empty(n, dtype=torch.int64, device=dev)
    v = torch.empty(n, dtype=torch.int64, device=dev)
    gpu.dev_gen_uniform(k, v, seed=1, key_bits=63)
    ws = gpu.alloc_ws(n)
    ok = torch.empty(n, dtype=torch.int64, device=dev)
    ov = torch.empty(n, dtype=torch.int64, device=dev)
    lib = gpu.lib()
    buf = (ctypes.c_ulonglong * 8)()
    # warmup + reset
    if args.op == "reduce":
        gpu.dev_sort_reduce(k, v, gpu.OP_SUM_I64, ok, ov, ws)
    else:
        ok.copy_(k); ov.copy_(v); gpu.dev_sort_pairs(ok, ov, ws)
    torch.cuda.synchronize()
    lib.vega_phase_prof_read(buf, 1)
    for _ in range(args.reps):
        if args.op == "reduce":
            gpu.dev_sort_reduce(k, v, gpu.OP_SUM_I64, ok, ov, ws)
        else:
            ok.copy_(k); ov.copy_(v); gpu.dev_sort_pairs(ok, ov, ws)
    torch.cuda.synchronize()
    rc = lib.vega_phase_prof_read(buf, 1)
    if rc != 0:
        print(f"vega_phase_prof_read rc={rc}", file=sys.stderr)
        sys.exit(1)
    total = sum(buf[:6])
    print(f"rows={n} reps={args.reps} op={args.op} total wave-cycles={total}")
    for i, name in enumerate(PHASES):
        print(f"  {i} {name:<15} {buf[i]:>16}  {100.0 * buf[i] / max(total, 1):6.2f}%")
    print(f"  lookback walk iterations={buf[6]}  publish stalls={buf[7]}")


if __name__ == "__main__":
    main()
