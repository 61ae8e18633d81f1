#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (kernel dispatch stats and,
when present, PMC counter totals per kernel). Used to distill gpurun_out/
profiles into the committed profiles/*.txt evidence.

Usage: python tools/rocpd_summary.py <results.db> [...]
"""
import sqlite3
import sys


def table(c, base):
    row = c.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE ?",
        (base + "%",)).fetchone()
    return row[0] if row else None


def summarize(path):
    print(f"==== {path}")
    c = sqlite3.connect(path)
    t_kd = table(c, "rocpd_kernel_dispatch")
    t_ks = table(c, "rocpd_info_kernel_symbol")
    t_s = table(c, "rocpd_string")
    t_pe = table(c, "rocpd_pmc_event")
    t_pi = table(c, "rocpd_info_pmc")

    cols_ks = [x[1] for x in c.execute(f"PRAGMA table_info({t_ks})")]
    name_col = "display_name" if "display_name" in cols_ks else "kernel_name"
    cols_kd = [x[1] for x in c.execute(f"PRAGMA table_info({t_kd})")]
    # resolve kernel display name: string table id or inline text
    row = c.execute(f"SELECT typeof({name_col}) FROM {t_ks} LIMIT 1").fetchone()
    name_is_id = row and row[0] in ("integer",)
    name_expr = (f"(SELECT string FROM {t_s} s WHERE s.id=ks.{name_col})"
                 if name_is_id else f"ks.{name_col}")

    q = f"""
      SELECT {name_expr} AS kname, COUNT(*) n,
             SUM(kd.end-kd.start)/1e6 tot_ms, AVG(kd.end-kd.start)/1e6 avg_ms,
             MIN(kd.end-kd.start)/1e6 min_ms, MAX(kd.end-kd.start)/1e6 max_ms
      FROM {t_kd} kd JOIN {t_ks} ks ON kd.kernel_id = ks.id
      GROUP BY kname ORDER BY tot_ms DESC"""
    print(f"{'kernel':64s} {'n':>5s} {'total_ms':>10s} {'avg_ms':>9s} {'min_ms':>9s} {'max_ms':>9s}")
    for kname, n, tot, avg, mn, mx in c.execute(q):
        kn = (kname or "?").split("(")[0][:64]
        print(f"{kn:64s} {n:5d} {tot:10.3f} {avg:9.4f} {mn:9.4f} {mx:9.4f}")

    if t_pe and c.execute(f"SELECT COUNT(*) FROM {t_pe}").fetchone()[0]:
        cols_pe = [x[1] for x in c.execute(f"PRAGMA table_info({t_pe})")]
        cols_pi = [x[1] for x in c.execute(f"PRAGMA table_info({t_pi})")]
        print("\nPMC totals per kernel (value summed over dispatches; "
              "NOTE gfx950 FETCH_SIZE under-reports wide coalesced reads 2x):")
        pi_name = "name" if "name" in cols_pi else ("symbol" if "symbol" in cols_pi else cols_pi[1])
        link = "dispatch_id" if "dispatch_id" in cols_pe else "event_id"
        try:
            q2 = f"""
              SELECT {name_expr} kname,
                     (SELECT {pi_name} FROM {t_pi} pi WHERE pi.id = pe.pmc_id) cname,
                     COUNT(*) n, SUM(pe.value) total
              FROM {t_pe} pe
              JOIN {t_kd} kd ON pe.{link} = kd.id
              JOIN {t_ks} ks ON kd.kernel_id = ks.id
              GROUP BY kname, cname ORDER BY total DESC"""
            for kname, cname, n, total in c.execute(q2):
                kn = (kname or "?").split("(")[0][:56]
                print(f"{kn:56s} {str(cname):12s} n={n:5d} sum={total:,.0f}")
        except Exception as e:
            print("  pmc join failed:", e)
            print("  pe cols:", cols_pe, "pi cols:", cols_pi)


if __name__ == "__main__":
    for p in sys.argv[1:]:
        summarize(p)
