"""Summarize a rocprofv3 results.db: per-kernel time breakdown and, when
the run collected PMC counters (--pmc, separate invocation — never
combined with trace domains), per-kernel counter totals.

Usage:
  rocprofv3 --kernel-trace --stats -d out -o run -- python bench.py ...
  python benchmarks/pmc_summarize.py out [kernel-name-filter]
"""

import glob
import sqlite3
import sys


def main():
    root = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out"
    filt = sys.argv[2] if len(sys.argv) > 2 else ""
    dbs = glob.glob(f"{root}/**/*results.db", recursive=True) + \
        glob.glob(f"{root}/*results.db")
    if not dbs:
        print(f"no results.db under {root}")
        return
    con = sqlite3.connect(dbs[0])
    cur = con.cursor()
    rows = cur.execute(
        "SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,"
        " AVG(k.end-k.start)/1e3"
        " FROM rocpd_kernel_dispatch k"
        " JOIN rocpd_info_kernel_symbol s ON k.kernel_id = s.id"
        " GROUP BY 1 ORDER BY 3 DESC").fetchall()
    tot = sum(r[2] for r in rows)
    print(f"{dbs[0]}  total GPU ms: {tot:.1f}")
    for name, calls, ms, us in rows:
        if filt and filt not in name:
            continue
        print(f"{name[:70]:70s} {calls:5d} {ms:9.2f}ms {us:8.1f}us "
              f"{100 * ms / max(tot, 1e-9):5.1f}%")

    # PMC tables are uuid-suffixed; join through dispatch if present
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    pmc = [t for t in tables if t.startswith("rocpd_pmc_event")]
    info = [t for t in tables if t.startswith("rocpd_info_pmc")]
    if pmc and info:
        cols = [c[1] for c in cur.execute(f"PRAGMA table_info({pmc[0]})")]
        print(f"\npmc table {pmc[0]} columns: {cols}")
        for r in cur.execute(f"SELECT * FROM {pmc[0]} LIMIT 5"):
            print("  ", r)


if __name__ == "__main__":
    main()
