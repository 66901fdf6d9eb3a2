"""Summarize a rocprofv3 rocpd SQLite DB: per-kernel total/avg time.

ROCm 7.2's rocprofv3 writes results as a rocpd database (no stdout
--stats table); this prints the per-kernel dispatch summary the older
CSV gave us. Usage: python scripts/rocpd_stats.py <results.db> [top_n]
"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next((t for t in tables if "kernel_dispatch" in t), None)
    if disp is None:
        print("tables:", tables)
        sys.exit(1)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
    # find the kernel-name link: either a direct name column or a
    # kernel_id -> info_kernel_symbol join
    if "name" in cols:
        q = f"SELECT name, COUNT(*), SUM(end-start), AVG(end-start) FROM {disp} GROUP BY name"
    else:
        sym = next((t for t in tables if "kernel_symbol" in t), None)
        scols = [r[1] for r in cur.execute(f"PRAGMA table_info({sym})")]
        name_c = ("display_name" if "display_name" in scols
                  else "kernel_name" if "kernel_name" in scols else "name")
        key = ("kernel_id" if "kernel_id" in cols else "kernel_symbol_id")
        skey = ("id" if "id" in scols else "kernel_id")
        q = (f"SELECT s.{name_c}, COUNT(*), SUM(d.end-d.start), "
             f"AVG(d.end-d.start) FROM {disp} d JOIN {sym} s "
             f"ON d.{key}=s.{skey} GROUP BY s.{name_c}")
    rows = sorted(cur.execute(q), key=lambda r: -(r[2] or 0))[:top]
    total = sum(r[2] or 0 for r in rows)
    print(f"{'kernel':<58} {'calls':>6} {'total_us':>10} {'avg_us':>8}")
    for name, calls, tot, avg in rows:
        nm = name.split("(")[0][:57]
        print(f"{nm:<58} {calls:>6} {(tot or 0)/1e3:>10.1f} {(avg or 0)/1e3:>8.1f}")
    print(f"total (top {top}): {total/1e3:.1f} us")


if __name__ == "__main__":
    main()
