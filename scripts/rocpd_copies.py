"""Attribute __amd_rocclr_copyBuffer dispatches in a rocpd kernel trace:
counts, grid sizes, and the kernel preceding each copy on its queue.
Usage: python scripts/rocpd_copies.py <results.db>"""
import sqlite3
import sys
from collections import Counter


def main():
    con = sqlite3.connect(sys.argv[1])
    cur = con.cursor()
    T = {t.split("_0000")[0].replace("rocpd_", ""): t for t in
         [r[0] for r in cur.execute(
             "SELECT name FROM sqlite_master WHERE type='table'")]}
    sym = dict(cur.execute(
        f"SELECT id, display_name FROM {T['info_kernel_symbol']}"))
    rows = list(cur.execute(
        f"SELECT kernel_id, queue_id, start, end, grid_size_x "
        f"FROM {T['kernel_dispatch']} ORDER BY start"))
    t1 = max(r[3] for r in rows)
    t0 = t1 - (t1 - min(r[2] for r in rows)) * 0.3
    tail = [r for r in rows if r[2] >= t0]
    cop = [i for i, r in enumerate(tail)
           if "copyBuffer" in (sym.get(r[0]) or "")]
    print("copies in window:", len(cop), "of", len(tail), "dispatches")
    print("grid_x:", Counter(tail[i][4] for i in cop).most_common(8))
    print("queues:", Counter(tail[i][1] for i in cop).most_common(8))
    prev_on_q = Counter()
    for i in cop:
        q = tail[i][1]
        for j in range(i - 1, -1, -1):
            if (tail[j][1] == q
                    and "copyBuffer" not in (sym.get(tail[j][0]) or "")):
                prev_on_q[(sym.get(tail[j][0]) or "?")
                          .split("(")[0][:44]] += 1
                break
    print("preceded by (same queue):", prev_on_q.most_common(10))


if __name__ == "__main__":
    main()
