"""Timeline-gap analysis of a rocprofv3 rocpd SQLite DB.

Answers "is the step launch/dependency-bound or kernel-busy-bound?":
per HW queue prints busy vs span vs gap time, and overall the union
busy (any queue active) vs wall span. Large union-gap fraction =>
dispatch/dependency bubbles (fusion / graph layout work); small =>
kernel time itself is the bound.

Usage: python scripts/rocpd_gaps.py <results.db> [--tail-frac F]
  --tail-frac F  analyze only the last F fraction of the trace span
                 (default 0.5: skips warmup/capture noise)
"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    tail_frac = 0.5
    if "--tail-frac" in sys.argv:
        tail_frac = float(sys.argv[sys.argv.index("--tail-frac") + 1])
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next((t for t in tables if "kernel_dispatch" in t), None)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
    qcol = next((c for c in ("queue_id", "queue", "stream_id") if c in cols),
                None)
    rows = list(cur.execute(
        f"SELECT {qcol}, start, end FROM {disp} ORDER BY start"))
    if not rows:
        print("no dispatches")
        return
    t0 = min(r[1] for r in rows)
    t1 = max(r[2] for r in rows)
    lo = t1 - (t1 - t0) * tail_frac
    rows = [r for r in rows if r[1] >= lo]
    t0 = min(r[1] for r in rows)
    span = t1 - t0
    print(f"analyzing tail {tail_frac:.0%}: {len(rows)} dispatches, "
          f"span {span/1e6:.3f} ms")

    def union_busy(intervals):
        intervals = sorted(intervals)
        busy = 0
        ce = -1
        for s, e in intervals:
            if s > ce:
                busy += e - s
                ce = e
            elif e > ce:
                busy += e - ce
                ce = e
        return busy

    queues = {}
    for q, s, e in rows:
        queues.setdefault(q, []).append((s, e))
    print(f"{'queue':>8} {'disp':>6} {'busy_us':>10} {'gap_us':>10} "
          f"{'span_us':>10} {'busy%':>6}")
    for q, iv in sorted(queues.items()):
        b = union_busy(iv)
        qs = max(e for _, e in iv) - min(s for s, _ in iv)
        print(f"{str(q):>8} {len(iv):>6} {b/1e3:>10.1f} {(qs-b)/1e3:>10.1f} "
              f"{qs/1e3:>10.1f} {100.0*b/max(qs,1):>5.1f}%")
    ub = union_busy([(s, e) for _, s, e in rows])
    print(f"\nunion busy (any queue active): {ub/1e3:.1f} us "
          f"({100.0*ub/span:.1f}% of span); idle bubbles: "
          f"{(span-ub)/1e3:.1f} us")
    # top gap contributors on the busiest queue
    qmain = max(queues, key=lambda q: union_busy(queues[q]))
    iv = sorted(queues[qmain])
    gaps = []
    for i in range(1, len(iv)):
        g = iv[i][0] - iv[i - 1][1]
        if g > 0:
            gaps.append((g, iv[i - 1][1] - t0))
    gaps.sort(reverse=True)
    print(f"\nlargest gaps on queue {qmain} (us, at_ms_into_window):")
    for g, at in gaps[:15]:
        print(f"  {g/1e3:9.2f}  @{at/1e6:8.3f}")


def union_gap_attribution(db, tail_frac=0.3, top=12):
    """Largest ALL-queues-idle gaps with the kernels ending before and
    starting after each gap."""
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if "kernel_dispatch" in t)
    sym_t = next(t for t in tables if "info_kernel_symbol" in t)
    sym = dict(cur.execute(f"SELECT id, display_name FROM {sym_t}"))
    rows = list(cur.execute(
        f"SELECT kernel_id, start, end FROM {disp} ORDER BY start"))
    t1 = max(r[2] for r in rows)
    t0 = t1 - (t1 - min(r[1] for r in rows)) * tail_frac
    tail = [r for r in rows if r[1] >= t0]
    # union timeline
    iv = sorted((s, e, k) for k, s, e in tail)
    gaps = []
    ce = iv[0][1]
    last_k = iv[0][2]
    for s, e, k in iv[1:]:
        if s > ce:
            gaps.append((s - ce, last_k, k, ce))
        if e > ce:
            ce = e
            last_k = k
    gaps.sort(reverse=True)
    nm = lambda k: (sym.get(k) or "?").split("(")[0][:40]
    print("largest union-idle gaps (us, before -> after):")
    for g, kb, ka, at in gaps[:top]:
        print(f"  {g/1e3:8.2f}  {nm(kb)} -> {nm(ka)}")


if __name__ == "__main__":
    if "--attr" in sys.argv:
        union_gap_attribution(sys.argv[1])
    else:
        main()
