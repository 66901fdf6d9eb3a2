"""Per-kernel PMC counter summary from a rocprofv3 rocpd DB.

Usage: python scripts/rocpd_pmc.py <results.db>
Prints, per kernel symbol, the SUM of each collected counter and the
derived stall split when the SQ wait counters are present:
  parked%  = SQ_WAIT_ANY / SQ_WAVE_CYCLES        (waitcnt/barrier)
  issue%   = SQ_WAIT_INST_ANY / SQ_WAVE_CYCLES   (pipe/RAW stall)
  active%  = SQ_ACTIVE_INST_ANY / SQ_WAVE_CYCLES
  mfma%    = SQ_VALU_MFMA_BUSY_CYCLES / (SQ_WAVE_CYCLES/4)
"""
import sqlite3
import sys
from collections import defaultdict


def main():
    con = sqlite3.connect(sys.argv[1])
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    ctab = next((t for t in tables if "counter" in t.lower()), None)
    if ctab is None:
        print("no counter table; tables:", tables)
        return
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({ctab})")]
    print(f"# table {ctab}: {cols}", file=sys.stderr)
    disp = next((t for t in tables if "kernel_dispatch" in t), None)
    sym = next((t for t in tables if "kernel_symbol" in t), None)
    scols = [r[1] for r in cur.execute(f"PRAGMA table_info({sym})")]
    name_c = ("display_name" if "display_name" in scols else "kernel_name")
    dcols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
    key = "kernel_id" if "kernel_id" in dcols else "kernel_symbol_id"
    skey = "id" if "id" in scols else "kernel_id"
    # counter table: expect dispatch_id, counter_id/name, value
    did = next(c for c in cols if "dispatch" in c)
    val = next(c for c in cols if c in ("value", "counter_value"))
    cid = next((c for c in cols if c in ("counter_id", "counter_name",
                                         "name", "id") and c != did), None)
    # map counter id -> name if an info table exists
    cname = {}
    itab = next((t for t in tables if "info" in t and "counter" in t), None)
    if itab:
        icols = [r[1] for r in cur.execute(f"PRAGMA table_info({itab})")]
        nm = next((c for c in icols if "name" in c), None)
        idc = next((c for c in icols if c == "id"), icols[0])
        if nm:
            for i, n in cur.execute(f"SELECT {idc}, {nm} FROM {itab}"):
                cname[i] = n
    agg = defaultdict(lambda: defaultdict(float))
    q = (f"SELECT s.{name_c}, c.{cid}, SUM(c.{val}) FROM {ctab} c "
         f"JOIN {disp} d ON c.{did} = d.id "
         f"JOIN {sym} s ON d.{key} = s.{skey} GROUP BY 1, 2")
    for name, c, v in cur.execute(q):
        agg[name.split("(")[0][:48]][cname.get(c, str(c))] = v
    for name, d in sorted(agg.items(),
                          key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0)):
        wc = d.get("SQ_WAVE_CYCLES", 0)
        line = name.ljust(50)
        if wc:
            for k, lbl in (("SQ_WAIT_ANY", "parked"),
                           ("SQ_WAIT_INST_ANY", "issue"),
                           ("SQ_ACTIVE_INST_ANY", "active")):
                if k in d:
                    line += f" {lbl}={100*d[k]/wc:5.1f}%"
            if "SQ_VALU_MFMA_BUSY_CYCLES" in d:
                line += f" mfma={100*d['SQ_VALU_MFMA_BUSY_CYCLES']/(wc/4):5.1f}%"
            line += f" waves_cyc={wc:.3g}"
        else:
            line += " " + " ".join(f"{k}={v:.3g}" for k, v in d.items())
        print(line)


if __name__ == "__main__":
    main()
