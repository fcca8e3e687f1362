#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-kernel stats
table (total us, calls, us/call, share) — the profiles/kernel_stats_*
evidence format. Usage: python rocpd_stats.py <results.db>"""

import sqlite3
import sys


def main(path):
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    # locate the kernel dispatch table + name column across rocpd
    # schema revisions
    cand = [t for t in tables if "kernel_dispatch" in t]
    if not cand:
        print("tables:", tables)
        raise SystemExit("no kernel_dispatch table")
    kd = cand[0]
    cols = [r[1] for r in db.execute(f"PRAGMA table_info({kd})")]
    start = "start" if "start" in cols else "start_timestamp"
    end = "end" if "end" in cols else "end_timestamp"
    # kernel name: via kernel_id -> kernel info table -> string table
    ki = [t for t in tables if "kernel" in t and "info" in t.replace(
        "_", "")] or [t for t in tables if t.endswith("kernel")]
    rows = None
    for name_join in (
        f"SELECT s.string AS name, ({end}-{start}) AS dur FROM {kd} k "
        f"JOIN rocpd_info_kernel_symbol ks ON k.kernel_id = ks.id "
        f"JOIN rocpd_string s ON ks.kernel_name_id = s.id",
        f"SELECT ks.kernel_name AS name, ({end}-{start}) AS dur "
        f"FROM {kd} k JOIN rocpd_info_kernel_symbol ks "
        f"ON k.kernel_id = ks.id",
    ):
        try:
            rows = db.execute(name_join).fetchall()
            break
        except sqlite3.OperationalError as e:
            err = e
    if rows is None:
        # dump candidate schemas for manual inspection
        for t in tables:
            print(t, [r[1] for r in db.execute(f"PRAGMA table_info({t})")])
        raise SystemExit(f"no join worked: {err}")
    agg = {}
    for name, dur in rows:
        short = name.split("(")[0].split(".")[0]
        tot, cnt = agg.get(short, (0, 0))
        agg[short] = (tot + dur, cnt + 1)
    total = sum(t for t, _ in agg.values()) or 1
    print(f"{'kernel':<28} {'total_ms':>10} {'calls':>7} "
          f"{'us/call':>9} {'share':>6}")
    for name, (tot, cnt) in sorted(agg.items(), key=lambda x: -x[1][0]):
        print(f"{name:<28} {tot/1e6:>10.3f} {cnt:>7} "
              f"{tot/cnt/1e3:>9.1f} {100*tot/total:>5.1f}%")


if __name__ == "__main__":
    main(sys.argv[1])
