#!/usr/bin/env python3
"""Summarize a rocprofv3 --sys-trace rocpd database: merged timeline of
memory copies and kernels over the last few ms (pipeline overlap check).

Usage: python benchmarks/trace_summary.py <dir-with-*.db> [window_ms]
"""

import glob
import sqlite3
import sys


def main():
    d = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof_c2"
    window_ms = float(sys.argv[2]) if len(sys.argv) > 2 else 8.0
    db = glob.glob(f"{d}/*.db")[0]
    c = sqlite3.connect(db)
    c.row_factory = sqlite3.Row
    tabs = [t for (t,) in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    mc = [t for t in tabs if 'memory_copy' in t][0]
    kd = [t for t in tabs if 'kernel_dispatch' in t][0]
    r = c.execute(f"SELECT * FROM {mc} LIMIT 1").fetchone()
    print("copy cols:", list(dict(r).keys()) if r else None)
    r = c.execute(f"SELECT * FROM {kd} LIMIT 1").fetchone()
    print("kern cols:", list(dict(r).keys()) if r else None)
    copies = [dict(x) for x in c.execute(f"SELECT * FROM {mc} ORDER BY start")]
    kerns = [dict(x) for x in c.execute(f"SELECT * FROM {kd} ORDER BY start")]
    print(f"{len(copies)} copies, {len(kerns)} kernels")
    if not copies:
        return
    t1 = max(x['end'] for x in copies)
    t0 = t1 - window_ms * 1e6
    strs = {}
    for t in tabs:
        if 'rocpd_string' in t:
            for row in c.execute(f"SELECT * FROM {t}"):
                rd = dict(row)
                strs[rd.get('id')] = rd.get('string', rd.get('value', ''))
    evs = []
    for x in copies:
        if x['end'] >= t0:
            sz = x.get('size', 0) or 0
            q = x.get('queue_id', '?')
            st = x.get('stream_id', '?')
            name = strs.get(x.get('name_id'), '')
            evs.append((x['start'], x['end'],
                        f"COPY {sz>>20:3d}MB q={q} s={st} {str(name)[:44]}"))
    ksyms = {}
    for t in tabs:
        if 'kernel_symbol' in t:
            for row in c.execute(f"SELECT * FROM {t}"):
                rd = dict(row)
                ksyms[rd.get('id')] = rd.get('display_name',
                                             rd.get('kernel_name', ''))
    for x in kerns:
        if x['end'] >= t0:
            q = x.get('queue_id', '?')
            st = x.get('stream_id', '?')
            nm = ksyms.get(x.get('kernel_id'), '')
            evs.append((x['start'], x['end'],
                        f"KERN q={q} s={st} {str(nm)[:44]}"))
    evs.sort()
    base = evs[0][0]
    for s, e, dsc in evs[:120]:
        print(f"{(s-base)/1e6:8.3f} +{(e-s)/1e6:6.3f} ms  {dsc}")


if __name__ == "__main__":
    main()
