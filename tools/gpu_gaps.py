"""Summarize GPU idle gaps from a rocprofv3 kernel-trace SQLite db.

Usage: python tools/gpu_gaps.py <rocprofv3 .db file>

Prints total busy vs idle time on the kernel timeline and attributes each
idle gap to the kernel that preceded it (i.e. what the host was doing after
that kernel finished before the next launch reached the GPU).  Schema is
discovered dynamically (rocprofv3 table names vary across versions).
"""
import sqlite3
import sys
from collections import defaultdict


def find_dispatch_rows(db):
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    # prefer a view/table with name+start+end directly
    for t in tables:
        if "kernel" not in t.lower():
            continue
        cols = [c[1].lower() for c in db.execute("PRAGMA table_info(%s)" % t)]
        name_col = next((c for c in cols if c in
                         ("display_name", "kernel_name", "name")), None)
        start_col = next((c for c in cols if c in ("start", "start_timestamp",
                                                   "begin_ns", "start_ns")), None)
        end_col = next((c for c in cols if c in ("end", "end_timestamp",
                                                 "end_ns")), None)
        if name_col and start_col and end_col:
            return db.execute("SELECT %s,%s,%s FROM %s ORDER BY %s" %
                              (name_col, start_col, end_col, t, start_col)).fetchall()
    raise SystemExit("no kernel dispatch table found; tables: %s" % tables)


def main(path):
    db = sqlite3.connect(path)
    rows = find_dispatch_rows(db)
    if not rows:
        print("no dispatches found")
        return
    busy = 0
    gaps = defaultdict(float)
    counts = defaultdict(int)
    prev_end = None
    prev_name = None
    t0, t1 = rows[0][1], rows[-1][2]
    for name, start, end in rows:
        busy += end - start
        if prev_end is not None and start > prev_end:
            g = start - prev_end
            gaps[prev_name] += g
            counts[prev_name] += 1
        if prev_end is None or end > prev_end:
            prev_end = end
            prev_name = str(name).split("(")[0]
    span = t1 - t0
    print("timeline span %.1f ms, busy %.1f ms (%.0f%%), idle %.1f ms" %
          (span / 1e6, busy / 1e6, 100.0 * busy / span, (span - busy) / 1e6))
    print("idle attributed to the kernel preceding each gap:")
    for name, g in sorted(gaps.items(), key=lambda kv: -kv[1])[:12]:
        print("  after %-28s %8.1f ms in %5d gaps" % (name, g / 1e6, counts[name]))


if __name__ == "__main__":
    main(sys.argv[1])
