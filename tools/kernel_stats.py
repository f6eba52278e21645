"""Per-kernel stats summary from a rocprofv3 kernel-trace SQLite db.

Usage: python tools/kernel_stats.py <results.db>
"""
import sqlite3
import sys
from collections import defaultdict


def find_rows(db):
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    for t in tables:
        if "kernel" not in t.lower():
            continue
        cols = [c[1].lower() for c in db.execute("PRAGMA table_info(%s)" % t)]
        name = next((c for c in cols if c in
                     ("display_name", "kernel_name", "name")), None)
        start = next((c for c in cols if c in ("start", "start_timestamp",
                                               "begin_ns", "start_ns")), None)
        end = next((c for c in cols if c in ("end", "end_timestamp", "end_ns")),
                   None)
        if name and start and end:
            return db.execute("SELECT %s,%s,%s FROM %s" % (name, start, end, t))
    raise SystemExit("no kernel dispatch table found")


def main(path):
    db = sqlite3.connect(path)
    agg = defaultdict(lambda: [0, 0.0])
    for name, s, e in find_rows(db):
        k = str(name).split("(")[0]
        agg[k][0] += 1
        agg[k][1] += (e - s) / 1e6
    total = sum(v[1] for v in agg.values())
    print("%-40s %8s %12s %10s %7s" % ("kernel", "calls", "total_ms",
                                       "avg_us", "pct"))
    for k, (n, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
        print("%-40s %8d %12.2f %10.1f %6.1f%%" %
              (k, n, ms, ms / n * 1000, 100.0 * ms / total))
    print("TOTAL busy: %.2f ms over %d dispatches" %
          (total, sum(v[0] for v in agg.values())))


if __name__ == "__main__":
    main(sys.argv[1])
