"""Per-kernel PMC counter totals from a rocprofv3 --pmc SQLite db.

Usage: python tools/pmc_traffic.py <results.db> [label]

Sums every counter in the db per kernel symbol (rocprofv3 one counter set
per pass; FETCH_SIZE / WRITE_SIZE units are KB per the MI355X microarch
guide).  Schema discovered dynamically.
"""
import sqlite3
import sys
from collections import defaultdict


def cols(db, t):
    return [c[1].lower() for c in db.execute("PRAGMA table_info(%s)" % t)]


def main(path, label=""):
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    ev = next(t for t in tables if t == "rocpd_pmc_event")
    evc = cols(db, ev)
    # find the dispatch link column and the value column
    val = next(c for c in evc if "value" in c)
    link = next((c for c in evc if "dispatch" in c or "corr" in c or
                 c.endswith("_id")), None)
    if link is None:
        print("rocpd_pmc_event columns:", evc)
        for t in tables:
            if "pmc" in t.lower() or "dispatch" in t.lower():
                print(t, cols(db, t))
        raise SystemExit("no link column")
    disp = next(t for t in tables if "kernel_dispatch" in t.lower()
                and not t.endswith("_info"))
    dc = cols(db, disp)
    did = next((c for c in dc if c in ("id", "dispatch_id", "correlation_id")),
               None)
    if did is None:
        print(disp, "columns:", dc)
        raise SystemExit("no dispatch id column")
    kid = next(c for c in dc if "kernel" in c and "id" in c)
    sym = next(t for t in tables if "kernel_symbol" in t.lower())
    sc = cols(db, sym)
    sname = next(c for c in sc if c in ("display_name", "kernel_name", "name"))
    sid = next(c for c in sc if c == "id")
    q = ("SELECT s.%s, SUM(e.%s) FROM %s e JOIN %s d ON e.%s = d.%s "
         "JOIN %s s ON d.%s = s.%s GROUP BY s.%s" %
         (sname, val, ev, disp, link, did, sym, kid, sid, sname))
    agg = defaultdict(float)
    for name, v in db.execute(q):
        agg[str(name).split("(")[0]] += v or 0
    total = sum(agg.values())
    print("%-40s %14s" % ("kernel [%s]" % label, "counter_sum"))
    for k, v in sorted(agg.items(), key=lambda kv: -kv[1]):
        print("%-40s %14.0f" % (k, v))
    print("TOTAL %47.0f" % total)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else "")
