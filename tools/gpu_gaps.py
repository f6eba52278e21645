"""Summarize GPU idle gaps from a rocprofv3 kernel-trace SQLite db.

Usage: python tools/gpu_gaps.py <rocprofv3 .db file>

Prints total busy vs idle time on the kernel timeline and attributes each
idle gap to the kernel that preceded it (i.e. what the host was doing after
that kernel finished before the next launch reached the GPU).
"""
import sqlite3
import sys
from collections import defaultdict


def main(path):
    db = sqlite3.connect(path)
    # rocpd schema: dispatch rows joined to kernel symbols for names
    rows = db.execute(
        """SELECT s.display_name, d.start, d.end
           FROM rocpd_kernel_dispatch_information d
           JOIN rocpd_info_kernel_symbol s
             ON d.kernel_id = s.id AND d.guid = s.guid
           ORDER BY d.start""").fetchall()
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
            prev_name = name.split("(")[0]
    span = t1 - t0
    print("timeline span %.1f ms, busy %.1f ms (%.0f%%), idle %.1f ms" %
          (span / 1e6, busy / 1e6, 100.0 * busy / span, (span - busy) / 1e6))
    print("idle attributed to the kernel preceding each gap:")
    for name, g in sorted(gaps.items(), key=lambda kv: -kv[1])[:12]:
        print("  after %-28s %8.1f ms in %5d gaps" % (name, g / 1e6, counts[name]))


if __name__ == "__main__":
    main(sys.argv[1])
