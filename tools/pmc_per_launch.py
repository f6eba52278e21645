"""Build profiles/pmc_per_launch.json from rocprofv3 --pmc SQLite dbs.

Usage:
  python tools/pmc_per_launch.py FETCH_db WRITE_db out.json [note]

FETCH_db / WRITE_db are results databases of two SEPARATE rocprofv3 passes
(--pmc FETCH_SIZE and --pmc WRITE_SIZE cannot share a pass).  Output maps
bench.py kernel labels -> average HBM bytes per launch:

  FETCH_SIZE / WRITE_SIZE are reported in KB and derive from TCC_EA0 request
  counters (MI355X_MICROARCH.md HBM section).  gfx950 FETCH_SIZE reports 1/2
  of the true bytes for wide coalesced streaming reads; per the guide it is
  doubled here (raw values kept alongside).  WRITE_SIZE is uncalibrated and
  reported as-is.
"""
import json
import re
import sqlite3
import sys
from collections import defaultdict

# mangled k_* symbol -> bench.py kernel-stats label (kbegin names)
LABEL = {
    "k_verify_usize": "verify_checksum",
    "k_decompress": "decompress",
    "k_count_entries": "count_entries",
    "k_decode_entries": "decode_entries",
    "k_merge_tiled": "merge_pair",
    "k_merge_pair": "merge_pair",
    "k_mark_heads": "mark_heads",
    "k_group_fsm": "group_fsm",
    "k_gather_survivors": "gather_survivors",
    "k_shared_prefix": "shared_prefix",
    "k_plan_next": "plan_next",
    "k_emit": "emit",
    "k_compress": "compress",
    "k_checksum": "checksum",
    "k_pack": "pack",
    "k_block_stats": "block_stats",
}


def cols(db, t):
    return [c[1].lower() for c in db.execute("PRAGMA table_info(%s)" % t)]


def per_kernel(path):
    """-> {label: (sum_counter_kb, n_launches)}"""
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    ev = next(t for t in tables if t == "rocpd_pmc_event")
    evc = cols(db, ev)
    val = next(c for c in evc if "value" in c)
    link = next(c for c in evc if "dispatch" in c or "corr" in c or
                c.endswith("_id"))
    disp = next(t for t in tables if "kernel_dispatch" in t.lower()
                and not t.endswith("_info"))
    dc = cols(db, disp)
    did = next(c for c in dc if c in ("id", "dispatch_id", "correlation_id"))
    kid = next(c for c in dc if "kernel" in c and "id" in c)
    sym = next(t for t in tables if "kernel_symbol" in t.lower())
    sc = cols(db, sym)
    sname = next(c for c in sc if c in ("display_name", "kernel_name", "name"))
    sid = next(c for c in sc if c == "id")
    q = ("SELECT s.%s, SUM(e.%s), COUNT(DISTINCT e.%s) FROM %s e "
         "JOIN %s d ON e.%s = d.%s JOIN %s s ON d.%s = s.%s GROUP BY s.%s" %
         (sname, val, link, ev, disp, link, did, sym, kid, sid, sname))
    agg = defaultdict(lambda: [0.0, 0])
    for name, v, n in db.execute(q):
        m = re.search(r"k_[a-z_0-9]+", str(name))
        if not m:
            continue
        label = LABEL.get(m.group(0))
        if not label:
            continue
        agg[label][0] += v or 0
        agg[label][1] += n or 0
    return agg


def main(fetch_db, write_db, out_path, note=""):
    f = per_kernel(fetch_db)
    w = per_kernel(write_db)
    kernels = {}
    for label in sorted(set(f) | set(w)):
        fs, fn = f.get(label, (0.0, 0))
        ws, wn = w.get(label, (0.0, 0))
        n = max(fn, wn, 1)
        kernels[label] = {
            "launches": n,
            # KB -> bytes; fetch doubled per the gfx950 calibration note
            "fetch_bytes_per_launch": round(fs * 1024 * 2 / max(fn, 1)) if fn else None,
            "fetch_bytes_per_launch_raw": round(fs * 1024 / max(fn, 1)) if fn else None,
            "write_bytes_per_launch": round(ws * 1024 / max(wn, 1)) if wn else None,
        }
    doc = {
        "source": {"fetch_db": fetch_db, "write_db": write_db},
        "correction": "FETCH_SIZE x2 per MI355X_MICROARCH.md (gfx950 reports "
                      "1/2 bytes for wide coalesced streaming reads; other "
                      "widths and WRITE_SIZE uncalibrated)",
        "units": "bytes per kernel launch (counter KB x 1024)",
        "note": note,
        "kernels": kernels,
    }
    with open(out_path, "w") as fp:
        json.dump(doc, fp, indent=1)
    print(json.dumps({k: v for k, v in kernels.items()}, indent=1))


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], sys.argv[3],
         sys.argv[4] if len(sys.argv) > 4 else "")
