"""Adversarial workload-shape sweep: GPU worker vs oracle, bit-exact.

Runs a grid of job shapes (value sizes, run counts, compression, file-size
targets, snapshots, bottommost) beyond the fixed test fixtures and
bit-compares every output SST against the oracle.  GPU-box evidence tool
(not part of the pytest suite — it takes minutes).
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle  # noqa: E402
import toplingdb_amd as dcw  # noqa: E402


def run_case(tag, runs_n, entries, vlen, comp, tfs, snaps, bottom, seq0=1):
    tmp = tempfile.mkdtemp(prefix="sweep_", dir="/dev/shm")
    runs = []
    for r in range(runs_n):
        p = os.path.join(tmp, "in%d.sst" % r)
        dcw.gen_sst(p, seed=7 * (r + 1) + entries, num_entries=entries,
                    value_len=vlen, seq_base=seq0 + r * entries,
                    compression=comp)
        runs.append([p])
    og = os.path.join(tmp, "g")
    oo = os.path.join(tmp, "o")
    os.makedirs(og)
    os.makedirs(oo)
    kw = dict(compression=comp, target_file_size=tfs, snapshots=snaps,
              bottommost_level=1 if bottom else 0)
    rg = dcw.execute(dcw.make_job(runs, og, **kw))
    ro = oracle.execute(oracle.make_job(runs, oo, **kw))
    assert len(rg["files"]) == len(ro["files"]), (tag, rg, ro)
    for fg, fo in zip(rg["files"], ro["files"]):
        with open(fg["path"], "rb") as a, open(fo["path"], "rb") as b:
            da, db = a.read(), b.read()
        assert da == db, "%s: %s differs (%d vs %d bytes)" % (
            tag, fg["path"], len(da), len(db))
    print("OK %-28s files=%d entries=%d" %
          (tag, len(rg["files"]), rg["out_entries"]), flush=True)


def main():
    dcw.init(0)
    cases = [
        ("tiny-values", 3, 50000, 8, 1, 4 << 20, [], True),
        ("large-values", 2, 20000, 2048, 1, 16 << 20, [], True),
        ("huge-values-raw", 2, 8000, 8192, 0, 32 << 20, [], True),
        ("single-run", 1, 120000, 100, 1, 4 << 20, [], True),
        ("many-runs-nocomp", 8, 25000, 100, 0, 8 << 20, [], True),
        ("snapshots-mid", 4, 40000, 100, 1, 8 << 20, [40000, 90000], True),
        ("snapshots-notbottom", 4, 40000, 100, 1, 8 << 20, [40000], False),
        ("tiny-files", 2, 60000, 100, 1, 256 << 10, [], True),
        ("one-entry-runs", 3, 1, 100, 0, 1 << 20, [], True),
        ("mixed-comp-in-raw-out", 3, 30000, 100, 0, 4 << 20, [], True),
        ("value-len-one", 2, 50000, 1, 1, 4 << 20, [], True),
        ("not-bottom-raw", 3, 30000, 100, 0, 4 << 20, [], False),
    ]
    for c in cases:
        run_case(*c)
    print("SWEEP PASSED: %d shapes" % len(cases))
    dcw.shutdown()


if __name__ == "__main__":
    main()
