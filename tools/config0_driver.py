"""config0_driver — BASELINE.json configs[0]: the db_bench-style
fillrandom+compact stand-in (tools/db_bench_tool.cc:3468 fillrandom, :3590
compact), exercising the dcompact executor seam end-to-end on CPU — no GPU
(SURVEY.md §7 step 3).

Flow (mirrors the DB host's write path + remote compaction dispatch):
 1. fillrandom: N random Put(key16, value100) ops into an in-memory
    memtable; flush every `memtable_entries` ops into a sorted L0 run
    (newest version of a key wins within a memtable, like the real
    memtable's sequence ordering).
 2. compact: L0 -> L1 through the WORKER BOUNDARY (dcw_job_desc /
    dcw_job_result structs), served by the CPU oracle worker — the same
    struct layout a GPU box serves via libdcw.so.  Then L1 -> L2
    (bottommost) the same way.
 3. verify: the final L2 stream equals the newest-version view of all puts.

Usage: python tools/config0_driver.py [--keys 1000000] [--verify]
Prints one JSON line with fillrandom ops/s and compact MB/s.
"""
import argparse
import json
import os
import random
import shutil
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import oracle  # the CPU dcompact worker for this no-GPU config


def fillrandom(workdir, n_keys, memtable_entries, seed=0x746F706C696E6721):
    """returns (l0_runs, expected_newest: dict, put_ops, bytes_written)"""
    rnd = random.Random(seed)
    memtable = {}
    seq = 0
    runs = []
    expected = {}
    t0 = time.time()

    def flush():
        if not memtable:
            return
        entries = sorted(memtable.items())
        kvs = [(oracle.make_ikey(k, s, 1), v) for k, (s, v) in entries]
        p = os.path.join(workdir, "l0_%05d.sst" % len(runs))
        with open(p, "wb") as f:
            f.write(oracle.build_sst(
                kvs, oracle.default_table_opts(compression=1,
                                               orig_file_number=len(runs) + 1)))
        runs.append([p])
        memtable.clear()

    for _ in range(n_keys):
        k = b"%016d" % rnd.randrange(n_keys * 4)
        seq += 1
        v = (b"v%014d" % seq) * 7  # ~105 B, trimmed
        v = v[:100]
        memtable[k] = (seq, v)
        expected[k] = (seq, v)
        if len(memtable) >= memtable_entries:
            flush()
    flush()
    dt = time.time() - t0
    bytes_written = sum(os.path.getsize(r[0]) for r in runs)
    return runs, expected, n_keys / dt, bytes_written


def compact_through_seam(runs, outdir, bottommost, target_file_size=64 << 20):
    """One compaction job through the dcw boundary structs, served by the
    CPU worker (oracle.execute == dcw_execute's contract)."""
    os.makedirs(outdir, exist_ok=True)
    jd = oracle.make_job(runs, outdir, compression=1,
                         target_file_size=target_file_size,
                         bottommost_level=1 if bottommost else 0)
    t0 = time.time()
    r = oracle.execute(jd)
    dt = time.time() - t0
    return r, dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--keys", type=int, default=1_000_000)
    ap.add_argument("--memtable-entries", type=int, default=125_000)
    ap.add_argument("--workdir", default="/tmp/dcw_config0")
    ap.add_argument("--verify", action="store_true")
    args = ap.parse_args()

    shutil.rmtree(args.workdir, ignore_errors=True)
    os.makedirs(args.workdir)

    runs, expected, put_ops, l0_bytes = fillrandom(
        args.workdir, args.keys, args.memtable_entries)

    # L0 -> L1 (all L0 runs merged; like CompactRange's first stage)
    l1_dir = os.path.join(args.workdir, "L1")
    r1, t1 = compact_through_seam(runs, l1_dir, bottommost=False)
    l1_runs = [[f["path"] for f in r1["files"]]]  # one sorted level run
    # L1 -> L2 (bottommost)
    l2_dir = os.path.join(args.workdir, "L2")
    r2, t2 = compact_through_seam(l1_runs, l2_dir, bottommost=True)

    verified = None
    if args.verify:
        stream = []
        for f in r2["files"]:
            stream += oracle.read_sst(open(f["path"], "rb").read())
        got = {ik[:-8]: v for ik, v in stream}
        want = {k: v for k, (s, v) in expected.items()}
        verified = got == want
        assert len(stream) == len(expected), (len(stream), len(expected))
        assert verified, "L2 view != newest-version view of all puts"

    out = {
        "config": "db_bench-style fillrandom+compact, CPU dcompact worker "
                  "(BASELINE.json configs[0], plumbing, no GPU)",
        "keys": args.keys,
        "fillrandom_ops_per_s": round(put_ops),
        "l0_runs": len(runs),
        "l0_bytes": l0_bytes,
        "compact_l0l1_mbps": round(r1["in_bytes"] / t1 / 1e6, 2),
        "compact_l1l2_mbps": round(r2["in_bytes"] / t2 / 1e6, 2),
        "l2_files": len(r2["files"]),
        "l2_entries": r2["out_entries"],
        "verified": verified,
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
