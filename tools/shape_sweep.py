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


def run_gp_case(tag, mcb):
    import random
    tmp = tempfile.mkdtemp(prefix="sweepgp_", dir="/dev/shm")
    runs = []
    for r in range(2):
        p = os.path.join(tmp, "in%d.sst" % r)
        dcw.gen_sst(p, seed=31 + r, num_entries=50000,
                    seq_base=1 + r * 50000, compression=1)
        runs.append([p])
    rnd = random.Random(mcb & 0xFFFF)
    gps = []
    lo = b"\x00" * 16
    for g in range(48):
        hi = bytes([5 * g + rnd.randrange(1, 5)]) + bytes(
            rnd.randrange(256) for _ in range(15))
        if hi <= lo:
            continue
        gps.append((lo, hi, rnd.choice([1 << 20, 1 << 27, 1 << 31])))
        lo = (hi + b"\x01")[:16]
    og = os.path.join(tmp, "g")
    oo = os.path.join(tmp, "o")
    os.makedirs(og)
    os.makedirs(oo)
    kw = dict(compression=1, target_file_size=1 << 20,
              max_compaction_bytes=mcb, grandparents=gps)
    rg = dcw.execute(dcw.make_job(runs, og, **kw))
    ro = oracle.execute(oracle.make_job(runs, oo, **kw))
    assert len(rg["files"]) == len(ro["files"]), tag
    for fg, fo in zip(rg["files"], ro["files"]):
        with open(fg["path"], "rb") as a, open(fo["path"], "rb") as b:
            assert a.read() == b.read(), (tag, fg["path"])
    print("OK %-28s files=%d" % (tag, len(rg["files"])), flush=True)


def run_concurrent_staged_case():
    import concurrent.futures
    tmp = tempfile.mkdtemp(prefix="sweepc_", dir="/dev/shm")
    runs = []
    for r in range(4):
        p = os.path.join(tmp, "in%d.sst" % r)
        dcw.gen_sst(p, seed=77 + r, num_entries=60000,
                    seq_base=1 + r * 60000, compression=1)
        runs.append([p])
    jd0 = dcw.make_job(runs, tmp, compression=1)
    h = dcw.stage_inputs(jd0)
    dirs = []
    for i in range(4):
        d = os.path.join(tmp, "g%d" % i)
        os.makedirs(d)
        dirs.append(d)
    kw = dict(compression=1, target_file_size=8 << 20)
    with concurrent.futures.ThreadPoolExecutor(4) as ex:
        futs = []
        for i, d in enumerate(dirs):
            jd = dcw.make_job(runs, d, staged_handle=h, **kw)
            futs.append(ex.submit(dcw.execute, jd))
        results = [f.result() for f in futs]
    dcw.release_staged(h)
    oo = os.path.join(tmp, "o")
    os.makedirs(oo)
    ro = oracle.execute(oracle.make_job(runs, oo, **kw))
    for i, rg in enumerate(results):
        assert len(rg["files"]) == len(ro["files"])
        for fg, fo in zip(rg["files"], ro["files"]):
            with open(fg["path"], "rb") as a, open(fo["path"], "rb") as b:
                assert a.read() == b.read(), ("concurrent-staged", i, fg["path"])
    print("OK %-28s 4 concurrent jobs, shared staged inputs" %
          "concurrent-staged", flush=True)


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
    run_gp_case("grandparents-mcb-2g", 2 << 30)
    run_gp_case("grandparents-mcb-256m", 256 << 20)
    run_concurrent_staged_case()
    print("SWEEP PASSED: %d shapes" % (len(cases) + 3))
    dcw.shutdown()


def run_r2_feature_fuzz(seed, n_cases=20, start_exec=0, scale=1):
    """Round-2 feature-mix fuzz: random jobs drawing from {BBT, DZT} x
    {no filter, bloom} x {uniform16, mixed<=48 keys} x {snappy, zstd
    inputs} x {plain, tombstoned (bottommost envelope)} x {SST runs,
    flush stream}; every output file bit-compared GPU vs oracle."""
    import random
    rnd = random.Random(seed)
    for c in range(n_cases):
        tmp = tempfile.mkdtemp(prefix="sweepf_", dir="/dev/shm")
        mixed_keys = rnd.random() < 0.35
        flush = rnd.random() < 0.25
        dzt = (not mixed_keys) and rnd.random() < 0.35
        bloom = rnd.choice([0, 10000, 15500]) if not dzt else 0
        zstd_in = (not flush) and rnd.random() < 0.3
        tomb = ((not mixed_keys) and (not flush) and rnd.random() < 0.3)
        bottom = 1 if tomb else rnd.choice([0, 1])
        snaps = [] if tomb else (sorted(rnd.sample(range(1, 50000), 2))
                                 if rnd.random() < 0.4 else [])
        n_runs = rnd.choice([1, 2, 3])
        nent = rnd.choice([3000, 12000, 30000]) * scale
        tag = "fuzz%d dzt=%d bloom=%d mk=%d z=%d t=%d fl=%d b=%d" % (
            c, dzt, bloom, mixed_keys, zstd_in, tomb, flush, bottom)
        print("RUN", tag, flush=True)
        kw = dict(compression=rnd.choice([0, 1]),
                  target_file_size=rnd.choice([1 << 20, 64 << 20]),
                  snapshots=snaps, bottommost_level=bottom,
                  output_table_factory=1 if dzt else 0,
                  bloom_millibits_per_key=bloom)
        seq = 1
        def gen_kvs(n):
            nonlocal seq
            out = []
            for _ in range(n):
                if mixed_keys:
                    ln = rnd.choice([6, 16, 24, 44])
                    uk = bytes(rnd.getrandbits(8) for _ in range(ln))
                else:
                    uk = b"k%015d" % rnd.randrange(n * 4)
                t = 0 if rnd.random() < 0.15 else 1
                out.append((uk, seq, t, b"" if t == 0 else b"v%d" % seq))
                seq += 1
            out.sort(key=lambda e: (e[0], -e[1]))
            dedup = []
            for e in out:
                if dedup and dedup[-1][0] == e[0] and dedup[-1][1] == e[1]:
                    continue
                dedup.append(e)
            return [(oracle.make_ikey(k, s2, t2), v) for k, s2, t2, v in dedup]
        og = os.path.join(tmp, "g")
        oo = os.path.join(tmp, "o")
        os.makedirs(og)
        os.makedirs(oo)
        if c < start_exec:
            # advance the rng identically without executing
            if flush:
                gen_kvs(nent)
            else:
                for r in range(n_runs):
                    gen_kvs(nent)
                    if tomb:
                        for _ in range(rnd.randrange(1, 4)):
                            rnd.randrange(nent * 4)
                            rnd.randrange(1, nent)
                            rnd.randrange(100)
                    rnd.choice([0, 1]) if not zstd_in else None
            continue
        if flush:
            es = gen_kvs(nent)
            rg = dcw.execute(dcw.make_job([], og, flush_entries=es, **kw))
            ro = oracle.execute(oracle.make_job([], oo, flush_entries=es, **kw))
        else:
            runs = []
            for r in range(n_runs):
                es = gen_kvs(nent)
                ts = []
                if tomb:
                    for _ in range(rnd.randrange(1, 4)):
                        a = rnd.randrange(nent * 4)
                        b2 = a + rnd.randrange(1, nent)
                        ts.append((b"k%015d" % a, b"k%015d" % b2,
                                   seq + rnd.randrange(100)))
                opts = oracle.default_table_opts(
                    compression=7 if zstd_in else rnd.choice([0, 1]))
                p = os.path.join(tmp, "in%d.sst" % r)
                with open(p, "wb") as f:
                    f.write(oracle.build_sst(es, opts, tombstones=ts))
                runs.append([p])
            rg = dcw.execute(dcw.make_job(runs, og, **kw))
            ro = oracle.execute(oracle.make_job(runs, oo, **kw))
        assert rg["out_entries"] == ro["out_entries"], tag
        assert len(rg["files"]) == len(ro["files"]), tag
        for fg, fo in zip(rg["files"], ro["files"]):
            with open(fg["path"], "rb") as a, open(fo["path"], "rb") as b:
                assert a.read() == b.read(), (tag, fg["path"])
        print("OK %-48s files=%d entries=%d" %
              (tag, len(rg["files"]), rg["out_entries"]), flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "r2fuzz":
        dcw.init(0)
        run_r2_feature_fuzz(int(sys.argv[2]) if len(sys.argv) > 2 else 1,
                            int(sys.argv[3]) if len(sys.argv) > 3 else 20,
                            int(sys.argv[4]) if len(sys.argv) > 4 else 0,
                            int(sys.argv[5]) if len(sys.argv) > 5 else 1)
        dcw.shutdown()
        print("R2 FUZZ DONE")
    else:
        main()
