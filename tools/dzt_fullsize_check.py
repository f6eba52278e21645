"""Full-size BASELINE configs[3] parity evidence: L2->L3 compaction of
~4 GiB input into DcwZipTable, GPU vs oracle, whole files bit-compared.
Run on a GPU box:  python tools/dzt_fullsize_check.py
Prints one JSON line; keep the output under profiles/.
"""
import json
import os
import shutil
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import oracle
import toplingdb_amd as dcw

WORK = "/dev/shm/dzt_full"


def main():
    runs_n = int(os.environ.get("DZT_RUNS", "8"))
    entries = int(os.environ.get("DZT_ENTRIES", "7_300_000"))  # ~512 MiB/run
    shutil.rmtree(WORK, ignore_errors=True)
    os.makedirs(WORK)
    dcw.init(0)
    t0 = time.time()
    runs = []
    for r in range(runs_n):
        p = os.path.join(WORK, "in_%d.sst" % r)
        dcw.gen_sst(p, seed=4000 + r, num_entries=entries,
                    seq_base=1 + r * entries, compression=1)
        runs.append([p])
    in_bytes = sum(os.path.getsize(r[0]) for r in runs)
    t_gen = time.time() - t0

    og = os.path.join(WORK, "gpu")
    oo = os.path.join(WORK, "orc")
    os.makedirs(og)
    os.makedirs(oo)
    t0 = time.time()
    rg = dcw.execute(dcw.make_job(runs, og, compression=1, bottommost_level=1,
                                  output_table_factory=1,
                                  target_file_size=512 << 20))
    t_gpu = time.time() - t0
    t0 = time.time()
    ro = oracle.execute(oracle.make_job(runs, oo, compression=1,
                                        bottommost_level=1,
                                        output_table_factory=1,
                                        target_file_size=512 << 20))
    t_cpu = time.time() - t0

    assert rg["out_entries"] == ro["out_entries"]
    assert len(rg["files"]) == len(ro["files"])
    for fg, fo in zip(rg["files"], ro["files"]):
        a = open(fg["path"], "rb").read()
        b = open(fo["path"], "rb").read()
        assert a == b, "mismatch %s (%d vs %d bytes)" % (
            fg["path"], len(a), len(b))
    print(json.dumps({
        "check": "BASELINE configs[3]: ~4GiB L2->L3 into DcwZipTable, "
                 "GPU vs oracle bit-compare",
        "in_bytes": in_bytes,
        "out_files": len(rg["files"]),
        "out_bytes": rg["out_bytes"],
        "out_entries": rg["out_entries"],
        "bit_identical": True,
        "gpu_s": round(t_gpu, 2),
        "gpu_mbps": round(in_bytes / t_gpu / 1e6, 1),
        "cpu_oracle_s": round(t_cpu, 2),
        "cpu_oracle_mbps": round(in_bytes / t_cpu / 1e6, 1),
        "gen_s": round(t_gen, 1),
    }))
    dcw.shutdown()
    shutil.rmtree(WORK, ignore_errors=True)


if __name__ == "__main__":
    main()
