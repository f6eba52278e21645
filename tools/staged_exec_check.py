"""Minimal staged-execute harness with progress markers.

Runs gen -> stage_inputs -> two staged executes on cuda:0 with a marker
printed after each step; useful for bisecting runtime issues on a GPU box
(pairs with DCW_SEGV_TRACE=1 for a native backtrace on crash)."""
import faulthandler
import os
import sys
import tempfile

faulthandler.enable()
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import toplingdb_amd as dcw

print("M1 init", flush=True)
dcw.init(0)
d = tempfile.mkdtemp(dir="/dev/shm")
runs = []
for r in range(2):
    p = os.path.join(d, "in%d.sst" % r)
    dcw.gen_sst(p, seed=1 + r, num_entries=20000, seq_base=1 + r * 20000,
                compression=1)
    runs.append([p])
outd = os.path.join(d, "out")
os.makedirs(outd)
print("M2 gen done", flush=True)
jd = dcw.make_job(runs, outd, compression=1)
h = dcw.stage_inputs(jd)
print("M3 staged", h, flush=True)
jd2 = dcw.make_job(runs, outd, compression=1, staged_handle=h)
r1 = dcw.execute(jd2)
print("M4 exec1 files=%d" % len(r1["files"]), flush=True)
jd3 = dcw.make_job(runs, outd, compression=1, staged_handle=h)
r2 = dcw.execute(jd3)
print("M5 exec2 files=%d" % len(r2["files"]), flush=True)
dcw.release_staged(h)
dcw.shutdown()
print("M6 done", flush=True)
