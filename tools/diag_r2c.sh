#!/bin/bash
# Round-2 diagnostics on the GPU box: host read bandwidth, kernel-trace
# stats, sys-trace gaps/copies, PMC FETCH/WRITE per-launch traffic, and
# the shim GPU end-to-end test.
set -u
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
mkdir -p gpurun_out

echo "== host read bw =="
timeout 240 python tools/read_bw.py > gpurun_out/read_bw.txt 2>&1
tail -9 gpurun_out/read_bw.txt

echo "== kernel-trace + stats (default jif) =="
rocprofv3 --kernel-trace --stats -d gpurun_out/kt -o kt -- \
  python bench.py --steps 8 --warmup 3 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/diag_kt.json 2> gpurun_out/diag_kt.log
python - <<'EOF'
import json
d = json.load(open('gpurun_out/diag_kt.json'))
print("kt bench:", d["value"], "MB/s")
EOF

echo "== sys-trace (copies + kernels) =="
rocprofv3 --sys-trace -d gpurun_out/st -o st -- \
  python bench.py --steps 5 --warmup 2 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/diag_st.json 2> gpurun_out/diag_st.log
DB=$(find gpurun_out/st -name "*.db" | head -1)
python tools/gpu_gaps.py "$DB" > gpurun_out/gaps_r2.txt 2>&1 || true
tail -20 gpurun_out/gaps_r2.txt

echo "== PMC FETCH/WRITE (1 job, jif1) =="
rocprofv3 --pmc FETCH_SIZE -d gpurun_out/pmcf -o f -- \
  python bench.py --steps 1 --warmup 1 --jobs-in-flight 1 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/pmcf.json 2> gpurun_out/pmcf.log
rocprofv3 --pmc WRITE_SIZE -d gpurun_out/pmcw -o w -- \
  python bench.py --steps 1 --warmup 1 --jobs-in-flight 1 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/pmcw.json 2> gpurun_out/pmcw.log
FDB=$(find gpurun_out/pmcf -name "*.db" | head -1)
WDB=$(find gpurun_out/pmcw -name "*.db" | head -1)
python tools/pmc_per_launch.py "$FDB" "$WDB" gpurun_out/pmc_per_launch.json \
  "r2 capture: 2 jobs (steps1+warmup1, jif1), decode word-window build" \
  > gpurun_out/pmc_summary.txt 2>&1
tail -30 gpurun_out/pmc_summary.txt

echo "== shim gpu end-to-end =="
LD_LIBRARY_PATH=toplingdb_amd timeout 240 shim/_build/shim_selftest gpu /dev/shm/shimw \
  > gpurun_out/shim_gpu.txt 2>&1
tail -2 gpurun_out/shim_gpu.txt
