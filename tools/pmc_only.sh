#!/bin/bash
set -u
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
mkdir -p gpurun_out
rocprofv3 --pmc FETCH_SIZE -d /tmp/pmcf -o f -- \
  python bench.py --steps 1 --warmup 1 --jobs-in-flight 1 --skip-cpu-baseline --skip-hbm-resident \
  > /tmp/pmcf.json 2> /tmp/pmcf.log
rocprofv3 --pmc WRITE_SIZE -d /tmp/pmcw -o w -- \
  python bench.py --steps 1 --warmup 1 --jobs-in-flight 1 --skip-cpu-baseline --skip-hbm-resident \
  > /tmp/pmcw.json 2> /tmp/pmcw.log
FDB=$(find /tmp/pmcf -name "*.db" | head -1)
WDB=$(find /tmp/pmcw -name "*.db" | head -1)
python tools/pmc_per_launch.py "$FDB" "$WDB" gpurun_out/pmc_per_launch.json \
  "r2-final capture: 2 jobs (steps1+warmup1, jif1), u16 tables + fast decoder" \
  > gpurun_out/pmc_summary.txt 2>&1 || cat gpurun_out/pmc_summary.txt
cp /tmp/pmcf.json gpurun_out/pmc_bench_f.json 2>/dev/null
# keep only small artifacts for the merge
echo done
