#!/bin/bash
# Same-box sweep of host-side knobs for the end-to-end number.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
B="python bench.py --steps 12 --warmup 4 --skip-cpu-baseline --skip-hbm-resident"
run() {
  name=$1; shift
  env "$@" $B > gpurun_out/hs_$name.json 2> gpurun_out/hs_$name.log
  python -c "import json;d=json.load(open('gpurun_out/hs_$name.json'));print('$name: %.0f MB/s (%.1f ms/job)'%(d['value'],d['ms_per_step']))" 2>&1
}
run base      DCW_NOP=1
run tails64   DCW_TAIL_WORKERS=64
run tails96   DCW_TAIL_WORKERS=96
run segs8     DCW_READ_SEGS=8
run t64s8     DCW_TAIL_WORKERS=64 DCW_READ_SEGS=8
run jif14     DCW_TAIL_WORKERS=64 DCW_NOP=1 && true
python bench.py --steps 14 --warmup 4 --jobs-in-flight 14 --skip-cpu-baseline --skip-hbm-resident > gpurun_out/hs_jif14.json 2> gpurun_out/hs_jif14.log
python -c "import json;d=json.load(open('gpurun_out/hs_jif14.json'));print('jif14: %.0f MB/s'%d['value'])"
DCW_TAIL_WORKERS=64 python bench.py --steps 14 --warmup 4 --jobs-in-flight 12 --skip-cpu-baseline --skip-hbm-resident > gpurun_out/hs_jif12t64.json 2> gpurun_out/hs_jif12t64.log
python -c "import json;d=json.load(open('gpurun_out/hs_jif12t64.json'));print('jif12+t64: %.0f MB/s'%d['value'])"
