#!/bin/bash
# A/B the compress/decompress kernel variants on the GPU box (one call).
# Writes per-variant bench JSON + a parity check per variant combo into
# gpurun_out/.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
B="python bench.py --jobs-in-flight 1 --steps 6 --warmup 2 --skip-cpu-baseline --skip-hbm-resident"

for cv in 0 2 3; do
  DCW_COMPRESS_V=$cv $B > gpurun_out/ab_comp$cv.json 2> gpurun_out/ab_comp$cv.log
  echo "comp$cv: $(python -c "import json;d=json.load(open('gpurun_out/ab_comp$cv.json'));k=d['kernels'];print('%.1f MB/s compress %.2fms/job decomp %.2fms/job'%(d['value'],k['compress']['ms']/d['steps'],k['decompress']['ms']/d['steps']))" 2>&1)"
done
for dv in 1 2; do
  DCW_COMPRESS_V=2 DCW_DECOMP_V=$dv $B > gpurun_out/ab_dec$dv.json 2> gpurun_out/ab_dec$dv.log
  echo "dec$dv: $(python -c "import json;d=json.load(open('gpurun_out/ab_dec$dv.json'));k=d['kernels'];print('%.1f MB/s compress %.2fms/job decomp %.2fms/job'%(d['value'],k['compress']['ms']/d['steps'],k['decompress']['ms']/d['steps']))" 2>&1)"
done

# parity under each variant combo
for combo in "2 0" "3 0" "2 1" "2 2"; do
  set -- $combo
  DCW_COMPRESS_V=$1 DCW_DECOMP_V=$2 python -m pytest \
    tests/test_worker_gpu.py::test_8way_snappy_filecuts \
    tests/test_worker_gpu.py::test_oversize_incompressible_entry -x -q \
    > gpurun_out/ab_parity_c$1_d$2.log 2>&1
  echo "parity c$1 d$2: $(tail -1 gpurun_out/ab_parity_c$1_d$2.log)"
done

# phase attribution at the default jobs-in-flight (end-to-end primary)
DCW_PHASE_DEBUG=1 python bench.py --steps 12 --warmup 4 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/ab_jif10.json 2> gpurun_out/ab_jif10.log
echo "jif10: $(python -c "import json;d=json.load(open('gpurun_out/ab_jif10.json'));print(d['value'],'MB/s')" 2>&1)"
tail -4 gpurun_out/ab_jif10.log
