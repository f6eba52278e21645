#!/bin/bash
# Round-end confirmation on a fresh box: full GPU suite, smoke, and the
# bench family.  Outputs into gpurun_out/ for committing under profiles/.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

python -m pytest tests -m gpu -q > gpurun_out/re_pytest.log 2>&1
echo "suite: $(tail -1 gpurun_out/re_pytest.log)"
python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/re_smoke.log 2>&1
echo "smoke: $(tail -1 gpurun_out/re_smoke.log)"

python bench.py > gpurun_out/re_bench_default.json 2> gpurun_out/re_bench_default.log
# kernel-trace cross-check of the same command (contract: rocprof average
# duration for the dominant kernel must agree with the HIP-event timing)
export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -f csv -d /tmp/kt -o kt -- \
  python bench.py --steps 3 --warmup 1 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/re_kt_bench.json 2> /tmp/kt_run.log
for f in $(find /tmp/kt -name "*stats*.csv" 2>/dev/null); do cp "$f" gpurun_out/re_kt_$(basename $f); done
python - <<'EOF'
import json
d = json.loads([l for l in open('gpurun_out/re_bench_default.json') if l.strip().startswith('{')][-1])
print("default: %.0f MB/s (hbm %.0f, cpu %d cores %.0f => %.2fx; traffic %s)" % (
    d['value'], d['hbm_resident']['value'], d['cpu_baseline']['cores'],
    d['cpu_baseline']['value'], d['value']/d['cpu_baseline']['value'],
    d['roofline']['traffic']))
EOF
python bench.py --jobs-in-flight 1 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/re_bench_seq.json 2> gpurun_out/re_bench_seq.log
python - <<'EOF'
import json
d = json.loads([l for l in open('gpurun_out/re_bench_seq.json') if l.strip().startswith('{')][-1])
print("sequential: %.0f MB/s (%.1f ms/job)" % (d['value'], d['ms_per_step']))
EOF
python bench.py --table-factory dzt --steps 8 --warmup 3 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/re_bench_dzt.json 2> gpurun_out/re_bench_dzt.log
python - <<'EOF'
import json
d = json.loads([l for l in open('gpurun_out/re_bench_dzt.json') if l.strip().startswith('{')][-1])
print("dzt: %.0f MB/s" % d['value'])
EOF
python bench.py --job-mix mixed --mix-jobs 8 --skip-cpu-baseline --skip-hbm-resident \
  > gpurun_out/re_bench_mix.json 2> gpurun_out/re_bench_mix.log
python - <<'EOF'
import json
d = json.loads([l for l in open('gpurun_out/re_bench_mix.json') if l.strip().startswith('{')][-1])
print("mixed: %.0f MB/s over %s jobs" % (d['value'], d['config']['mix_jobs']))
EOF
