cd $GRAFT_REPO_ROOT
timeout 300 python -m pytest tests/test_gpu_parity.py -k "op_gemm" -q 2>&1 | tail -2
for G in 0 1; do
echo "== CAKE_GEMM256=$G =="
CAKE_GEMM256=$G timeout 200 python - <<'EOF'
import sys, numpy as np
sys.path.insert(0, ".")
import cake_amd, ctypes, time
# resident-buffer GEMM timing via the stats-free op path is transfer-bound;
# use engine prefill gemm timing instead: bench prefill only
EOF
CAKE_GEMM256=$G timeout 300 python bench.py --steps 4 --warmup 2 --prefill-len 2048 --no-cpu-baseline --stats-steps 0 2>&1 | grep prefill:
done
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o g256 -- python bench.py --steps 2 --warmup 1 --prefill-len 2048 --no-cpu-baseline --stats-steps 0 > /dev/null 2>&1
python3 - <<'EOF'
import csv
for r in csv.DictReader(open('gpurun_out/prof/g256_kernel_stats.csv')):
    if float(r['Percentage']) > 2:
        print(r['Name'].split('(')[0][:46], r['Calls'], round(float(r['AverageNs'])/1e3,1),'us', r['Percentage'],'%')
EOF
