cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 200 python - <<'EOF' > gpurun_out/gemmtf.log 2>&1
import sys, time, numpy as np
sys.path.insert(0, ".")
import cake_amd
for MNK in [(4096,4096,4096), (2048,28672,4096), (2048,6144,4096)]:
    M,N,K = MNK
    x = (np.random.default_rng(0).standard_normal((M,K))*0.1).astype(np.float32)
    w = (np.random.default_rng(1).standard_normal((N,K))*0.1).astype(np.float32)
    cake_amd.op_linear(x, w)
    t0=time.perf_counter(); cake_amd.op_linear(x, w); el=time.perf_counter()-t0
    print(M,N,K,"wall(with transfers)", round(el*1e3,1), "ms")
EOF
tail -5 gpurun_out/gemmtf.log
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o longctx -- python bench.py --steps 24 --warmup 4 --prompt-len 2048 --prefill-len 0 --no-cpu-baseline --stats-steps 0 --max-seq 4096 > gpurun_out/longctx.log 2>&1
grep decode: gpurun_out/longctx.log
timeout 200 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o gemm4k -- python - <<'EOF' > /dev/null 2>&1
import sys, numpy as np
sys.path.insert(0, ".")
import cake_amd
x = (np.random.default_rng(0).standard_normal((4096,4096))*0.1).astype(np.float32)
w = (np.random.default_rng(1).standard_normal((4096,4096))*0.1).astype(np.float32)
for _ in range(5): cake_amd.op_linear(x, w)
EOF
python3 - <<'EOF'
import csv
for f in ('longctx','gemm4k'):
    print('==',f,'==')
    for r in csv.DictReader(open(f'gpurun_out/prof/{f}_kernel_stats.csv')):
        if float(r['Percentage']) > 1.0:
            print(r['Name'].split('(')[0][:50], r['Calls'], round(float(r['AverageNs'])/1e3,2),'us', r['Percentage'])
EOF
