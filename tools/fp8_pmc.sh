cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
mkdir -p gpurun_out/prof
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY --output-format csv -d gpurun_out/prof -o fp8pmc -- python bench.py --model qwen3-32b-fp8 --steps 4 --warmup 2 --prefill-len 0 --no-cpu-baseline --stats-steps 0 --no-graph > gpurun_out/fp8pmc.log 2>&1
python3 - <<'EOF'
import csv
from collections import defaultdict
agg = defaultdict(lambda: defaultdict(float))
cnt = defaultdict(int)
for row in csv.DictReader(open('gpurun_out/prof/fp8pmc_counter_collection.csv')):
    k = row["Kernel_Name"].split("(")[0][:40]
    agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
    cnt[(k, row["Counter_Name"])] += 1
for k, d in sorted(agg.items(), key=lambda x: -x[1].get("SQ_WAVE_CYCLES", 0))[:8]:
    wc = d.get("SQ_WAVE_CYCLES", 1)
    print(f"{k:42s} waves_cyc={wc/1e6:9.1f}M wait={d.get('SQ_WAIT_ANY',0)/wc*100:5.1f}% issue_stall={d.get('SQ_WAIT_INST_ANY',0)/wc*100:5.1f}% active={d.get('SQ_ACTIVE_INST_ANY',0)/wc*100:5.1f}%")
EOF
