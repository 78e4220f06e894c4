cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
mkdir -p gpurun_out/prof
echo "=== bf16 decode wait/stall PMC ==="
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY --output-format csv -d gpurun_out/prof -o bf16pmc -- python bench.py --steps 4 --warmup 2 --prefill-len 0 --no-cpu-baseline --stats-steps 0 --no-graph > /dev/null 2>&1
python3 - <<'EOF'
import csv
from collections import defaultdict
agg = defaultdict(lambda: defaultdict(float))
for row in csv.DictReader(open('gpurun_out/prof/bf16pmc_counter_collection.csv')):
    k = row["Kernel_Name"].split("(")[0][:40]
    agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
for k, d in sorted(agg.items(), key=lambda x: -x[1].get("SQ_WAVE_CYCLES", 0))[:7]:
    wc = d.get("SQ_WAVE_CYCLES", 1)
    print(f"{k:40s} wait={d.get('SQ_WAIT_ANY',0)/wc*100:5.1f}% stall={d.get('SQ_WAIT_INST_ANY',0)/wc*100:5.1f}% active={d.get('SQ_ACTIVE_INST_ANY',0)/wc*100:5.1f}%")
EOF
echo "=== serve smoke (real engine, qwen3-0.6b) ==="
timeout 120 python -m cake_amd.serve --model qwen3-0.6b --port 8731 --max-seq 512 > gpurun_out/serve.log 2>&1 &
SPID=$!
sleep 30
curl -s -m 20 http://127.0.0.1:8731/v1/models | head -c 200; echo
curl -s -m 30 -X POST http://127.0.0.1:8731/v1/completions -H 'Content-Type: application/json' -d '{"prompt_token_ids": [1,2,3,4], "max_tokens": 8}' | head -c 300; echo
curl -s -m 30 -X POST http://127.0.0.1:8731/v1/chat/completions -H 'Content-Type: application/json' -d '{"prompt_token_ids": [5,6,7], "max_tokens": 4, "temperature": 0.8, "stream": true}' | head -c 300; echo
kill $SPID 2>/dev/null
echo "=== triple gpu suite ==="
for i in 1 2 3; do timeout 300 python -m pytest tests -m gpu -q 2>&1 | tail -1; done
