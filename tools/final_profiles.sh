cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
mkdir -p gpurun_out/prof
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o final_decode -- python bench.py --steps 64 --warmup 8 --prefill-len 0 --no-cpu-baseline --stats-steps 0 > gpurun_out/final_decode.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o final_prefill -- python bench.py --steps 2 --warmup 1 --prefill-len 2048 --no-cpu-baseline --stats-steps 0 > gpurun_out/final_prefill.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o final_fp8 -- python bench.py --model qwen3-32b-fp8 --steps 24 --warmup 4 --prefill-len 0 --no-cpu-baseline --stats-steps 0 > gpurun_out/final_fp8.log 2>&1
grep decode: gpurun_out/final_decode.log gpurun_out/final_fp8.log; grep prefill: gpurun_out/final_prefill.log
echo "=== soak: 8B 4096 steps + 70B 256 steps ==="
timeout 500 python bench.py --steps 4096 --warmup 16 --prefill-len 0 --no-cpu-baseline --stats-steps 0 --max-seq 8192 2>&1 | grep decode:
timeout 500 python bench.py --model llama3-70b --steps 256 --warmup 8 --prefill-len 0 --no-cpu-baseline --stats-steps 0 2>&1 | grep decode:
