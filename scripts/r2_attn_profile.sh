#!/bin/bash
# GPU call #3: attention fwd v3 (packed P^T + tr16 + exp2 + defer-max)
# numerics + microbench, and re-profile with the 4-vs-2 probe +
# re-validate the cost model.
set -x
cd "$(dirname "$0")/.."
OUT=gpurun_out/r2c
mkdir -p $OUT gpurun_out/profiles_r2

echo "== attention numerics ==" | tee $OUT/summary.txt
timeout 600 python -m pytest tests/test_attn_bwd_gpu.py tests/test_ops_gpu.py \
    -m gpu -q 2>&1 | tail -4 | tee -a $OUT/summary.txt

echo "== attention microbench (v3) ==" | tee -a $OUT/summary.txt
timeout 300 python scripts/attn_bench.py --json $OUT/attn_v3.json \
    2>&1 | tee -a $OUT/summary.txt

echo "== re-profile (4-vs-2 probe) ==" | tee -a $OUT/summary.txt
timeout 400 python -m metis_amd.profiler.profile_model \
    --model gpt2-small --bs 1,2,4,8 --out gpurun_out/profiles_r2/gpt2-small \
    2>&1 | tail -4
timeout 900 python -m metis_amd.profiler.profile_model \
    --model gpt3-2.7b --bs 1,2,4,8,16 --iters 6 --out gpurun_out/profiles_r2/gpt3-2.7b \
    2>&1 | tail -5
timeout 700 python -m metis_amd.profiler.profile_model \
    --model llama3-8b --bs 1,2,4 --iters 5 --out gpurun_out/profiles_r2/llama3-8b \
    2>&1 | tail -3

echo "== validation sweeps (new probe) ==" | tee -a $OUT/summary.txt
MODEL=gpt2-small GBS=8 PROFILE_DIR=gpurun_out/profiles_r2/gpt2-small \
    timeout 600 python scripts/validate_cost_model.py 2>&1 | tail -6 | tee -a $OUT/summary.txt
MODEL=gpt2-small GBS=12 PROFILE_DIR=gpurun_out/profiles_r2/gpt2-small \
    timeout 600 python scripts/validate_cost_model.py 2>&1 | tail -7 | tee -a $OUT/summary.txt
MODEL=gpt3-2.7b GBS=16 MAX_BS=16 STEPS=5 PROFILE_DIR=gpurun_out/profiles_r2/gpt3-2.7b \
    timeout 900 python scripts/validate_cost_model.py 2>&1 | tail -7 | tee -a $OUT/summary.txt

echo "== flagship bench (attn v3) ==" | tee -a $OUT/summary.txt
timeout 600 python bench.py --steps 6 --warmup 3 2>$OUT/bench.err | tail -1 | tee -a $OUT/summary.txt
echo DONE | tee -a $OUT/summary.txt
