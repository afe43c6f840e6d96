#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
OUT=gpurun_out/r2g
mkdir -p $OUT gpurun_out/profiles_r2/gpt3-moe-1.3b-8e

echo "== MoE profiles (tp1 + emulated ep2/4) ==" | tee $OUT/summary.txt
timeout 500 python -m metis_amd.profiler.profile_model \
  --model gpt3-moe-1.3b-8e --bs 1,2,4 --iters 5 \
  --out gpurun_out/profiles_r2/gpt3-moe-1.3b-8e 2>&1 | tail -3 | tee -a $OUT/summary.txt
for TP in 2 4; do
  timeout 500 python -m metis_amd.profiler.profile_model \
    --model gpt3-moe-1.3b-8e --bs 1,2,4 --iters 4 --emulate-tp $TP \
    --out gpurun_out/profiles_r2/gpt3-moe-1.3b-8e 2>&1 | tail -2 | tee -a $OUT/summary.txt
done

echo "== qkv_rope A/B on llama3-8b ==" | tee -a $OUT/summary.txt
timeout 500 python -m metis_amd.cli.plan_runner --model llama3-8b \
  --plans "1,1,1,1,2" --steps 4 --warmup 2 --out $OUT/l8_default.json \
  2>&1 | tail -2 | tee -a $OUT/summary.txt
METIS_QKV_ROPE=1 timeout 500 python -m metis_amd.cli.plan_runner \
  --model llama3-8b --plans "1,1,1,1,2" --steps 4 --warmup 2 \
  --out $OUT/l8_qkvrope.json 2>&1 | tail -2 | tee -a $OUT/summary.txt
echo DONE | tee -a $OUT/summary.txt
