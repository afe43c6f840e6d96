#!/bin/bash
# Round-2 first GPU call: regression + A/B evidence for the staged
# opt-in paths + attention microbench baseline. Everything lands in
# gpurun_out/r2ab/.
set -x
cd "$(dirname "$0")/.."
OUT=gpurun_out/r2ab
mkdir -p $OUT

echo "== GPU pytest (incl. gated experimental) ==" | tee $OUT/summary.txt
METIS_EXPERIMENTAL=1 timeout 600 python -m pytest tests -m gpu -q \
    2>&1 | tail -4 | tee -a $OUT/summary.txt

echo "== attention microbench (baseline) ==" | tee -a $OUT/summary.txt
timeout 300 python scripts/attn_bench.py --json $OUT/attn_base.json \
    2>&1 | tee -a $OUT/summary.txt

echo "== bench.py A/B: default vs FC1 epilogue ==" | tee -a $OUT/summary.txt
timeout 600 python bench.py --steps 6 --warmup 3 \
    > $OUT/bench_default.json 2> $OUT/bench_default.err
tail -1 $OUT/bench_default.json | tee -a $OUT/summary.txt
METIS_FC1_EPILOGUE=1 timeout 600 python bench.py --steps 6 --warmup 3 \
    > $OUT/bench_fc1.json 2> $OUT/bench_fc1.err
tail -1 $OUT/bench_fc1.json | tee -a $OUT/summary.txt

echo "== llama plan A/B: default vs qkv+rope ==" | tee -a $OUT/summary.txt
timeout 300 python -m metis_amd.cli.plan_runner --model llama3-1b \
    --plans "1,1,1,2,8" --steps 6 --warmup 2 --out $OUT/llama_default.json \
    2>&1 | tail -2 | tee -a $OUT/summary.txt
METIS_QKV_ROPE=1 timeout 300 python -m metis_amd.cli.plan_runner \
    --model llama3-1b --plans "1,1,1,2,8" --steps 6 --warmup 2 \
    --out $OUT/llama_qkvrope.json 2>&1 | tail -2 | tee -a $OUT/summary.txt

echo "== decode A/B: sdpa vs decode kernel ==" | tee -a $OUT/summary.txt
timeout 300 python scripts/decode_bench.py 2>&1 | tee -a $OUT/summary.txt
METIS_DECODE_KERNEL=1 timeout 300 python scripts/decode_bench.py \
    2>&1 | tee -a $OUT/summary.txt

echo DONE | tee -a $OUT/summary.txt
