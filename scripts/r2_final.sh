#!/bin/bash
# GPU call: full validation sweep + flagship bench + rocprof kernel
# stats for the committed record.
set -x
cd "$GRAFT_REPO_ROOT"
OUT=gpurun_out/r2f
mkdir -p $OUT

echo "== full GPU pytest ==" | tee $OUT/summary.txt
timeout 700 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee -a $OUT/summary.txt

echo "== attention microbench (reverted r1 kernels) ==" | tee -a $OUT/summary.txt
timeout 300 python scripts/attn_bench.py --json $OUT/attn_r1_confirm.json \
    2>&1 | tee -a $OUT/summary.txt

echo "== decode bench (default-on kernel) ==" | tee -a $OUT/summary.txt
timeout 300 python scripts/decode_bench.py 2>&1 | tee -a $OUT/summary.txt

echo "== flagship bench ==" | tee -a $OUT/summary.txt
timeout 700 python bench.py --steps 8 --warmup 3 2>$OUT/bench.err | tail -1 | tee -a $OUT/summary.txt

echo "== rocprof kernel stats (flagship step) ==" | tee -a $OUT/summary.txt
(cd /tmp && export TMPDIR=/tmp && timeout 700 rocprofv3 --kernel-trace --stats \
    -d $GRAFT_REPO_ROOT/$OUT/prof -o flagship \
    -- python $GRAFT_REPO_ROOT/bench.py --steps 3 --warmup 2 \
    > $GRAFT_REPO_ROOT/$OUT/prof_bench.log 2>&1)
ls $OUT/prof* | tee -a $OUT/summary.txt
echo DONE | tee -a $OUT/summary.txt
