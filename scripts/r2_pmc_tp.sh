#!/bin/bash
# GPU call #4: PMC counters on the attention kernels (what binds them)
# + emulated tp>1 profiles + llama/2.7b 8-GPU plan-search printouts.
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT"
OUT=gpurun_out/r2d
mkdir -p $OUT gpurun_out/profiles_r2

echo "== rocprof PMC: wait/issue/LDS buckets ==" | tee $OUT/summary.txt
(cd /tmp && timeout 420 rocprofv3 \
    --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
          SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE \
    -d $GRAFT_REPO_ROOT/$OUT/pmc1 -o att \
    -- python $GRAFT_REPO_ROOT/scripts/attn_pmc.py) 2>&1 | tail -3 | tee -a $OUT/summary.txt
(cd /tmp && timeout 420 rocprofv3 \
    --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_ACTIVE_INST_VALU \
          SQ_ACTIVE_INST_LDS SQ_WAIT_INST_LDS \
    -d $GRAFT_REPO_ROOT/$OUT/pmc2 -o att2 \
    -- python $GRAFT_REPO_ROOT/scripts/attn_pmc.py) 2>&1 | tail -3 | tee -a $OUT/summary.txt

echo "== emulated tp profiles (gpt3-2.7b) ==" | tee -a $OUT/summary.txt
for TP in 2 4 8; do
  timeout 500 python -m metis_amd.profiler.profile_model \
    --model gpt3-2.7b --bs 1,2,4,8,16 --iters 5 --emulate-tp $TP \
    --out gpurun_out/profiles_r2/gpt3-2.7b 2>&1 | tail -2
done
echo "== emulated tp profiles (llama3-8b) ==" | tee -a $OUT/summary.txt
for TP in 2 4 8; do
  timeout 500 python -m metis_amd.profiler.profile_model \
    --model llama3-8b --bs 1,2,4 --iters 4 --emulate-tp $TP \
    --out gpurun_out/profiles_r2/llama3-8b 2>&1 | tail -2
done
echo "== emulated tp profiles (gpt2-small, for search tests) ==" | tee -a $OUT/summary.txt
for TP in 2 4; do
  timeout 300 python -m metis_amd.profiler.profile_model \
    --model gpt2-small --bs 1,2,4,8 --iters 6 --emulate-tp $TP \
    --out gpurun_out/profiles_r2/gpt2-small 2>&1 | tail -2
done

echo "== 8-GPU plan search over the measured+emulated profiles ==" | tee -a $OUT/summary.txt
timeout 300 python - <<'EOF' 2>&1 | tee -a $OUT/summary.txt
from metis_amd.config import ModelConfig
from metis_amd.cli.plan_search import best_plan
from metis_amd.models.gpt import MODEL_SPECS as G
from metis_amd.models.llama import LLAMA_SPECS as L
for name, spec, gbs in [("gpt3-2.7b", G["gpt3-2.7b"], 128),
                        ("llama3-8b", L["llama3-8b"], 64)]:
    mc = ModelConfig(spec.name, spec.profile_num_layers, spec.hidden_size,
                     spec.seq_length, spec.vocab_size)
    for n in (2, 4, 8):
        found = best_plan(f"gpurun_out/profiles_r2/{name}", mc, n, gbs // (8 // n))
        print(name, f"N={n}", "->", found)
EOF
echo DONE | tee -a $OUT/summary.txt
