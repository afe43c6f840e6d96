#!/bin/bash
# Full MI355X profiling sweep on one GPU box (run under gpurun).
#
# Produces the planner's inputs:
#   profiles/mi355x/DeviceType.MI355X_tp1_bs{1,2,4,8}.json     (config #2)
#   profiles/mi355x_lc/DeviceType.MI355X_LC_tp1_bs{...}.json   (config #5,
#       clock-capped MI355X emulating a second, slower device type)
#
# TP>1 profiles need a multi-GPU box (the driver's round-end environment):
#   torchrun --nproc-per-node T -m metis_amd.profiler.profile_model ...
set -e
cd "$(dirname "$0")/.."

MODEL=${MODEL:-gpt2-small}
BS=${BS:-1,2,4,8}
ITERS=${ITERS:-8}
CAP_MHZ=${CAP_MHZ:-900}

echo "== full-clock profile ($MODEL, bs=$BS) =="
python -m metis_amd.profiler.profile_model --model "$MODEL" --bs "$BS" \
    --device-type MI355X --out profiles/mi355x --iters "$ITERS" --warmup 3

# The second (slower) device type is emulated by restricting the process
# to half the CUs (rocm-smi clock/power caps are rejected on this pool;
# HSA_CU_MASK gives a genuine compute-capacity asymmetry).
echo "== half-CU profile (MI355X_LC via HSA_CU_MASK) =="
HSA_CU_MASK=0:0-127 python -m metis_amd.profiler.profile_model \
    --model "$MODEL" --bs "$BS" \
    --device-type MI355X_LC --out "profiles/mi355x_lc/$MODEL" \
    --iters "$ITERS" --warmup 3

echo "profiles written:"
ls profiles/mi355x profiles/mi355x_lc 2>/dev/null
