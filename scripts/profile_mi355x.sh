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

echo "== clock-capped profile (MI355X_LC @ ${CAP_MHZ} MHz) =="
if rocm-smi --setperfdeterminism "$CAP_MHZ" >/dev/null 2>&1; then
    python -m metis_amd.profiler.profile_model --model "$MODEL" --bs "$BS" \
        --device-type MI355X_LC --out profiles/mi355x_lc --iters "$ITERS" --warmup 3
    rocm-smi --resetperfdeterminism >/dev/null 2>&1 || true
else
    echo "rocm-smi clock capping unavailable; skipping MI355X_LC profile"
fi

echo "profiles written:"
ls profiles/mi355x profiles/mi355x_lc 2>/dev/null
