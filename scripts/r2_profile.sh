#!/bin/bash
# Round-2 GPU call #2: re-profile with the accumulation-marginal
# extension keys, then run the cost-model validation sweeps in all
# three estimator modes. Profiles land in gpurun_out/profiles_r2/ and
# are committed into profiles/mi355x/ back home.
set -x
cd "$(dirname "$0")/.."
OUT=gpurun_out/profiles_r2
mkdir -p $OUT

timeout 300 python -m metis_amd.profiler.profile_model \
    --model gpt2-small --bs 1,2,4,8 --out $OUT/gpt2-small \
    2>&1 | tail -5
timeout 600 python -m metis_amd.profiler.profile_model \
    --model gpt3-2.7b --bs 1,2,4,8,16 --iters 6 --out $OUT/gpt3-2.7b \
    2>&1 | tail -6
timeout 600 python -m metis_amd.profiler.profile_model \
    --model llama3-8b --bs 1,2,4 --iters 5 --out $OUT/llama3-8b \
    2>&1 | tail -4

echo "== validation sweeps =="
MODEL=gpt2-small GBS=8 PROFILE_DIR=$OUT/gpt2-small \
    timeout 600 python scripts/validate_cost_model.py 2>&1 | tail -8
MODEL=gpt2-small GBS=12 PROFILE_DIR=$OUT/gpt2-small \
    timeout 600 python scripts/validate_cost_model.py 2>&1 | tail -8
MODEL=gpt3-2.7b GBS=16 MAX_BS=16 STEPS=5 PROFILE_DIR=$OUT/gpt3-2.7b \
    timeout 900 python scripts/validate_cost_model.py 2>&1 | tail -8

echo "== hipblaslt epilogue probe =="
timeout 300 python - <<'EOF' 2>&1 | tail -12
import torch, metis_amd._hip_ops as ext
for (m, n, k) in [(4096, 4096, 4096), (8192, 10240, 2560), (32768, 10240, 2560)]:
    x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    try:
        y, aux = ext.lt_fc1_forward(x, w, b)
        ref = torch.nn.functional.gelu(x.float() @ w.float().T + b.float())
        err = (y.float() - ref).abs().max().item()
        print(f"GELU_AUX_BIAS {m}x{n}x{k}: OK maxerr {err:.4f}")
    except RuntimeError as e:
        print(f"GELU_AUX_BIAS {m}x{n}x{k}: {str(e)[:80]}")
EOF
echo DONE
