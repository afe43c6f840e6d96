#!/bin/bash
# Round-2: validate the opt-in kernel paths on a GPU box, then flip the
# defaults that win. One gpurun call per block; each also runs a short
# A/B of the wired path vs the default.
set -e
cd "$(dirname "$0")/.."

echo "== gated GPU numerics (all three opt-in paths) =="
METIS_EXPERIMENTAL=1 python -m pytest \
    tests/test_fused_mlp.py tests/test_ops_gpu.py -m gpu -q

echo "== A/B: hipBLASLt GELU epilogue fc1 (METIS_FC1_EPILOGUE) =="
python bench.py --steps 4 --warmup 2 > gpurun_out/ab_default.json
METIS_FC1_EPILOGUE=1 python bench.py --steps 4 --warmup 2 \
    > gpurun_out/ab_fc1_epilogue.json

echo "== A/B: fused qkv+rope (Llama) =="
python -m metis_amd.cli.plan_runner --model llama3-1b \
    --plans "1,1,1,2,4" --steps 3 --out gpurun_out/ab_llama_default.json
METIS_QKV_ROPE=1 python -m metis_amd.cli.plan_runner --model llama3-1b \
    --plans "1,1,1,2,4" --steps 3 --out gpurun_out/ab_llama_qkvrope.json

echo "(METIS_VP_CE needs tp>1: validate numerics via the gated kernel"
echo " test above; A/B on the 8-GPU node with torchrun tp=2)"
