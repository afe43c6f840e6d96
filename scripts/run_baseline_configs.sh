#!/bin/bash
# How each BASELINE.json config is produced with this framework.
# Configs 1 runs anywhere; 2-5 need an MI355X (single GPU via gpurun;
# tp>1 / multi-GPU pieces need the 8-GPU node, marked [8-GPU]).
set -e
cd "$(dirname "$0")/.."

echo "=== Config 1: cost_homo_cluster on bundled-style profiles (pure CPU) ==="
# (the reference's bundled A100 samples live outside this repo; the
#  committed measured MI355X profiles play the same role)
python3 cost_homo_cluster.py --model_name gpt2-small --num_layers 14 --gbs 32 \
  --hidden_size 768 --sequence_length 1024 --vocab_size 51200 \
  --hostfile_path tests/data/mi355x_single_node/hostfile \
  --clusterfile_path tests/data/mi355x_single_node/clusterfile.json \
  --profile_data_path profiles/mi355x/gpt2-small \
  --max_profiled_tp_degree 1 --max_profiled_batch_size 8 \
  --comm_model alpha_beta --top_k 5

echo "=== Config 2: GPT-2-small profile on 1 MI355X + estimator validation ==="
echo "  python3 -m metis_amd.profiler.profile_model --model gpt2-small --bs 1,2,4,8"
echo "  MODEL=gpt2-small python3 scripts/validate_cost_model.py"
echo "  (committed results: profiles/mi355x/gpt2-small, profiles/validation/)"

echo "=== Config 3: GPT-3 2.7B homogeneous search [8-GPU for execution] ==="
python3 cost_homo_cluster.py --model_name gpt3-2.7b --num_layers 34 --gbs 128 \
  --hidden_size 2560 --sequence_length 2048 --vocab_size 51200 \
  --hostfile_path tests/data/mi355x_single_node/hostfile \
  --clusterfile_path tests/data/mi355x_single_node/clusterfile.json \
  --profile_data_path profiles/mi355x/gpt3-2.7b \
  --max_profiled_tp_degree 1 --max_profiled_batch_size 16 \
  --comm_model alpha_beta --top_k 5
echo "  best-plan execution: python bench.py --gpus 8 --plan-search  [8-GPU]"

echo "=== Config 4: Llama-3-8B profile + best-plan run [8-GPU] ==="
echo "  python3 -m metis_amd.profiler.profile_model --model llama3-8b --bs 1,2"
echo "  torchrun --nproc-per-node 8 -m metis_amd.cli.plan_runner \\"
echo "      --model llama3-8b --plans '8,1,1,2,16' --out measured.json  [8-GPU]"
echo "  (committed 1-GPU profile: profiles/mi355x/llama3-8b)"

echo "=== Config 5: hetero plan on emulated mixed cluster ==="
echo "  (MI355X_LC = HSA_CU_MASK=0:0-127 half-CU emulation; see"
echo "   scripts/profile_mi355x.sh; committed ranked output:"
echo "   results/mi355x_hetero_27b_cumask_gbs16.txt)"
mkdir -p /tmp/metis_mix27
cp profiles/mi355x/gpt3-2.7b/*.json profiles/mi355x_lc/gpt3-2.7b/*.json /tmp/metis_mix27/
python3 cost_het_cluster.py --model_name gpt3-2.7b --num_layers 34 --gbs 16 \
  --hidden_size 2560 --sequence_length 2048 --vocab_size 51200 \
  --hostfile_path tests/data/mi355x_hetero/hostfile \
  --clusterfile_path tests/data/mi355x_hetero/clusterfile.json \
  --profile_data_path /tmp/metis_mix27 \
  --max_profiled_tp_degree 1 --max_profiled_batch_size 8 \
  --min_group_scale_variance 1 --max_permute_len 4 --top_k 5
