#!/bin/bash
# Round-2 multi-GPU measurement plan for the 8-GPU MI355X node.
# (single-GPU gpurun boxes cannot run these; kept as the prepared recipe)
set -e
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== 1. RCCL calibration -> clusterfile + alpha_beta inputs =="
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
    -m metis_amd.profiler.comm_bench --out profiles/comm_bench_8gpu.json

echo "== 2. tp>1 profiles (planner tp axis; also = MoE ep axis) =="
for TP in 2 4 8; do
  torchrun --nnodes=1 --nproc-per-node $TP --master-addr 127.0.0.1 \
      -m metis_amd.profiler.profile_model --model gpt3-2.7b --bs 1,2,4,8
done

echo "== 3. hetero plan EXECUTION vs uniform (BASELINE config 5) =="
# planner's 13/21 split from results/mi355x_hetero_27b_cumask_gbs16.txt;
# CU-mask half the ranks to emulate the mixed cluster
HSA_CU_MASK=4:0-127,5:0-127,6:0-127,7:0-127 \
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
    -m metis_amd.cli.plan_runner --model gpt3-2.7b \
    --plans "4,1,2,8,64" --layer-partition 0,13,34 \
    --out gpurun_out/het_exec_planned.json
HSA_CU_MASK=4:0-127,5:0-127,6:0-127,7:0-127 \
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
    -m metis_amd.cli.plan_runner --model gpt3-2.7b \
    --plans "4,1,2,8,64" \
    --out gpurun_out/het_exec_uniform.json

echo "== 4. multi-GPU cost-model validation =="
torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
    -m metis_amd.cli.plan_runner --model gpt3-2.7b \
    --plans "8,1,1,16,128;4,2,1,16,128;4,1,2,16,128;2,2,2,16,128" \
    --out gpurun_out/measured_8gpu.json
python3 cost_homo_cluster.py --model_name gpt3-2.7b --num_layers 34 \
    --gbs 128 --hidden_size 2560 --sequence_length 2048 --vocab_size 51200 \
    --hostfile_path tests/data/mi355x_single_node/hostfile \
    --clusterfile_path tests/data/mi355x_single_node/clusterfile.json \
    --profile_data_path profiles/mi355x/gpt3-2.7b \
    --max_profiled_tp_degree 8 --max_profiled_batch_size 16 \
    --evaluation_data_path gpurun_out/measured_8gpu.json
