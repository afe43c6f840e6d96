#!/bin/bash
# KEY=VALUE wrapper for the homogeneous-cluster planner
# (reference-compatible: scripts/cost_homo_cluster.sh). Same presets and
# variables as cost_het_cluster.sh.
set -e
cd "$(dirname "$0")/.."

for kv in "$@"; do export "$kv"; done

MODEL_NAME=${MODEL_NAME:-GPT}
MODEL_SIZE=${MODEL_SIZE:-1.5B}
NUM_LAYERS=${NUM_LAYERS:-10}
GBS=${GBS:-128}
HOME_DIR=${HOME_DIR:-$PWD}

case "$MODEL_NAME $MODEL_SIZE" in
  "GPT 1.5B") HIDDEN=4096; SEQ=1024; VOCAB=51200; HEADS=32 ;;
  "GPT 2.7B") HIDDEN=2560; SEQ=2048; VOCAB=51200; HEADS=32 ;;
  "GPT 6.7B") HIDDEN=4096; SEQ=2048; VOCAB=51200; HEADS=32 ;;
  *) echo "unknown preset: $MODEL_NAME $MODEL_SIZE"; exit 1 ;;
esac
HIDDEN=${HIDDEN_SIZE:-$HIDDEN}; SEQ=${SEQUENCE_LENGTH:-$SEQ}
VOCAB=${VOCAB_SIZE:-$VOCAB}

HOSTFILE=${HOSTFILE:-$HOME_DIR/hostfile}
CLUSTERFILE=${CLUSTERFILE:-$HOME_DIR/clusterfile.json}
PROFILE_DIR=${PROFILE_DIR:-$HOME_DIR/profile}
LOG_DIR=${LOG_DIR:-$HOME_DIR/logs}
mkdir -p "$LOG_DIR"
LOG="$LOG_DIR/${MODEL_NAME}_${MODEL_SIZE}_homo_$(date +%Y%m%d_%H%M%S).log"

python3 cost_homo_cluster.py \
  --model_name="$MODEL_NAME" --model_size="$MODEL_SIZE" \
  --num_layers="$NUM_LAYERS" --gbs="$GBS" \
  --hidden_size="$HIDDEN" --sequence_length="$SEQ" \
  --vocab_size="$VOCAB" --attention_head_size="$((HIDDEN / HEADS))" \
  --hostfile_path="$HOSTFILE" --clusterfile_path="$CLUSTERFILE" \
  --profile_data_path="$PROFILE_DIR" \
  --max_profiled_tp_degree="${MAX_PROFILED_TP_DEGREE:-8}" \
  --max_profiled_batch_size="${MAX_PROFILED_BATCH_SIZE:-16}" \
  ${EVALUATION_DATA_PATH:+--evaluation_data_path="$EVALUATION_DATA_PATH"} \
  | tee "$LOG"
echo "log: $LOG"
