#!/bin/bash
# Single-model continual baseline entrypoint (win-1 / win-2 / all):
# same 22 positional arguments as the reference
# fedml_experiments/distributed/fedavg_cont_one/run_fedavg_distributed_pytorch.sh
# (:3-24) — RETRAIN_DATA in slot 19 instead of the drift-algorithm pair.

set -e

CLIENT_NUM=$1
WORKER_NUM=$2
SERVER_NUM=$3
GPU_NUM_PER_SERVER=$4
MODEL=$5
DISTRIBUTION=$6
ROUND=$7
EPOCH=$8
BATCH_SIZE=$9
LR=${10}
DATASET=${11}
DATA_DIR=${12}
SAMPLE_NUM=${13}
NOISE_PROB=${14}
CI=${15}
TRAIN_ITER=${16}
RESET_MODELS=${17}
DRIFT_TOGETHER=${18}
RETRAIN_DATA=${19}
TIME_STRETCH=${20}
DUMMY_ARG=${21}
CHANGE_POINTS=${22}

SCRIPT_DIR="$(cd "$(dirname "$0")" && pwd)"

NPROC=$(python3 - <<'EOF'
import torch
print(torch.cuda.device_count() if torch.cuda.is_available() else 1)
EOF
)
if [ "$GPU_NUM_PER_SERVER" -gt 0 ] && [ "$NPROC" -gt "$GPU_NUM_PER_SERVER" ]; then
    NPROC=$GPU_NUM_PER_SERVER
fi

python3 "$SCRIPT_DIR/prepare_data.py" \
  --dataset "$DATASET" --data_dir "$DATA_DIR" --sample_num "$SAMPLE_NUM" \
  --noise_prob "$NOISE_PROB" --partition_method "$DISTRIBUTION" \
  --client_num_in_total "$CLIENT_NUM" --client_num_per_round "$WORKER_NUM" \
  --batch_size "$BATCH_SIZE" --train_iteration "$TRAIN_ITER" \
  --drift_together "$DRIFT_TOGETHER" --time_stretch "$TIME_STRETCH" \
  --dummy_arg "$DUMMY_ARG" --change_points "${CHANGE_POINTS:-rand}"

python3 -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
  --nnodes=1 --nproc-per-node "$NPROC" \
  "$SCRIPT_DIR/main_timeline.py" \
  --model "$MODEL" --dataset "$DATASET" --data_dir "$DATA_DIR" \
  --noise_prob "$NOISE_PROB" \
  --client_num_in_total "$CLIENT_NUM" --client_num_per_round "$WORKER_NUM" \
  --comm_round "$ROUND" --epochs "$EPOCH" --batch_size "$BATCH_SIZE" \
  --lr "$LR" --ci "$CI" --total_train_iteration "$TRAIN_ITER" \
  --reset_models "$RESET_MODELS" --drift_together "$DRIFT_TOGETHER" \
  --report_client 1 --concept_drift_algo single \
  --retrain_data "$RETRAIN_DATA" --time_stretch "$TIME_STRETCH" \
  --dummy_arg "$DUMMY_ARG" --change_points "${CHANGE_POINTS:-rand}"
