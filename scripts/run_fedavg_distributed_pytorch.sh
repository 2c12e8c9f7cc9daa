#!/bin/bash
# Reference-compatible entrypoint: same 24 positional arguments as
# /fedml_experiments/distributed/fedavg_cont_ens/run_fedavg_distributed_pytorch.sh
# in microsoft/FedDrift (see its :3-26), driving the MI355X engine instead:
# one process per GPU via torchrun (RCCL over xGMI), whole timeline in one
# launch, per-iteration checkpoints preserved.
#
# Usage (canonical FedDrift SEA-4 run, reference README.md:45-48):
#   ./run_fedavg_distributed_pytorch.sh 10 10 1 4 fnn homo 200 5 500 0.01 sea \
#       ./data 100 0 0 10 4 0 0 softcluster H_A_C_1_10_0 1 0 A

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
CONCEPT_NUM=${17}
RESET_MODELS=${18}
DRIFT_TOGETHER=${19}
CL_ALGO=${20}
CL_ALGO_ARG=${21}
TIME_STRETCH=${22}
DUMMY_ARG=${23}
CHANGE_POINTS=${24}

SCRIPT_DIR="$(cd "$(dirname "$0")" && pwd)"

# number of engine processes = number of GPUs on this node (1 on CPU)
NPROC=$(python3 - <<'EOF'
import torch
print(torch.cuda.device_count() if torch.cuda.is_available() else 1)
EOF
)
if [ "$GPU_NUM_PER_SERVER" -gt 0 ] && [ "$NPROC" -gt "$GPU_NUM_PER_SERVER" ]; then
    NPROC=$GPU_NUM_PER_SERVER
fi

python3 "$SCRIPT_DIR/prepare_data.py" \
  --dataset "$DATASET" \
  --data_dir "$DATA_DIR" \
  --sample_num "$SAMPLE_NUM" \
  --noise_prob "$NOISE_PROB" \
  --partition_method "$DISTRIBUTION" \
  --client_num_in_total "$CLIENT_NUM" \
  --client_num_per_round "$WORKER_NUM" \
  --batch_size "$BATCH_SIZE" \
  --train_iteration "$TRAIN_ITER" \
  --drift_together "$DRIFT_TOGETHER" \
  --time_stretch "$TIME_STRETCH" \
  --dummy_arg "$DUMMY_ARG" \
  --change_points "${CHANGE_POINTS:-rand}"

python3 -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
  --nnodes=1 --nproc-per-node "$NPROC" \
  "$SCRIPT_DIR/main_timeline.py" \
  --model "$MODEL" \
  --dataset "$DATASET" \
  --data_dir "$DATA_DIR" \
  --noise_prob "$NOISE_PROB" \
  --client_num_in_total "$CLIENT_NUM" \
  --client_num_per_round "$WORKER_NUM" \
  --comm_round "$ROUND" \
  --epochs "$EPOCH" \
  --batch_size "$BATCH_SIZE" \
  --lr "$LR" \
  --ci "$CI" \
  --total_train_iteration "$TRAIN_ITER" \
  --concept_num "$CONCEPT_NUM" \
  --reset_models "$RESET_MODELS" \
  --drift_together "$DRIFT_TOGETHER" \
  --report_client 1 \
  --retrain_data win-1 \
  --concept_drift_algo "$CL_ALGO" \
  --concept_drift_algo_arg "$CL_ALGO_ARG" \
  --time_stretch "$TIME_STRETCH" \
  --dummy_arg "$DUMMY_ARG" \
  --change_points "${CHANGE_POINTS:-rand}"
