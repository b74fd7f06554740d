#!/usr/bin/env bash
# Centralized (non-federated) training — reference centralised_training.sh
# (composer launcher -> torchrun; DDP grad sync is a bucketed RCCL
# all-reduce inside the trainer).
set -euo pipefail
N_GPUS=${N_GPUS:-1}
export PHOTON_SAVE_PATH=${PHOTON_SAVE_PATH:-runs/centralised}
mkdir -p "$PHOTON_SAVE_PATH"
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m photon_amd.hydra_resolver llm_config=mpt-125m "$@"
if [ "$N_GPUS" -gt 1 ]; then
  torchrun --nnodes=1 --nproc-per-node "$N_GPUS" --master-addr 127.0.0.1 \
      -m photon_amd.centralised_train
else
  python -m photon_amd.centralised_train
fi
