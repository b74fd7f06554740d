#!/usr/bin/env bash
# Federated MPT-125M example — 8 clients x 1 MI355X on one node.
# The MI355X-native replacement for the reference's
# ray-head + flower-superlink + server-app + client-app process stack
# (fed_125m_example.sh / photon_llm_125M.sh): ONE torchrun over RCCL.
set -euo pipefail
N_GPUS=${N_GPUS:-8}
export PHOTON_SAVE_PATH=${PHOTON_SAVE_PATH:-runs/fed_125m}
mkdir -p "$PHOTON_SAVE_PATH"
export HSA_ENABLE_IPC_MODE_LEGACY=0

# resolve + dump the single source-of-truth config (hydra_resolver contract)
python -m photon_amd.hydra_resolver \
    llm_config=mpt-125m \
    fl.n_total_clients=8 fl.n_clients_per_round=8 fl.n_rounds=10 \
    llm_config.local_steps=500ba \
    photon.checkpoint=true \
    "$@"

torchrun --nnodes=1 --nproc-per-node "$N_GPUS" --master-addr 127.0.0.1 \
    -m photon_amd.fed_train
