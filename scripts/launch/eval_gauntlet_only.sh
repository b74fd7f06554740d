#!/usr/bin/env bash
# ICL/gauntlet evaluation only — reference eval_gauntlet_only.sh.
# Evaluates a checkpoint on the configured ICL tasks without training.
set -euo pipefail
export PHOTON_SAVE_PATH=${PHOTON_SAVE_PATH:-runs/eval}
mkdir -p "$PHOTON_SAVE_PATH"
python -m photon_amd.hydra_resolver llm_config=mpt-125m \
    centralized.eval_only=true "$@"
python -m photon_amd.centralised_train
