#!/usr/bin/env bash
# Multi-node client launch — reference photon_llm_125M_client_only.sh.
# The MI355X-native runtime has no separate superlink/server-app processes:
# every rank runs the symmetric RCCL round loop, so a "client-only" node is
# just additional torchrun ranks pointed at the rendezvous master.
#
#   MASTER_ADDR=10.0.0.1 NODE_RANK=1 NNODES=2 ./fed_125m_client_only.sh
set -euo pipefail
cd "$(dirname "$0")/../.."
export PHOTON_SAVE_PATH=${PHOTON_SAVE_PATH:-runs/fed125m}
mkdir -p "$PHOTON_SAVE_PATH"
NPROC=${NPROC_PER_NODE:-$(python -c 'import torch; print(max(torch.cuda.device_count(),1))')}
python -m torch.distributed.run \
    --nnodes "${NNODES:-2}" --node-rank "${NODE_RANK:?set NODE_RANK}" \
    --nproc-per-node "$NPROC" \
    --master-addr "${MASTER_ADDR:?set MASTER_ADDR}" \
    --master-port "${MASTER_PORT:-29500}" \
    -m photon_amd.fed_train llm_config=mpt-125m "$@"
