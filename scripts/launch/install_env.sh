#!/usr/bin/env bash
# Environment build — reference install_env.sh (poetry + flash-attn source
# build). MI355X-native equivalent: compile the in-tree HIP extension for
# gfx950 (the only build step; PyTorch-ROCm and RCCL ship with the image).
set -euo pipefail
cd "$(dirname "$0")/../.."
export PYTORCH_ROCM_ARCH=${PYTORCH_ROCM_ARCH:-gfx950}
python setup.py build_ext --inplace
python -c "import photon_amd.ops as o; o._try_load(); print('HIP extension:', 'ok' if o.hip_ext() else 'MISSING')"
