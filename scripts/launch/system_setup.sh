#!/usr/bin/env bash
# System sanity — reference system_setup.sh (CUDA 12.4/cuDNN/pyenv).
# MI355X equivalent: verify the ROCm stack + environment the runtime needs.
set -euo pipefail
echo "== ROCm =="; ls /opt/rocm/.info/version 2>/dev/null && cat /opt/rocm/.info/version || hipcc --version | head -1
echo "== GPUs =="; rocm-smi --showid 2>/dev/null | head -20 || echo "no GPU visible (CPU container)"
echo "== PyTorch =="; python - << 'PY'
import torch
print("torch", torch.__version__, "| hip", torch.version.hip, "| cuda avail", torch.cuda.is_available())
PY
echo "== IPC mode =="; echo "HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-unset} (must be 0 for multi-process RCCL)"
