"""photon_amd — an MI355X-native federated LLM pre-training engine.

A from-scratch rebuild of the capabilities of relogu/photon (MLSys'25) designed
for AMD Instinct MI355X (gfx950, CDNA4) nodes:

* each federated client owns one MI355X GPU (one process per GPU),
* round aggregation is an RCCL weighted all-reduce over xGMI instead of the
  reference's gRPC + NumPy streaming aggregation (photon/strategy/aggregation.py:19-87),
* parameter broadcast is an RCCL broadcast of HBM-resident tensors instead of
  the reference's SHM/Ray/S3 comm stacks (photon/server/s3_utils.py:730-1115),
* the MPT decoder's hot ops (flash attention with fused ALiBi, LayerNorm,
  cross-entropy over the 50368 vocab, fused AdamW/ADOPT, grad-norm clipping)
  are hand-written CDNA4 HIP kernels on MFMA with LDS tiling,
* PyTorch-ROCm is the only framework layer: no Triton, no CUDA shims,
  no flash_attn wheel.

The Hydra config surface (photon/conf/*) and checkpoint formats
(ep{e}-ba{b}-rank{r}.pt client checkpoints; server round .npz + state.bin)
are kept compatible so reference runs can be restored.
"""

# -- GEMM autotuning ---------------------------------------------------------
# hipBLASLt/rocBLAS solution selection via PyTorch TunableOp, READ-ONLY from
# the pre-tuned MI355X table (photon_amd/tuned/tunableop_mi355x.csv, +3-5%
# end-to-end measured). Opt out with PHOTON_NO_TUNABLEOP=1; re-tune with
# PYTORCH_TUNABLEOP_TUNING=1.
import os as _os

if _os.environ.get("PHOTON_NO_TUNABLEOP", "0") != "1":
    _tuned = _os.path.join(_os.path.dirname(__file__), "tuned",
                           "tunableop_mi355x.csv")
    if _os.path.exists(_tuned):
        _os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
        _os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
        _os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _tuned)


__version__ = "0.1.0"
