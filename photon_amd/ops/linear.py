"""Fused linear / MLP through hipBLASLt epilogues (MI355X).

Replaces the reference's cuBLAS GEMM + separate torch bias/GELU kernels
(SURVEY.md §2.3 "cuBLAS GEMM" row) with single-kernel epilogue GEMMs:

* forward:  BIAS epilogue (bias in the GEMM)
* backward: BGRADB — the bias gradient rides the dW GEMM, removing every
  `at::native::reduce_kernel` bias-grad column reduce from the step
  (VERDICT r01 "GEMM-side fusion"). The DGELU/AUX epilogues are probed
  unsupported on gfx950's hipBLASLt (scripts/lt_epi_probe.hip), so GELU
  stays a torch op between the two fused MLP GEMMs.

GELU flavor: hipBLASLt implements the tanh approximation, so the eager
fallback uses ``nn.GELU(approximate="tanh")`` to keep CPU/GPU paths
bit-comparable (difference vs exact GELU is below bf16 resolution).

Autocast: ``custom_fwd(cast_inputs=bf16)`` casts activations AND (master)
fp32 weights on entry, so weight gradients flow back through autograd's
cast nodes to fp32 automatically.
"""

from __future__ import annotations

import os

import torch

from . import hip_ext

# PHOTON_LT_MODE:
#   "cpp" (default) — C++ autograd Function (zero Python overhead): torch
#       tuned GEMMs + the HIP bias-grad column reduce in backward
#       (ops/hip/fused_linear_fn.cpp + bias_grad.hip);
#   "dbk"           — same structure as a Python autograd.Function
#       (measured: ~30 ms/step of Python Function overhead at 125M —
#       kept only as the measurement record);
#   "all"/"dw"      — hipblaslt epilogue experiments (measured REJECTED:
#       BGRADB disables split-K on the reduction-heavy dW GEMMs, MPT-1B
#       step 1704 -> 3515 ms; kept for re-evaluation on newer hipblaslt);
#   "off"           — plain torch linears.
import functools


@functools.lru_cache(maxsize=1)
def _lt_mode() -> str:
    # cached: read once per process (384 linear calls per 125M step)
    return os.environ.get("PHOTON_LT_MODE", "cpp")

_CUSTOM_FWD = torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
_CUSTOM_BWD = torch.amp.custom_bwd(device_type="cuda")


def lt_available(x: torch.Tensor) -> bool:
    ext = hip_ext()
    if ext is None or not x.is_cuda:
        return False
    mode = _lt_mode()
    if mode == "off":
        return False
    need = "fused_linear" if mode == "cpp" else "lt_linear_fwd"
    return hasattr(ext, need)


class _LtLinearFn(torch.autograd.Function):
    @staticmethod
    @_CUSTOM_FWD
    def forward(ctx, x, w, b):
        if _lt_mode() == "all":
            y, _ = hip_ext().lt_linear_fwd(x, w, b, False)
        else:  # torch (TunableOp-tuned) forward GEMM
            y = torch.nn.functional.linear(x, w, b)
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    @_CUSTOM_BWD
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        ext = hip_ext()
        mode = _lt_mode()
        if mode == "all":
            dx = ext.lt_linear_bwd_dx(dy, w, None, False)[0]
        else:
            dx = dy @ w
        if mode in ("all", "dw"):
            dw, db = ext.lt_linear_bwd_dw(x, dy, ctx.has_bias)
        else:  # "dbk": tuned torch dW GEMM + HIP bias-grad reduce
            dw = dy.t() @ x
            db = ext.bias_grad(dy) if ctx.has_bias else None
        return dx, dw, (db if ctx.has_bias else None)


def lt_linear(x: torch.Tensor, weight: torch.Tensor, bias=None) -> torch.Tensor:
    """y = x @ W^T + b with the HIP bias-grad backward."""
    if _lt_mode() == "cpp":
        return hip_ext().fused_linear(x, weight, bias)
    shp = x.shape
    y = _LtLinearFn.apply(x.reshape(-1, shp[-1]), weight, bias)
    return y.reshape(*shp[:-1], weight.shape[0])


def lt_mlp(x, w_up, b_up, w_down, b_down, act=None) -> torch.Tensor:
    """up -> GELU(tanh) -> down. The GEMMs carry BIAS epilogues forward and
    BGRADB backward; the GELU itself stays a torch op because hipBLASLt on
    gfx950 ships no kernels for the AUX/DGELU epilogues (probed:
    scripts/lt_epi_probe.hip — GELU_AUX*/DGELU*/BGRADA all return 0 algos
    in the forward layout; BGRADB works in the dW layout)."""
    h = lt_linear(x, w_up, b_up)
    h = torch.nn.functional.gelu(h, approximate="tanh")
    return lt_linear(h, w_down, b_down)
