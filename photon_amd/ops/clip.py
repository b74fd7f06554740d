"""Global-norm gradient clipping — multi-tensor HIP L2-norm + scale kernels.

Replaces the reference's composer ``algorithms.gradient_clipping`` (norm 1.0,
mpt-125m.yaml:65-68). One kernel computes per-tensor partial sums of squares
(fp32 accumulation) into a single reduction buffer; a second scales every
grad by clip_norm/total_norm when needed.
"""

from __future__ import annotations

import torch

from . import hip_ext, use_hip


def clip_grad_norm_(params, max_norm: float) -> torch.Tensor:
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return torch.zeros(())
    if use_hip(grads[0]):
        ext = hip_ext()
        total = ext.multi_tensor_l2norm(grads)  # scalar fp32 tensor
        ext.multi_tensor_scale_clip(grads, total, float(max_norm))
        return total
    total = torch.norm(
        torch.stack([g.detach().float().norm(2) for g in grads]), 2
    )
    scale = max_norm / (total + 1e-6)
    if scale < 1.0:
        for g in grads:
            g.mul_(scale.to(g.dtype))
    return total
