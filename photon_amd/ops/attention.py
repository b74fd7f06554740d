"""Causal flash attention with fused ALiBi — CDNA4 HIP kernel + SDPA fallback.

Replaces the reference's flash-attn 2.6.3 CUDA kernels (install_env.sh:71,
mpt-125m.yaml:28). The HIP kernel is a tiled online-softmax attention on
MFMA (bf16 32x32x16) with K/V staged through LDS and the ALiBi bias applied
in-register; `impl="torch"` keeps the PyTorch SDPA path (the reference's
``attn_impl: torch``).
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from . import use_hip, hip_ext


def alibi_slopes(n_heads: int, alibi_bias_max: float = 8.0) -> torch.Tensor:
    """ALiBi slopes, llm-foundry/MPT convention.

    slope_h = 1 / 2**(h_ceil_pow2-normalized exponent); for n_heads not a
    power of two, MPT computes slopes for the next power of two and takes
    the odd-indexed entries first (llm-foundry attention.py gen_slopes).
    """
    next_pow2 = 2 ** math.ceil(math.log2(n_heads))
    m = torch.arange(1, next_pow2 + 1, dtype=torch.float32)
    m = m * (alibi_bias_max / next_pow2)
    slopes = 1.0 / torch.pow(2.0, m)
    if next_pow2 != n_heads:
        # interleave: odd-index slopes first, then even, truncate
        slopes = torch.cat([slopes[1::2], slopes[0::2]])[:n_heads]
    return slopes


def alibi_bias(
    slopes: torch.Tensor, seq_len: int, device=None, dtype=torch.float32
) -> torch.Tensor:
    """Dense [1, H, S, S] ALiBi bias b[h, i, j] = -slope_h * (i - j), causal
    part only (upper triangle is masked separately)."""
    pos = torch.arange(seq_len, device=device, dtype=dtype)
    rel = pos.view(1, 1, -1, 1) - pos.view(1, 1, 1, -1)  # i - j
    return (-slopes.to(device=device, dtype=dtype).view(1, -1, 1, 1)) * rel


class _FlashAttentionQKV(torch.autograd.Function):
    """Packed-qkv path: takes the fused Wqkv output [B, S, 3*H*dh] and
    returns [B, S, H*dh] — no transpose/contiguous copies on either side
    (profiles/r01: the chunk/cat copies were ~3% of step time)."""

    @staticmethod
    def forward(ctx, qkv, n_heads, slopes, causal):
        ext = hip_ext()
        o, lse = ext.attn_fwd_qkv(qkv, n_heads, slopes, causal)
        ctx.save_for_backward(qkv, slopes, o, lse)
        ctx.causal = causal
        ctx.n_heads = n_heads
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, slopes, o, lse = ctx.saved_tensors
        ext = hip_ext()
        dqkv = ext.attn_bwd_qkv(
            do.contiguous(), qkv, ctx.n_heads, slopes, o, lse, ctx.causal
        )
        return dqkv, None, None, None


def flash_attention_qkv(
    qkv: torch.Tensor, n_heads: int, slopes: torch.Tensor, causal: bool = True,
    impl: str = "flash",
) -> torch.Tensor:
    """Attention on the packed [B, S, 3*H*dh] Wqkv output; returns
    [B, S, H*dh]. Falls back to the reshape + SDPA path off-GPU."""
    B, S, three_hd = qkv.shape
    dh = three_hd // (3 * n_heads)
    if impl == "flash" and qkv.dtype == torch.bfloat16 and use_hip(qkv) and dh in (64, 128):
        return _FlashAttentionQKV.apply(qkv, n_heads, slopes, causal)
    q, k, v = qkv.view(B, S, 3, n_heads, dh).permute(2, 0, 3, 1, 4).unbind(0)
    out = flash_attention(q, k, v, slopes, causal=causal, impl=impl)
    return out.transpose(1, 2).reshape(B, S, n_heads * dh)


class _FlashAttentionHIP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, slopes, causal):
        ext = hip_ext()
        o, lse = ext.attn_fwd(q, k, v, slopes, causal)
        ctx.save_for_backward(q, k, v, slopes, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, slopes, o, lse = ctx.saved_tensors
        ext = hip_ext()
        dq, dk, dv = ext.attn_bwd(do.contiguous(), q, k, v, slopes, o, lse, ctx.causal)
        return dq, dk, dv, None, None


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    slopes: torch.Tensor,
    causal: bool = True,
    impl: str = "flash",
) -> torch.Tensor:
    """Attention over [B, H, S, dh] tensors with ALiBi bias.

    impl="flash": HIP kernel on GPU (falls back to SDPA on CPU);
    impl="torch": always SDPA with a dense bias (reference fallback path).
    """
    if impl == "flash" and use_hip(q):
        return _FlashAttentionHIP.apply(
            q.contiguous(), k.contiguous(), v.contiguous(), slopes, causal
        )
    return sdpa_attention(q, k, v, slopes, causal)


def sdpa_attention(q, k, v, slopes, causal=True):
    S = q.shape[-2]
    bias = alibi_bias(slopes, S, device=q.device, dtype=torch.float32)
    if causal:
        mask = torch.full((S, S), float("-inf"), device=q.device, dtype=torch.float32)
        mask = torch.triu(mask, diagonal=1)
        bias = bias + mask
    return F.scaled_dot_product_attention(q, k, v, attn_mask=bias.to(q.dtype))


def reference_attention_fp32(q, k, v, slopes, causal=True):
    """Plain fp32 PyTorch reference for kernel numerics tests."""
    q, k, v = q.float(), k.float(), v.float()
    S = q.shape[-2]
    scale = 1.0 / math.sqrt(q.shape[-1])
    scores = torch.matmul(q, k.transpose(-1, -2)) * scale
    scores = scores + alibi_bias(slopes, S, device=q.device)
    if causal:
        mask = torch.triu(
            torch.ones(S, S, dtype=torch.bool, device=q.device), diagonal=1
        )
        scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    return torch.matmul(p, v)
