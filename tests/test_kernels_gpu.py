"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference.

All tests are @pytest.mark.gpu (run on a real MI355X via gpurun / the
driver's round-end pass). Tolerances account for bf16 inputs with fp32
accumulation.
"""

import math

import pytest
import torch

gpu = pytest.mark.gpu

pytestmark = gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


@pytest.fixture(scope="module")
def ext():
    from photon_amd.ops import hip_ext

    e = hip_ext()
    assert e is not None, "HIP extension must be built on GPU boxes"
    return e


# ---------------------------------------------------------------------------
# LayerNorm
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("D", [768, 2048])
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_layernorm_fwd(dev, ext, D, dtype):
    torch.manual_seed(0)
    x = torch.randn(64, D, device=dev, dtype=dtype)
    w = torch.randn(D, device=dev) * 0.5 + 1.0
    b = torch.randn(D, device=dev) * 0.1
    y, mean, rstd = ext.layernorm_fwd(x, w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(x.float(), (D,), w, b, 1e-5)
    # bf16 output-store rounding is ~2^-8 relative; compare relative to |ref|
    tol = 1e-2 if dtype == torch.bfloat16 else 1e-5
    denom = float(ref.abs().max().clamp_min(1.0))
    assert (y.float() - ref).abs().max() / denom < tol
    assert (mean - x.float().mean(-1)).abs().max() < 1e-4


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_layernorm_bwd(dev, ext, dtype):
    torch.manual_seed(1)
    D = 768
    x = torch.randn(128, D, device=dev, dtype=dtype)
    w = (torch.randn(D, device=dev) * 0.5 + 1.0).requires_grad_(True)
    b = (torch.randn(D, device=dev) * 0.1).requires_grad_(True)
    dy = torch.randn_like(x)

    y, mean, rstd = ext.layernorm_fwd(x, w.detach(), b.detach(), 1e-5)
    dx, dw, db = ext.layernorm_bwd(dy, x, w.detach(), mean, rstd)

    xr = x.float().detach().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xr, (D,), w, b, 1e-5)
    ref.backward(dy.float())
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert (dx.float() - xr.grad).abs().max() < tol
    assert (dw - w.grad).abs().max() / w.grad.abs().max() < 2e-2
    assert (db - b.grad).abs().max() / b.grad.abs().max() < 2e-2


def test_layernorm_bwd_deterministic(dev, ext):
    torch.manual_seed(2)
    x = torch.randn(256, 768, device=dev, dtype=torch.bfloat16)
    w = torch.ones(768, device=dev)
    dy = torch.randn_like(x)
    y, mean, rstd = ext.layernorm_fwd(x, w, None, 1e-5)
    out1 = ext.layernorm_bwd(dy, x, w, mean, rstd)
    out2 = ext.layernorm_bwd(dy, x, w, mean, rstd)
    for a, b in zip(out1, out2):
        assert torch.equal(a, b)


# ---------------------------------------------------------------------------
# Cross-entropy
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("V", [50368, 1000])
@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_ce_fwd_bwd(dev, ext, V, dtype):
    torch.manual_seed(3)
    N = 64
    logits = (torch.randn(N, V, device=dev) * 3).to(dtype)
    targets = torch.randint(0, V, (N,), device=dev)
    ref_logits = logits.float().clone()
    losses = ext.ce_fwd_bwd_inplace(logits, targets)

    ref_losses = torch.nn.functional.cross_entropy(
        ref_logits, targets, reduction="none"
    )
    assert (losses - ref_losses).abs().max() < 2e-2

    dl_ref = torch.softmax(ref_logits, -1)
    dl_ref[torch.arange(N, device=dev), targets] -= 1.0
    tol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    assert (logits.float() - dl_ref).abs().max() < tol


def test_fused_linear_ce_end_to_end(dev, ext):
    from photon_amd.ops.cross_entropy import (
        fused_cross_entropy,
        reference_cross_entropy_fp32,
    )

    torch.manual_seed(4)
    N, D, V = 512, 256, 50368
    h = (torch.randn(N, D, device=dev) * 0.1).to(torch.bfloat16).requires_grad_(True)
    w = (torch.randn(V, D, device=dev) * 0.1).to(torch.bfloat16).requires_grad_(True)
    t = torch.randint(0, V, (N,), device=dev)
    loss = fused_cross_entropy(h, w, t, impl="fused")
    ref = reference_cross_entropy_fp32(h.detach(), w.detach(), t)
    assert abs(float(loss) - float(ref)) < 5e-2
    loss.backward()
    h2 = h.detach().float().requires_grad_(True)
    w2 = w.detach().float().requires_grad_(True)
    ref2 = reference_cross_entropy_fp32(h2, w2, t)
    ref2.backward()
    assert (h.grad.float() - h2.grad).abs().max() < 5e-2
    assert (w.grad.float() - w2.grad).abs().max() < 5e-2


# ---------------------------------------------------------------------------
# Optimizers
# ---------------------------------------------------------------------------
def test_adamw_kernel_vs_torch(dev, ext):
    torch.manual_seed(5)
    shapes = [(768,), (768, 768), (50368, 768), (64,)]
    ps = [torch.randn(s, device=dev) for s in shapes]
    gs = [torch.randn(s, device=dev) for s in shapes]
    ms = [torch.zeros(s, device=dev) for s in shapes]
    vs = [torch.zeros(s, device=dev) for s in shapes]
    ps_ref = [p.clone() for p in ps]

    lr, b1, b2, eps = 1e-3, 0.9, 0.95, 1e-8
    for step in range(1, 4):
        bc1, bc2 = 1 - b1**step, 1 - b2**step
        ext.adamw_step(ps, gs, ms, vs, lr, b1, b2, eps, 0.0, bc1, bc2, [])
    # torch reference
    ms_r = [torch.zeros_like(p) for p in ps_ref]
    vs_r = [torch.zeros_like(p) for p in ps_ref]
    for step in range(1, 4):
        bc1, bc2 = 1 - b1**step, 1 - b2**step
        for p, g, m, v in zip(ps_ref, gs, ms_r, vs_r):
            m.mul_(b1).add_(g, alpha=1 - b1)
            v.mul_(b2).addcmul_(g, g, value=1 - b2)
            p.addcdiv_(m / bc1, (v / bc2).sqrt() + eps, value=-lr)
    for a, b in zip(ps, ps_ref):
        assert (a - b).abs().max() < 1e-5


def test_adopt_kernel_vs_cpu_impl(dev, ext):
    from photon_amd.ops.optim import ADOPT

    torch.manual_seed(6)
    shape = (1024,)
    p_gpu = torch.nn.Parameter(torch.randn(shape, device=dev))
    p_cpu = torch.nn.Parameter(p_gpu.detach().cpu().clone())
    o_gpu = ADOPT([p_gpu], lr=0.01)
    o_cpu = ADOPT([p_cpu], lr=0.01)
    for _ in range(4):
        g = torch.randn(shape)
        p_gpu.grad = g.to(dev)
        p_cpu.grad = g.clone()
        o_gpu.step()
        o_cpu.step()
    assert (p_gpu.detach().cpu() - p_cpu.detach()).abs().max() < 1e-5


def test_clip_kernel_vs_torch(dev, ext):
    from photon_amd.ops.clip import clip_grad_norm_

    torch.manual_seed(7)
    ps = [torch.nn.Parameter(torch.randn(128, 128, device=dev)) for _ in range(3)]
    for p in ps:
        p.grad = torch.randn_like(p) * 3
    ref_grads = [p.grad.clone() for p in ps]
    total = clip_grad_norm_(ps, 1.0)
    ref_total = torch.norm(torch.stack([g.norm(2) for g in ref_grads]), 2)
    assert abs(float(total) - float(ref_total)) < 1e-3
    for p, g in zip(ps, ref_grads):
        assert (p.grad - g / (ref_total + 1e-6)).abs().max() < 1e-5


# ---------------------------------------------------------------------------
# Flash attention
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("S", [128, 256, 300, 2048])
def test_attn_fwd_vs_fp32_ref(dev, ext, D, S):
    from photon_amd.ops.attention import alibi_slopes, reference_attention_fp32

    torch.manual_seed(8)
    B, H = 2, 4
    q = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    slopes = alibi_slopes(H).to(dev)
    o, lse = ext.attn_fwd(q, k, v, slopes, True)
    ref = reference_attention_fp32(q, k, v, slopes, causal=True)
    err = (o.float() - ref).abs().max()
    assert err < 3e-2, f"attn fwd max err {err}"
    # LSE sanity: exp(lse) positive & finite
    assert torch.isfinite(lse).all()


def test_attn_fwd_noncausal(dev, ext):
    from photon_amd.ops.attention import alibi_slopes, reference_attention_fp32

    torch.manual_seed(9)
    B, H, S, D = 1, 2, 192, 64
    q = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
    slopes = alibi_slopes(H).to(dev)
    o, _ = ext.attn_fwd(q, k, v, slopes, False)
    ref = reference_attention_fp32(q, k, v, slopes, causal=False)
    assert (o.float() - ref).abs().max() < 3e-2


@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("S", [128, 300, 512])
def test_attn_bwd_vs_fp32_ref(dev, ext, D, S):
    from photon_amd.ops.attention import alibi_slopes, reference_attention_fp32

    torch.manual_seed(10)
    B, H = 2, 2
    q = (torch.randn(B, H, S, D, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, H, S, D, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, H, S, D, device=dev) * 0.5).to(torch.bfloat16)
    slopes = alibi_slopes(H).to(dev)
    do = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)

    o, lse = ext.attn_fwd(q, k, v, slopes, True)
    dq, dk, dv = ext.attn_bwd(do, q, k, v, slopes, o, lse, True)

    qr = q.float().detach().requires_grad_(True)
    kr = k.float().detach().requires_grad_(True)
    vr = v.float().detach().requires_grad_(True)
    ref = reference_attention_fp32(qr, kr, vr, slopes, causal=True)
    ref.backward(do.float())
    for got, want, name in ((dq, qr.grad, "dq"), (dk, kr.grad, "dk"), (dv, vr.grad, "dv")):
        err = (got.float() - want).abs().max()
        scale = want.abs().max().clamp_min(1.0)
        assert err / scale < 5e-2, f"{name} rel err {err/scale} (abs {err})"


def test_attn_autograd_path(dev, ext):
    """flash_attention() Function end-to-end under autocast."""
    from photon_amd.ops.attention import alibi_slopes, flash_attention, sdpa_attention

    torch.manual_seed(11)
    B, H, S, D = 2, 4, 256, 64
    q = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    k = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    v = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    slopes = alibi_slopes(H).to(dev)
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        out = flash_attention(q.to(torch.bfloat16), k.to(torch.bfloat16),
                              v.to(torch.bfloat16), slopes, impl="flash")
    out.float().sum().backward()
    assert q.grad is not None


# ---------------------------------------------------------------------------
# End-to-end model on GPU
# ---------------------------------------------------------------------------
def test_model_train_step_flash_vs_torch(dev, ext):
    """One training step with HIP kernels ~ the torch-op path."""
    from photon_amd.models.mpt import MPTCausalLM, MPTConfig

    torch.manual_seed(12)
    ids = torch.randint(0, 50368, (2, 256), device=dev)

    losses = {}
    for impl, loss_impl in (("flash", "fused"), ("torch", "torch")):
        torch.manual_seed(12)
        cfg = MPTConfig(d_model=256, n_heads=4, n_layers=2, max_seq_len=256,
                        vocab_size=50368, attn_impl=impl, loss_impl=loss_impl)
        m = MPTCausalLM(cfg).to(dev)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            out = m(ids, labels=ids)
        out["loss"].backward()
        losses[impl] = float(out["loss"])
        grad_norm = torch.stack(
            [p.grad.norm() for p in m.parameters() if p.grad is not None]
        ).norm()
        assert torch.isfinite(grad_norm)
    assert abs(losses["flash"] - losses["torch"]) < 0.05, losses


@pytest.mark.gpu
def test_attn_qkv_packed_matches_separate():
    """Packed [B,S,3HD] path must match the separate-tensor path (fwd+bwd)."""
    from photon_amd.ops.attention import alibi_slopes, flash_attention, flash_attention_qkv

    torch.manual_seed(11)
    B, H, S, D = 2, 4, 256, 64
    qkv = torch.randn(B, S, 3 * H * D, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    slopes = alibi_slopes(H).to("cuda")
    out_p = flash_attention_qkv(qkv, H, slopes, causal=True)
    g = torch.randn_like(out_p)
    out_p.backward(g)
    dqkv_p = qkv.grad.clone()

    qkv2 = qkv.detach().clone().requires_grad_(True)
    q, k, v = qkv2.view(B, S, 3, H, D).permute(2, 0, 3, 1, 4).unbind(0)
    out_s = flash_attention(q.contiguous(), k.contiguous(), v.contiguous(),
                            slopes, causal=True)
    out_s = out_s.transpose(1, 2).reshape(B, S, H * D)
    out_s.backward(g)
    assert torch.allclose(out_p.float(), out_s.float(), atol=1e-3), (
        (out_p.float() - out_s.float()).abs().max()
    )
    assert torch.allclose(dqkv_p.float(), qkv2.grad.float(), atol=1e-3), (
        (dqkv_p.float() - qkv2.grad.float()).abs().max()
    )


@pytest.mark.gpu
def test_ce_kernel_odd_vocab():
    """CE kernel remainder path: vocab not a multiple of 8."""
    from photon_amd.ops import hip_ext

    torch.manual_seed(3)
    x = torch.randn(64, 1003, device="cuda", dtype=torch.bfloat16)
    t = torch.randint(0, 1003, (64,), device="cuda")
    ref = torch.nn.functional.cross_entropy(
        x.float(), t, reduction="none"
    )
    losses = hip_ext().ce_fwd_bwd_inplace(x.clone(), t)
    assert torch.allclose(losses, ref, atol=2e-2), (losses - ref).abs().max()


@pytest.mark.gpu
def test_layernorm_d2560():
    """LN kernel at MPT-3B's d_model (2560 = 10 chunks/wave)."""
    from photon_amd.ops.layernorm import FusedLayerNorm

    torch.manual_seed(4)
    ln = FusedLayerNorm(2560).to("cuda")
    x = torch.randn(128, 2560, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ln(x)
    ref = torch.nn.functional.layer_norm(
        x.float(), (2560,), ln.weight.float(), ln.bias.float(), ln.eps
    )
    assert (y.float() - ref).abs().max() < 2e-2
    y.sum().backward()
    assert x.grad is not None and ln.weight.grad is not None


@pytest.mark.gpu
def test_adamw_kernel_master_weights():
    """Fused AdamW with bf16 params + fp32 masters matches the fp32 path."""
    from photon_amd.ops import hip_ext

    torch.manual_seed(5)
    dev = "cuda"
    ext = hip_ext()
    p32 = torch.randn(1000, device=dev)
    g = torch.randn(1000, device=dev)
    m = torch.zeros(1000, device=dev)
    v = torch.zeros(1000, device=dev)
    pb = p32.to(torch.bfloat16).clone()
    master = pb.float().clone()
    m2 = torch.zeros(1000, device=dev)
    v2 = torch.zeros(1000, device=dev)
    gb = g.to(torch.bfloat16)
    lr, b1, b2, eps = 1e-2, 0.9, 0.95, 1e-8
    # reference: fp32 path fed the SAME bf16-rounded inputs
    pref = master.clone()
    gref = gb.float()
    ext.adamw_step([pref], [gref], [m, v][:1], [v], lr, b1, b2, eps, 0.0,
                   1 - b1, 1 - b2, [])
    ext.adamw_step([pb], [gb], [m2], [v2], lr, b1, b2, eps, 0.0,
                   1 - b1, 1 - b2, [master])
    assert torch.allclose(master, pref, atol=1e-6), (master - pref).abs().max()
    assert torch.equal(pb, master.to(torch.bfloat16))


# ---------------------------------------------------------------------------
# hipBLASLt fused linear / MLP epilogues
# ---------------------------------------------------------------------------

def test_lt_linear_vs_torch(dev, ext):
    torch.manual_seed(21)
    M, K, N = 512, 256, 384
    x = (torch.randn(M, K, device=dev) * 0.5).to(torch.bfloat16).requires_grad_()
    w = (torch.randn(N, K, device=dev) * 0.05).to(torch.bfloat16).requires_grad_()
    b = torch.randn(N, device=dev).to(torch.bfloat16).requires_grad_()
    from photon_amd.ops.linear import lt_linear

    y = lt_linear(x, w, b)
    dy = torch.randn_like(y) * 0.1
    y.backward(dy)

    xr = x.detach().float().requires_grad_()
    wr = w.detach().float().requires_grad_()
    br = b.detach().float().requires_grad_()
    yr = torch.nn.functional.linear(xr, wr, br)
    yr.backward(dy.float())

    assert (y.float() - yr).abs().max() < 0.1, "fwd mismatch"
    assert (x.grad.float() - xr.grad).abs().max() < 0.1
    assert (w.grad.float() - wr.grad).abs().max() / wr.grad.abs().max() < 0.05
    assert (b.grad.float() - br.grad).abs().max() / br.grad.abs().max() < 0.05


def test_lt_mlp_vs_torch(dev, ext):
    torch.manual_seed(22)
    M, K, H = 512, 256, 1024
    x = (torch.randn(M, K, device=dev) * 0.5).to(torch.bfloat16).requires_grad_()
    wu = (torch.randn(H, K, device=dev) * 0.05).to(torch.bfloat16).requires_grad_()
    bu = torch.randn(H, device=dev).mul(0.1).to(torch.bfloat16).requires_grad_()
    wd = (torch.randn(K, H, device=dev) * 0.05).to(torch.bfloat16).requires_grad_()
    bd = torch.randn(K, device=dev).mul(0.1).to(torch.bfloat16).requires_grad_()
    from photon_amd.ops.linear import lt_mlp

    y = lt_mlp(x, wu, bu, wd, bd)
    dy = torch.randn_like(y) * 0.1
    y.backward(dy)

    xr = x.detach().float().requires_grad_()
    wur = wu.detach().float().requires_grad_()
    bur = bu.detach().float().requires_grad_()
    wdr = wd.detach().float().requires_grad_()
    bdr = bd.detach().float().requires_grad_()
    h = torch.nn.functional.gelu(
        torch.nn.functional.linear(xr, wur, bur), approximate="tanh"
    )
    yr = torch.nn.functional.linear(h, wdr, bdr)
    yr.backward(dy.float())

    assert (y.float() - yr).abs().max() < 0.15, "mlp fwd mismatch"
    assert (x.grad.float() - xr.grad).abs().max() < 0.1
    for got, ref in [(wu.grad, wur.grad), (wd.grad, wdr.grad),
                     (bu.grad, bur.grad), (bd.grad, bdr.grad)]:
        rel = (got.float() - ref).abs().max() / ref.abs().max().clamp_min(1e-6)
        assert rel < 0.06, f"grad rel err {rel}"


def test_model_forward_uses_lt_path(dev, ext):
    """End-to-end: a tiny MPT block fwd+bwd on GPU runs through the fused
    path (lt_available) and produces finite grads."""
    from photon_amd.models import build_model

    torch.manual_seed(23)
    cfg = {
        "model": {
            "d_model": 128, "n_heads": 2, "n_layers": 2,
            "expansion_ratio": 4, "max_seq_len": 128, "vocab_size": 512,
            "attn_config": {"attn_impl": "flash"},
        }
    }
    m = build_model(cfg).to(dev).to(torch.bfloat16)
    ids = torch.randint(0, 512, (2, 128), device=dev)
    out = m(ids, labels=ids)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_bias_grad_kernel(dev, ext):
    torch.manual_seed(31)
    for M, N in [(4096, 768), (65536, 3072), (1000, 2048), (16384, 8192)]:
        dy = (torch.randn(M, N, device=dev) * 0.1).to(torch.bfloat16)
        db = ext.bias_grad(dy)
        ref = dy.float().sum(0)
        rel = (db.float() - ref).abs().max() / ref.abs().max().clamp_min(1e-6)
        assert rel < 0.02, (M, N, rel)
        # deterministic: same input -> bit-identical output
        assert torch.equal(db, ext.bias_grad(dy))


def test_cpp_fused_linear_grads(dev, ext):
    """C++ autograd fused_linear vs torch F.linear: values and all grads,
    both direct bf16 and under autocast from fp32 masters."""
    torch.manual_seed(41)
    M, K, N = 256, 128, 192
    for autocast_mode in (False, True):
        dt = torch.float32 if autocast_mode else torch.bfloat16
        x = (torch.randn(M, K, device=dev) * 0.5).to(dt).requires_grad_()
        w = (torch.randn(N, K, device=dev) * 0.05).to(dt).requires_grad_()
        b = torch.randn(N, device=dev).to(dt).requires_grad_()
        xr = x.detach().clone().requires_grad_()
        wr = w.detach().clone().requires_grad_()
        br = b.detach().clone().requires_grad_()
        ctx = (
            torch.autocast("cuda", dtype=torch.bfloat16)
            if autocast_mode
            else torch.autocast("cuda", enabled=False)
        )
        with ctx:
            y = ext.fused_linear(x, w, b)
            yr = torch.nn.functional.linear(xr, wr, br)
        dy = torch.randn_like(y.float()) * 0.1
        y.backward(dy.to(y.dtype))
        yr.backward(dy.to(yr.dtype))
        assert (y.float() - yr.float()).abs().max() < 1e-3
        assert (x.grad - xr.grad).abs().max() < 1e-2, autocast_mode
        relw = (w.grad - wr.grad).abs().max() / wr.grad.abs().max()
        assert relw < 0.03, (autocast_mode, relw)
        relb = (b.grad - br.grad).abs().max() / br.grad.abs().max()
        assert relb < 0.03, (autocast_mode, relb)


def test_layernorm_fused_residual_add(dev, ext):
    """Fused s=x+r + LN forward and the residual-grad add-through backward
    vs the eager two-op reference (bf16, bit-matching rounding of s)."""
    torch.manual_seed(51)
    N, D = 512, 768
    x = (torch.randn(N, D, device=dev) * 0.5).to(torch.bfloat16).requires_grad_()
    r = (torch.randn(N, D, device=dev) * 0.5).to(torch.bfloat16).requires_grad_()
    w = torch.randn(D, device=dev).requires_grad_()
    b = torch.randn(D, device=dev).requires_grad_()
    from photon_amd.ops.layernorm import _LayerNormAddHIP

    s, y = _LayerNormAddHIP.apply(x, r, w, b, 1e-5)
    ds = (torch.randn_like(s.float()) * 0.1).to(torch.bfloat16)
    dy = (torch.randn_like(y.float()) * 0.1).to(torch.bfloat16)
    torch.autograd.backward([s, y], [ds, dy])

    xr = x.detach().clone().requires_grad_()
    rr = r.detach().clone().requires_grad_()
    wr = w.detach().clone().requires_grad_()
    br = b.detach().clone().requires_grad_()
    sr = xr + rr
    yr = torch.nn.functional.layer_norm(sr.float(), (D,), wr, br, 1e-5)
    torch.autograd.backward([sr, yr], [ds, dy.float()])

    assert torch.equal(s, (x.detach() + r.detach())), "s must be the bf16 sum"
    # y is bf16 with values up to ~4: 2 ulp = 0.03 at that magnitude
    assert (y.float() - yr).abs().max() < 5e-2
    assert (x.grad.float() - xr.grad.float()).abs().max() < 3e-2
    assert torch.equal(x.grad, r.grad), "residual pair shares the gradient"
    assert (w.grad - wr.grad).abs().max() / wr.grad.abs().max() < 0.03
    assert (b.grad - br.grad).abs().max() / br.grad.abs().max() < 0.03


def test_model_block_fused_flow_matches_eager(dev, ext):
    """The pending-add block flow must produce the same loss/grads as the
    naive x + attn(ln(x)) composition (GPU bf16)."""
    from photon_amd.models import build_model

    cfg = {
        "model": {"d_model": 256, "n_heads": 4, "n_layers": 3,
                  "expansion_ratio": 2, "max_seq_len": 128,
                  "vocab_size": 512,
                  "attn_config": {"attn_impl": "flash"}}
    }
    torch.manual_seed(52)
    m = build_model(cfg).to(dev).to(torch.bfloat16)
    ids = torch.randint(0, 512, (2, 128), device=dev)
    out = m(ids, labels=ids)
    out["loss"].backward()
    # eager reference: run the same weights through the naive composition
    import copy as _copy

    m2 = build_model(cfg).to(dev).to(torch.bfloat16)
    m2.load_state_dict(m.state_dict())
    x = m2.transformer.wte(ids)
    for blk in m2.transformer.blocks:
        x = x + blk.attn(blk.norm_1(x))
        x = x + blk.ffn(blk.norm_2(x))
    x = m2.transformer.norm_f(x)
    logits = torch.nn.functional.linear(x, m2.transformer.wte.weight)
    ref_loss = torch.nn.functional.cross_entropy(
        logits[:, :-1].reshape(-1, 512).float(), ids[:, 1:].reshape(-1)
    )
    assert abs(float(out["loss"]) - float(ref_loss)) < 5e-2
    for p in m.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()
