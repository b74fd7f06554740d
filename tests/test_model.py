"""MPT model tests: shapes, causality, ALiBi, loss correctness, naming contract."""

import math

import pytest
import torch

from photon_amd.models.mpt import MPTCausalLM, MPTConfig
from photon_amd.ops.attention import alibi_slopes, reference_attention_fp32, sdpa_attention
from photon_amd.ops.cross_entropy import fused_cross_entropy, reference_cross_entropy_fp32


def tiny(vocab=128, **kw):
    d = dict(d_model=32, n_heads=4, n_layers=2, max_seq_len=32, vocab_size=vocab,
             attn_impl="torch")
    d.update(kw)
    return MPTCausalLM(MPTConfig(**d))


def test_forward_shapes():
    m = tiny()
    ids = torch.randint(0, 128, (2, 16))
    out = m(ids)
    assert out["logits"].shape == (2, 16, 128)
    out = m(ids, labels=ids)
    assert out["loss"].ndim == 0


def test_causality():
    """Changing a future token must not change past logits."""
    m = tiny()
    m.eval()
    ids = torch.randint(0, 128, (1, 16))
    with torch.no_grad():
        l1 = m(ids)["logits"]
        ids2 = ids.clone()
        ids2[0, 10] = (ids2[0, 10] + 1) % 128
        l2 = m(ids2)["logits"]
    assert torch.allclose(l1[0, :10], l2[0, :10], atol=1e-5)
    assert not torch.allclose(l1[0, 10:], l2[0, 10:], atol=1e-5)


def test_alibi_slopes_power_of_two():
    s = alibi_slopes(8)
    expect = torch.tensor([2.0 ** (-(i + 1)) for i in range(8)])
    assert torch.allclose(s, expect)


def test_alibi_slopes_non_power_of_two():
    s = alibi_slopes(12)
    assert s.shape == (12,)
    assert (s > 0).all() and (s <= 1).all()
    # MPT convention: odd-index slopes of the next pow2 come first
    s16 = alibi_slopes(16)
    assert torch.allclose(s[:8], s16[1::2])


def test_sdpa_matches_reference():
    torch.manual_seed(1)
    B, H, S, dh = 2, 4, 32, 16
    q, k, v = (torch.randn(B, H, S, dh) for _ in range(3))
    slopes = alibi_slopes(H)
    out = sdpa_attention(q, k, v, slopes)
    ref = reference_attention_fp32(q, k, v, slopes)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def test_fused_ce_matches_reference():
    torch.manual_seed(2)
    N, D, V = 64, 32, 97
    h = torch.randn(N, D, requires_grad=True)
    w = torch.randn(V, D, requires_grad=True)
    t = torch.randint(0, V, (N,))
    loss = fused_cross_entropy(h, w, t, impl="torch")
    ref = reference_cross_entropy_fp32(h.detach(), w.detach(), t)
    assert torch.allclose(loss, ref, atol=1e-5)
    loss.backward()
    h2 = h.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    ref2 = reference_cross_entropy_fp32(h2, w2, t)
    ref2.backward()
    assert torch.allclose(h.grad, h2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)


def test_loss_near_log_vocab_at_init():
    m = tiny(vocab=512)
    ids = torch.randint(0, 512, (2, 32))
    loss = float(m(ids, labels=ids)["loss"])
    assert abs(loss - math.log(512)) < 0.5


def test_param_naming_contract():
    """All trainable params live under `transformer` (filter-key contract)."""
    m = tiny()
    names = [n for n, p in m.named_parameters() if p.requires_grad]
    assert all("transformer" in n for n in names)
    assert any("wte" in n for n in names)
    assert any("norm_1" in n for n in names)
    assert any("Wqkv" in n for n in names)
    assert any("out_proj" in n for n in names)
    assert any("up_proj" in n for n in names)
    assert any("down_proj" in n for n in names)
    assert any("norm_f" in n for n in names)


def test_tied_lm_head():
    m = tiny()
    ids = torch.randint(0, 128, (1, 8))
    logits = m(ids)["logits"]
    manual = m.transformer(ids) @ m.transformer.wte.weight.t()
    assert torch.allclose(logits, manual, atol=1e-5)


def test_resize_vocab():
    from photon_amd.models.mpt import MPTCausalLM, MPTConfig, resize_vocab

    torch.manual_seed(0)
    m = MPTCausalLM(MPTConfig(d_model=64, n_heads=2, n_layers=1,
                              max_seq_len=32, vocab_size=100,
                              attn_impl="torch", loss_impl="torch"))
    old_rows = m.transformer.wte.weight[:100].clone()
    resize_vocab(m, 140)
    assert m.transformer.wte.num_embeddings == 140
    assert torch.equal(m.transformer.wte.weight[:100], old_rows)
    ids = torch.randint(0, 140, (2, 16))
    out = m(ids, labels=ids)
    assert torch.isfinite(out["loss"])
