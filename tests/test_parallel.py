"""Tests for photon_amd.parallel: TP linears (2-rank gloo numerics vs the
unsharded model), bucketed DDP grad sync, FSDP wrap no-op paths."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from photon_amd.models.mpt import MPTCausalLM, MPTConfig
from photon_amd.parallel import BucketedGradSync, apply_fsdp, apply_tensor_parallel
from photon_amd.parallel.tp import MPT_TP_PLAN


def tiny_model(seed=0):
    torch.manual_seed(seed)
    return MPTCausalLM(
        MPTConfig(d_model=64, n_heads=4, n_layers=2, max_seq_len=32,
                  vocab_size=128, attn_impl="torch", loss_impl="torch")
    )


def test_tp_world1_is_noop():
    model = tiny_model()
    assert apply_tensor_parallel(model, 0, 1) == []


def test_fsdp_falsy_config_noop():
    model = tiny_model()
    assert apply_fsdp(model, None) is model
    assert apply_fsdp(model, {}) is model


def test_ddp_sync_world1_noop():
    model = tiny_model()
    ids = torch.randint(0, 128, (2, 16))
    model(ids, labels=ids)["loss"].backward()
    g0 = next(p.grad for p in model.parameters() if p.grad is not None).clone()
    BucketedGradSync(world_size=1)(model)
    g1 = next(p.grad for p in model.parameters() if p.grad is not None)
    assert torch.equal(g0, g1)


# ---------------------------------------------------------------------------
# 2-rank gloo: TP forward/backward equals the unsharded model
# ---------------------------------------------------------------------------
def _tp_worker(rank, world, port, out_dir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model = tiny_model(seed=0)  # same init on both ranks
        replaced = apply_tensor_parallel(model, rank, world)
        assert len(replaced) == 2 * 4  # 4 linears per block x 2 blocks
        torch.manual_seed(123)
        ids = torch.randint(0, 128, (2, 16))
        out = model(ids, labels=ids)
        out["loss"].backward()
        torch.save(
            {"loss": out["loss"].detach(),
             "wte_grad": model.transformer.wte.weight.grad.clone()},
            os.path.join(out_dir, f"tp_{rank}.pt"),
        )
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_tp_two_rank_matches_unsharded(tmp_path):
    ref = tiny_model(seed=0)
    torch.manual_seed(123)
    ids = torch.randint(0, 128, (2, 16))
    out = ref(ids, labels=ids)
    out["loss"].backward()
    ref_loss = out["loss"].detach()
    ref_wte_grad = ref.transformer.wte.weight.grad.clone()

    ctx = mp.get_context("spawn")
    from tests.conftest import free_port
    port = free_port()
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, port, str(tmp_path)))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    for r in range(2):
        got = torch.load(tmp_path / f"tp_{r}.pt")
        assert torch.allclose(got["loss"], ref_loss, atol=1e-5), (
            got["loss"], ref_loss
        )
        # wte is replicated: its grad must equal the unsharded grad
        assert torch.allclose(got["wte_grad"], ref_wte_grad, atol=1e-5)


# ---------------------------------------------------------------------------
# 2-rank gloo: DDP bucketed sync averages gradients
# ---------------------------------------------------------------------------
def _ddp_worker(rank, world, port, out_dir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model = tiny_model(seed=0)
        torch.manual_seed(1000 + rank)  # different data per rank
        ids = torch.randint(0, 128, (2, 16))
        model(ids, labels=ids)["loss"].backward()
        pre = model.transformer.wte.weight.grad.clone()
        BucketedGradSync(bucket_bytes=1 << 16)(model)
        post = model.transformer.wte.weight.grad.clone()
        torch.save({"pre": pre, "post": post},
                   os.path.join(out_dir, f"ddp_{rank}.pt"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ddp_two_rank_sync(tmp_path):
    ctx = mp.get_context("spawn")
    from tests.conftest import free_port
    port = free_port()
    procs = [
        ctx.Process(target=_ddp_worker, args=(r, 2, port, str(tmp_path)))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    a = torch.load(tmp_path / "ddp_0.pt")
    b = torch.load(tmp_path / "ddp_1.pt")
    # post-sync grads identical on both ranks and equal to the mean of pre
    assert torch.allclose(a["post"], b["post"], atol=1e-7)
    assert torch.allclose(a["post"], (a["pre"] + b["pre"]) / 2, atol=1e-6)


def test_tp_swap_disables_fused_linear_path(tiny_cfg):
    """TP replaces the projection Linears; the fused lt_linear path must
    detect the swap and fall back to module forwards (otherwise the TP
    collectives would be silently skipped on GPU)."""
    import torch.nn as nn

    from photon_amd.models import build_model
    from photon_amd.parallel.tp import apply_tensor_parallel

    m = build_model(tiny_cfg["llm_config"])
    apply_tensor_parallel(m, rank=0, world=2)  # world=1 is a no-op
    blk = m.transformer.blocks[0]
    assert type(blk.attn.Wqkv) is not nn.Linear
    assert type(blk.ffn.up_proj) is not nn.Linear
    # forward must route through the swapped modules (shape check only:
    # without an initialized group the row-parallel all-reduce is local)
    ids = __import__("torch").randint(0, 512, (1, 32))
    out = m(ids)
    assert out["logits"].shape[-1] == tiny_cfg["llm_config"]["model"]["vocab_size"]
