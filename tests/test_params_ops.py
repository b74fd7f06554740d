"""Tests for fed/params_ops.py + fed/store.py: momenta payload round-trip,
layer personalization/randomization, freezing, checkers, object store."""

import pytest
import torch

from photon_amd.conf import compose, config_yaml_dir
from photon_amd.fed.flat import FlatParams
from photon_amd.fed.params_ops import (
    freeze_blocks,
    get_optimizer_momenta,
    join_payload,
    l2_norm,
    layer_l2_norms,
    manipulate_pre_training,
    parameters_checker,
    personalize_layers,
    post_process_client_result,
    randomize_layers,
    set_optimizer_state,
    split_payload,
)
from photon_amd.fed.store import LocalStore, get_store
from photon_amd.models.mpt import MPTCausalLM, MPTConfig
from photon_amd.train import Trainer


def tiny_model():
    torch.manual_seed(0)
    return MPTCausalLM(
        MPTConfig(d_model=64, n_heads=2, n_layers=2, max_seq_len=32,
                  vocab_size=128, attn_impl="torch", loss_impl="torch")
    )


def tiny_trainer(model):
    llm = {
        "optimizer": {"name": "decoupled_adamw", "lr": 1e-3,
                      "betas": [0.9, 0.95], "eps": 1e-8, "weight_decay": 0.0},
        "scheduler": {"schedulers": {"lr": {"name": "cosine_with_warmup",
                                            "t_warmup": "1ba"}}},
        "max_duration": "100ba",
        "precision": "fp32",
        "device_train_microbatch_size": 2,
        "global_train_batch_size": 2,
    }
    return Trainer(model, llm, device="cpu")


def test_payload_join_split_roundtrip():
    p = torch.randn(10)
    m1 = torch.randn(10)
    m2 = torch.randn(10)
    pay = join_payload(p, m1, m2)
    assert pay.numel() == 30
    p2, a, b = split_payload(pay, 10, momenta=True)
    assert torch.equal(p2, p) and torch.equal(a, m1) and torch.equal(b, m2)
    p3, a3, b3 = split_payload(p, 10, momenta=False)
    assert a3 is None and torch.equal(p3, p)


def test_momenta_export_import_roundtrip():
    model = tiny_model()
    trainer = tiny_trainer(model)
    layout = FlatParams(model)
    # take one step so the optimizer has momenta
    ids = torch.randint(0, 128, (2, 16))
    out = model(ids, labels=ids)
    out["loss"].backward()
    trainer.optimizer.step()
    m1, m2 = get_optimizer_momenta(trainer, layout)
    assert float(m1.abs().sum()) > 0
    # wipe and re-import
    trainer.optimizer.state.clear()
    set_optimizer_state(trainer, layout, m1, m2, step=7)
    m1b, m2b = get_optimizer_momenta(trainer, layout)
    assert torch.allclose(m1, m1b) and torch.allclose(m2, m2b)
    params = dict(model.named_parameters())
    some_p = params[layout.names[0]]
    assert trainer.optimizer.state[some_p]["step"] == 7


def test_personalize_and_randomize_layers():
    model = tiny_model()
    layout = FlatParams(model)
    layout.copy_from_model(model)
    incoming = layout.clone_flat()
    local = incoming + 1.0
    chosen = personalize_layers(layout, incoming, local, ["wte"])
    assert chosen, "wte should match"
    views = dict(zip(layout.names, layout.layer_views_of(incoming)))
    lviews = dict(zip(layout.names, layout.layer_views_of(local)))
    for n in layout.names:
        if n in chosen:
            assert torch.equal(views[n], lviews[n])
        else:
            assert not torch.equal(views[n], lviews[n])
    flat2 = layout.clone_flat()
    r1 = randomize_layers(layout, flat2, ["norm_1"], seed=3)
    assert r1
    flat3 = layout.clone_flat()
    randomize_layers(layout, flat3, ["norm_1"], seed=3)
    assert torch.equal(flat2, flat3), "same seed => same randomization"


def test_freeze_blocks():
    model = tiny_model()
    # frozen only: matching params freeze, everything else stays trainable
    touched = freeze_blocks(model, frozen=["blocks.0"])
    assert touched
    for n, p in model.named_parameters():
        if ".blocks.0." in n:
            assert not p.requires_grad, n
        else:
            assert p.requires_grad, n
    for p in model.parameters():
        p.requires_grad_(True)
    # unfrozen given: the COMPLEMENT freezes (reference photon/utils.py:368-387
    # — "name not in unfrozen_layers" freezes)
    freeze_blocks(model, frozen=None, unfrozen=["blocks.0.norm_1"])
    for n, p in model.named_parameters():
        if "blocks.0.norm_1" in n:
            assert p.requires_grad, n
        else:
            assert not p.requires_grad, n
    for p in model.parameters():
        p.requires_grad_(True)


def test_match_names_boundary_anchoring():
    from photon_amd.fed.params_ops import _match_names

    names = [
        "transformer.blocks.1.attn.Wqkv.weight",
        "transformer.blocks.10.attn.Wqkv.weight",
        "transformer.blocks.12.norm_1.weight",
    ]
    # 'blocks.1' must not swallow blocks.10/12
    assert _match_names(names, ["blocks.1"]) == [names[0]]
    # a bare index only matches a whole dotted component
    assert _match_names(names, ["1"]) == [names[0]]
    assert _match_names(names, ["10"]) == [names[1]]
    # globs still work
    assert _match_names(names, ["*norm_1*"]) == [names[2]]
    # exact names
    assert _match_names(names, [names[1]]) == [names[1]]


def test_parameters_checker():
    a = torch.randn(5)
    parameters_checker(a, a.clone(), equal=True)
    with pytest.raises(AssertionError):
        parameters_checker(a, a + 1, equal=True)
    with pytest.raises(AssertionError):
        parameters_checker(a, a.clone(), equal=False)


def test_post_process_momenta_payload():
    model = tiny_model()
    trainer = tiny_trainer(model)
    layout = FlatParams(model)
    layout.copy_from_model(model)
    g = layout.clone_flat()
    local = g - 0.1
    payload, metrics = post_process_client_result(
        layout, g, local, 4.0, trainer=trainer, aggregate_momenta=True,
        report_layer_norms=True,
    )
    assert payload.numel() == 3 * layout.total
    assert "l2_norm_pseudo_gradient_client" in metrics
    assert abs(metrics["l2_norm_pseudo_gradient_client"] -
               l2_norm(torch.full_like(g, 0.1))) < 1e-4
    assert set(metrics["layer_pseudo_grad_norms"]) == set(layout.names)


def test_manipulate_pre_training_momenta_and_random():
    model = tiny_model()
    layout = FlatParams(model)
    layout.copy_from_model(model)
    p = layout.clone_flat()
    m1 = torch.ones_like(p)
    m2 = 2 * torch.ones_like(p)
    fl = {"aggregate_momenta": True, "random_layers": ["norm_f"], "seed": 5}
    params, a, b = manipulate_pre_training(join_payload(p, m1, m2), layout, fl, cid=0)
    assert torch.equal(a, m1) and torch.equal(b, m2)
    assert not torch.equal(params, p), "random_layers must change the payload"


def test_local_store_roundtrip(tmp_path):
    store = LocalStore(tmp_path)
    store.write_bytes("run/server/3/state.bin", b"hello")
    assert store.exists("run/server/3/state.bin")
    assert store.read_bytes("run/server/3/state.bin") == b"hello"
    assert store.list("run") == ["run/server/3/state.bin"]
    store.delete("run")
    assert store.list("run") == []


def test_get_store_local(tmp_path):
    cfg = {"comm_stack": {"s3": False, "shm": True},
           "photon": {"saving_path": str(tmp_path)}}
    store = get_store(cfg)
    assert isinstance(store, LocalStore)


def test_fed_round_with_momenta(tmp_path):
    """Two federated rounds with aggregate_momenta on a tiny model: the
    second round must import the first round's aggregated momenta."""
    from photon_amd.fed.runtime import Comm
    from photon_amd.fed.server import FedServer

    cfg = compose(config_yaml_dir(), "base", [
        "llm_config=mpt-125m",
    ])
    cfg = cfg.to_plain() if hasattr(cfg, "to_plain") else cfg
    llm = cfg["llm_config"]
    llm["model"].update({"d_model": 64, "n_heads": 2, "n_layers": 2,
                         "max_seq_len": 32, "vocab_size": 128})
    llm["model"]["attn_config"]["attn_impl"] = "torch"
    llm["global_train_batch_size"] = 2
    llm["device_train_microbatch_size"] = 2
    llm["local_steps"] = "2ba"
    llm["precision"] = "fp32"
    cfg["fl"].update({
        "n_total_clients": 2, "n_clients_per_round": 2, "n_rounds": 2,
        "aggregate_momenta": True, "reset_optimizer": False,
        "eval_period": 0,
    })
    cfg["photon"]["saving_path"] = str(tmp_path)
    server = FedServer(cfg, Comm(0, 1), device="cpu")
    server.initialize()
    server.run_round(1)
    assert float(server.client_m1.abs().sum()) > 0, "momenta aggregated"
    server.run_round(2)


def test_master_weights_cpu_matches_fp32_math():
    """bf16 param + fp32 master (PURE mixed precision): the master follows
    exact fp32 AdamW math; the bf16 param is its rounded copy."""
    from photon_amd.ops.optim import DecoupledAdamW

    torch.manual_seed(0)
    p_bf = torch.nn.Parameter(torch.randn(64, dtype=torch.bfloat16))
    ref = p_bf.detach().float().clone()
    m = torch.zeros(64)
    v = torch.zeros(64)
    opt = DecoupledAdamW([p_bf], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                         weight_decay=0.01)
    for step in range(1, 4):
        g = torch.randn(64, dtype=torch.bfloat16)
        p_bf.grad = g.clone()
        opt.step()
        # reference fp32 math on the same bf16 grads
        gf = g.float()
        m.mul_(0.9).add_(gf, alpha=0.1)
        v.mul_(0.95).addcmul_(gf, gf, value=0.05)
        bc1, bc2 = 1 - 0.9**step, 1 - 0.95**step
        ref.mul_(1 - 1e-2 * 0.01)
        ref.addcdiv_(m / bc1, (v / bc2).sqrt().add_(1e-8), value=-1e-2)
        master = opt.state[p_bf]["master"]
        assert torch.allclose(master, ref, atol=1e-6), (master - ref).abs().max()
        assert torch.equal(p_bf.detach(), master.to(torch.bfloat16))


def test_sync_masters_exact_fp32():
    from photon_amd.ops.optim import DecoupledAdamW

    p_bf = torch.nn.Parameter(torch.randn(8, dtype=torch.bfloat16))
    opt = DecoupledAdamW([p_bf], lr=1e-2)
    src = torch.randn(8)  # exact fp32 global params
    opt.sync_masters([p_bf], [src])
    assert torch.equal(opt.state[p_bf]["master"], src)
    assert torch.equal(p_bf.detach(), src.to(torch.bfloat16))


def test_fed_round_momenta_plus_personalization(tmp_path):
    """Combination: aggregate_momenta + personalized_layers + partial
    participation in one round loop (feature interaction check)."""
    from photon_amd.fed.runtime import Comm
    from photon_amd.fed.server import FedServer

    cfg = compose(config_yaml_dir(), "base", ["llm_config=mpt-125m"]).to_plain()
    llm = cfg["llm_config"]
    llm["model"].update({"d_model": 64, "n_heads": 2, "n_layers": 2,
                         "max_seq_len": 32, "vocab_size": 128})
    llm["model"]["attn_config"]["attn_impl"] = "torch"
    llm.update({"global_train_batch_size": 2, "device_train_microbatch_size": 2,
                "local_steps": "2ba", "precision": "fp32"})
    cfg["fl"].update({
        "n_total_clients": 4, "n_clients_per_round": 2, "n_rounds": 2,
        "aggregate_momenta": True, "reset_optimizer": False,
        "personalized_layers": ["norm_f"], "eval_period": 0,
    })
    cfg["photon"]["checkpoint"] = False
    cfg["photon"]["saving_path"] = str(tmp_path)
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    m1 = srv.run_round(1)
    m2 = srv.run_round(2)
    assert m1["server/sampled_clients"] == 2
    assert float(srv.client_m1.abs().sum()) > 0
    assert torch.isfinite(srv.strategy.params).all()


def test_dma_swizzle_source_permutation():
    """The LDS-DMA source-offset formula must be a bijection of each 1 KiB
    window whose inverse is the row-local XOR swizzle — i.e. for every
    destination chunk, reading global byte dest^field(row) and landing it
    linearly reproduces exactly the swizzled image the compute-side reads
    expect (mirrors attn_kernels.h dma_voff + swz/swz_field)."""

    def swz_field(r, d64):
        r &= 15
        if d64:
            return (((r >> 1) & 1) << 2) | ((r >> 2) & 1) | (((r >> 3) & 1) << 1)
        return ((r & 3) << 2) | (r >> 2)

    for D in (64, 128):
        rowstride = D * 2
        d64 = D == 64
        img_bytes = 64 * rowstride  # one staged [64][D] bf16 tile
        # forward map used by the COMPUTE side: byte -> byte ^ field(row)<<4
        def swz(b):
            return b ^ (swz_field((b // rowstride) & 15, d64) << 4)

        # DMA side: dest chunk (16B) at d reads source chunk s = d^field
        for win in range(0, img_bytes, 1024):
            seen = set()
            for lane_byte in range(win, win + 1024, 16):
                row = lane_byte // rowstride
                src = lane_byte ^ (swz_field(row & 15, d64) << 4)
                # row-preserving: source stays in the same row
                assert src // rowstride == row, (D, lane_byte)
                # the value landing at `lane_byte` must be what the
                # compute-side read of unswizzled byte `src` expects:
                assert swz(src) == lane_byte, (D, lane_byte)
                seen.add(src)
            # bijection within the window
            assert len(seen) == 64 and all(win <= s < win + 1024 for s in seen)
