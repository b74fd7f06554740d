"""Trainer tests: scheduler math, grad accumulation, checkpoints, optimizers."""

import math

import pytest
import torch

from photon_amd.data.shards import StatefulLoader
from photon_amd.data.synthetic import SyntheticTokenDataset
from photon_amd.models.mpt import MPTCausalLM, MPTConfig
from photon_amd.ops.clip import clip_grad_norm_
from photon_amd.ops.optim import ADOPT, DecoupledAdamW
from photon_amd.train import Trainer
from photon_amd.train.scheduler import CosineWithWarmup


def make_trainer(tiny_llm_config, tmp_path, **kw):
    torch.manual_seed(0)
    model = MPTCausalLM(
        MPTConfig(d_model=64, n_heads=4, n_layers=2, max_seq_len=64, vocab_size=512,
                  attn_impl="torch")
    )
    ds = SyntheticTokenDataset(64, vocab_size=512, seed=9)
    return Trainer(
        model, tiny_llm_config,
        train_loader=StatefulLoader(ds, 2),
        eval_loader=StatefulLoader(ds, 2),
        device="cpu", save_folder=str(tmp_path), **kw,
    )


def test_scheduler_values():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=1.0)
    sch = CosineWithWarmup(opt, "10ba", "110ba", alpha_f=0.1)
    assert sch.alpha(0) == 0.0
    assert sch.alpha(5) == 0.5
    assert sch.alpha(10) == 1.0
    assert abs(sch.alpha(60) - (0.1 + 0.9 * 0.5)) < 1e-9  # cosine midpoint
    assert abs(sch.alpha(110) - 0.1) < 1e-9
    assert sch.alpha(200) == pytest.approx(0.1)


def test_grad_accum_count(tiny_llm_config, tmp_path):
    tr = make_trainer(tiny_llm_config, tmp_path)
    # global 4 / world 1 / micro 2 -> 2 microbatches
    assert tr.grad_accum == 2
    tr.world_size = 2
    assert tr.grad_accum == 1


def test_fit_advances_timestamp(tiny_llm_config, tmp_path):
    tr = make_trainer(tiny_llm_config, tmp_path)
    tr.fit("3ba")
    assert tr.timestamp.batch == 3
    assert tr.timestamp.sample == 3 * 4
    assert tr.timestamp.token == 3 * 4 * 64


def test_checkpoint_roundtrip(tiny_llm_config, tmp_path):
    tr = make_trainer(tiny_llm_config, tmp_path)
    tr.fit("2ba")
    p = tr.save_checkpoint()
    assert p.name == "ep0-ba2-rank0.pt"

    tr2 = make_trainer(tiny_llm_config, tmp_path)
    tr2.load_checkpoint(p)
    assert tr2.timestamp.batch == 2
    assert tr2.train_loader.samples_consumed == tr.train_loader.samples_consumed
    for (n1, p1), (n2, p2) in zip(
        tr.model.named_parameters(), tr2.model.named_parameters()
    ):
        assert torch.equal(p1, p2), n1


def test_checkpoint_ignore_keys(tiny_llm_config, tmp_path):
    tr = make_trainer(tiny_llm_config, tmp_path)
    tr.fit("2ba")
    p = tr.save_checkpoint()
    tr2 = make_trainer(tiny_llm_config, tmp_path)
    tr2.load_checkpoint(p, load_ignore_keys=["*optim*", "*dataset_state*"])
    assert not tr2.optimizer.state  # optimizer state dropped
    assert tr2.train_loader.samples_consumed == 0
    assert tr2.timestamp.batch == 2


def test_resume_determinism(tiny_llm_config, tmp_path):
    """fit(4) == fit(2) + checkpoint + restore + fit(2)."""
    tr_a = make_trainer(tiny_llm_config, tmp_path)
    tr_a.fit("4ba")

    tr_b = make_trainer(tiny_llm_config, tmp_path)
    tr_b.fit("2ba")
    p = tr_b.save_checkpoint()
    tr_c = make_trainer(tiny_llm_config, tmp_path)
    tr_c.load_checkpoint(p)
    tr_c.fit("2ba")
    for (n1, p1), (n2, p2) in zip(
        tr_a.model.named_parameters(), tr_c.model.named_parameters()
    ):
        assert torch.allclose(p1, p2, atol=1e-6), n1


def test_clip_grad_norm_matches_torch():
    torch.manual_seed(4)
    ps = [torch.nn.Parameter(torch.randn(10, 10)) for _ in range(3)]
    for p in ps:
        p.grad = torch.randn_like(p) * 5
    grads_ref = [p.grad.clone() for p in ps]
    total = clip_grad_norm_(ps, 1.0)
    ref_total = torch.norm(torch.stack([g.norm(2) for g in grads_ref]), 2)
    assert torch.allclose(total, ref_total, atol=1e-5)
    scale = 1.0 / (ref_total + 1e-6)
    for p, g in zip(ps, grads_ref):
        assert torch.allclose(p.grad, g * scale, atol=1e-5)


def test_adamw_matches_torch_adamw():
    torch.manual_seed(5)
    p1 = torch.nn.Parameter(torch.randn(32))
    p2 = torch.nn.Parameter(p1.detach().clone())
    g = torch.randn(32)
    o1 = DecoupledAdamW([p1], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.0)
    o2 = torch.optim.AdamW([p2], lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.0)
    for _ in range(5):
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_adopt_first_step_no_update():
    p = torch.nn.Parameter(torch.ones(4))
    o = ADOPT([p], lr=0.1)
    p.grad = torch.ones(4)
    o.step()
    assert torch.equal(p.detach(), torch.ones(4))  # v init only
    p.grad = torch.ones(4)
    o.step()
    assert not torch.equal(p.detach(), torch.ones(4))


def test_adopt_oracle():
    """ADOPT update vs a NumPy oracle of the clipped rule."""
    import numpy as np

    torch.manual_seed(6)
    x0 = torch.randn(16)
    p = torch.nn.Parameter(x0.clone())
    o = ADOPT([p], lr=0.05, betas=(0.9, 0.9999), eps=1e-6)
    grads = [torch.randn(16) for _ in range(4)]
    for g in grads:
        p.grad = g.clone()
        o.step()

    x = x0.numpy().astype(np.float64)
    m = np.zeros(16)
    v = None
    for t, g in enumerate([g.numpy().astype(np.float64) for g in grads], start=1):
        if t == 1:
            v = g * g
            continue
        clip = (t - 1) ** 0.25
        c = np.clip(g / np.maximum(np.sqrt(v), 1e-6), -clip, clip)
        m = 0.9 * m + 0.1 * c
        x = x - 0.05 * m
        v = 0.9999 * v + 0.0001 * g * g
    assert torch.allclose(p.detach(), torch.tensor(x, dtype=torch.float32), atol=1e-5)


def test_momenta_export_import():
    p = torch.nn.Parameter(torch.randn(8))
    o = DecoupledAdamW([p], lr=1e-3)
    p.grad = torch.randn(8)
    o.step()
    m1, m2 = o.export_momenta([p])
    o2 = DecoupledAdamW([torch.nn.Parameter(torch.randn(8))], lr=1e-3)
    o2.import_momenta(list(o2.param_groups[0]["params"]), m1, m2, step=5)
    pp = o2.param_groups[0]["params"][0]
    assert torch.equal(o2.state[pp]["exp_avg"], m1[0])
    assert o2.state[pp]["step"] == 5


def test_monitors_and_profiler(tiny_llm_config, tmp_path):
    """speed/lr/memory monitors populate metrics; profiler writes a trace."""
    cfg = dict(tiny_llm_config)
    cfg["callbacks"] = {
        "speed_monitor": {"window_size": 4},
        "lr_monitor": {},
        "memory_monitor": {},
        "runtime_estimator": {},
        "optimizer_monitor": {"interval": 1},
    }
    cfg["profiler"] = {"folder": str(tmp_path / "traces"),
                       "schedule": {"wait": 0, "warmup": 0, "active": 2}}
    tr = make_trainer(cfg, tmp_path)
    tr.fit("3ba")
    keys = tr.metrics.keys()
    assert "throughput/tokens_per_sec" in keys
    assert any(k.startswith("lr-") for k in keys)
    assert "optimizer/l2_norm_grad" in keys
    assert list((tmp_path / "traces").glob("*.json*")), "chrome trace written"


def test_checkpoint_retention(tiny_llm_config, tmp_path):
    cfg = dict(tiny_llm_config)
    cfg["save_num_checkpoints_to_keep"] = 2
    tr = make_trainer(cfg, tmp_path)
    for _ in range(4):
        tr.fit("1ba")
        tr.save_checkpoint()
    kept = sorted(tmp_path.glob("ep*-ba*-rank0.pt"))
    assert len(kept) == 2
    assert kept[-1].name == "ep0-ba4-rank0.pt"


def test_microbatch_auto(tiny_llm_config, tmp_path):
    cfg = dict(tiny_llm_config)
    cfg["device_train_microbatch_size"] = "auto"
    tr = make_trainer(cfg, tmp_path)
    assert isinstance(tr.microbatch, int) and tr.microbatch >= 1


def test_optimizer_bucket_with_master_only_state():
    """ADVICE m2: after reset_optimizer clears state and sync_masters
    re-seeds only {'master': ...}, the first step() must initialize
    exp_avg/exp_avg_sq instead of crashing with KeyError."""
    import torch

    from photon_amd.ops.optim import DecoupledAdamW

    p = torch.nn.Parameter(torch.randn(8))
    opt = DecoupledAdamW([p], lr=1e-3)
    # one normal step to build state, then a fed-style reset + master seed
    p.grad = torch.randn(8)
    opt.step()
    opt.state.clear()
    opt.sync_masters([p], [p.detach().to(torch.float32).clone()])
    p.grad = torch.randn(8)
    opt.step()  # must not raise
    st = opt.state[p]
    assert "exp_avg" in st and st["step"] == 1


def test_centralized_split_eval(tiny_cfg, tmp_path, monkeypatch):
    """centralized.split_eval reports a per-stream CE for every configured
    stream (reference centralised_train.py:74 surface)."""
    import copy

    from photon_amd import centralised_train as ct

    cfg = copy.deepcopy(tiny_cfg)
    cfg.setdefault("centralized", {})
    cfg["centralized"]["split_eval"] = True
    cfg["dataset"]["val"] = {
        "root_local": "", "split": "validation",
        "streams": [
            {"client_streams": {f"c{i}": {"local": f"client_{i}"}}}
            for i in range(3)
        ],
    }
    from photon_amd.data import build_eval_loader, build_train_loader
    from photon_amd.models import build_model
    from photon_amd.train import Trainer

    model = build_model(cfg["llm_config"])
    tr = Trainer(model, cfg["llm_config"],
                 train_loader=build_train_loader(cfg, client_id=0),
                 eval_loader=build_eval_loader(cfg, client_id=0),
                 device="cpu")
    out = ct.run_split_eval(cfg, tr)
    assert set(out) == {
        f"metrics/eval/LanguageCrossEntropy_stream_{i}" for i in range(3)
    }
