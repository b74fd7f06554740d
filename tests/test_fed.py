"""Federated runtime tests: sampling, round loop, failure budget, resume,
and 2-rank gloo equivalence (the distributed path must be correct by
construction — SURVEY.md drives multi-GPU checks from CPU gloo tests)."""

import copy
import os

import pytest
import torch
import torch.multiprocessing as mp

from photon_amd.fed.flat import FlatParams
from photon_amd.fed.runtime import Comm, assign_clients_to_ranks, sample_clients
from photon_amd.fed.server import FedServer, TooManyFailuresError, weighted_loss_avg


def test_sample_clients_deterministic():
    a = sample_clients(1337, 5, 8, 4)
    b = sample_clients(1337, 5, 8, 4)
    c = sample_clients(1337, 6, 8, 4)
    assert a == b
    assert len(a) == 4 and all(0 <= x < 8 for x in a)
    # successive rounds differ (with overwhelming probability for this seed)
    assert a != c or sample_clients(1337, 7, 8, 4) != a


def test_assignment_round_robin():
    out = assign_clients_to_ranks([0, 1, 2, 3, 4], 2)
    assert out == {0: [0, 2, 4], 1: [1, 3]}


def test_weighted_loss_avg():
    assert weighted_loss_avg([(2.0, 1.0), (4.0, 3.0)]) == pytest.approx(3.5)


def test_flatparams_roundtrip(tiny_cfg):
    from photon_amd.models import build_model

    torch.manual_seed(0)
    m = build_model(tiny_cfg["llm_config"])
    fp = FlatParams(m)
    fp.copy_from_model(m)
    arrays = fp.to_ndarrays()
    assert len(arrays) == len(fp.names)
    fp2 = FlatParams(m)
    fp2.from_ndarrays(arrays)
    assert torch.equal(fp.flat, fp2.flat)
    # names sorted
    assert fp.names == sorted(fp.names)
    fp2.flat.add_(1.0)
    fp2.copy_to_model(m)
    fp3 = FlatParams(m).copy_from_model(m)
    assert torch.allclose(fp3.flat, fp2.flat)


def test_single_process_rounds_and_resume(tiny_cfg):
    srv = FedServer(tiny_cfg, Comm(0, 1), "cpu")
    hist = srv.run(2)
    assert len(hist.losses_distributed) == 2
    srv2 = FedServer(tiny_cfg, Comm(0, 1), "cpu")
    srv2.initialize()
    assert srv2.start_round == 3
    assert torch.allclose(srv2.strategy.params, srv.strategy.params)


def test_failure_budget(tiny_cfg):
    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    # sabotage: client fit raises
    def boom(*a, **k):
        raise RuntimeError("injected client failure")

    srv.client.fit = boom
    with pytest.raises(TooManyFailuresError):
        srv.run_round(1)


def test_failure_tolerated_with_budget(tiny_cfg):
    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = False
    cfg["fl"]["accept_failures_cnt"] = 1
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    real_fit = srv.client.fit
    calls = {"n": 0}

    def flaky(cid, *a, **k):
        calls["n"] += 1
        if cid == 1:
            raise RuntimeError("injected")
        return real_fit(cid, *a, **k)

    srv.client.fit = flaky
    metrics = srv.run_round(1)
    assert metrics["server/failures"] == 1


def test_partial_participation(tiny_cfg):
    cfg = copy.deepcopy(tiny_cfg)
    cfg["fl"]["n_total_clients"] = 4
    cfg["fl"]["n_clients_per_round"] = 2
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    m = srv.run_round(1)
    assert m["server/sampled_clients"] == 2


# ---------------------------------------------------------------------------
# 2-process gloo tests (world_size 2 over 127.0.0.1)
# ---------------------------------------------------------------------------

def _dist_worker(rank, world, port, cfg, out_dir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        srv = FedServer(cfg, Comm(rank, world), "cpu")
        srv.run(2)
        torch.save(srv.strategy.params, os.path.join(out_dir, f"params_{rank}.pt"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_two_rank_gloo_equivalence(tiny_cfg, tmp_path):
    """2 ranks x 1 client each must equal 1 rank x 2 clients bit-for-bit
    (both sides deterministic; aggregation is the same weighted average)."""
    cfg1 = copy.deepcopy(tiny_cfg)
    cfg1["photon"]["checkpoint"] = False
    cfg1["photon"]["saving_path"] = str(tmp_path / "a")
    srv = FedServer(cfg1, Comm(0, 1), "cpu")
    srv.run(2)
    single = srv.strategy.params.clone()

    cfg2 = copy.deepcopy(tiny_cfg)
    cfg2["photon"]["checkpoint"] = False
    cfg2["photon"]["saving_path"] = str(tmp_path / "b")
    ctx = mp.get_context("spawn")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir, exist_ok=True)
    from tests.conftest import free_port
    port = free_port()
    procs = [
        ctx.Process(target=_dist_worker, args=(r, 2, port, cfg2, out_dir))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    results = {
        r: torch.load(os.path.join(out_dir, f"params_{r}.pt")) for r in range(2)
    }
    # both ranks hold identical global params (replicated server-opt)
    assert torch.equal(results[0], results[1])
    # and they match the single-process run
    assert torch.allclose(results[0], single, atol=1e-6), (
        (results[0] - single).abs().max()
    )


def test_client_checkpoint_skip_and_load(tiny_cfg, tmp_path):
    """Reference mid-round resume: a second fit over the same rounds loads
    the existing client checkpoint instead of re-training
    (llm_config_functions.py:642-764)."""
    import copy

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    m1 = srv.run_round(1)
    assert "client/fit_skipped_from_checkpoint" not in m1
    # new server over the same save dir: round 1 must be skipped-from-ckpt
    srv2 = FedServer(copy.deepcopy(cfg), Comm(0, 1), "cpu")
    srv2.initialize()
    m2 = srv2.run_round(1)
    assert m2.get("client/fit_skipped_from_checkpoint") == 1.0


def test_restore_run_uuid(tiny_cfg, tmp_path):
    """Cross-run restore: a new run_uuid picks up the old run's latest
    server round (s3_utils.py:275-345,1478-1608 semantics)."""
    import copy

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    cfg["run_uuid"] = "old_run"
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.run(2)
    old_params = srv.strategy.params.clone()

    cfg2 = copy.deepcopy(tiny_cfg)
    cfg2["photon"]["checkpoint"] = False
    cfg2["photon"]["saving_path"] = str(tmp_path)
    cfg2["photon"]["restore_run_uuid"] = "old_run"
    cfg2["photon"]["resume_round"] = -1
    cfg2["run_uuid"] = "new_run"
    srv2 = FedServer(cfg2, Comm(0, 1), "cpu")
    srv2.initialize()
    assert srv2.start_round == 3
    assert torch.allclose(srv2.strategy.params, old_params, atol=1e-6)


@pytest.mark.timeout(600)
def test_fed_training_reduces_loss(tiny_cfg):
    """End-to-end learning check: 6 federated rounds on a tiny model must
    reduce eval loss (the reference's artifact-evaluation criterion —
    perplexity dropping over rounds, docs/artifact_evaluation.tex:133-139)."""
    import copy

    cfg = copy.deepcopy(tiny_cfg)
    cfg["llm_config"]["model"].update({"d_model": 128, "n_heads": 4})
    cfg["llm_config"]["local_steps"] = "10ba"
    cfg["llm_config"]["optimizer"]["lr"] = 3e-3
    cfg["fl"].update({"n_rounds": 5, "eval_period": 1})
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    first = None
    last = None
    for r in range(1, 6):
        srv.run_round(r)
        loss = srv.evaluate_round(r)
        first = first if first is not None else loss
        last = loss
    assert last < first - 0.03, (first, last)


def test_round_checkpoint_retention(tiny_cfg, tmp_path):
    import copy

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    cfg["photon"]["save_num_rounds_to_keep"] = 2
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.run(3)
    from photon_amd.fed.server_ckpt import obtain_sorted_rounds

    rounds = obtain_sorted_rounds(tmp_path, cfg["run_uuid"], srv.strategy.state_keys)
    assert rounds == [2, 3]


def test_resume_restores_client_state(tiny_cfg, tmp_path):
    import copy

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.run(2)
    steps = {cid: st.steps_done for cid, st in srv.client.client_states.items()}
    assert steps

    cfg2 = copy.deepcopy(cfg)
    cfg2["photon"]["resume_round"] = -1
    srv2 = FedServer(cfg2, Comm(0, 1), "cpu")
    srv2.initialize()
    assert {c: s.steps_done for c, s in srv2.client.client_states.items()} == steps
    assert srv2.server_steps_cumulative == srv.server_steps_cumulative


def test_npz_roundtrip_property(tiny_cfg):
    """FlatParams .npz save/load round-trips bit-exactly for several
    random payloads (the server-checkpoint wire format)."""
    import numpy as np
    import tempfile

    from photon_amd.models import build_model

    model = build_model(tiny_cfg["llm_config"])
    layout = FlatParams(model)
    for seed in range(3):
        torch.manual_seed(seed)
        flat = torch.randn_like(layout.flat)
        with tempfile.TemporaryDirectory() as d:
            p = os.path.join(d, "x.npz")
            layout.save_npz(p, flat)
            arrays = layout.load_npz(p)
            back = torch.cat([torch.from_numpy(np.ascontiguousarray(a)).reshape(-1)
                              for a in arrays])
            assert torch.equal(back, flat)


def test_eval_covers_every_sampled_client(tiny_cfg):
    """VERDICT missing #5: on 1 rank with n sampled clients, evaluate_round
    must evaluate EVERY sampled client (reference node_manager_app.py:594-725),
    and split_eval reports a per-cid loss keyed by the real cid."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["fl"]["n_total_clients"] = 4
    cfg["fl"]["n_clients_per_round"] = 3
    cfg["fl"]["split_eval"] = True
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    evaluated = []
    real_eval = srv.client.evaluate

    def spy(cid, *a, **k):
        evaluated.append(cid)
        return real_eval(cid, *a, **k)

    srv.client.evaluate = spy
    srv.run_round(1)
    srv.evaluate_round(1)
    sampled = sample_clients(cfg["seed"], 1, 4, 3)
    assert sorted(evaluated) == sampled
    # split_eval reported a per-cid metric for each sampled cid
    per_cid = {
        k for k in srv.history.metrics_distributed
        if k.startswith("metrics/eval/LanguageCrossEntropy_client_")
    }
    assert per_cid == {
        f"metrics/eval/LanguageCrossEntropy_client_{c}" for c in sampled
    }


def test_eval_full_split_not_coerced(tiny_cfg):
    """eval_subset_num_batches=-1 must reach the trainer as -1 (full split),
    not be coerced to 8 in the fed path (VERDICT weak #4)."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["llm_config"]["eval_subset_num_batches"] = -1
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    seen = []
    real_eval = srv.client.evaluate

    def spy(cid, flat, layout, subset):
        seen.append(subset)
        return real_eval(cid, flat, layout, subset)

    srv.client.evaluate = spy
    srv.run_round(1)
    srv.evaluate_round(1)
    assert seen and all(s == -1 for s in seen)


def test_aggregate_momenta_survive_resume(tiny_cfg, tmp_path):
    """ADVICE m3: aggregated client momenta are persisted with the server
    round checkpoint and restored on resume (no silent zero-momenta
    broadcast after resume)."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["fl"]["aggregate_momenta"] = True
    cfg["fl"]["reset_optimizer"] = False
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.run(2)
    assert float(srv.client_m1.abs().sum()) > 0, "momenta should be non-zero"
    m1, m2 = srv.client_m1.clone(), srv.client_m2.clone()

    cfg2 = copy.deepcopy(cfg)
    cfg2["photon"]["resume_round"] = -1
    srv2 = FedServer(cfg2, Comm(0, 1), "cpu")
    srv2.initialize()
    assert torch.equal(srv2.client_m1, m1)
    assert torch.equal(srv2.client_m2, m2)


def test_icl_eval_runs_on_current_global_params(tiny_cfg):
    """ADVICE h1: the ICL/gauntlet eval must see the CURRENT global params
    (strategy.params), not the stale layout.flat init snapshot."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    srv.run_round(1)
    # simulate the ICL path's model refresh: patch run_icl_eval to capture
    # the model's params at call time
    captured = {}

    def fake_icl(cfg_, model, device):
        captured["flat"] = FlatParams(model).copy_from_model(model).flat.clone()
        return {"icl/fake": 1.0}

    import photon_amd.centralised_train as ct

    orig = ct.run_icl_eval
    ct.run_icl_eval = fake_icl
    try:
        srv.cfg["icl_tasks_config"] = {"icl_tasks": [{"label": "fake"}]}
        srv.evaluate_round(1)
    finally:
        ct.run_icl_eval = orig
        srv.cfg.pop("icl_tasks_config", None)
    assert "flat" in captured
    assert torch.allclose(captured["flat"], srv.strategy.params, atol=1e-6)


# ---------------------------------------------------------------------------
# Hung-rank watchdog (SURVEY §7 hard-part 3)
# ---------------------------------------------------------------------------

def _watchdog_worker(rank, world, port, cfg, out_dir, flag_path):
    import time as _time

    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    srv = FedServer(cfg, Comm(rank, world), "cpu")
    if rank == 1:
        def hang(*a, **k):
            open(flag_path, "w").close()
            _time.sleep(600)

        srv.client.fit = hang
    srv.initialize()
    m1 = srv.run_round(1)
    m2 = srv.run_round(2)  # post-rebuild round must also work
    torch.save(
        {
            "params": srv.strategy.params,
            "world": srv.comm.world_size,
            "failures_r1": m1["server/failures"],
            "failures_r2": m2["server/failures"],
        },
        os.path.join(out_dir, f"wd_{rank}.pt"),
    )
    if dist.is_initialized():
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_watchdog_survives_killed_rank(tiny_cfg, tmp_path):
    """A rank killed mid-fit must not deadlock the round: the survivor
    detects it via the fit-timeout watchdog, rebuilds the process group
    alone, counts the dead rank's clients as failures, and completes this
    round AND the next (reference worker.py:437-448 + fit_utils.py:198-288
    analogue)."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = False
    cfg["photon"]["fit_timeout_s"] = 6
    cfg["fl"]["accept_failures_cnt"] = 0  # requeue must fully recover
    from tests.conftest import free_port

    port = free_port()
    flag = str(tmp_path / "rank1_fitting")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir, exist_ok=True)
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(
            target=_watchdog_worker, args=(r, 2, port, cfg, out_dir, flag)
        )
        for r in range(2)
    ]
    for p in procs:
        p.start()
    # wait until rank 1 is inside its (hung) fit, then SIGKILL it
    import time as _time

    deadline = _time.time() + 60
    while not os.path.exists(flag) and _time.time() < deadline:
        _time.sleep(0.1)
    assert os.path.exists(flag), "rank 1 never reached its fit"
    procs[1].kill()
    procs[0].join(timeout=180)
    assert procs[0].exitcode == 0, "survivor rank must complete both rounds"
    procs[1].join(timeout=30)
    out = torch.load(os.path.join(out_dir, "wd_0.pt"))
    assert out["world"] == 1, "group must be rebuilt without the dead rank"
    # the dead rank's client is REQUEUED onto the survivor and recovers
    # (reference node_manager_app.py:574-579) -> zero failures
    assert out["failures_r1"] == 0.0, "requeued client must recover"
    assert out["failures_r2"] == 0.0
    assert torch.isfinite(out["params"]).all()


def test_restore_cent_from_composer_pt(tiny_cfg, tmp_path):
    """VERDICT missing #3: a federated run bootstraps from a Composer-keyed
    ep{e}-ba{b}-rank0.pt checkpoint, params bit-equal to the source
    (reference init_utils.py:43-125)."""
    from photon_amd.models import build_model

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["saving_path"] = str(tmp_path)
    # write a synthetic Composer checkpoint from a randomly-initialized model
    torch.manual_seed(99)
    src_model = build_model(cfg["llm_config"])
    for p in src_model.parameters():
        p.data.add_(torch.randn_like(p) * 0.01)
    cent_dir = tmp_path / "cent_run"
    cent_dir.mkdir(parents=True)
    torch.save(
        {"state": {"model": src_model.state_dict(),
                   "optimizers": {}, "timestamp": {"batch": 120}}},
        cent_dir / "ep0-ba120-rank0.pt",
    )
    torch.save(
        {"state": {"model": {k: v * 0 for k, v in
                             src_model.state_dict().items()}}},
        cent_dir / "ep0-ba60-rank0.pt",
    )

    cfg["photon"]["restore_cent_run_uuid"] = "cent_run"
    cfg["photon"]["restore_cent_run_batches"] = 120
    cfg["photon"]["checkpoint"] = False
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.initialize()
    ref = FlatParams(src_model).copy_from_model(src_model)
    assert torch.equal(srv.strategy.params, ref.flat)
    # and the desired-batches selector is honored (ba60 is all-zeros)
    cfg2 = copy.deepcopy(cfg)
    cfg2["photon"]["restore_cent_run_batches"] = 60
    srv2 = FedServer(cfg2, Comm(0, 1), "cpu")
    srv2.initialize()
    assert float(srv2.strategy.params.abs().sum()) == 0.0


@pytest.mark.timeout(300)
def test_watchdog_multi_survivor_agreement(tiny_cfg, tmp_path):
    """3 ranks, rank 2 killed mid-fit: BOTH survivors must agree on the
    alive set via the decider protocol (first atomic claim seals it),
    rebuild to world 2, requeue the dead rank's client, and end the round
    with identical replicated params."""
    cfg = copy.deepcopy(tiny_cfg)
    cfg["fl"]["n_total_clients"] = 3
    cfg["fl"]["n_clients_per_round"] = 3
    cfg["photon"]["checkpoint"] = False
    cfg["photon"]["fit_timeout_s"] = 6
    cfg["fl"]["accept_failures_cnt"] = 0
    from tests.conftest import free_port

    port = free_port()
    flag = str(tmp_path / "rank2_fitting")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir, exist_ok=True)
    ctx = mp.get_context("spawn")

    def worker(rank):
        return ctx.Process(
            target=_watchdog_worker_n,
            args=(rank, 3, port, cfg, out_dir, flag, 2),
        )

    procs = [worker(r) for r in range(3)]
    for p in procs:
        p.start()
    import time as _time

    deadline = _time.time() + 60
    while not os.path.exists(flag) and _time.time() < deadline:
        _time.sleep(0.1)
    assert os.path.exists(flag), "rank 2 never reached its fit"
    procs[2].kill()
    for r in (0, 1):
        procs[r].join(timeout=180)
        assert procs[r].exitcode == 0, f"survivor rank {r} failed"
    procs[2].join(timeout=30)
    o0 = torch.load(os.path.join(out_dir, "wd_0.pt"))
    o1 = torch.load(os.path.join(out_dir, "wd_1.pt"))
    assert o0["world"] == 2 and o1["world"] == 2
    assert o0["failures_r1"] == 0.0, "requeue must recover the dead client"
    assert torch.equal(o0["params"], o1["params"]), "replicas must agree"


def _watchdog_worker_n(rank, world, port, cfg, out_dir, flag_path, hang_rank):
    import time as _time

    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    srv = FedServer(cfg, Comm(rank, world), "cpu")
    if rank == hang_rank:
        def hang(*a, **k):
            open(flag_path, "w").close()
            _time.sleep(600)

        srv.client.fit = hang
    srv.initialize()
    m1 = srv.run_round(1)
    m2 = srv.run_round(2)
    torch.save(
        {"params": srv.strategy.params, "world": srv.comm.world_size,
         "failures_r1": m1["server/failures"],
         "failures_r2": m2["server/failures"]},
        os.path.join(out_dir, f"wd_{rank}.pt"),
    )
    if dist.is_initialized():
        dist.destroy_process_group()


def test_time_offset_accumulates_across_resume(tiny_cfg, tmp_path):
    """state.bin time_offset carries cumulative wall time across resumes
    (reference s3_utils.py:374-389)."""
    import pickle

    cfg = copy.deepcopy(tiny_cfg)
    cfg["photon"]["checkpoint"] = True
    cfg["photon"]["saving_path"] = str(tmp_path)
    srv = FedServer(cfg, Comm(0, 1), "cpu")
    srv.run(2)
    with open(tmp_path / cfg["run_uuid"] / "server" / "2" / "state.bin",
              "rb") as f:
        t1 = pickle.load(f)["time_offset"]
    assert t1 > 0

    cfg2 = copy.deepcopy(cfg)
    cfg2["photon"]["resume_round"] = -1
    srv2 = FedServer(cfg2, Comm(0, 1), "cpu")
    srv2.initialize()
    assert srv2.time_offset == t1
    srv2.run_round(3)
    # force a checkpoint of round 3 via the run loop bookkeeping
    from photon_amd.fed.server_ckpt import upload_server_checkpoint

    upload_server_checkpoint(
        srv2.saving_path, srv2.run_uuid, 3, srv2.strategy, srv2.layout,
        srv2.history.state(), {}, srv2.server_steps_cumulative,
        time_offset=srv2.time_offset + 1.0,
    )
    with open(tmp_path / cfg["run_uuid"] / "server" / "3" / "state.bin",
              "rb") as f:
        t2 = pickle.load(f)["time_offset"]
    assert t2 > t1
