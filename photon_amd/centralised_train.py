"""Centralized (non-federated) training — reference photon/centralised_train.py.

BASELINE config 1 is this path on CPU (attn_impl=torch, fp32, 2 steps).
Launched directly (1 GPU / CPU) or under torchrun for multi-GPU data
parallelism, where gradient sync is an explicit bucketed all-reduce hook
over RCCL (the reference relied on Composer's DDP).

Usage: python -m photon_amd.centralised_train [overrides...]
       (reads $PHOTON_SAVE_PATH/config.yaml if present, else composes)
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

import torch
import torch.distributed as dist

from .conf import compose, config_yaml_dir, load_resolved, validate
from .data import build_eval_loader, build_train_loader
from .fed.flat import FlatParams
from .fed.runtime import Comm, init_distributed
from .history import History
from .models import build_model
from .parallel import BucketedGradSync, apply_fsdp, apply_tensor_parallel
from .train import Trainer


def make_grad_sync_hook(comm: Comm):
    """Bucketed gradient all-reduce (mean) over RCCL after local backward
    (photon_amd.parallel.ddp — FORCED_SYNC semantics)."""
    if not comm.is_distributed:
        return None
    return BucketedGradSync(world_size=comm.world_size)


def run_split_eval(cfg, trainer) -> dict:
    """centralized.split_eval (reference centralised_train.py:74): evaluate
    every client stream separately and report per-stream CE alongside the
    concatenated-stream eval."""
    from .data import build_eval_loader

    streams = (cfg.get("dataset", {}).get("val", {}) or {}).get("streams") \
        or (cfg.get("dataset", {}).get("train", {}) or {}).get("streams") \
        or []
    out = {}
    saved = trainer.eval_loader
    subset = int(cfg["llm_config"].get("eval_subset_num_batches", -1))
    try:
        for cid in range(len(streams)):
            trainer.eval_loader = build_eval_loader(cfg, client_id=cid)
            m = trainer.eval(subset)
            ce = m.get("metrics/eval/LanguageCrossEntropy")
            if ce is not None:
                out[f"metrics/eval/LanguageCrossEntropy_stream_{cid}"] = ce
    finally:
        trainer.eval_loader = saved
    return out


def run_icl_eval(cfg, model, device) -> dict:
    """Config-gated ICL + gauntlet evaluation (icl_tasks_config /
    eval_gauntlet_config groups, both `empty` by default)."""
    icl_cfg = cfg.get("icl_tasks_config") or {}
    tasks = icl_cfg.get("icl_tasks")
    if not tasks:
        return {}
    from .data.convert import load_tokenizer
    from .eval import evaluate_icl_tasks, gauntlet_composite

    tok = load_tokenizer(icl_cfg.get("tokenizer_path"))
    results = evaluate_icl_tasks(
        model, tasks, tok, device=device,
        max_seq_len=int(cfg["llm_config"].get("max_seq_len", 2048)),
        limit_examples=icl_cfg.get("limit_examples"),
    )
    gauntlet = (cfg.get("eval_gauntlet_config") or {}).get("eval_gauntlet")
    results.update(gauntlet_composite(results, gauntlet))
    return results


def main(argv: list[str] | None = None):
    overrides = list(sys.argv[1:] if argv is None else argv)
    save_path = Path(os.environ.get("PHOTON_SAVE_PATH", "."))
    cfg_file = save_path / "config.yaml"
    if cfg_file.exists() and not overrides:
        cfg = load_resolved(cfg_file)
    else:
        cfg = validate(compose(config_yaml_dir(), "base", overrides))

    rank, world = init_distributed()
    comm = Comm(rank, world)
    device = (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available()
        else torch.device("cpu")
    )

    llm = cfg["llm_config"]
    torch.manual_seed(int(llm.get("seed", 17)))
    model = build_model(llm)

    # parallelism config (reference parallelism_config={"fsdp","tp"},
    # trainer_utils.py:1641-1648): TP shards linears/heads; FSDP wraps
    # blocks; both degenerate to no-ops at world 1 / falsy config.
    tp_cfg = llm.get("tp_config") or {}
    tp_degree = int(tp_cfg.get("degree", 1) or 1)
    if tp_degree > 1:
        apply_tensor_parallel(model, rank=rank % tp_degree, world=tp_degree)
    model = apply_fsdp(model, llm.get("fsdp_config"), device=None)

    cent = cfg.get("centralized", {}) or {}
    stream_id = cent.get("stream_id")
    run_uuid = str(cfg.get("run_uuid", "run"))
    trainer = Trainer(
        model,
        llm,
        train_loader=build_train_loader(cfg, client_id=stream_id),
        eval_loader=build_eval_loader(cfg, client_id=stream_id),
        device=device,
        world_size=world,
        grad_sync_hook=make_grad_sync_hook(comm),
        run_name=run_uuid,
        save_folder=str(save_path / run_uuid / "checkpoints"),
        rank=rank,
    )
    # make all ranks start from rank-0's init
    layout = FlatParams(model, filter_key=None, device=device)
    layout.copy_from_model(model)
    comm.broadcast_flat(layout.flat, src=0)
    layout.copy_to_model(model)
    if cent.get("store_init_model", False) and rank == 0:
        (save_path / run_uuid).mkdir(parents=True, exist_ok=True)
        layout.save_npz(save_path / run_uuid / "init_parameters.npz")

    # pretrained init: Composer .pt checkpoint, or .npz parameter dump
    # (reference centralised_train.py:98-117, incl. the WTE-only transplant).
    pretrained = cfg.get("pretrained_model_path")
    if pretrained:
        if str(pretrained).endswith(".npz"):
            pt_layout = FlatParams(model, filter_key=None, device=device)
            arrays = pt_layout.load_npz(pretrained)
            if cent.get("wte_only", False):
                params = dict(model.named_parameters())
                for n, a in zip(pt_layout.names, arrays):
                    if "wte" in n:
                        with torch.no_grad():
                            params[n].data.copy_(torch.from_numpy(a).to(params[n].dtype))
            else:
                pt_layout.from_ndarrays(arrays)
                pt_layout.copy_to_model(model)
        else:
            trainer.load_checkpoint(pretrained)

    history = History(run_dir=save_path / run_uuid if rank == 0 else None,
                      use_wandb=bool(cfg.get("use_wandb", False)))

    if cent.get("eval_only", False):
        metrics = trainer.eval()
        if cent.get("split_eval", False):
            metrics.update(run_split_eval(cfg, trainer))
        metrics.update(run_icl_eval(cfg, model, device))
        if rank == 0:
            history.add_metrics_centralized(0, metrics)
            print(metrics)
        return trainer

    if llm.get("eval_first", False):
        trainer.eval()

    duration = llm.get("max_duration", "100ba")
    metrics = trainer.fit(duration)
    eval_metrics = trainer.eval(int(llm.get("eval_subset_num_batches", -1)))
    if cent.get("split_eval", False):
        eval_metrics.update(run_split_eval(cfg, trainer))
    eval_metrics.update(run_icl_eval(cfg, model, device))
    if rank == 0:
        history.add_metrics_centralized(trainer.timestamp.batch, {**metrics, **eval_metrics})
        if cent.get("store_final_model", False):
            out_layout = FlatParams(model, filter_key=None, device=device)
            out_layout.copy_from_model(model)
            out_layout.save_npz(save_path / run_uuid / "final_parameters.npz")
        trainer.save_checkpoint()
        print(
            f"[centralised] done: batches={trainer.timestamp.batch} "
            f"loss={metrics.get('loss/train/total'):.4f} "
            f"eval_ce={eval_metrics.get('metrics/eval/LanguageCrossEntropy'):.4f}"
        )
    return trainer


if __name__ == "__main__":
    main()
