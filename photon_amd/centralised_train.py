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
from .train import Trainer


def make_grad_sync_hook(comm: Comm):
    """Bucketed gradient all-reduce (mean) over RCCL after local backward."""
    if not comm.is_distributed:
        return None

    def hook(model: torch.nn.Module) -> None:
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        if not grads:
            return
        # one flat all-reduce per dtype bucket; ~bucket the lot (model sizes
        # here are small enough that a single flat is the fastest on xGMI)
        flat = torch.cat([g.reshape(-1) for g in grads])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
        flat.div_(comm.world_size)
        off = 0
        for g in grads:
            g.copy_(flat[off : off + g.numel()].view_as(g))
            off += g.numel()

    return hook


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

    pretrained = cfg.get("pretrained_model_path")
    if pretrained:
        trainer.load_checkpoint(pretrained)

    history = History(run_dir=save_path / run_uuid if rank == 0 else None,
                      use_wandb=bool(cfg.get("use_wandb", False)))

    if cent.get("eval_only", False):
        metrics = trainer.eval()
        if rank == 0:
            history.add_metrics_centralized(0, metrics)
            print(metrics)
        return trainer

    if llm.get("eval_first", False):
        trainer.eval()

    duration = llm.get("max_duration", "100ba")
    metrics = trainer.fit(duration)
    eval_metrics = trainer.eval(int(llm.get("eval_subset_num_batches", -1)))
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
