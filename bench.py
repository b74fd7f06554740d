"""Flagship benchmark: federated MPT-125M pre-training throughput on MI355X.

Measures the BASELINE.json headline metric — tokens/sec over the whole node
with N federated clients (one per GPU) training MPT-125M on synthetic
C4-shaped tokens (seq 2048, vocab 50368, bf16) — including the per-round
RCCL aggregation + server-opt update inside the timed region.

One bench "step" = one optimization batch per client (global_train_batch_size
samples via gradient accumulation, the reference's per-client batch semantics:
every client trains with the full global batch — SURVEY §2.2 LocalSGD row).
Every `--local-steps` steps a federated round boundary runs: weighted
all-reduce of the flat parameter buffer + FedNesterov server update.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--local-steps", type=int, default=4,
                    help="steps between federated round boundaries")
    ap.add_argument("--model", default="mpt-125m")
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--global-batch", type=int, default=256)
    ap.add_argument("--microbatch", type=int, default=32)
    ap.add_argument("--attn", default="flash", choices=["flash", "torch"])
    ap.add_argument("--hip-graphs", action=argparse.BooleanOptionalAction,
                    default=False,
                    help="capture the microbatch fwd+bwd in a hipGraph "
                         "(falls back to eager on capture failure; "
                         "box-dependent +-1-2%% at 125M, -2%% at 1B -> "
                         "off by default)")
    ap.add_argument("--master-weights", action=argparse.BooleanOptionalAction,
                    default=True,
                    help="bf16 model weights + fp32 optimizer masters "
                         "(PURE mixed precision; measured +1%% vs autocast "
                         "weight-cast caching)")
    args = ap.parse_args()

    from photon_amd.conf import compose, config_yaml_dir
    from photon_amd.data.shards import StatefulLoader
    from photon_amd.data.synthetic import SyntheticTokenDataset
    from photon_amd.fed.flat import FlatParams
    from photon_amd.fed.runtime import Comm, init_distributed
    from photon_amd.fed.strategies import dispatch_strategy
    from photon_amd.models import build_model
    from photon_amd.train import Trainer

    from photon_amd.fed.rccl_tuning import apply_rccl_env

    rccl_env = apply_rccl_env(cfg=None)

    # ---- rendezvous / environment preflight (fail fast, loudly) ----------
    import sys

    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    env_world = int(os.environ.get("WORLD_SIZE", 1))
    problems = []
    if torch.cuda.is_available():
        n_dev = torch.cuda.device_count()
        if local_rank >= n_dev:
            problems.append(f"LOCAL_RANK {local_rank} >= visible GPUs {n_dev}")
        if env_world > 1 and os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY") != "0":
            # dmabuf IPC is required on this host driver; legacy IPC fails
            # with hipIpcGetMemHandle: invalid argument
            os.environ["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
            print("[preflight] forced HSA_ENABLE_IPC_MODE_LEGACY=0",
                  file=sys.stderr, flush=True)
    if env_world > 1:
        addr = os.environ.get("MASTER_ADDR", "")
        if addr not in ("127.0.0.1", "localhost") and env_world > 1:
            print(f"[preflight] MASTER_ADDR={addr!r} (expect 127.0.0.1 "
                  "single-node)", file=sys.stderr, flush=True)
    if problems:
        raise SystemExit("[preflight] " + "; ".join(problems))

    rank, world = init_distributed()
    if world == 1 and args.gpus > 1:
        raise SystemExit("multi-GPU bench must be launched via torchrun")
    comm = Comm(rank, world)
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", torch.cuda.current_device()) if use_cuda else torch.device("cpu")
    if rank == 0 and world > 1:
        print(f"[preflight] world={world} backend="
              f"{dist.get_backend() if dist.is_initialized() else 'none'} "
              f"rccl_env={rccl_env}", file=sys.stderr, flush=True)

    cfg = compose(config_yaml_dir(), "base", [f"llm_config={args.model}"])
    llm = cfg["llm_config"].to_plain()
    llm["max_seq_len"] = args.seq_len
    llm["model"]["max_seq_len"] = args.seq_len
    llm["model"]["attn_config"]["attn_impl"] = args.attn
    llm["global_train_batch_size"] = args.global_batch
    llm["device_train_microbatch_size"] = args.microbatch
    llm["precision"] = "amp_bf16" if use_cuda else "fp32"
    if args.hip_graphs:
        llm["use_hip_graphs"] = True
    if args.master_weights:
        llm["master_weights"] = True

    torch.manual_seed(17)
    model = build_model(llm)
    ds = SyntheticTokenDataset(
        args.seq_len, vocab_size=int(llm["model"]["vocab_size"]),
        seed=1337, client_id=rank,
    )
    trainer = Trainer(
        model, llm,
        train_loader=StatefulLoader(ds, args.microbatch),
        device=device,
        world_size=1,  # each client trains its own full global batch
    )

    layout = FlatParams(model, device=device)
    layout.copy_from_model(model)
    comm.broadcast_flat(layout.flat, src=0)
    layout.copy_to_model(model)
    strategy = dispatch_strategy(
        "NESTOROV", layout, {"server_learning_rate": 0.7, "server_momentum": 0.7}
    )
    strategy.initialize(layout.flat)

    tokens_per_step = args.global_batch * args.seq_len  # per client

    def one_step(step_idx: int) -> None:
        mbs = [trainer.train_loader.next_batch() for _ in range(trainer.grad_accum)]
        trainer.train_batch(mbs)
        if (step_idx + 1) % args.local_steps == 0:
            # federated round boundary: weighted all-reduce + server opt
            local = layout.copy_from_model(model).flat.clone()
            n_i = float(tokens_per_step * args.local_steps)
            local.mul_(n_i)
            avg, total = comm.weighted_average_(local, n_i)
            strategy.update(avg, server_round=1 + step_idx // args.local_steps,
                            n_clients=world)
            views = layout.layer_views_of(strategy.params)
            params = dict(model.named_parameters())
            with torch.no_grad():
                for n, vw in zip(layout.names, views):
                    params[n].data.copy_(vw.to(params[n].dtype))

    def sync() -> None:
        comm.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        one_step(i)
    sync()
    t0 = time.time()
    for i in range(args.warmup, args.warmup + args.steps):
        one_step(i)
    sync()
    elapsed = time.time() - t0
    # MAX over ranks (all ranks hit the same barriers; elapsed is rank-local)
    per_rank = comm.all_gather_scalars(elapsed)
    elapsed = max(per_rank)

    if rank == 0:
        if world > 1:
            # per-rank spread: DVFS imbalance / straggler diagnosis for the
            # scaling run (NOTES_NEXT_ROUND §Scaling)
            ms = [e / args.steps * 1000.0 for e in per_rank]
            print(json.dumps({
                "per_rank_ms_per_step": [round(x, 2) for x in ms],
                "rank_spread_pct": round(
                    (max(ms) - min(ms)) / max(ms) * 100.0, 2
                ),
            }), file=sys.stderr, flush=True)
        total_tokens = tokens_per_step * args.steps * world
        result = {
            "metric": "tokens_per_sec_whole_node",
            "value": total_tokens / elapsed,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic (C4-shaped Zipf tokens, random-init weights)",
            "config": {
                "model": args.model,
                "global_batch": args.global_batch,
                "seq_len": args.seq_len,
                "parallelism": f"fed_dp{world}",
                "local_steps": args.local_steps,
                "attn_impl": args.attn,
                "strategy": "NESTOROV",
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
