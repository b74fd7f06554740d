"""Federated training entry — the reference's flower-superlink +
server-app + client-app process trio (scripts/photon_llm_125M.sh:137-164)
collapsed to ONE torchrun command: every rank is a federated client owning
one MI355X, rank 0 doubles as server bookkeeper, aggregation is the RCCL
weighted all-reduce (photon_amd.fed.server).

Usage:
    # 8 federated clients on one node (the BASELINE config-2 topology)
    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        -m photon_amd.fed_train [hydra-style overrides...]

    # plumbing run on CPU, 1 rank
    python -m photon_amd.fed_train llm_config=mpt-125m fl.n_rounds=1

Reads $PHOTON_SAVE_PATH/config.yaml when present (the hydra_resolver dump
contract, photon/hydra_resolver.py:30-39), else composes from overrides.
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

import torch

from .conf import compose, config_yaml_dir, load_resolved, validate
from .fed.runtime import Comm, init_distributed
from .fed.server import FedServer


def main(argv: list[str] | None = None):
    overrides = list(sys.argv[1:] if argv is None else argv)
    save_path = Path(os.environ.get("PHOTON_SAVE_PATH", "."))
    cfg_file = save_path / "config.yaml"
    if cfg_file.exists() and not overrides:
        cfg = load_resolved(cfg_file)
    else:
        cfg = validate(compose(config_yaml_dir(), "base", overrides))

    from .fed.rccl_tuning import apply_rccl_env

    apply_rccl_env(cfg)
    rank, world = init_distributed()
    comm = Comm(rank, world)
    device = (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available()
        else torch.device("cpu")
    )
    server = FedServer(cfg, comm, device)
    server.run()
    if rank == 0:
        print(f"[fed] finished {server.n_rounds} rounds "
              f"(run_uuid={server.run_uuid})")
    return server


if __name__ == "__main__":
    main()
