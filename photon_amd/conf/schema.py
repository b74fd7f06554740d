"""Typed config schema — parity with reference photon/conf/base_schema.py.

The reference registers dataclasses with Hydra's ConfigStore
(base_schema.py:345-398); here the same structure is expressed as
dataclasses used for validation of the composed tree. The ``llm_config``
subtree stays deliberately schema-free (base_schema.py:336-341) and is
interpreted by the trainer layer.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

from .engine import ConfigError, DictConfig


@dataclass
class CommStack:
    # reference: photon/conf/base_schema.py:12-28 (s3/shm/ray); the rebuild adds
    # the native RCCL data plane and retires shm/ray (kept as flags for surface
    # compatibility — they must be false).
    rccl: bool = True
    s3: bool = False
    shm: bool = False
    ray: bool = False


@dataclass
class Photon:
    # reference: photon/conf/base_schema.py:61-97
    n_nodes: int = 1
    refresh_period: int = 50
    checkpoint: bool = False
    restore_run_uuid: str | None = None
    restore_cent_run_uuid: str | None = None
    restore_cent_run_batches: int | None = None
    copy_client_checkpoints: bool = True
    resume_round: int = -1
    saving_path: str | None = None
    comm_stack: CommStack = field(default_factory=CommStack)


@dataclass
class FL:
    # reference: photon/conf/base_schema.py:145-239
    n_total_clients: int = 8
    n_clients_per_round: int = 8
    n_rounds: int = 200
    reset_checkpoint: bool = False
    reset_optimizer: bool = True
    reset_dataset_state: bool = False
    reset_timestamp: bool = False
    resize_vocab: int | None = None
    use_unigram_metrics: bool = False
    allow_unigram_metrics_failures: bool = False
    n_local_epochs: int = 1
    n_local_steps: int = 0
    random_layers: list = field(default_factory=list)
    random_init_freq: int = 0
    truly_random_init: bool = True
    personalized_layers: list = field(default_factory=list)
    frozen_layers: list | None = None
    unfrozen_layers: list | None = None
    ignore_failed_rounds: bool = False
    accept_failures_cnt: int = 0
    eval_period: int = 1
    split_eval: bool = False
    strategy_name: str = "NESTOROV"
    strategy_kwargs: dict = field(default_factory=dict)
    set_trainer_params_filter_keys: bool = True
    set_trainer_key_to_filter: str = "transformer"
    aggregate_momenta: bool = False
    use_noise_scale_metric: bool = False
    noise_scale_beta: float = 0.99


@dataclass
class Centralized:
    store_init_model: bool = False
    store_final_model: bool = False
    stream_id: int | None = None
    eval_only: bool = False
    split_eval: bool = False
    reset_timestamp: bool = False


KNOWN_STRATEGIES = ("FEDAVG", "NESTOROV", "MOM", "FEDADAM", "FEDYOGI")


def validate(cfg: DictConfig) -> DictConfig:
    """Validate the composed tree (cheap structural checks, like the pydantic
    wire-validators in photon/clients/configs.py)."""
    for key in ("run_uuid", "seed", "photon", "fl", "llm_config"):
        if key not in cfg:
            raise ConfigError(f"config missing top-level key {key!r}")
    fl = cfg.fl
    if fl.n_clients_per_round > fl.n_total_clients:
        raise ConfigError("fl.n_clients_per_round > fl.n_total_clients")
    if str(fl.strategy_name).upper() not in KNOWN_STRATEGIES:
        raise ConfigError(
            f"unknown fl.strategy_name {fl.strategy_name!r}; known: {KNOWN_STRATEGIES}"
        )
    cs = cfg.photon.comm_stack
    if cs.get("shm") or cs.get("ray"):
        raise ConfigError(
            "comm_stack.shm/ray are reference-only transports; the MI355X rebuild "
            "uses the RCCL data plane (photon.comm_stack.rccl)"
        )
    model = cfg.llm_config.model
    if model.d_model % model.n_heads != 0:
        raise ConfigError("d_model must be divisible by n_heads")
    return cfg


def duration_to_batches(text: Any) -> int:
    """Parse Composer-style durations: '500ba' -> 500. Accepts ints."""
    if isinstance(text, int):
        return text
    s = str(text).strip()
    if s.endswith("ba"):
        return int(s[:-2])
    if s.endswith("ep"):
        raise ConfigError("epoch durations are not supported; use 'Nba'")
    return int(s)
