"""Wire-format configs — the reference's pydantic validators.

Re-implements photon/clients/configs.py: typed ``FitConfig`` /
``EvaluateConfig`` / ``CentralizedConfig`` models (:55-214, :289-425,
:488-573) and the flat ConfigsRecord (de)serialization the reference sends
over gRPC: lists/dicts become ``str()`` on the wire and are parsed back
with ``ast.literal_eval`` on receive (:140-214).

In the MI355X runtime there is no gRPC — per-round instructions are
replicated deterministic state — but the typed validation layer survives:
it is what checks an operator-supplied config before a round starts, and
round-trips through flat records for logging/checkpoint metadata.
"""

from __future__ import annotations

import ast
from typing import Any

from pydantic import BaseModel, Field, field_validator


def _parse_seq(v):
    if isinstance(v, str):
        parsed = ast.literal_eval(v)
        return parsed
    return v


class _RecordModel(BaseModel):
    """Base with the flat-record round-trip."""

    def to_record(self) -> dict[str, Any]:
        """Flat dict with lists/dicts stringified (ConfigsRecord contract)."""
        out = {}
        for k, v in self.model_dump().items():
            if isinstance(v, (list, dict, tuple)):
                out[k] = str(v)
            else:
                out[k] = v
        return out

    @classmethod
    def from_record(cls, record: dict[str, Any]):
        return cls(**record)


class FitConfig(_RecordModel):
    """Per-round fit instruction (reference FitConfig :55-214)."""

    server_round: int = 1
    client_ids: list[int] = Field(default_factory=list)
    local_steps: int = 500
    server_steps_cumulative: int = 0
    reset_optimizer: bool = True
    reset_dataset_state: bool = False
    aggregate_momenta: bool = False
    set_trainer_params_filter_keys: bool = True
    set_trainer_key_to_filter: str = "transformer"
    frozen_layers: list[str] = Field(default_factory=list)
    unfrozen_layers: list[str] = Field(default_factory=list)
    personalized_layers: list[str] = Field(default_factory=list)
    random_layers: list[str] = Field(default_factory=list)
    client_state: dict = Field(default_factory=dict)

    @field_validator(
        "client_ids", "frozen_layers", "unfrozen_layers",
        "personalized_layers", "random_layers", "client_state",
        mode="before",
    )
    @classmethod
    def _lists_from_str(cls, v):
        return _parse_seq(v)

    @field_validator("local_steps", mode="before")
    @classmethod
    def _steps(cls, v):
        if isinstance(v, str) and v.endswith("ba"):
            return int(v[:-2])
        return v


class EvaluateConfig(_RecordModel):
    """Per-round evaluate instruction (reference EvaluateConfig :289-425)."""

    server_round: int = 1
    client_ids: list[int] = Field(default_factory=list)
    eval_subset_num_batches: int = -1
    split_eval: bool = False
    client_state: dict = Field(default_factory=dict)

    @field_validator("client_ids", "client_state", mode="before")
    @classmethod
    def _lists_from_str(cls, v):
        return _parse_seq(v)


class CentralizedConfig(_RecordModel):
    """Centralized-run config (reference CentralizedConfig :488-573)."""

    run_uuid: str = "run"
    stream_id: int | None = None
    eval_first: bool = False
    use_pretrained: bool = False
    pretrained_path: str | None = None
    wte_only: bool = False
    saving_path: str = "checkpoints"


def get_fit_config(cfg: dict, server_round: int, client_ids: list[int],
                   server_steps_cumulative: int = 0) -> FitConfig:
    """Build the round instruction from the resolved photon config
    (reference get_photon_fit_config_fn :217-286)."""
    fl = cfg.get("fl", {})
    llm = cfg.get("llm_config", {})
    steps = llm.get("local_steps", 500)
    if isinstance(steps, str) and steps.endswith("ba"):
        steps = int(steps[:-2])
    return FitConfig(
        server_round=server_round,
        client_ids=list(client_ids),
        local_steps=int(steps),
        server_steps_cumulative=server_steps_cumulative,
        reset_optimizer=bool(fl.get("reset_optimizer", True)),
        reset_dataset_state=bool(fl.get("reset_dataset_state", False)),
        aggregate_momenta=bool(fl.get("aggregate_momenta", False)),
        set_trainer_params_filter_keys=bool(
            fl.get("set_trainer_params_filter_keys", True)
        ),
        set_trainer_key_to_filter=str(
            fl.get("set_trainer_key_to_filter", "transformer")
        ),
        frozen_layers=list(fl.get("frozen_layers") or []),
        unfrozen_layers=list(fl.get("unfrozen_layers") or []),
        personalized_layers=list(fl.get("personalized_layers") or []),
        random_layers=list(fl.get("random_layers") or []),
    )


def get_evaluate_config(cfg: dict, server_round: int,
                        client_ids: list[int]) -> EvaluateConfig:
    llm = cfg.get("llm_config", {})
    return EvaluateConfig(
        server_round=server_round,
        client_ids=list(client_ids),
        eval_subset_num_batches=int(llm.get("eval_subset_num_batches", -1)),
        split_eval=bool(cfg.get("fl", {}).get("split_eval", False)),
    )
