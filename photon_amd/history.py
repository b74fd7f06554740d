"""Run history — the reference's WandbHistory (photon/wandb_history.py:12-70)
without the hard wandb dependency.

Every metric is recorded keyed by round, mirrored to a JSONL file under the
run dir, and forwarded to wandb when it is importable and ``use_wandb`` is
set (it is absent in this image, so the JSONL sink is the primary record).
"""

from __future__ import annotations

import json
import time
from pathlib import Path


class History:
    def __init__(self, run_dir: str | Path | None = None, use_wandb: bool = False,
                 wandb_setup: dict | None = None, suffix: str = ""):
        self.losses_distributed: list[tuple[int, float]] = []
        self.metrics_distributed: dict[str, list[tuple[int, float]]] = {}
        self.metrics_centralized: dict[str, list[tuple[int, float]]] = {}
        self._jsonl = None
        if run_dir is not None:
            p = Path(run_dir)
            p.mkdir(parents=True, exist_ok=True)
            self._jsonl = open(p / f"history{suffix}.jsonl", "a")
        self._wandb = None
        if use_wandb:
            try:
                import wandb  # noqa: F401

                self._wandb = wandb.init(**(wandb_setup or {}))
            except ImportError:
                self._wandb = None

    def _emit(self, record: dict) -> None:
        if self._jsonl is not None:
            self._jsonl.write(json.dumps(record) + "\n")
            self._jsonl.flush()
        if self._wandb is not None:
            rnd = record.pop("round", None)
            self._wandb.log(record, step=rnd)

    def add_loss_distributed(self, server_round: int, loss: float) -> None:
        self.losses_distributed.append((server_round, loss))
        self._emit({"round": server_round, "loss_distributed": loss, "t": time.time()})

    def add_metrics_distributed(self, server_round: int, metrics: dict) -> None:
        flat = _flatten(metrics)
        for k, v in flat.items():
            self.metrics_distributed.setdefault(k, []).append((server_round, v))
        self._emit({"round": server_round, **flat, "t": time.time()})

    def add_metrics_centralized(self, server_round: int, metrics: dict) -> None:
        flat = _flatten(metrics)
        for k, v in flat.items():
            self.metrics_centralized.setdefault(k, []).append((server_round, v))
        self._emit({"round": server_round, "centralized": True, **flat, "t": time.time()})

    def state(self) -> dict:
        return {
            "losses_distributed": self.losses_distributed,
            "metrics_distributed": self.metrics_distributed,
            "metrics_centralized": self.metrics_centralized,
        }

    def load_state(self, state: dict) -> None:
        self.losses_distributed = [tuple(x) for x in state.get("losses_distributed", [])]
        self.metrics_distributed = {
            k: [tuple(x) for x in v] for k, v in state.get("metrics_distributed", {}).items()
        }
        self.metrics_centralized = {
            k: [tuple(x) for x in v] for k, v in state.get("metrics_centralized", {}).items()
        }


def _flatten(metrics: dict, prefix: str = "") -> dict:
    out = {}
    for k, v in metrics.items():
        key = f"{prefix}{k}"
        if isinstance(v, dict):
            out.update(_flatten(v, key + "/"))
        elif isinstance(v, (int, float)):
            out[key] = float(v)
        elif isinstance(v, (list, tuple)) and all(isinstance(x, (int, float)) for x in v):
            out[key] = [float(x) for x in v]
    return out
