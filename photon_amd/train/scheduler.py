"""LR schedulers — cosine_with_warmup, Composer duration semantics ("100ba")."""

from __future__ import annotations

import math

from ..conf.schema import duration_to_batches


class CosineWithWarmup:
    """alpha(t): linear 0->1 over t_warmup, then cosine 1->alpha_f by t_max.

    Multiplies each param group's base lr; stepped once per optimization batch.
    """

    def __init__(self, optimizer, t_warmup, t_max, alpha_f: float = 0.1):
        self.optimizer = optimizer
        self.t_warmup = duration_to_batches(t_warmup)
        self.t_max = duration_to_batches(t_max)
        self.alpha_f = alpha_f
        self.base_lrs = [g["lr"] for g in optimizer.param_groups]
        self.last_batch = 0

    def alpha(self, t: int) -> float:
        if self.t_warmup > 0 and t < self.t_warmup:
            return t / self.t_warmup
        span = max(self.t_max - self.t_warmup, 1)
        frac = min(max((t - self.t_warmup) / span, 0.0), 1.0)
        return self.alpha_f + (1 - self.alpha_f) * 0.5 * (1 + math.cos(math.pi * frac))

    def step(self, batch: int | None = None) -> float:
        t = self.last_batch + 1 if batch is None else batch
        self.last_batch = t
        a = self.alpha(t)
        for g, base in zip(self.optimizer.param_groups, self.base_lrs):
            g["lr"] = base * a
        return a

    def state_dict(self) -> dict:
        return {"last_batch": self.last_batch, "base_lrs": self.base_lrs}

    def load_state_dict(self, state: dict) -> None:
        self.last_batch = int(state["last_batch"])
        self.base_lrs = list(state["base_lrs"])
        self.step(self.last_batch)


def build_scheduler(cfg: dict, optimizer):
    sch = cfg["scheduler"]["schedulers"]["lr"]
    name = str(sch.get("name", "cosine_with_warmup"))
    if name != "cosine_with_warmup":
        raise ValueError(f"unknown scheduler {name!r}")
    t_max = sch.get("t_max", cfg.get("max_duration", "1000ba"))
    return CosineWithWarmup(
        optimizer,
        sch.get("t_warmup", "100ba"),
        t_max,
        float(sch.get("alpha_f", 0.1)),
    )
