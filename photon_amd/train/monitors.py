"""Training monitors — the reference's default callback set.

Re-implements the Composer callbacks photon enables by default
(mpt-125m.yaml:98-109): speed_monitor (throughput + MFU, window 20),
lr_monitor, memory_monitor, runtime_estimator, optimizer_monitor.
Metric names match the reference's so dashboards carry over.

MFU prices against the MI355X dense bf16 peak (2.5 PFLOP/s — never the
2:1-sparsity marketing figure, MI355X_MICROARCH.md) with the standard
decoder flops model: 6*N params + causal attention term per token.
"""

from __future__ import annotations

import time
from collections import deque

import torch

MI355X_BF16_DENSE_PEAK = 2.5e15  # FLOP/s, dense (no sparsity)


def flops_per_token(n_params: int, n_layers: int, d_model: int, seq_len: int) -> float:
    """fwd+bwd FLOPs per token: 6N dense + 6*L*S*d causal attention."""
    return 6.0 * n_params + 6.0 * n_layers * seq_len * d_model


class SpeedMonitor:
    """throughput/samples_per_sec, throughput/tokens_per_sec, throughput/mfu
    over a sliding window (reference window_size=20)."""

    def __init__(self, window_size: int = 20, peak_flops: float | None = None):
        self.window: deque[tuple[float, int, int]] = deque(maxlen=window_size)
        self.peak = peak_flops or MI355X_BF16_DENSE_PEAK

    def batch_end(self, trainer, loss) -> dict:
        now = time.time()
        ts = trainer.timestamp
        self.window.append((now, ts.sample, ts.token))
        out = {}
        if len(self.window) >= 2:
            t0, s0, tok0 = self.window[0]
            t1, s1, tok1 = self.window[-1]
            dt = max(t1 - t0, 1e-9)
            tokens_per_sec = (tok1 - tok0) / dt
            out["throughput/samples_per_sec"] = (s1 - s0) / dt
            out["throughput/tokens_per_sec"] = tokens_per_sec
            m = trainer.model
            cfg = getattr(m, "cfg", None)
            if cfg is not None and trainer.device.type == "cuda":
                fpt = flops_per_token(
                    sum(p.numel() for p in m.parameters()),
                    cfg.n_layers, cfg.d_model, cfg.max_seq_len,
                )
                out["throughput/mfu"] = tokens_per_sec * fpt / self.peak
        return out


class LRMonitor:
    def batch_end(self, trainer, loss) -> dict:
        return {
            f"lr-{type(trainer.optimizer).__name__}/group{i}": g["lr"]
            for i, g in enumerate(trainer.optimizer.param_groups)
        }


class MemoryMonitor:
    def batch_end(self, trainer, loss) -> dict:
        if trainer.device.type != "cuda":
            return {}
        return {
            "memory/allocated_mem": torch.cuda.memory_allocated() / 2**30,
            "memory/reserved_mem": torch.cuda.memory_reserved() / 2**30,
            "memory/peak_allocated_mem": torch.cuda.max_memory_allocated() / 2**30,
        }


class RuntimeEstimator:
    """time/remaining estimate from the observed batch rate."""

    def __init__(self):
        self.t0 = None
        self.b0 = None

    def batch_end(self, trainer, loss) -> dict:
        now = time.time()
        b = trainer.timestamp.batch
        if self.t0 is None:
            self.t0, self.b0 = now, b
            return {}
        done = b - self.b0
        if done <= 0:
            return {}
        rate = (now - self.t0) / done
        remaining = max(trainer.max_duration - b, 0)
        return {"time/remaining_estimate": remaining * rate}


class OptimizerMonitor:
    """L2 norm of the update moments (reference optimizer_monitor)."""

    def __init__(self, interval: int = 10):
        self.interval = interval

    def batch_end(self, trainer, loss) -> dict:
        if trainer.timestamp.batch % self.interval != 0:
            return {}
        sq_m, sq_g = 0.0, 0.0
        for group in trainer.optimizer.param_groups:
            for p in group["params"]:
                st = trainer.optimizer.state.get(p)
                if st and "exp_avg" in st:
                    sq_m += float(st["exp_avg"].float().pow(2).sum())
                if p.grad is not None:
                    sq_g += float(p.grad.float().pow(2).sum())
        return {
            "optimizer/l2_norm_first_moment": sq_m**0.5,
            "optimizer/l2_norm_grad": sq_g**0.5,
        }


_MONITORS = {
    "speed_monitor": SpeedMonitor,
    "lr_monitor": LRMonitor,
    "memory_monitor": MemoryMonitor,
    "runtime_estimator": RuntimeEstimator,
    "optimizer_monitor": OptimizerMonitor,
}


def build_monitors(callbacks_cfg) -> list:
    """Build from the llm_config.callbacks subtree (names as keys, kwargs as
    values — Hydra surface parity); unknown names are ignored with a note."""
    out = []
    for name, kwargs in (callbacks_cfg or {}).items():
        cls = _MONITORS.get(str(name))
        if cls is None:
            continue
        kw = dict(kwargs) if isinstance(kwargs, dict) else {}
        kw.pop("window_size", None) if False else None
        try:
            out.append(cls(**kw))
        except TypeError:
            out.append(cls())
    return out
