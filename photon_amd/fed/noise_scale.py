"""Gradient-noise-scale estimator — reference FedSimpleNoiseScale
(photon/strategy/metrics.py:123-267).

Simple noise scale (McCandlish et al.) from federated pseudo-gradients:
with B_small = per-client batch and B_big = round total,

    |G_big|^2 est  = (B_big*|g_big|^2 - B_small*mean|g_i|^2) / (B_big - B_small)
    trace(S) est   = (mean|g_i|^2 - |g_big|^2) / (1/B_small - 1/B_big)
    noise_scale    = trace / |G|^2,  EMA-smoothed with de-bias
    (metrics.py:22-50 EMA; estimators at metrics.py:123-267)
"""

from __future__ import annotations

import torch


class _EMA:
    def __init__(self, beta: float):
        self.beta = beta
        self.value = 0.0
        self.count = 0

    def update(self, x: float) -> float:
        self.count += 1
        self.value = self.beta * self.value + (1 - self.beta) * x
        return self.value / (1 - self.beta**self.count)  # de-biased


class FedSimpleNoiseScale:
    def __init__(self, beta: float = 0.99):
        self.ema_trace = _EMA(beta)
        self.ema_gsq = _EMA(beta)

    def update(self, per_client: list[tuple[float, float]], fedavg_minus=None,
               comm=None) -> dict[str, float]:
        """per_client: local (n_i, |g_i|^2) pairs; reduced across ranks via comm."""
        if comm is not None:
            local_n = sum(n for n, _ in per_client)
            local_sq = sum(n * s for n, s in per_client)
            local_cnt = float(len(per_client))
            n_tot = sum(comm.all_gather_scalars(local_n))
            sq_tot = sum(comm.all_gather_scalars(local_sq))
            cnt = sum(comm.all_gather_scalars(local_cnt))
        else:
            n_tot = sum(n for n, _ in per_client)
            sq_tot = sum(n * s for n, s in per_client)
            cnt = float(len(per_client))
        if cnt == 0 or n_tot == 0:
            return {}
        mean_gi_sq = sq_tot / n_tot
        b_small = n_tot / cnt
        b_big = n_tot
        if fedavg_minus is None:
            return {}
        # |g_big|^2: norm of the aggregated pseudo-gradient is computed by the
        # caller's strategy; here fedavg_minus is the aggregated *average*
        # params; use the strategy's pseudo-grad norm if provided as tensor
        g_big_sq = float(torch.dot(fedavg_minus.flatten(), fedavg_minus.flatten())) if isinstance(fedavg_minus, torch.Tensor) else float(fedavg_minus)
        denom = b_big - b_small
        if denom <= 0:
            return {}
        g2_est = (b_big * g_big_sq - b_small * mean_gi_sq) / denom
        trace_est = (mean_gi_sq - g_big_sq) / (1.0 / b_small - 1.0 / b_big)
        g2_ema = self.ema_gsq.update(g2_est)
        tr_ema = self.ema_trace.update(trace_est)
        scale = tr_ema / g2_ema if g2_ema != 0 else float("nan")
        return {
            "noise_scale/simple": scale,
            "noise_scale/trace_estimate": tr_ema,
            "noise_scale/gsq_estimate": g2_ema,
        }
