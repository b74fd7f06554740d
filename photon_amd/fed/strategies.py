"""Server optimization strategies (FedOpt family) on HBM-resident flat buffers.

Behavioral parity with the reference update rules (verified against NumPy
oracles in tests/test_strategies.py):

* FEDAVG   (fedavg_eff.py:318-324):  x <- x - slr*(x - avg)
* NESTOROV (fednestorov.py:323-331): m <- mu*m + g; g' = g + mu*m; x <- x - slr*g'
* MOM      (fedmom.py:263-278):      v_new = x - slr*g; x <- (1+mu)*v_new - mu*v_old
* FEDADAM  (fedadam.py:295-314):     m <- b1*m + (1-b1)*g; v <- b2*v + (1-b2)*g^2;
                                     x <- x + eta * m_hat / (sqrt(v_hat) + tau)
                                     (sic: the reference ADDS along the
                                     x-minus-avg pseudo-gradient — verified
                                     against fedadam.py:291-316, and FedYogi
                                     does the same at fedyogi.py:297-316;
                                     parity preserves the exact rules —
                                     NESTOROV, the default, subtracts)
* FEDYOGI  (fedyogi.py:299-320):     v += (1-b2)*g^2*sign(g^2 - v); x as FedAdam

where g = x - sf*avg, sf = scaling_fn(n_clients) in {1, linear, sqrt}
(fedavg_eff.py:190-198,296) and avg is the sample-weighted client average.
Bias correction uses the server round (1-indexed), as the reference does.

Everything operates on flat fp32 torch tensors (CPU or HBM): the server-opt
update runs as one fused elementwise pass on the all-reduced buffer — on
GPU, a single HIP-friendly torch op chain over 1 flat tensor (bandwidth
bound), replacing the reference's per-layer NumPy loop.

Metrics parity: per-layer and global L2 norms of pseudo-gradient, momentum,
fedavg result and model (fedavg_eff.py:308-364 and equivalents).
"""

from __future__ import annotations

import math
from typing import Callable

import torch

from .flat import FlatParams

STATE_PARAMS = "current_server_parameters"
STATE_M1 = "current_momentum_vector"
STATE_M2 = "current_second_momentum_vector"


def _scaling_fn(name: str | None) -> Callable[[int], float]:
    if name is None:
        return lambda n: 1.0
    if name == "linear":
        return lambda n: float(n)
    if name == "sqrt":
        return lambda n: math.sqrt(n)
    raise ValueError("Scaling function must be either 'linear' or 'sqrt'.")


def _l2(t: torch.Tensor) -> float:
    return float(torch.linalg.vector_norm(t.float()))


class Strategy:
    """Base: holds the global flat params and optional momenta."""

    name = "BASE"
    state_keys: tuple[str, ...] = (STATE_PARAMS,)

    def __init__(self, layout: FlatParams, **kwargs):
        self.layout = layout
        self.params = layout.clone_flat()  # global model (flat fp32)
        self.m1: torch.Tensor | None = None
        self.m2: torch.Tensor | None = None
        self.scaling = _scaling_fn(kwargs.pop("scaling_fn", None))
        self.kwargs = kwargs

    # momenta initialization (strategy/utils.py:13-54 semantics)
    def initialize(self, params_flat: torch.Tensor, m1=None, m2=None) -> None:
        self.params.copy_(params_flat)
        if STATE_M1 in self.state_keys:
            self.m1 = m1.clone() if m1 is not None else torch.zeros_like(self.params)
        if STATE_M2 in self.state_keys:
            self.m2 = m2.clone() if m2 is not None else torch.zeros_like(self.params)

    def update(self, fedavg_flat: torch.Tensor, server_round: int,
               n_clients: int) -> dict[str, float]:
        """Apply the server-opt update in place on self.params; return metrics."""
        raise NotImplementedError

    # -- shared metric collection ------------------------------------------
    def _metrics(self, g: torch.Tensor, extra: dict | None = None) -> dict:
        out = {
            "l2_norm_pseudo_gradient": _l2(g),
            "l2_norm_model": _l2(self.params),
        }
        views_g = self.layout.layer_views_of(g)
        out["layerwise_l2_norms_pseudo_gradient"] = [_l2(v) for v in views_g]
        if self.m1 is not None:
            out["l2_norm_momentum_vector"] = _l2(self.m1)
        if self.m2 is not None:
            out["l2_norm_second_momentum_vector"] = _l2(self.m2)
        if extra:
            out.update(extra)
        return out

    # -- state for server checkpoints --------------------------------------
    def state_tensors(self) -> dict[str, torch.Tensor]:
        out = {STATE_PARAMS: self.params}
        if self.m1 is not None:
            out[STATE_M1] = self.m1
        if self.m2 is not None:
            out[STATE_M2] = self.m2
        return out

    def load_state_tensors(self, state: dict[str, torch.Tensor]) -> None:
        self.params.copy_(state[STATE_PARAMS])
        if STATE_M1 in self.state_keys and STATE_M1 in state:
            self.m1 = state[STATE_M1].to(self.params.device).clone()
        if STATE_M2 in self.state_keys and STATE_M2 in state:
            self.m2 = state[STATE_M2].to(self.params.device).clone()


class FedAvgEfficient(Strategy):
    name = "FEDAVG"
    state_keys = (STATE_PARAMS,)

    def __init__(self, layout, server_learning_rate: float = 1.0, **kw):
        super().__init__(layout, **kw)
        self.slr = float(server_learning_rate)

    @torch.no_grad()
    def update(self, fedavg_flat, server_round, n_clients):
        sf = self.scaling(n_clients)
        g = self.params - sf * fedavg_flat
        self.params.sub_(g, alpha=self.slr)
        return self._metrics(g)


class FedNesterov(Strategy):
    name = "NESTOROV"
    state_keys = (STATE_PARAMS, STATE_M1)

    def __init__(self, layout, server_learning_rate: float = 0.7,
                 server_momentum: float = 0.7, **kw):
        super().__init__(layout, **kw)
        self.slr = float(server_learning_rate)
        self.mu = float(server_momentum)

    @torch.no_grad()
    def update(self, fedavg_flat, server_round, n_clients):
        sf = self.scaling(n_clients)
        g = self.params - sf * fedavg_flat
        self.m1.mul_(self.mu).add_(g)
        g = g + self.mu * self.m1
        self.params.sub_(g, alpha=self.slr)
        return self._metrics(g)


class FedMom(Strategy):
    name = "MOM"
    state_keys = (STATE_PARAMS, STATE_M1)

    def __init__(self, layout, server_learning_rate: float = 1.0,
                 server_momentum: float = 0.9, **kw):
        super().__init__(layout, **kw)
        self.slr = float(server_learning_rate)
        self.mu = float(server_momentum)

    @torch.no_grad()
    def update(self, fedavg_flat, server_round, n_clients):
        sf = self.scaling(n_clients)
        g = self.params - sf * fedavg_flat
        v_new = self.params - self.slr * g
        self.params.copy_((1 + self.mu) * v_new - self.mu * self.m1)
        self.m1.copy_(v_new)
        return self._metrics(g)


class FedAdam(Strategy):
    name = "FEDADAM"
    state_keys = (STATE_PARAMS, STATE_M1, STATE_M2)

    def __init__(self, layout, eta: float = 1e-1, beta_1: float = 0.9,
                 beta_2: float = 0.99, tau: float = 1e-9, **kw):
        super().__init__(layout, **kw)
        self.eta, self.b1, self.b2, self.tau = (
            float(eta), float(beta_1), float(beta_2), float(tau),
        )

    @torch.no_grad()
    def update(self, fedavg_flat, server_round, n_clients):
        sf = self.scaling(n_clients)
        g = self.params - sf * fedavg_flat
        self.m1.mul_(self.b1).add_(g, alpha=1 - self.b1)
        self.m2.mul_(self.b2).addcmul_(g, g, value=1 - self.b2)
        bc1 = 1.0 / (1 - self.b1**server_round)
        bc2 = 1.0 / (1 - self.b2**server_round)
        self.params.add_(
            self.eta * (self.m1 * bc1) / ((self.m2 * bc2).sqrt() + self.tau)
        )
        return self._metrics(g)


class FedYogi(Strategy):
    name = "FEDYOGI"
    state_keys = (STATE_PARAMS, STATE_M1, STATE_M2)

    def __init__(self, layout, eta: float = 1e-2, beta_1: float = 0.9,
                 beta_2: float = 0.99, tau: float = 1e-3, **kw):
        super().__init__(layout, **kw)
        self.eta, self.b1, self.b2, self.tau = (
            float(eta), float(beta_1), float(beta_2), float(tau),
        )

    @torch.no_grad()
    def update(self, fedavg_flat, server_round, n_clients):
        sf = self.scaling(n_clients)
        g = self.params - sf * fedavg_flat
        self.m1.mul_(self.b1).add_(g, alpha=1 - self.b1)
        g2 = g * g
        self.m2.add_((1 - self.b2) * g2 * torch.sign(g2 - self.m2))
        bc1 = 1.0 / (1 - self.b1**server_round)
        bc2 = 1.0 / (1 - self.b2**server_round)
        self.params.add_(
            self.eta * (self.m1 * bc1) / ((self.m2 * bc2).sqrt() + self.tau)
        )
        return self._metrics(g)


_REGISTRY = {
    "FEDAVG": FedAvgEfficient,
    "NESTOROV": FedNesterov,
    "MOM": FedMom,
    "FEDADAM": FedAdam,
    "FEDYOGI": FedYogi,
}


def dispatch_strategy(name: str, layout: FlatParams, kwargs: dict | None = None) -> Strategy:
    """Reference dispatcher semantics (photon/strategy/dispatcher.py:44-165)."""
    cls = _REGISTRY.get(str(name).upper())
    if cls is None:
        raise ValueError(f"unknown strategy {name!r}; known: {sorted(_REGISTRY)}")
    return cls(layout, **dict(kwargs or {}))
