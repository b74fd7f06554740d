"""Optimizers: DecoupledAdamW and ADOPT, with fused multi-tensor HIP kernels.

Replaces the reference's torch fused AdamW + llm-foundry "adopt" optimizer
(mpt-125m.yaml:58-63). The GPU step is one multi-tensor HIP
kernel per dtype bucket updating p/m/v in a single pass (HBM-bound: read
p,g,m,v + write p,m,v). State tensors are fp32.

Both optimizers expose the momenta import/export hooks the federated
``aggregate_momenta`` path needs (reference set_optimizer_state,
photon/clients/utils.py:257-402): `export_momenta()` / `import_momenta()`
including the `step` injection that keeps Adam bias correction consistent.
"""

from __future__ import annotations

import math
from typing import Iterable

import torch
from torch.optim.optimizer import Optimizer

from . import hip_ext, use_hip


def _flat_grads_ok(params):
    return [p for p in params if p.grad is not None]


class _FusedStepMixin:
    """Shared fused multi-tensor step plumbing."""

    def _bucket(self, group):
        """Return (params, grads, m, v, masters) lists for params with
        grads. Non-fp32 params get an fp32 MASTER copy (PURE mixed
        precision: bf16 weights in the model, exact update state here)."""
        ps, gs, ms, vs, ws = [], [], [], [], []
        any_master = False
        for p in group["params"]:
            if p.grad is None:
                continue
            state = self.state[p]
            # NOT `len(state)==0`: sync_masters may have seeded only
            # {'master': ...} before the first step.
            if "exp_avg" not in state:
                state.setdefault("step", 0)
                state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
            if p.dtype != torch.float32 and "master" not in state:
                state["master"] = p.detach().to(torch.float32).clone()
            ps.append(p)
            gs.append(p.grad)
            ms.append(state["exp_avg"])
            vs.append(state["exp_avg_sq"])
            master = state.get("master")
            ws.append(master if master is not None else None)
            any_master = any_master or master is not None
        if any_master:
            # HIP path needs a uniform list; fp32 params use themselves
            ws = [w if w is not None else p.detach() for w, p in zip(ws, ps)]
        else:
            ws = []
        return ps, gs, ms, vs, ws

    def _bump_steps(self, ps):
        for p in ps:
            self.state[p]["step"] += 1
        return self.state[ps[0]]["step"] if ps else 0

    # -- federated momenta import/export hooks ------------------------------
    def export_momenta(self, order: list[torch.nn.Parameter]):
        """(exp_avg list, exp_avg_sq list) in the given parameter order."""
        m1, m2 = [], []
        for p in order:
            st = self.state.get(p, {})
            m1.append(st.get("exp_avg", torch.zeros_like(p, dtype=torch.float32)))
            m2.append(st.get("exp_avg_sq", torch.zeros_like(p, dtype=torch.float32)))
        return m1, m2

    def sync_masters(self, order: list[torch.nn.Parameter],
                     fp32_views: list[torch.Tensor]) -> None:
        """Overwrite fp32 masters (and the bf16 params) with exact fp32
        values — the federated round's parameter set must not round-trip
        through bf16 before entering the master copy."""
        for p, src in zip(order, fp32_views):
            st = self.state.setdefault(p, {})
            if p.dtype != torch.float32:
                st["master"] = src.detach().to(p.device, torch.float32) \
                    .view_as(p).clone()
            p.data.copy_(src.view_as(p).to(p.dtype))

    def import_momenta(
        self,
        order: list[torch.nn.Parameter],
        m1: list[torch.Tensor],
        m2: list[torch.Tensor],
        step: int | None = None,
    ):
        """Inject aggregated momenta (and optionally the step counter used
        for bias correction) — reference set_optimizer_state semantics."""
        for p, a, b in zip(order, m1, m2):
            st = self.state.setdefault(p, {})
            st["exp_avg"] = a.to(p.device, torch.float32).view_as(p).clone()
            st["exp_avg_sq"] = b.to(p.device, torch.float32).view_as(p).clone()
            if step is not None:
                st["step"] = int(step)
            elif "step" not in st:
                st["step"] = 0


class DecoupledAdamW(Optimizer, _FusedStepMixin):
    """AdamW with weight decay decoupled from the gradient update
    (Composer DecoupledAdamW semantics: decay scaled by lr)."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            ps, gs, ms, vs, ws = self._bucket(group)
            if not ps:
                continue
            step = self._bump_steps(ps)
            lr, (b1, b2), eps, wd = (
                group["lr"], group["betas"], group["eps"], group["weight_decay"],
            )
            bc1 = 1 - b1**step
            bc2 = 1 - b2**step
            if use_hip(ps[0]):
                hip_ext().adamw_step(ps, gs, ms, vs, lr, b1, b2, eps, wd,
                                     bc1, bc2, ws)
                continue
            for i, (p, g, m, v) in enumerate(zip(ps, gs, ms, vs)):
                gf = g.float()
                m.mul_(b1).add_(gf, alpha=1 - b1)
                v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
                denom = (v / bc2).sqrt_().add_(eps)
                target = ws[i] if ws else p
                if wd != 0.0:
                    target.mul_(1 - lr * wd)
                if ws:
                    target.addcdiv_(m / bc1, denom, value=-lr)
                    p.data.copy_(target.to(p.dtype))
                else:
                    p.addcdiv_((m / bc1).to(p.dtype), denom.to(p.dtype), value=-lr)
        return loss


class ADOPT(Optimizer, _FusedStepMixin):
    """ADOPT (Taniguchi et al., 2024): Adam variant that normalizes the
    gradient by the PREVIOUS second moment before the momentum update,
    with the clipped update rule (clip value step**0.25).

      v_0 = g_0^2
      t>=1:  c_t = clip(g_t / max(sqrt(v_{t-1}), eps), +-step^0.25)
             m_t = b1 m_{t-1} + (1-b1) c_t
             p_t = p_{t-1} - lr * m_t
             v_t = b2 v_{t-1} + (1-b2) g_t^2
    """

    def __init__(self, params, lr=6e-4, betas=(0.9, 0.9999), eps=1e-6, weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            ps, gs, ms, vs, ws = self._bucket(group)
            if not ps:
                continue
            step = self._bump_steps(ps)
            lr, (b1, b2), eps, wd = (
                group["lr"], group["betas"], group["eps"], group["weight_decay"],
            )
            clip = (step - 1) ** 0.25 if step > 1 else 1.0
            if use_hip(ps[0]):
                hip_ext().adopt_step(ps, gs, ms, vs, lr, b1, b2, eps, wd,
                                     float(clip), step, ws)
                continue
            for i, (p, g, m, v) in enumerate(zip(ps, gs, ms, vs)):
                gf = g.float()
                if step == 1:
                    # v_0 = g_0^2; no parameter update on the first step
                    v.copy_(gf * gf)
                    continue
                c = gf / v.sqrt().clamp_min(eps)
                c.clamp_(-clip, clip)
                m.mul_(b1).add_(c, alpha=1 - b1)
                target = ws[i] if ws else p
                if wd != 0.0:
                    target.mul_(1 - lr * wd)
                if ws:
                    target.add_(m, alpha=-lr)
                    p.data.copy_(target.to(p.dtype))
                else:
                    p.add_(m.to(p.dtype), alpha=-lr)
                v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
        return loss


def build_optimizer(cfg: dict, params: Iterable[torch.nn.Parameter]) -> Optimizer:
    name = str(cfg.get("name", "decoupled_adamw")).lower()
    kwargs = dict(
        lr=float(cfg.get("lr", 1e-4)),
        betas=tuple(cfg.get("betas", (0.9, 0.95))),
        eps=float(cfg.get("eps", 1e-8)),
        weight_decay=float(cfg.get("weight_decay", 0.0)),
    )
    if name == "adopt":
        return ADOPT(params, **kwargs)
    if name in ("decoupled_adamw", "adamw"):
        return DecoupledAdamW(params, **kwargs)
    raise ValueError(f"unknown optimizer {name!r}")
